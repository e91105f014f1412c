"""Evaluation suites (reference api/pkg evaluations: LLM-judged app
tests with persisted runs)."""
import pytest
from fastapi.testclient import TestClient

from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import MockClient, ProviderManager
from helix_amd.store import Store


def _stack(tmp_path, responses):
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    cfg.filestore.path = str(tmp_path / "fs")
    store = Store(":memory:")
    pm = ProviderManager(store)
    pm.register("mock", MockClient(responses=responses))
    app = create_app(cfg, store=store, providers=pm)
    client = TestClient(app)
    key = client.post("/api/v1/users", json={"username": "eva"},
                      headers={"Authorization": "Bearer admin-key"}
                      ).json()["api_key"]
    H = {"Authorization": f"Bearer {key}"}
    app_id = client.post("/api/v1/apps", json={"config": {
        "name": "eval-app", "assistants": [{"name": "a"}]}},
        headers=H).json()["id"]
    return client, H, store, app_id


def test_suite_run_pass_and_fail(tmp_path):
    # responses: answer1, judge1(YES), answer2, judge2(NO)
    client, H, store, app_id = _stack(tmp_path, [
        "Paris is the capital.", "YES", "I do not know.", "NO"])
    suite = client.post(f"/api/v1/apps/{app_id}/evaluation-suites", json={
        "name": "geo", "tests": [
            {"name": "capital", "steps": [
                {"prompt": "Capital of France?",
                 "expected_output": "Paris"}]},
            {"name": "hard", "steps": [
                {"prompt": "Answer the unanswerable",
                 "expected_output": "42"}]},
        ]}, headers=H).json()
    assert suite["id"]
    assert client.get("/api/v1/evaluation-suites",
                      headers=H).json()[0]["name"] == "geo"
    run = client.post(f"/api/v1/evaluation-suites/{suite['id']}/runs",
                      headers=H).json()
    assert run["state"] == "complete"
    assert run["total"] == 2 and run["passed"] == 1
    r0, r1 = run["results"]
    assert r0["passed"] is True and "Paris" in r0["answer"]
    assert r1["passed"] is False
    # persisted and fetchable
    got = client.get(f"/api/v1/evaluation-runs/{run['id']}",
                     headers=H).json()
    assert got["passed"] == 1
    assert client.get("/api/v1/evaluation-runs/nope",
                      headers=H).status_code == 404


def test_suite_run_survives_provider_error(tmp_path):
    def boom(req):
        raise RuntimeError("provider down")
    client, H, _, app_id = _stack(tmp_path, [boom])
    suite = client.post(f"/api/v1/apps/{app_id}/evaluation-suites", json={
        "name": "s", "tests": [{"name": "t", "steps": [
            {"prompt": "p", "expected_output": "x"}]}]},
        headers=H).json()
    run = client.post(f"/api/v1/evaluation-suites/{suite['id']}/runs",
                      headers=H).json()
    assert run["state"] == "complete"
    assert run["passed"] == 0
    assert "error" in run["results"][0]["verdict"]
