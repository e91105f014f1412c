"""helix.yaml parsing + CLI-level behaviors."""
import textwrap

import pytest

from helix_amd.server.apps import parse_helix_yaml


def test_parse_plain_form(tmp_path):
    y = textwrap.dedent("""
        name: My App
        description: demo
        assistants:
          - name: default
            model: llama3-8b
            provider: helix
            system_prompt: Be helpful.
            temperature: 0.5
            knowledge:
              - name: docs
                source:
                  text: "inline knowledge"
            apis:
              - name: petstore
                url: https://petstore.example.com
                schema: '{"paths": {}}'
            tests:
              - name: smoke
                steps:
                  - prompt: hello
                    expected_output: a greeting
    """)
    cfg = parse_helix_yaml(y)
    assert cfg.name == "My App"
    a = cfg.assistants[0]
    assert a.model == "llama3-8b"
    assert a.temperature == 0.5
    assert a.knowledge[0].name == "docs"
    assert a.apis[0].name == "petstore"
    assert a.tests[0].steps[0]["prompt"] == "hello"


def test_parse_crd_form():
    y = textwrap.dedent("""
        apiVersion: app.aispec.org/v1alpha1
        kind: AIApp
        metadata:
          name: crd-app
        spec:
          description: from crd
          assistants:
            - name: a
              model: m
    """)
    cfg = parse_helix_yaml(y)
    assert cfg.name == "crd-app"
    assert cfg.assistants[0].model == "m"


def test_file_ref_inlining(tmp_path):
    (tmp_path / "prompt.txt").write_text("You are from a file.")
    y = textwrap.dedent("""
        name: f
        assistants:
          - name: a
            model: m
            system_prompt: file://prompt.txt
    """)
    cfg = parse_helix_yaml(y, base_dir=str(tmp_path))
    assert cfg.assistants[0].system_prompt == "You are from a file."


def test_cli_version():
    from typer.testing import CliRunner
    from helix_amd.cli import app
    res = CliRunner().invoke(app, ["version"])
    assert res.exit_code == 0
    assert "helix_amd" in res.output


def test_cli_doctor_runs():
    import subprocess
    import sys
    r = subprocess.run([sys.executable, "-m", "helix_amd.cli", "doctor"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "extension built" in r.stdout


def test_helix_yaml_triggers_register(tmp_path):
    """helix.yaml `triggers:` are reconciled into the trigger manager
    on app create/update/delete (reference apply behavior)."""
    from fastapi.testclient import TestClient

    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        key = auth.create_api_key(auth.create_user("u")["id"])
        H = {"Authorization": f"Bearer {key}"}
        r = client.post("/api/v1/apps", headers=H, json={"config": {
            "name": "cronapp", "assistants": [{"name": "a"}],
            "triggers": [{"kind": "cron",
                          "config": {"schedule": "0 9 * * *",
                                     "prompt": "daily digest"}}]}})
        assert r.status_code == 200, r.text
        aid = r.json()["id"]
        trs = client.get("/api/v1/triggers", headers=H).json()
        assert [t["kind"] for t in trs] == ["cron"]
        assert trs[0]["app_id"] == aid
        # update swaps the trigger set atomically
        client.put(f"/api/v1/apps/{aid}", headers=H, json={"config": {
            "name": "cronapp", "assistants": [{"name": "a"}],
            "triggers": [{"kind": "webhook", "config": {}}]}})
        trs = client.get("/api/v1/triggers", headers=H).json()
        assert [t["kind"] for t in trs] == ["webhook"]
        # bad cron rejected with 400
        r = client.put(f"/api/v1/apps/{aid}", headers=H, json={"config": {
            "name": "cronapp", "assistants": [{"name": "a"}],
            "triggers": [{"kind": "cron",
                          "config": {"schedule": "junk"}}]}})
        assert r.status_code == 400
        # delete clears app-config triggers
        client.delete(f"/api/v1/apps/{aid}", headers=H)
        assert client.get("/api/v1/triggers", headers=H).json() == []


def test_app_scoped_api_keys(tmp_path):
    """App API keys (reference client/app.go:44 GetAppAPIKeys):
    requests made with an app key default to that app's assistant."""
    import json as _json

    from fastapi.testclient import TestClient

    from helix_amd.server.app import create_app
    from helix_amd.server.config import ServerConfig
    from helix_amd.server.providers import MockClient, ProviderManager
    from helix_amd.store import Store
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    cfg.filestore.path = str(tmp_path / "fs")
    store = Store(":memory:")
    pm = ProviderManager(store)
    pm.register("mock", MockClient())
    app = create_app(cfg, store=store, providers=pm)
    with TestClient(app) as client:
        auth = app.state.auth
        me = auth.create_user("owner-ak")
        key = auth.create_api_key(me["id"])
        H = {"Authorization": f"Bearer {key}"}
        r = client.post("/api/v1/apps", headers=H, json={"config": {
            "name": "sys-app", "assistants": [
                {"name": "a", "system_prompt": "ALWAYS SAY BANANA"}]}})
        aid = r.json()["id"]
        akey = client.post(f"/api/v1/apps/{aid}/keys", headers=H,
                           json={}).json()["key"]
        r = client.post("/v1/chat/completions",
                        headers={"Authorization": f"Bearer {akey}"},
                        json={"model": "mock-model",
                              "messages": [{"role": "user",
                                            "content": "hi"}]})
        assert r.status_code == 200, r.text
        calls = store.list("llm_calls", limit=5)
        assert any("BANANA" in _json.dumps(c) for c in calls)
        # listing shows the key (truncated), other users cannot mint
        assert len(client.get(f"/api/v1/apps/{aid}/keys",
                              headers=H).json()) == 1
        intruder = auth.create_api_key(
            auth.create_user("intruder-ak")["id"])
        r = client.post(f"/api/v1/apps/{aid}/keys",
                        headers={"Authorization": f"Bearer {intruder}"},
                        json={})
        assert r.status_code in (403, 404)
