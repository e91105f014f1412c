"""RBAC / orgs / runner profiles / oauth tests (reference
compatibility_test.go + authz + gpucloud scenarios at unit scale)."""
import pytest
from fastapi.testclient import TestClient

from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import MockClient, ProviderManager
from helix_amd.server.runner_profiles import (ProfileGPURequirement,
                                              compatibility,
                                              filter_compatible,
                                              RunnerProfile)
from helix_amd.server.types import GPUStatus
from helix_amd.store import Store


@pytest.fixture()
def stack(tmp_path):
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.filestore.path = str(tmp_path / "fs")
    store = Store(":memory:")
    pm = ProviderManager(store)
    pm.register("mock", MockClient())
    app = create_app(cfg, store=store, providers=pm)
    client = TestClient(app)
    keys = {}
    for name in ("alice", "bob"):
        r = client.post("/api/v1/users", json={"username": name},
                        headers={"Authorization": "Bearer admin-key"})
        keys[name] = r.json()["api_key"]
        keys[name + "_id"] = r.json()["id"]
    return app, client, keys, store


def H(key):
    return {"Authorization": f"Bearer {key}"}


MI355 = GPUStatus(index=0, vendor="amd", arch="cdna4", name="MI355X",
                  total_memory=288 << 30)
A100 = GPUStatus(index=0, vendor="nvidia", arch="ampere", name="A100",
                 total_memory=80 << 30)


def test_compatibility_matrix():
    req = ProfileGPURequirement(count=1, vendor="amd",
                                architectures=["cdna4"],
                                min_vram_bytes=200 << 30)
    assert compatibility(req, [MI355])[0]
    ok, why = compatibility(req, [A100])
    assert not ok and "found 0" in why
    # count requirement
    req8 = ProfileGPURequirement(count=8, vendor="amd")
    assert not compatibility(req8, [MI355])[0]
    assert compatibility(req8, [MI355] * 8)[0]
    # model_match
    reqm = ProfileGPURequirement(model_match="mi355")
    assert compatibility(reqm, [MI355])[0]
    assert not compatibility(reqm, [A100])[0]


def test_filter_compatible():
    p1 = RunnerProfile(name="amd", gpu_requirement=ProfileGPURequirement(
        vendor="amd"))
    p2 = RunnerProfile(name="nv", gpu_requirement=ProfileGPURequirement(
        vendor="nvidia"))
    out = filter_compatible([p1, p2], [MI355])
    assert [p.name for p in out] == ["amd"]


def test_profile_assignment_flow(stack):
    app, client, keys, _ = stack
    # runner heartbeats with MI355X
    client.post("/api/v1/runner/heartbeat", json={
        "runner_id": "r1", "address": "http://r1:8090",
        "gpus": [MI355.model_dump()], "models": []},
        headers=H("runner-token"))
    # create a profile requiring cdna4
    r = client.post("/api/v1/runner-profiles", json={
        "name": "llama-profile",
        "models": [{"name": "llama3-8b", "preset": "llama3-8b"}],
        "gpu_requirement": {"count": 1, "vendor": "amd",
                            "architectures": ["cdna4"]}},
        headers=H("admin-key"))
    pid = r.json()["id"]
    # assign (compatible)
    r = client.post("/api/v1/runners/r1/assign-profile",
                    json={"profile_id": pid}, headers=H("admin-key"))
    assert r.status_code == 200
    # runner polls assignment
    r = client.get("/api/v1/runner/r1/assignment",
                   headers=H("runner-token"))
    assert r.json()["name"] == "llama-profile"
    # incompatible rejection (gpucloud incompatible_rejection scenario)
    r = client.post("/api/v1/runner-profiles", json={
        "name": "nv-profile",
        "gpu_requirement": {"count": 1, "vendor": "nvidia"}},
        headers=H("admin-key"))
    r2 = client.post("/api/v1/runners/r1/assign-profile",
                     json={"profile_id": r.json()["id"]},
                     headers=H("admin-key"))
    assert r2.status_code == 409
    # clear profile (clear_profile scenario)
    r = client.request("DELETE", "/api/v1/runners/r1/assignment",
                       headers=H("admin-key"))
    assert r.json()["ok"]
    assert client.get("/api/v1/runner/r1/assignment",
                      headers=H("runner-token")).json() == {}


def test_org_team_grants(stack):
    app, client, keys, store = stack
    rbac = app.state.rbac
    r = client.post("/api/v1/organizations", json={"name": "acme"},
                    headers=H(keys["alice"]))
    oid = r.json()["id"]
    assert client.get("/api/v1/organizations",
                      headers=H(keys["alice"])).json()[0]["role"] == "owner"
    # team with bob
    r = client.post(f"/api/v1/organizations/{oid}/teams",
                    json={"name": "eng"}, headers=H(keys["alice"]))
    tid = r.json()["id"]
    client.post(f"/api/v1/teams/{tid}/members",
                json={"user_id": keys["bob_id"]}, headers=H(keys["alice"]))
    # grant team access to a resource
    client.post("/api/v1/access-grants", json={
        "resource_type": "app", "resource_id": "app_1",
        "role": "member", "team_id": tid}, headers=H(keys["alice"]))
    assert rbac.authorize(keys["bob_id"], "app", "app_1", "use")
    assert not rbac.authorize(keys["bob_id"], "app", "app_1", "admin")
    assert rbac.authorize(keys["alice_id"], "app", "app_1", "admin",
                          resource_owner=keys["alice_id"])


def test_oauth_flow(stack):
    app, client, keys, _ = stack
    client.post("/api/v1/oauth/providers", json={
        "name": "github", "client_id": "cid", "client_secret": "sec"},
        headers=H("admin-key"))
    r = client.get("/api/v1/oauth/github/authorize-url",
                   params={"redirect_uri": "http://cb"},
                   headers=H(keys["alice"]))
    assert "github.com/login/oauth/authorize" in r.json()["url"]
    assert "client_id=cid" in r.json()["url"]
    # store a token directly and read it back via manager
    client.post("/api/v1/oauth/github/token",
                json={"access_token": "tok123", "expires_in": 3600},
                headers=H(keys["alice"]))
    import asyncio
    token = asyncio.run(app.state.oauth.get_valid_token(
        keys["alice_id"], "github"))
    assert token == "tok123"


def test_provider_endpoints_resolution(stack):
    app, client, keys, _ = stack
    client.post("/api/v1/provider-endpoints", json={
        "name": "myvllm", "base_url": "http://my:8000/v1",
        "api_key": "k"}, headers=H(keys["alice"]))
    r = client.get("/api/v1/provider-endpoints", headers=H(keys["alice"]))
    assert r.json()[0]["name"] == "myvllm"
    pm = app.state.providers
    provider, model = pm.resolve("myvllm/some-model", "mock",
                                 keys["alice_id"])
    assert provider == "myvllm" and model == "some-model"
    client2 = pm.get_client("myvllm", keys["alice_id"])
    assert client2.provider == "myvllm"
