"""Org runtime: positions/bots/streams with message fan-out to bots
(reference "helix-org" api/pkg/org runtime)."""
import pytest
from fastapi.testclient import TestClient

from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import MockClient, ProviderManager
from helix_amd.store import Store


@pytest.fixture()
def stack(tmp_path):
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    cfg.filestore.path = str(tmp_path / "fs")
    store = Store(":memory:")
    pm = ProviderManager(store)
    pm.register("mock", MockClient())
    app = create_app(cfg, store=store, providers=pm)
    client = TestClient(app)
    r = client.post("/api/v1/users", json={"username": "alice"},
                    headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    return client, key, store


def H(key):
    return {"Authorization": f"Bearer {key}"}


def _mk_org(client, key):
    return client.post("/api/v1/organizations", json={"name": "acme"},
                       headers=H(key)).json()["id"]


def test_position_bot_stream_crud(stack):
    client, key, _ = stack
    oid = _mk_org(client, key)
    pos = client.post(f"/api/v1/organizations/{oid}/positions",
                      json={"name": "support", "role": "worker",
                            "system_prompt": "Be helpful."},
                      headers=H(key)).json()
    assert pos["org_id"] == oid
    bots = client.post(f"/api/v1/organizations/{oid}/bots",
                       json={"name": "helper", "position_id": pos["id"]},
                       headers=H(key)).json()
    assert bots["position_id"] == pos["id"]
    strm = client.post(f"/api/v1/organizations/{oid}/streams",
                       json={"name": "general"}, headers=H(key)).json()
    assert client.get(f"/api/v1/organizations/{oid}/positions",
                      headers=H(key)).json()[0]["name"] == "support"
    assert client.get(f"/api/v1/organizations/{oid}/bots",
                      headers=H(key)).json()[0]["name"] == "helper"
    assert client.get(f"/api/v1/organizations/{oid}/streams",
                      headers=H(key)).json()[0]["name"] == "general"
    # bad position -> 400
    r = client.post(f"/api/v1/organizations/{oid}/bots",
                    json={"name": "x", "position_id": "nope"},
                    headers=H(key))
    assert r.status_code == 400
    # non-member denied
    r2 = client.post("/api/v1/users", json={"username": "mallory"},
                     headers={"Authorization": "Bearer admin-key"})
    mkey = r2.json()["api_key"]
    r = client.get(f"/api/v1/organizations/{oid}/bots", headers=H(mkey))
    assert r.status_code == 403
    assert strm["org_id"] == oid


def test_stream_message_fans_out_to_subscribed_bots(stack):
    client, key, store = stack
    oid = _mk_org(client, key)
    pos = client.post(f"/api/v1/organizations/{oid}/positions",
                      json={"name": "support",
                            "system_prompt": "Answer tersely."},
                      headers=H(key)).json()
    bot = client.post(f"/api/v1/organizations/{oid}/bots",
                      json={"name": "helper", "position_id": pos["id"]},
                      headers=H(key)).json()
    strm = client.post(f"/api/v1/organizations/{oid}/streams",
                       json={"name": "general"}, headers=H(key)).json()
    other = client.post(f"/api/v1/organizations/{oid}/streams",
                        json={"name": "random"}, headers=H(key)).json()
    client.post(f"/api/v1/bots/{bot['id']}/subscribe",
                json={"stream_id": strm["id"]}, headers=H(key))

    msg = client.post(f"/api/v1/streams/{strm['id']}/messages",
                      json={"text": "What is our refund policy?"},
                      headers=H(key)).json()
    assert len(msg["replies"]) == 1
    assert msg["replies"][0]["bot"] == "helper"
    assert msg["replies"][0]["text"] == "mock response"
    # bot session was created and turn counted
    b = store.get("org_bots", bot["id"])
    assert b["turns"] == 1
    # unsubscribed stream does not fan out
    msg2 = client.post(f"/api/v1/streams/{other['id']}/messages",
                       json={"text": "hello?"}, headers=H(key)).json()
    assert msg2["replies"] == []
    # message history is ordered and complete
    hist = client.get(f"/api/v1/streams/{strm['id']}/messages",
                      headers=H(key)).json()
    assert [m["text"] for m in hist] == ["What is our refund policy?"]


def test_org_primitives_via_mcp(stack):
    """Org streams exposed as MCP tools: list, post (bot replies), read."""
    client, key, store = stack
    oid = _mk_org(client, key)
    pos = client.post(f"/api/v1/organizations/{oid}/positions",
                      json={"name": "support"}, headers=H(key)).json()
    bot = client.post(f"/api/v1/organizations/{oid}/bots",
                      json={"name": "helper", "position_id": pos["id"]},
                      headers=H(key)).json()
    strm = client.post(f"/api/v1/organizations/{oid}/streams",
                       json={"name": "general"}, headers=H(key)).json()
    client.post(f"/api/v1/bots/{bot['id']}/subscribe",
                json={"stream_id": strm["id"]}, headers=H(key))
    app_id = client.post("/api/v1/apps", json={"config": {
        "name": "mcp-app", "assistants": [{"name": "a"}]}},
        headers=H(key)).json()["id"]

    def rpc(method, params=None, rid=1):
        return client.post(f"/api/v1/mcp/{app_id}", json={
            "jsonrpc": "2.0", "id": rid, "method": method,
            "params": params or {}}, headers=H(key)).json()

    tools = [t["name"] for t in rpc("tools/list")["result"]["tools"]]
    assert {"org_list_streams", "org_read_stream",
            "org_post_message"} <= set(tools)
    import json as _json
    r = rpc("tools/call", {"name": "org_list_streams", "arguments": {}})
    streams = _json.loads(r["result"]["content"][0]["text"])
    assert streams and streams[0]["name"] == "general"
    r = rpc("tools/call", {"name": "org_post_message", "arguments": {
        "stream_id": strm["id"], "text": "hello from mcp"}})
    posted = _json.loads(r["result"]["content"][0]["text"])
    assert posted["replies"][0]["text"] == "mock response"
    r = rpc("tools/call", {"name": "org_read_stream", "arguments": {
        "stream_id": strm["id"]}})
    hist = _json.loads(r["result"]["content"][0]["text"])
    assert hist[0]["text"] == "hello from mcp"


def _mk_bot(client, key, oid, name):
    pos = client.post(f"/api/v1/organizations/{oid}/positions",
                      json={"name": f"pos-{name}"},
                      headers=H(key)).json()
    return client.post(f"/api/v1/organizations/{oid}/bots",
                       json={"name": name, "position_id": pos["id"]},
                       headers=H(key)).json()


def test_reporting_lines_dag_and_chart(stack):
    """Cycle-guarded reporting-line DAG + chart view (reference
    org QA.md mental model: org_reporting_lines, ReactFlow chart)."""
    client, key, _ = stack
    oid = _mk_org(client, key)
    root = _mk_bot(client, key, oid, "b-root")
    eng = _mk_bot(client, key, oid, "b-eng")
    dev = _mk_bot(client, key, oid, "b-dev")
    # dev -> eng -> root
    r = client.put(f"/api/v1/bots/{eng['id']}/parents",
                   json={"parent_ids": [root["id"]]}, headers=H(key))
    assert r.status_code == 200
    client.put(f"/api/v1/bots/{dev['id']}/parents",
               json={"parent_ids": [eng["id"]]}, headers=H(key))
    # closing the cycle root -> dev is rejected
    r = client.put(f"/api/v1/bots/{root['id']}/parents",
                   json={"parent_ids": [dev["id"]]}, headers=H(key))
    assert r.status_code == 400 and "cycle" in r.json()["detail"]
    # self-report rejected
    r = client.put(f"/api/v1/bots/{root['id']}/parents",
                   json={"parent_ids": [root["id"]]}, headers=H(key))
    assert r.status_code == 400
    chart = client.get(f"/api/v1/organizations/{oid}/chart",
                       headers=H(key)).json()
    assert len(chart["nodes"]) == 3
    assert {"manager": root["id"], "report": eng["id"]} in chart["edges"]
    assert {"manager": eng["id"], "report": dev["id"]} in chart["edges"]
    # deleting eng cascades its lines (dev loses its manager)
    client.delete(f"/api/v1/bots/{eng['id']}", headers=H(key))
    chart = client.get(f"/api/v1/organizations/{oid}/chart",
                       headers=H(key)).json()
    assert len(chart["nodes"]) == 2 and chart["edges"] == []


def test_escalation_and_audit(stack):
    client, key, store = stack
    oid = _mk_org(client, key)
    mgr = _mk_bot(client, key, oid, "b-mgr")
    worker = _mk_bot(client, key, oid, "b-worker")
    client.put(f"/api/v1/bots/{worker['id']}/parents",
               json={"parent_ids": [mgr["id"]]}, headers=H(key))
    r = client.post(f"/api/v1/bots/{worker['id']}/escalate",
                    json={"text": "prod is down"}, headers=H(key))
    assert r.status_code == 200, r.text
    out = r.json()
    assert out and out[0]["manager"] == "b-mgr"
    assert out[0]["reply"]              # mock model replied
    # activation landed in the audit log
    audit = client.get(f"/api/v1/organizations/{oid}/audit",
                       headers=H(key)).json()
    assert audit and audit[0]["bot_name"] == "b-mgr"
    assert audit[0]["ok"] is True
    assert audit[0]["duration_ms"] >= 0
