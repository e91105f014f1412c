"""Platform services: catalog, usage/wallet, triggers, filestore,
Anthropic surface, weight IO."""
import asyncio
import time

import pytest
import torch
from fastapi.testclient import TestClient

from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import MockClient, ProviderManager
from helix_amd.server.triggers import CronSchedule
from helix_amd.store import Store


@pytest.fixture()
def stack(tmp_path):
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    cfg.filestore.path = str(tmp_path / "fs")
    store = Store(":memory:")
    pm = ProviderManager(store)
    mock = MockClient()
    pm.register("mock", mock)
    app = create_app(cfg, store=store, providers=pm)
    client = TestClient(app)
    r = client.post("/api/v1/users", json={"username": "alice"},
                    headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    return app, client, mock, key, store


def H(key):
    return {"Authorization": f"Bearer {key}"}


def test_model_catalog(stack):
    _, client, _, key, _ = stack
    r = client.get("/api/v1/helix-models", headers=H(key))
    ids = [m["id"] for m in r.json()]
    assert "llama3-8b" in ids and "bge-base" in ids
    r = client.get("/api/v1/model-info/llama3-8b", headers=H(key))
    assert r.json()["context_length"] == 8192
    # admin override
    r = client.put("/api/v1/model-info/llama3-8b",
                   json={"context_length": 16384},
                   headers=H("admin-key"))
    assert r.json()["context_length"] == 16384


def test_usage_metering_and_wallet(stack):
    app, client, _, key, store = stack
    uid = store.list("users")[0]["id"]
    client.post("/api/v1/wallet/topup", json={"owner": uid,
                                              "amount_usd": 10.0},
                headers=H("admin-key"))
    # mock-model has no price -> usage rows but no debit
    client.post("/v1/chat/completions", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "hi"}]}, headers=H(key))
    r = client.get("/api/v1/usage", headers=H(key))
    rows = r.json()
    assert rows and rows[0]["prompt_tokens"] == 7
    assert rows[0]["calls"] == 1
    w = client.get("/api/v1/wallet", headers=H(key)).json()
    assert w["balance_usd"] == 10.0


def test_cron_schedule():
    s = CronSchedule("*/15 9-17 * * 1-5")
    t = time.struct_time((2026, 9, 7, 10, 30, 0, 0, 250, -1))  # Monday
    assert s.matches(t)
    t2 = time.struct_time((2026, 9, 6, 10, 30, 0, 6, 249, -1))  # Sunday
    assert not s.matches(t2)
    assert not s.matches(time.struct_time((2026, 9, 7, 10, 7, 0, 0, 250, -1)))
    with pytest.raises(ValueError):
        CronSchedule("* * *")


def test_trigger_fire_and_webhook(stack):
    app, client, _, key, store = stack
    r = client.post("/api/v1/triggers", json={
        "kind": "webhook",
        "config": {"prompt": "handle the webhook"}}, headers=H(key))
    tid = r.json()["id"]
    r = client.post(f"/api/v1/webhooks/{tid}", json={"event": "push"})
    assert r.status_code == 200
    sid = r.json()["session_id"]
    its = store.list("interactions", parent=sid)
    assert its and its[0]["response_message"] == "mock response"
    assert "push" in its[0]["prompt_message"]


def test_trigger_cron_validation(stack):
    _, client, _, key, _ = stack
    r = client.post("/api/v1/triggers", json={
        "kind": "cron", "config": {"schedule": "bogus"}}, headers=H(key))
    assert r.status_code == 400
    r = client.post("/api/v1/triggers", json={
        "kind": "cron", "config": {"schedule": "0 9 * * *",
                                   "prompt": "daily"}}, headers=H(key))
    assert r.status_code == 200


def test_filestore_roundtrip(stack):
    _, client, _, key, _ = stack
    r = client.put("/api/v1/filestore/upload?path=docs/a.txt",
                   content=b"hello file", headers=H(key))
    assert r.status_code == 200
    assert r.json()["size"] == 10
    r = client.get("/api/v1/filestore/list?path=docs", headers=H(key))
    assert r.json()[0]["path"].endswith("a.txt")
    r = client.get("/api/v1/filestore/download?path=docs/a.txt",
                   headers=H(key))
    assert r.content == b"hello file"
    r = client.request("DELETE", "/api/v1/filestore?path=docs/a.txt",
                       headers=H(key))
    assert r.json()["ok"]


def test_filestore_path_escape(stack):
    _, client, _, key, _ = stack
    r = client.put("/api/v1/filestore/upload?path=../../etc/passwd",
                   content=b"x", headers=H(key))
    assert r.status_code >= 400


def test_anthropic_messages(stack):
    _, client, _, key, _ = stack
    r = client.post("/v1/messages", json={
        "model": "mock-model", "max_tokens": 32,
        "system": "be terse",
        "messages": [{"role": "user", "content": "hello"}]}, headers=H(key))
    assert r.status_code == 200
    body = r.json()
    assert body["type"] == "message"
    assert body["content"][0]["text"] == "mock response"
    assert body["stop_reason"] == "end_turn"
    assert body["usage"]["input_tokens"] == 7


def test_anthropic_streaming(stack):
    _, client, _, key, _ = stack
    with client.stream("POST", "/v1/messages", json={
        "model": "mock-model", "max_tokens": 32, "stream": True,
        "messages": [{"role": "user", "content": "hello"}]},
            headers=H(key)) as r:
        text = "".join(r.iter_text())
    assert "message_start" in text
    assert "content_block_delta" in text
    assert "message_stop" in text
    assert "mock response".split()[0] in text


def test_weight_io_roundtrip(tmp_path):
    from helix_amd.engine.weights import load_llama_weights, save_sharded
    from helix_amd.models.llama import PRESETS, LlamaForCausalLM
    torch.manual_seed(0)
    m1 = LlamaForCausalLM(PRESETS["tiny"]).float()
    m1.init_random(1)
    save_sharded(m1, str(tmp_path / "ckpt"), shard_bytes=1 << 20)
    m2 = LlamaForCausalLM(PRESETS["tiny"]).float()
    m2.init_random(2)
    load_llama_weights(m2, str(tmp_path / "ckpt"))
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(),
                                  m2.named_parameters()):
        assert torch.equal(p1, p2), n1


def test_weight_io_hf_names(tmp_path):
    """HF-style checkpoint names map onto fused projections."""
    from safetensors.torch import save_file
    from helix_amd.engine.weights import load_llama_weights
    from helix_amd.models.llama import PRESETS, LlamaForCausalLM
    torch.manual_seed(3)
    cfg = PRESETS["tiny"]
    m = LlamaForCausalLM(cfg).float()
    m.init_random(3)
    q, kv, h = cfg.q_size, cfg.kv_size, cfg.hidden_size
    sd = {}
    qw = torch.randn(q, h)
    kw = torch.randn(kv, h)
    vw = torch.randn(kv, h)
    sd["model.layers.0.self_attn.q_proj.weight"] = qw
    sd["model.layers.0.self_attn.k_proj.weight"] = kw
    sd["model.layers.0.self_attn.v_proj.weight"] = vw
    save_file(sd, str(tmp_path / "hf.safetensors"))
    load_llama_weights(m, str(tmp_path))
    fused = m.layers[0].attn.qkv_proj.weight.data
    assert torch.equal(fused[:q], qw)
    assert torch.equal(fused[q:q + kv], kw)
    assert torch.equal(fused[q + kv:], vw)


def test_quota_enforced(stack):
    app, client, _, key, store = stack
    app.state.cfg.daily_token_limit = 10
    # first call consumes 12 tokens (mock usage), second must 429
    r = client.post("/v1/chat/completions", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "a"}]}, headers=H(key))
    assert r.status_code == 200
    r = client.post("/v1/chat/completions", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "b"}]}, headers=H(key))
    assert r.status_code == 429
    assert "quota" in r.json()["error"]["message"]


def test_reasoning_effort_passthrough(stack):
    app, client, mock, key, _ = stack
    r = client.post("/api/v1/apps", json={"config": {
        "name": "re app",
        "assistants": [{"name": "a", "model": "mock-model",
                        "provider": "mock",
                        "reasoning_effort": "high"}]}}, headers=H(key))
    app_id = r.json()["id"]
    client.post("/v1/chat/completions", json={
        "app_id": app_id,
        "messages": [{"role": "user", "content": "x"}]}, headers=H(key))
    assert mock.calls[-1]["reasoning_effort"] == "high"


def test_prometheus_metrics(stack):
    _, client, _, key, _ = stack
    # generate one chat call so llm metrics have samples
    r = client.post("/v1/chat/completions", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "hi"}]}, headers=H(key))
    assert r.status_code == 200
    m = client.get("/metrics")
    assert m.status_code == 200
    text = m.text
    assert "helix_http_requests_total" in text
    assert "helix_llm_calls_total" in text
    assert 'model="mock-model"' in text
    assert "helix_runners_online" in text


def test_openapi_spec_served(stack):
    _, client, _, _, _ = stack
    r = client.get("/openapi.json")
    assert r.status_code == 200
    paths = r.json()["paths"]
    assert "/v1/chat/completions" in paths
    assert len(paths) > 80


def test_janitor_prunes_old_rows(stack):
    app, client, _, key, store = stack
    import time as _t
    H_admin = {"Authorization": "Bearer admin-key"}
    old_ms = int((_t.time() - 90 * 86400) * 1000)
    store.put("llm_calls", "old1", {"id": "old1", "created": old_ms})
    store.put("llm_calls", "new1", {"id": "new1",
                                    "created": int(_t.time() * 1000)})
    store.put("usage_metrics", "um1", {"id": "um1",
                                       "ts": _t.time() - 90 * 86400})
    r = client.post("/api/v1/admin/janitor",
                    json={"retention_days": 30}, headers=H_admin)
    assert r.status_code == 200
    pruned = r.json()
    assert pruned["llm_calls"] == 1 and pruned["usage_metrics"] == 1
    assert store.get("llm_calls", "old1") is None
    assert store.get("llm_calls", "new1") is not None
    # non-admin denied
    r = client.post("/api/v1/admin/janitor", json={}, headers=H(key))
    assert r.status_code in (401, 403)


def test_debug_threads(stack):
    _, client, _, _, _ = stack
    r = client.get("/debug/threads",
                   headers={"Authorization": "Bearer admin-key"})
    assert r.status_code == 200
    assert any("MainThread" in k for k in r.json())


def test_anthropic_tool_use_translation():
    """/v1/messages tool-use round-trip translation (reference
    api/pkg/anthropic proxies tool blocks)."""
    from helix_amd.server.anthropic_api import (anthropic_to_openai,
                                                openai_to_anthropic)
    areq = {
        "model": "m", "max_tokens": 64,
        "tools": [{"name": "get_weather",
                   "description": "Weather lookup",
                   "input_schema": {"type": "object", "properties": {
                       "city": {"type": "string"}}}}],
        "tool_choice": {"type": "any"},
        "messages": [
            {"role": "user", "content": "Weather in Paris?"},
            {"role": "assistant", "content": [
                {"type": "tool_use", "id": "toolu_1",
                 "name": "get_weather", "input": {"city": "Paris"}}]},
            {"role": "user", "content": [
                {"type": "tool_result", "tool_use_id": "toolu_1",
                 "content": "18C, sunny"}]},
        ],
    }
    oreq = anthropic_to_openai(areq)
    assert oreq["tools"][0]["function"]["name"] == "get_weather"
    assert oreq["tool_choice"] == "required"
    roles = [m["role"] for m in oreq["messages"]]
    assert roles == ["user", "assistant", "tool"]
    assert oreq["messages"][1]["tool_calls"][0]["function"]["name"] == \
        "get_weather"
    assert oreq["messages"][2]["tool_call_id"] == "toolu_1"

    oresp = {"model": "m", "choices": [{"finish_reason": "tool_calls",
             "message": {"content": "",
                         "tool_calls": [{"id": "call_9", "type": "function",
                                         "function": {"name": "get_weather",
                                                      "arguments":
                                                      '{"city": "Paris"}'}}]}}],
             "usage": {"prompt_tokens": 10, "completion_tokens": 5}}
    aresp = openai_to_anthropic(oresp)
    assert aresp["stop_reason"] == "tool_use"
    blk = aresp["content"][0]
    assert blk["type"] == "tool_use" and blk["input"] == {"city": "Paris"}


def test_knowledge_filestore_is_owner_scoped(stack, tmp_path):
    """A knowledge filestore source cannot traverse out of the owner's
    namespace (and reads the owner's files, matching upload paths)."""
    import asyncio
    app, client, _, key, store = stack
    # upload a file as the user
    r = client.put("/api/v1/filestore/upload?path=docs/n.txt",
                   content=b"alpha beta gamma", headers=H(key))
    assert r.status_code == 200
    kn = app.state.knowledge
    app.state.cfg.rag.embeddings_provider = "mock"   # MockClient embeds
    me = [u for u in store.list("users") if u["username"] == "alice"][0]
    doc = kn.create(me["id"], "k1", {"filestore": {"path": "docs/n.txt"}})
    asyncio.run(kn.reconcile_once())   # preparing -> pending
    asyncio.run(kn.reconcile_once())   # pending -> indexing -> ready
    assert kn.get(doc["id"])["state"] == "ready"
    # traversal attempt errors, never reads outside
    evil = kn.create(me["id"], "k2",
                     {"filestore": {"path": "../../../../etc/hostname"}})
    asyncio.run(kn.reconcile_once())
    asyncio.run(kn.reconcile_once())
    got = kn.get(evil["id"])
    assert got["state"] == "error"
    assert "escapes" in got["message"] or "No such" in got["message"] or \
        "not found" in got["message"].lower()


def test_filestore_prefix_sibling_rejected(tmp_path):
    """`/users/u1evil` must not pass the containment check for owner
    `u1` (plain startswith prefix bug)."""
    from helix_amd.server.filestore import FileStore
    fs = FileStore(str(tmp_path))
    fs.write("u1evil", "secret.txt", b"other tenant data")
    import pytest as _pt
    with _pt.raises((PermissionError, FileNotFoundError)):
        fs.read("u1", "../u1evil/secret.txt")


def test_webui_served(stack):
    _, client, _, _, _ = stack
    r = client.get("/")
    assert r.status_code == 200
    assert "<title>helix_amd</title>" in r.text
    for pane in ("Chat", "Apps", "Knowledge", "Runners", "Usage"):
        assert pane in r.text


def test_janitor_session_retention(stack):
    _, client, _, _, store = stack
    import time as _t
    H_admin = {"Authorization": "Bearer admin-key"}
    old_ms = int((_t.time() - 90 * 86400) * 1000)
    store.put("sessions", "old_s", {"id": "old_s", "updated": old_ms},
              owner="u")
    store.put("interactions", "old_i", {"id": "old_i",
                                        "session_id": "old_s"},
              owner="u", parent="old_s")
    store.put("sessions", "new_s", {"id": "new_s",
                                    "updated": int(_t.time() * 1000)},
              owner="u")
    r = client.post("/api/v1/admin/janitor",
                    json={"retention_days": 30,
                          "session_retention_days": 30}, headers=H_admin)
    pruned = r.json()
    assert pruned["sessions"] == 1 and pruned["interactions"] == 1
    assert store.get("sessions", "old_s") is None
    assert store.get("interactions", "old_i") is None
    assert store.get("sessions", "new_s") is not None
