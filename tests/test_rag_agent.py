"""RAG pipeline + agent loop tests (mock embeddings/LLM on CPU)."""
import asyncio
import json

import pytest
from fastapi.testclient import TestClient

from helix_amd.rag.chunker import chunk_text
from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import MockClient, ProviderManager
from helix_amd.server.types import new_id
from helix_amd.store import Store


class EmbedMock(MockClient):
    """Deterministic embeddings: bag-of-words hash so similar texts get
    similar vectors (enough for ranking tests)."""

    async def embeddings(self, req):
        import zlib
        inputs = req.get("input")
        inputs = [inputs] if isinstance(inputs, str) else list(inputs)
        data = []
        for i, text in enumerate(inputs):
            v = [0.0] * 16
            for w in text.lower().split():
                # deterministic hash (python str hash is per-process seeded)
                v[zlib.crc32(w.encode()) % 16] += 1.0
            data.append({"object": "embedding", "index": i, "embedding": v})
        return {"object": "list", "data": data, "model": req.get("model")}


@pytest.fixture()
def stack():
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    cfg.rag.embeddings_provider = "mock"
    store = Store(":memory:")
    pm = ProviderManager(store)
    mock = EmbedMock()
    pm.register("mock", mock)
    app = create_app(cfg, store=store, providers=pm)
    client = TestClient(app)
    r = client.post("/api/v1/users", json={"username": "alice"},
                    headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    return app, client, mock, key, store


def H(key):
    return {"Authorization": f"Bearer {key}"}


def test_chunker_overlap():
    text = "para one words here.\n\n" + ("x" * 1200) + "\n\nlast para."
    chunks = chunk_text(text, chunk_size=512, overlap=64)
    assert len(chunks) >= 3
    assert all(len(c["text"]) <= 600 for c in chunks)


def test_knowledge_lifecycle_and_query(stack):
    app, client, _, key, store = stack
    r = client.post("/api/v1/knowledge", json={
        "name": "docs",
        "source": {"text": "Helix is a private GenAI stack. "
                           "The MI355X has 288 GB of HBM3E memory. "
                           "Bananas are yellow."}}, headers=H(key))
    kid = r.json()["id"]
    assert r.json()["state"] == "preparing"

    # drive the reconciler synchronously
    kn = app.state.knowledge
    asyncio.run(kn.reconcile_once())   # preparing -> pending
    asyncio.run(kn.reconcile_once())   # pending -> indexing -> ready
    doc = client.get(f"/api/v1/knowledge/{kid}", headers=H(key)).json()
    assert doc["state"] == "ready", doc
    assert doc["chunks"] >= 1
    assert doc["version"] == 1

    r = client.post(f"/api/v1/knowledge/{kid}/query",
                    json={"query": "how much HBM3E memory"}, headers=H(key))
    results = r.json()
    assert results and "288 GB" in results[0]["text"]

    # refresh bumps version
    client.post(f"/api/v1/knowledge/{kid}/refresh", headers=H(key))
    asyncio.run(kn.reconcile_once())
    asyncio.run(kn.reconcile_once())
    doc = client.get(f"/api/v1/knowledge/{kid}", headers=H(key)).json()
    assert doc["version"] == 2


def test_rag_enrichment_in_chat(stack):
    app, client, mock, key, store = stack
    r = client.post("/api/v1/knowledge", json={
        "name": "kb", "source": {"text": "The secret code is OMEGA-7."}},
        headers=H(key))
    kid = r.json()["id"]
    kn = app.state.knowledge
    asyncio.run(kn.reconcile_once())
    asyncio.run(kn.reconcile_once())

    r = client.post("/api/v1/apps", json={"config": {
        "name": "kb app",
        "assistants": [{"name": "a", "model": "mock-model",
                        "provider": "mock",
                        "knowledge": [{"name": "kb"}]}]}}, headers=H(key))
    app_id = r.json()["id"]
    r = client.post("/v1/chat/completions", json={
        "app_id": app_id,
        "messages": [{"role": "user", "content": "what is the secret code"}]},
        headers=H(key))
    assert r.status_code == 200
    sent = mock.calls[-1]
    user_msg = [m for m in sent["messages"] if m["role"] == "user"][-1]
    assert "OMEGA-7" in user_msg["content"]          # context injected
    assert "<context>" in user_msg["content"]


def agent_tool_script(req):
    """First call: request calculator; second: final answer."""
    msgs = req["messages"]
    if any(m.get("role") == "tool" for m in msgs):
        tool_result = [m for m in msgs if m.get("role") == "tool"][-1]
        return {
            "id": new_id("chatcmpl"), "object": "chat.completion",
            "created": 0, "model": req["model"],
            "choices": [{"index": 0, "message": {
                "role": "assistant",
                "content": f"The answer is {tool_result['content']}"},
                "finish_reason": "stop"}],
            "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                      "total_tokens": 2}}
    return {
        "id": new_id("chatcmpl"), "object": "chat.completion",
        "created": 0, "model": req["model"],
        "choices": [{"index": 0, "message": {
            "role": "assistant", "content": "",
            "tool_calls": [{"id": "tc1", "type": "function", "function": {
                "name": "calculator",
                "arguments": json.dumps({"expression": "6*7"})}}]},
            "finish_reason": "tool_calls"}],
        "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                  "total_tokens": 2}}


def test_agent_loop_with_calculator(stack):
    app, client, _, key, store = stack
    scripted = MockClient(responses=[agent_tool_script, agent_tool_script])
    app.state.providers.register("mock", scripted)
    r = client.post("/api/v1/apps", json={"config": {
        "name": "agent app",
        "assistants": [{"name": "a", "model": "mock-model",
                        "provider": "mock", "agent_mode": True,
                        "calculator": {"enabled": True}}]}}, headers=H(key))
    app_id = r.json()["id"]
    r = client.post("/v1/chat/completions", json={
        "app_id": app_id,
        "messages": [{"role": "user", "content": "what is 6*7?"}]},
        headers=H(key))
    assert r.status_code == 200, r.text
    assert "42" in r.json()["choices"][0]["message"]["content"]
    # step info persisted
    steps = store.list("step_info", limit=10)
    assert any(s["step"] == "calculator" for s in steps)


def test_agent_streaming(stack):
    app, client, _, key, store = stack
    scripted = MockClient(responses=[agent_tool_script, agent_tool_script])
    app.state.providers.register("mock", scripted)
    r = client.post("/api/v1/apps", json={"config": {
        "name": "agent app",
        "assistants": [{"name": "a", "model": "mock-model",
                        "provider": "mock", "agent_mode": True,
                        "calculator": {"enabled": True}}]}}, headers=H(key))
    app_id = r.json()["id"]
    with client.stream("POST", "/v1/chat/completions", json={
        "app_id": app_id, "stream": True,
        "messages": [{"role": "user", "content": "what is 6*7?"}]},
            headers=H(key)) as resp:
        lines = [l for l in resp.iter_lines() if l.startswith("data: ")]
    chunks = [json.loads(l[6:]) for l in lines if l != "data: [DONE]"]
    text = "".join(c["choices"][0]["delta"].get("content", "")
                   for c in chunks if c.get("choices"))
    assert "42" in text


def test_agent_memory_skill(stack):
    app, client, _, key, store = stack

    def memory_script(req):
        msgs = req["messages"]
        if any(m.get("role") == "tool" for m in msgs):
            return "stored it"
        return {
            "id": new_id("c"), "object": "chat.completion", "created": 0,
            "model": req["model"],
            "choices": [{"index": 0, "message": {
                "role": "assistant", "content": "",
                "tool_calls": [{"id": "t1", "type": "function", "function": {
                    "name": "memory", "arguments": json.dumps(
                        {"action": "store",
                         "content": "user likes terse answers"})}}]},
                "finish_reason": "tool_calls"}],
            "usage": {}}
    scripted = MockClient(responses=[memory_script, memory_script])
    app.state.providers.register("mock", scripted)
    r = client.post("/api/v1/apps", json={"config": {
        "name": "mem app",
        "assistants": [{"name": "a", "model": "mock-model",
                        "provider": "mock", "agent_mode": True,
                        "memory": {"enabled": True}}]}}, headers=H(key))
    app_id = r.json()["id"]
    r = client.post("/v1/chat/completions", json={
        "app_id": app_id,
        "messages": [{"role": "user", "content": "remember I like terse"}]},
        headers=H(key))
    assert r.status_code == 200
    assert store.list("memories", limit=5)


def test_agent_tool_error_is_fed_back(stack_agent=None):
    """A skill that raises becomes a tool-error message to the model,
    not a failed turn."""
    import asyncio
    from helix_amd.agent.runner import AgentRunner
    from helix_amd.server.config import ServerConfig
    from helix_amd.server.providers import MockClient, ProviderManager
    from helix_amd.server.types import AssistantConfig
    from helix_amd.store import Store
    store = Store(":memory:")
    pm = ProviderManager(store)
    # 1st call: model calls the calculator with args that make it raise;
    # 2nd call: model answers using the error feedback
    toolcall = {"choices": [{"finish_reason": "tool_calls", "message": {
        "content": "", "tool_calls": [{"id": "c1", "type": "function",
            "function": {"name": "calculator",
                         "arguments": '{"expression": "__import__"}'}}]}}],
        "usage": {}}
    pm.register("mock", MockClient(responses=[toolcall, "recovered"]))
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    runner = AgentRunner(cfg, store, pm, None)
    asst = AssistantConfig(name="a", system_prompt="x",
                           calculator={"enabled": True})
    msgs, final = asyncio.run(runner._loop(
        asst, {"messages": [{"role": "user", "content": "calc"}]},
        "u1", {"owner": "u1"}))
    tool_msgs = [m for m in msgs if m.get("role") == "tool"]
    assert tool_msgs, msgs
    assert "error" in tool_msgs[0]["content"].lower() or \
        "invalid" in tool_msgs[0]["content"].lower() or \
        tool_msgs[0]["content"]
    assert msgs[-1]["role"] == "assistant"
    assert msgs[-1]["content"] == "recovered"
