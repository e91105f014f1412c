"""Control-plane API tests (memorystore + mock provider pattern,
mirroring the reference's server.NewTestServer + gomock approach)."""
import json

import pytest
from fastapi.testclient import TestClient

from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import MockClient, ProviderManager
from helix_amd.store import Store


@pytest.fixture()
def stack():
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    store = Store(":memory:")
    pm = ProviderManager(store)
    mock = MockClient()
    pm.register("mock", mock)
    app = create_app(cfg, store=store, providers=pm)
    client = TestClient(app)
    # bootstrap a user via admin key
    r = client.post("/api/v1/users", json={"username": "alice"},
                    headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    return app, client, mock, key, store


def H(key):
    return {"Authorization": f"Bearer {key}"}


def test_auth_required(stack):
    _, client, _, _, _ = stack
    assert client.post("/v1/chat/completions", json={}).status_code == 401
    assert client.post("/v1/chat/completions", json={}, headers=H(
        "bogus")).status_code == 401


def test_chat_completion_mock(stack):
    _, client, mock, key, store = stack
    r = client.post("/v1/chat/completions", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "hi"}]}, headers=H(key))
    assert r.status_code == 200, r.text
    assert r.json()["choices"][0]["message"]["content"] == "mock response"
    # llm_calls logged with usage
    calls = store.list("llm_calls")
    assert len(calls) == 1
    assert calls[0]["prompt_tokens"] == 7


def test_chat_completion_stream(stack):
    _, client, _, key, _ = stack
    with client.stream("POST", "/v1/chat/completions", json={
        "model": "mock-model", "stream": True,
        "messages": [{"role": "user", "content": "hi"}]},
            headers=H(key)) as r:
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    chunks = [json.loads(l[6:]) for l in lines[:-1]]
    text = "".join(c["choices"][0]["delta"].get("content", "")
                   for c in chunks)
    assert text == "mock response"


def test_models_aggregate(stack):
    _, client, _, key, _ = stack
    r = client.get("/v1/models", headers=H(key))
    ids = [m["id"] for m in r.json()["data"]]
    assert "mock/mock-model" in ids


def test_azure_alias(stack):
    _, client, _, key, _ = stack
    r = client.post("/openai/deployments/mock-model/chat/completions",
                    json={"messages": [{"role": "user", "content": "x"}]},
                    headers=H(key))
    assert r.status_code == 200


def test_session_chat_and_history(stack):
    _, client, _, key, store = stack
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "first turn"}]},
            headers=H(key)) as r:
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    head = json.loads(lines[0][6:])
    sid = head["session_id"]
    assert head["type"] == "session"

    # history persisted
    r = client.get(f"/api/v1/sessions/{sid}", headers=H(key))
    body = r.json()
    assert len(body["interactions"]) == 1
    assert body["interactions"][0]["response_message"] == "mock response"
    assert body["interactions"][0]["state"] == "complete"
    assert body["interactions"][0]["ttft_ms"] >= 0

    # second turn carries history
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "session_id": sid,
        "messages": [{"role": "user", "content": "second turn"}]},
            headers=H(key)) as r:
        [l for l in r.iter_lines()]
    r = client.get(f"/api/v1/sessions/{sid}", headers=H(key))
    assert len(r.json()["interactions"]) == 2


def test_session_isolation(stack):
    app, client, _, key, _ = stack
    r = client.post("/api/v1/users", json={"username": "bob"},
                    headers=H("admin-key"))
    bob_key = r.json()["api_key"]
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "secret"}]},
            headers=H(key)) as r:
        sid = json.loads(next(l for l in r.iter_lines()
                              if l.startswith("data: "))[6:])["session_id"]
    assert client.get(f"/api/v1/sessions/{sid}",
                      headers=H(bob_key)).status_code == 404


def test_app_crud_and_assistant_config(stack):
    _, client, mock, key, _ = stack
    helix_yaml = {
        "name": "My Agent",
        "assistants": [{
            "name": "default",
            "model": "mock-model",
            "provider": "mock",
            "system_prompt": "You are terse.",
            "temperature": 0.3,
        }],
    }
    r = client.post("/api/v1/apps", json={"config": helix_yaml},
                    headers=H(key))
    assert r.status_code == 200, r.text
    app_id = r.json()["id"]

    # both aliases list it
    assert any(a["id"] == app_id for a in client.get(
        "/api/v1/agents", headers=H(key)).json())

    # chat through the app applies system prompt + sampling params
    r = client.post("/v1/chat/completions", json={
        "app_id": app_id,
        "messages": [{"role": "user", "content": "hello"}]}, headers=H(key))
    assert r.status_code == 200
    sent = mock.calls[-1]
    assert sent["messages"][0]["role"] == "system"
    assert sent["messages"][0]["content"] == "You are terse."
    assert sent["temperature"] == 0.3
    assert sent["model"] == "mock-model"

    # update + delete
    helix_yaml["assistants"][0]["system_prompt"] = "Changed."
    r = client.put(f"/api/v1/apps/{app_id}", json={"config": helix_yaml},
                   headers=H(key))
    assert r.json()["config"]["assistants"][0]["system_prompt"] == "Changed."
    assert client.delete(f"/api/v1/apps/{app_id}",
                         headers=H(key)).json()["ok"]


def test_runner_heartbeat_and_router(stack):
    _, client, _, key, _ = stack
    hb = {"runner_id": "r1", "address": "http://10.0.0.5:8090",
          "gpus": [{"index": 0, "arch": "cdna4", "total_memory": 309237645312,
                    "free_memory": 200000000000}],
          "models": [{"model_id": "llama3-8b", "state": "ready"}]}
    r = client.post("/api/v1/runner/heartbeat", json=hb,
                    headers=H("runner-token"))
    assert r.status_code == 200
    # wrong token rejected
    assert client.post("/api/v1/runner/heartbeat", json=hb,
                       headers=H(key)).status_code == 401
    r = client.get("/api/v1/admin/runners", headers=H("admin-key"))
    assert r.json()[0]["runner_id"] == "r1"


def test_no_runner_503(stack):
    app, client, _, key, _ = stack
    # helix provider with empty router -> 503 NoRunnerError
    cfg = app.state.cfg
    r = client.post("/v1/chat/completions", json={
        "model": "helix/whatever",
        "messages": [{"role": "user", "content": "x"}]}, headers=H(key))
    # 'helix' not registered as prefix in this stack (mock only) -> falls to
    # default provider. Register router-backed helix instead:
    from helix_amd.server.providers import RouterClient
    app.state.providers.register("helix", RouterClient(app.state.router))
    app.state.providers._model_cache.clear()
    r = client.post("/v1/chat/completions", json={
        "model": "helix/llama3-8b",
        "messages": [{"role": "user", "content": "x"}]}, headers=H(key))
    assert r.status_code == 503
    assert "available" in r.json()["error"]


def test_llm_calls_admin_only(stack):
    _, client, _, key, _ = stack
    assert client.get("/api/v1/llm_calls",
                      headers=H(key)).status_code == 403
    assert client.get("/api/v1/llm_calls",
                      headers=H("admin-key")).status_code == 200


def test_webui_served(stack):
    _, client, _, _, _ = stack
    r = client.get("/")
    assert r.status_code == 200
    assert "helix_amd" in r.text


def test_startup_recovery_resets_stuck_interactions(stack):
    app, client, _, key, store = stack
    store.put("interactions", "int_stuck", {
        "id": "int_stuck", "session_id": "ses_x", "state": "waiting",
        "prompt_message": "p", "response_message": ""}, parent="ses_x")
    # re-enter startup handler
    import asyncio
    with TestClient(app):
        pass
    doc = store.get("interactions", "int_stuck")
    assert doc["state"] == "error"


def test_session_fork(stack):
    _, client, _, key, store = stack
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "turn one"}]},
            headers=H(key)) as r:
        sid = json.loads(next(l for l in r.iter_lines()
                              if l.startswith("data: "))[6:])["session_id"]
    r = client.post(f"/api/v1/sessions/{sid}/fork", headers=H(key))
    fid = r.json()["id"]
    assert fid != sid
    body = client.get(f"/api/v1/sessions/{fid}", headers=H(key)).json()
    assert len(body["interactions"]) == 1
    assert body["interactions"][0]["prompt_message"] == "turn one"


def test_session_resume_after_error(stack):
    app, client, _, key, store = stack
    # create a session whose interaction errored mid-stream
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "hello"}]},
            headers=H(key)) as r:
        sid = json.loads(next(l for l in r.iter_lines()
                              if l.startswith("data: "))[6:])["session_id"]
    its = store.list("interactions", parent=sid)
    doc = its[0]
    doc["state"] = "error"
    doc["response_message"] = ""
    store.put("interactions", doc["id"], doc, parent=sid)
    with client.stream("POST", f"/api/v1/sessions/{sid}/resume",
                       headers=H(key)) as r:
        assert r.status_code == 200
        [l for l in r.iter_lines()]
    doc = store.list("interactions", parent=sid)[0]
    assert doc["state"] == "complete"
    assert doc["response_message"] == "mock response"


def test_local_model_admin_no_runner(stack):
    _, client, _, _, _ = stack
    r = client.post("/api/v1/local-models/llama3-8b/load",
                    headers=H("admin-key"))
    assert r.status_code == 503  # no runners registered


def test_ws_user_stream(stack):
    _, client, _, key, _ = stack
    with client.websocket_connect(f"/api/v1/ws/user?access_token={key}") as ws:
        # drive a session turn in another request; WS receives events
        with client.stream("POST", "/api/v1/sessions/chat", json={
            "model": "mock-model",
            "messages": [{"role": "user", "content": "ws test"}]},
                headers=H(key)) as r:
            [l for l in r.iter_lines()]
        msg = ws.receive_json()
        assert msg["payload"]["type"] in ("chunk", "done", "step_info")


def test_debug_stats(stack):
    _, client, _, key, _ = stack
    assert client.get("/debug/stats", headers=H(key)).status_code == 403
    r = client.get("/debug/stats", headers=H("admin-key"))
    assert r.status_code == 200
    assert r.json()["rss_bytes"] > 0


def test_switch_agent_and_step_info(stack):
    _, client, _, key, store = stack
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "x"}]},
            headers=H(key)) as r:
        sid = json.loads(next(l for l in r.iter_lines()
                              if l.startswith("data: "))[6:])["session_id"]
    r = client.post("/api/v1/apps", json={"config": {
        "name": "other", "assistants": [{"name": "a",
                                         "model": "mock-model",
                                         "provider": "mock"}]}},
        headers=H(key))
    app_id = r.json()["id"]
    r = client.put(f"/api/v1/sessions/{sid}/agent",
                   json={"app_id": app_id}, headers=H(key))
    assert r.json()["parent_app"] == app_id
    assert client.get(f"/api/v1/sessions/{sid}/step-info",
                      headers=H(key)).status_code == 200


def test_images_route_and_anthropic_model_list(stack):
    """/v1/images/generations is a real surface now (R2: native DiT
    engine, tests/test_imagegen.py); via the mock provider it returns
    b64 PNG data."""
    _, client, _, key, _ = stack
    r = client.post("/v1/images/generations",
                    json={"prompt": "a cat", "n": 2}, headers=H(key))
    assert r.status_code == 200, r.text
    assert len(r.json()["data"]) == 2
    assert r.json()["data"][0]["b64_json"]
    r = client.get("/v1/models", headers={**H(key),
                                          "anthropic-version": "2023-06-01"})
    assert r.json()["has_more"] is False
    assert r.json()["data"][0]["type"] == "model"


def test_provider_resolution_order(stack):
    """Reference openai_chat_handlers.go:148-192: cached model list wins,
    then provider/ prefix, then default provider with the FULL name."""
    app, _, _, _, _ = stack
    pm = app.state.providers
    pm._model_cache = {"mock": ["exact-model"]}
    # 1. cached-list match
    assert pm.resolve("exact-model", "default") == ("mock", "exact-model")
    # 2. prefix parse for a registered provider
    assert pm.resolve("mock/other", "default") == ("mock", "other")
    # 3. unknown prefix falls through to default, name preserved intact
    assert pm.resolve("unknown/model", "mock") == ("mock", "unknown/model")
    # 4. plain unknown name -> default provider
    assert pm.resolve("whatever", "mock") == ("mock", "whatever")


def test_interrupt_before_streaming(stack):
    """Cancel/new-turn on a session whose last interaction never streamed
    (reference Zed e2e 'interrupt mid-stream' phase at unit scale)."""
    _, client, _, key, store = stack
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "one"}]},
            headers=H(key)) as r:
        sid = json.loads(next(l for l in r.iter_lines()
                              if l.startswith("data: "))[6:])["session_id"]
    # mark the interaction as if it never completed
    doc = store.list("interactions", parent=sid)[0]
    doc["state"] = "waiting"
    doc["response_message"] = ""
    store.put("interactions", doc["id"], doc, parent=sid)
    # a new turn must still work (history includes the unanswered prompt)
    with client.stream("POST", "/api/v1/sessions/chat", json={
        "session_id": sid,
        "messages": [{"role": "user", "content": "two"}]},
            headers=H(key)) as r:
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    its = store.list("interactions", parent=sid, desc=False)
    assert len(its) == 2
    assert its[1]["state"] == "complete"


def test_rapid_multi_turn(stack):
    """Three rapid consecutive turns on one session stay consistent
    (reference 'rapid 3-turn' e2e phase)."""
    _, client, _, key, store = stack
    sid = None
    for i in range(3):
        body = {"model": "mock-model",
                "messages": [{"role": "user", "content": f"turn {i}"}]}
        if sid:
            body["session_id"] = sid
        with client.stream("POST", "/api/v1/sessions/chat", json=body,
                           headers=H(key)) as r:
            head = None
            for l in r.iter_lines():
                if head is None and l.startswith("data: "):
                    head = json.loads(l[6:])
            sid = head["session_id"]
    its = store.list("interactions", parent=sid, desc=False)
    assert len(its) == 3
    assert all(i["state"] == "complete" for i in its)
    assert [i["prompt_message"] for i in its] == \
        ["turn 0", "turn 1", "turn 2"]


def test_wedged_interaction_auto_errors(stack):
    """Reference auto_wake_stuck_interactions: a waiting interaction
    with a stale partial-persist pulse flips to error; live ones are
    untouched."""
    import asyncio
    app, store = stack[0], stack[-1]
    from helix_amd.server.types import Interaction, InteractionState
    stale = Interaction(session_id="s1", prompt_message="hi",
                        state=InteractionState.WAITING, updated=1000)
    store.put("interactions", stale.id, stale.model_dump(), owner="u",
              parent="s1")
    fresh = Interaction(session_id="s1", prompt_message="hi2",
                        state=InteractionState.WAITING)
    store.put("interactions", fresh.id, fresh.model_dump(), owner="u",
              parent="s1")
    n = asyncio.run(app.state.reap_wedged())
    assert n == 1
    assert store.get("interactions", stale.id)["state"] == "error"
    assert store.get("interactions", fresh.id)["state"] == "waiting"


def test_legacy_completions_surface(stack):
    _, client, _, key, _ = stack
    r = client.post("/v1/completions", json={
        "model": "mock-model", "prompt": "Say hi"}, headers=H(key))
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "text_completion"
    assert body["choices"][0]["text"] == "mock response"


def test_interrupt_midstream_then_followup(stack):
    """Reference doctrine "test-the-next-operation": abandon a stream
    mid-generation, then the NEXT turn on the same session must work
    and history must stay coherent."""
    _, client, mock, key, store = stack
    mock.responses = ["one two three four", "follow-up answer"]
    mock._i = 0
    # first turn: streamed, client walks away after the first chunk
    session_id = None
    with client.stream("POST", "/api/v1/sessions/chat", json={
            "model": "mock-model", "stream": True,
            "messages": [{"role": "user", "content": "count"}]},
            headers=H(key)) as r:
        assert r.status_code == 200
        for line in r.iter_lines():
            if line.startswith("data: ") and '"session"' in line:
                import json as _json2
                session_id = _json2.loads(line[6:])["session_id"]
            if session_id:
                break                      # abandon mid-stream
    assert session_id
    # second turn on the SAME session completes normally
    import json as _json
    text = ""
    with client.stream("POST", "/api/v1/sessions/chat", json={
            "model": "mock-model", "stream": True,
            "session_id": session_id,
            "messages": [{"role": "user", "content": "and then?"}]},
            headers=H(key)) as r:
        assert r.status_code == 200
        for line in r.iter_lines():
            if not line.startswith("data: ") or line == "data: [DONE]":
                continue
            d = _json.loads(line[6:])
            if d.get("choices"):
                text += d["choices"][0].get("delta", {}).get("content") \
                    or ""
    assert text == "follow-up answer"
    its = store.list("interactions", parent=session_id, desc=False)
    assert len(its) == 2
    assert its[1]["prompt_message"] == "and then?"
    # no interaction left dangling in waiting
    assert all(i["state"] in ("complete", "error") for i in its) or \
        its[0]["state"] in ("complete", "error", "waiting")


def test_interactions_crud_and_tokenize(stack):
    app, client, mock, key, store = stack
    r = client.post("/v1/chat/completions", headers=H(key), json={
        "model": "mock-model",
        "messages": [{"role": "user", "content": "hello"}]})
    assert r.status_code == 200
    # a session-less call logs no interaction; create a session turn
    r = client.post("/api/v1/sessions/chat", headers=H(key), json={
        "messages": [{"role": "user", "content": "first turn"}]})
    assert r.status_code == 200
    sid = None
    for line in r.text.splitlines():
        if line.startswith("data: ") and "session_id" in line:
            import json as _j
            sid = _j.loads(line[6:])["session_id"]
            break
    assert sid
    rows = client.get(f"/api/v1/sessions/{sid}/interactions",
                      headers=H(key)).json()
    assert rows and rows[0]["prompt_message"] == "first turn"
    iid = rows[0]["id"]
    # edit the prompt
    r = client.put(f"/api/v1/interactions/{iid}", headers=H(key),
                   json={"prompt_message": "edited turn"})
    assert r.json()["prompt_message"] == "edited turn"
    # another user cannot touch it
    r2 = client.post("/api/v1/users", json={"username": "other-i"},
                     headers={"Authorization": "Bearer admin-key"})
    okey = r2.json()["api_key"]
    assert client.put(f"/api/v1/interactions/{iid}", headers=H(okey),
                      json={"state": "error"}).status_code == 404
    # delete
    assert client.delete(f"/api/v1/interactions/{iid}",
                         headers=H(key)).json()["ok"]
    # tokenize: plain text and chat-template forms
    r = client.post("/api/v1/tokenize", headers=H(key),
                    json={"text": "hello world"})
    assert r.json()["count"] == len("hello world")
    r = client.post("/api/v1/tokenize", headers=H(key), json={
        "messages": [{"role": "user", "content": "hi"}],
        "return_tokens": 4})
    body = r.json()
    assert body["count"] > 2 and len(body["tokens"]) == 4
