"""Anthropic native passthrough (weak #7; reference
api/pkg/anthropic/anthropic_proxy.go): raw body forwarded verbatim
(cache_control survives => prompt caching works), helix token swapped
for the provider key, thinking.type adaptive<->enabled retry, SSE
streamed through untouched; translation fallback when no Anthropic
endpoint is configured.
"""
import asyncio
import json

import pytest

from helix_amd.server.anthropic_api import AnthropicPassthrough


class FakeUpstream:
    """httpx.AsyncClient stand-in that records what it was sent."""

    def __init__(self, thinking_pickiness=None):
        self.calls = []
        self.thinking_pickiness = thinking_pickiness  # e.g. "enabled"

    class _Resp:
        def __init__(self, code, body):
            self.status_code = code
            self._body = body
            self.text = json.dumps(body)

        def json(self):
            return self._body

    async def post(self, url, json=None, headers=None):
        self.calls.append({"url": url, "body": json, "headers": headers})
        th = (json or {}).get("thinking", {})
        if self.thinking_pickiness and \
                th.get("type") not in (self.thinking_pickiness, None):
            return self._Resp(400, {"type": "error", "error": {
                "message": "thinking.type must be "
                           + self.thinking_pickiness}})
        return self._Resp(200, {
            "id": "msg_1", "type": "message", "role": "assistant",
            "content": [{"type": "text", "text": "pong"}],
            "usage": {"input_tokens": 3, "output_tokens": 1}})


def test_passthrough_preserves_body_and_swaps_auth():
    up = FakeUpstream()
    pt = AnthropicPassthrough("https://api.anthropic.test", "sk-prov",
                              http_client=up)
    body = {"model": "claude-x", "max_tokens": 10,
            "system": [{"type": "text", "text": "sys",
                        "cache_control": {"type": "ephemeral"}}],
            "messages": [{"role": "user", "content": "ping"}]}
    status, resp = asyncio.run(pt.forward(
        body, {"anthropic-version": "2024-01-01",
               "anthropic-beta": "prompt-caching-2024"}))
    assert status == 200 and resp["content"][0]["text"] == "pong"
    call = up.calls[0]
    # raw body verbatim — cache_control intact
    assert call["body"] is body or call["body"] == body
    assert call["body"]["system"][0]["cache_control"] == \
        {"type": "ephemeral"}
    # provider key, caller's version/beta preserved, no helix token
    assert call["headers"]["x-api-key"] == "sk-prov"
    assert call["headers"]["anthropic-version"] == "2024-01-01"
    assert call["headers"]["anthropic-beta"] == "prompt-caching-2024"
    assert "authorization" not in {k.lower() for k in call["headers"]}


def test_thinking_type_retry():
    up = FakeUpstream(thinking_pickiness="enabled")
    pt = AnthropicPassthrough("https://api.anthropic.test", "k",
                              http_client=up)
    body = {"model": "m", "thinking": {"type": "adaptive",
                                       "budget_tokens": 100},
            "messages": []}
    status, resp = asyncio.run(pt.forward(body))
    assert status == 200
    assert len(up.calls) == 2
    assert up.calls[1]["body"]["thinking"]["type"] == "enabled"
    assert up.calls[1]["body"]["thinking"]["budget_tokens"] == 100
    # no retry loop when the alternate also fails
    up2 = FakeUpstream(thinking_pickiness="never-matches")
    pt2 = AnthropicPassthrough("https://x", "k", http_client=up2)
    status, _ = asyncio.run(pt2.forward(body))
    assert status == 400 and len(up2.calls) == 2


def test_route_uses_passthrough_when_configured(tmp_path, monkeypatch):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    monkeypatch.setenv("HELIX_ANTHROPIC_BASE_URL", "https://api.anthropic.test")
    monkeypatch.setenv("HELIX_ANTHROPIC_API_KEY", "sk-prov")
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        key = auth.create_api_key(auth.create_user("u")["id"])
        # swap in the fake upstream before first use
        from helix_amd.server.anthropic_api import AnthropicPassthrough
        up = FakeUpstream()
        app.state.anthropic_passthrough = AnthropicPassthrough(
            "https://api.anthropic.test", "sk-prov", http_client=up)
        r = client.post("/v1/messages",
                        headers={"Authorization": f"Bearer {key}",
                                 "anthropic-version": "2023-06-01"},
                        json={"model": "claude-x", "max_tokens": 5,
                              "messages": [{"role": "user",
                                            "content": "hi"}]})
        assert r.status_code == 200, r.text
        assert r.json()["content"][0]["text"] == "pong"
        assert up.calls  # passthrough, not translation


def test_route_translates_without_endpoint(tmp_path, monkeypatch):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    monkeypatch.delenv("HELIX_ANTHROPIC_BASE_URL", raising=False)
    from helix_amd.server.providers import MockClient, ProviderManager
    from helix_amd.store import Store
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    store = Store(cfg.store.path)
    pm = ProviderManager(store)
    pm.register("mock", MockClient())
    app = create_app(cfg, store=store, providers=pm)
    with TestClient(app) as client:
        auth = app.state.auth
        key = auth.create_api_key(auth.create_user("u")["id"])
        r = client.post("/v1/messages",
                        headers={"Authorization": f"Bearer {key}"},
                        json={"model": "mock-model", "max_tokens": 5,
                              "messages": [{"role": "user",
                                            "content": "hello"}]})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["type"] == "message"
        assert body["content"][0]["type"] == "text"
