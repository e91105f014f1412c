"""Round-2 agent skills: browser, email, MCP client,
and the informative-or-actionable classification chain.

Reference behaviors: api/pkg/agent/skill/browser_skill.go (url+prompt,
readability, processOutput), email_sending_skill.go (user-only
delivery), skill/mcp/mcp_skill.go (tools/list -> per-tool skills),
api/pkg/tools/informative_or_actionable.go (JSON yes/no gate before
tool execution).
"""
import asyncio
import json

import pytest

from helix_amd.agent.skills import (BrowserSkill, EmailSkill, MCPClient,
                                    MCPToolSkill, build_mcp_skills)
from helix_amd.server.types import AssistantConfig


class FakeResponse:
    def __init__(self, status_code=200, text="", json_body=None):
        self.status_code = status_code
        self.text = text
        self._json = json_body

    def json(self):
        return self._json


class FakeHTTP:
    """Stands in for httpx.AsyncClient."""

    def __init__(self, routes=None, rpc_handler=None):
        self.routes = routes or {}
        self.rpc_handler = rpc_handler
        self.calls = []

    async def get(self, url, **kw):
        self.calls.append(("GET", url))
        if url in self.routes:
            return self.routes[url]
        return FakeResponse(404, "not found")

    async def post(self, url, json=None, **kw):
        self.calls.append(("POST", url, json))
        return FakeResponse(200, json_body=self.rpc_handler(json))


PAGE = """<html><head><title>Release notes</title></head><body>
<nav><a href="/old">old</a></nav>
<article><p>Version 2.4 adds fused decode attention and an xGMI
one-shot allreduce. The scheduler now supports chunked prefill with a
512-token quantum for long prompts.</p></article>
<footer>copyright</footer></body></html>"""


def test_browser_skill_fetch_and_extract():
    http = FakeHTTP(routes={"https://x.test/notes": FakeResponse(200, PAGE)})
    sk = BrowserSkill({"process_output": False}, http_client=http)
    out = asyncio.run(sk.execute(
        {"url": "https://x.test/notes", "prompt": "what changed?"}, {}))
    assert "fused decode attention" in out
    assert "Release notes" in out
    assert "copyright" not in out
    # cache: second call does not refetch
    asyncio.run(sk.execute(
        {"url": "https://x.test/notes", "prompt": "again"}, {}))
    assert len([c for c in http.calls if c[0] == "GET"]) == 1
    # scheme restriction
    out = asyncio.run(sk.execute(
        {"url": "file:///etc/passwd", "prompt": "x"}, {}))
    assert "only http(s)" in out


def test_browser_skill_process_output_llm():
    http = FakeHTTP(routes={"https://x.test/n": FakeResponse(200, PAGE)})

    async def llm(messages):
        assert "what changed" in messages[-1]["content"]
        return "Fused decode attention was added."

    sk = BrowserSkill({}, llm=llm, http_client=http)
    out = asyncio.run(sk.execute(
        {"url": "https://x.test/n", "prompt": "what changed?"}, {}))
    assert out == "Fused decode attention was added."


def test_email_skill_sends_to_user_only():
    sent = []

    class Email:
        host = "smtp.test"

        def send(self, to, subject, body):
            sent.append((to, subject, body))
            return True

    class Notif:
        email = Email()

    sk = EmailSkill(Notif(), "user@test.dev")
    out = asyncio.run(sk.execute(
        {"subject": "hi", "message": "body"}, {}))
    assert out == "Email sent"
    assert sent == [("user@test.dev", "hi", "body")]
    # the tool surface has no recipient parameter: agents cannot email
    # arbitrary addresses (reference email_sending_skill.go meta.UserEmail)
    assert "to" not in sk.parameters["properties"]
    # unconfigured provider degrades cleanly
    class NoEmail:
        email = None
    out = asyncio.run(EmailSkill(NoEmail(), "user@test.dev").execute(
        {"subject": "s", "message": "m"}, {}))
    assert "no email provider" in out


def _mcp_rpc(req):
    method = req["method"]
    if method == "initialize":
        return {"jsonrpc": "2.0", "id": req["id"],
                "result": {"protocolVersion": "2024-11-05"}}
    if method == "tools/list":
        return {"jsonrpc": "2.0", "id": req["id"], "result": {"tools": [
            {"name": "ticket lookup",
             "description": "Look up a ticket",
             "inputSchema": {"type": "object",
                             "properties": {"id": {"type": "string"}},
                             "required": ["id"]}}]}}
    if method == "tools/call":
        args = req["params"]["arguments"]
        return {"jsonrpc": "2.0", "id": req["id"], "result": {
            "content": [{"type": "text",
                         "text": f"ticket {args['id']}: open"}]}}
    return {"jsonrpc": "2.0", "id": req["id"],
            "error": {"message": f"unknown method {method}"}}


def test_mcp_client_skills():
    http = FakeHTTP(rpc_handler=_mcp_rpc)
    skills = asyncio.run(build_mcp_skills(
        {"url": "https://mcp.test/rpc"}, http_client=http))
    assert len(skills) == 1
    sk = skills[0]
    assert sk.name == "mcp_ticket_lookup"          # sanitized
    assert sk.parameters["required"] == ["id"]
    out = asyncio.run(sk.execute({"id": "T-1"}, {}))
    assert out == "ticket T-1: open"
    # initialize happened exactly once before the listing
    methods = [c[2]["method"] for c in http.calls if c[0] == "POST"]
    assert methods[0] == "initialize"
    assert methods.count("initialize") == 1


def test_mcp_error_surfaces():
    def rpc(req):
        if req["method"] == "initialize":
            return {"jsonrpc": "2.0", "id": req["id"], "result": {}}
        if req["method"] == "tools/call":
            return {"jsonrpc": "2.0", "id": req["id"], "result": {
                "isError": True,
                "content": [{"type": "text", "text": "boom"}]}}
        return {"jsonrpc": "2.0", "id": req["id"],
                "result": {"tools": []}}
    client = MCPClient("https://m.test", http_client=FakeHTTP(rpc_handler=rpc))
    sk = MCPToolSkill(client, {"name": "x"})
    assert "tool error: boom" in asyncio.run(sk.execute({}, {}))


# ---------------------------------------------------------------------------
class ScriptedClient:
    provider = "mock"
    model = ""

    def __init__(self, responses):
        self.responses = list(responses)
        self.requests = []

    async def chat(self, req):
        self.requests.append(req)
        content = self.responses.pop(0)
        return {"choices": [{"message": {"role": "assistant",
                                         "content": content},
                             "finish_reason": "stop"}],
                "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                          "total_tokens": 2}}


def _runner(tmp_path, client):
    from helix_amd.agent.runner import AgentRunner
    from helix_amd.server.config import load_config
    from helix_amd.store import Store

    class PM:
        def get_client(self, name, owner=None):
            return client
    cfg = load_config()
    store = Store(str(tmp_path / "db.sqlite"))
    return AgentRunner(cfg, store, PM(), None), store


def test_is_actionable_yes_no_and_retry(tmp_path):
    asst = AssistantConfig(name="a", calculator={"enabled": True})
    msgs = [{"role": "user", "content": "what is 2+2*3?"}]
    client = ScriptedClient(
        ['{"needs_tool": "yes", "api": "calculator", '
         '"justification": "arithmetic"}'])
    ar, _ = _runner(tmp_path, client)
    out = asyncio.run(ar.is_actionable(asst, msgs, "u1", {}))
    assert out["needs_tool"] == "yes" and out["api"] == "calculator"
    # the last user message carries the reference's json nudge
    sent = client.requests[0]["messages"]
    assert "Return the corresponding json" in sent[-1]["content"]

    # malformed response retried, then parsed
    client = ScriptedClient(
        ["sure, I think a tool is needed",
         '{"needs_tool": "no", "api": "", "justification": "chitchat"}'])
    ar, _ = _runner(tmp_path, client)
    out = asyncio.run(ar.is_actionable(asst, msgs, "u1", {}))
    assert out["needs_tool"] == "no"
    assert len(client.requests) == 2

    # no tools -> informative without an LLM call
    plain = AssistantConfig(name="p")
    client = ScriptedClient([])
    ar, _ = _runner(tmp_path, client)
    out = asyncio.run(ar.is_actionable(plain, msgs, "u1", {}))
    assert out["needs_tool"] == "no"
    assert client.requests == []
