"""MCP gateway (parity with the reference's MCP surface: apps' skills
exposed as MCP tools, server/mcp gateway + kodit MCP backend). Speaks
JSON-RPC 2.0 over HTTP POST: initialize, ping, tools/list, tools/call.
"""
from __future__ import annotations

import json
import logging
from typing import Any, Dict, Optional

log = logging.getLogger("helix_amd.mcp")

PROTOCOL_VERSION = "2024-11-05"


class _OrgTool:
    """Lightweight skill adapter exposing an org-runtime operation as an
    MCP tool (reference: "MCP surface for org primitives")."""

    def __init__(self, name, description, parameters, fn):
        self.name = name
        self.description = description
        self.parameters = parameters
        self._fn = fn

    async def execute(self, args, ctx):
        return await self._fn(args, ctx)


class MCPGateway:
    def __init__(self, store, agent_runner, code_intel=None, org_rt=None):
        self.store = store
        self.agent_runner = agent_runner
        self.code_intel = code_intel
        self.org_rt = org_rt

    def _org_tools(self, owner: str):
        if self.org_rt is None:
            return []
        rt = self.org_rt

        async def list_streams(args, ctx):
            out = []
            for m in self.store.list("memberships", limit=10000):
                if m.get("user_id") != owner:
                    continue
                for st in rt.list_streams(m.get("org_id", "")):
                    out.append({"id": st["id"], "name": st["name"],
                                "org_id": st["org_id"]})
            return json.dumps(out)

        async def read_stream(args, ctx):
            msgs = rt.stream_messages(args.get("stream_id", ""), limit=50)
            return json.dumps([{ "sender": m["sender"], "text": m["text"],
                                 "replies": m.get("replies", [])}
                               for m in msgs])

        async def post_message(args, ctx):
            msg = await rt.post_message(args.get("stream_id", ""), owner,
                                        args.get("text", ""))
            return json.dumps({"id": msg["id"],
                               "replies": msg.get("replies", [])})

        sid = {"type": "object", "properties": {
            "stream_id": {"type": "string"}}, "required": ["stream_id"]}
        post = {"type": "object", "properties": {
            "stream_id": {"type": "string"}, "text": {"type": "string"}},
            "required": ["stream_id", "text"]}
        return [
            _OrgTool("org_list_streams",
                     "List org streams you are a member of",
                     {"type": "object", "properties": {}}, list_streams),
            _OrgTool("org_read_stream",
                     "Read recent messages (and bot replies) of a stream",
                     sid, read_stream),
            _OrgTool("org_post_message",
                     "Post a message to an org stream; subscribed bots "
                     "reply", post, post_message),
        ]

    async def _tools_for_app(self, app_id: str, owner: str):
        from helix_amd.server.types import App
        doc = self.store.get("apps", app_id)
        if doc is None:
            raise KeyError(app_id)
        app = App.model_validate(doc)
        caller = self.store.get("users", owner) or {}
        if app.owner != owner and not app.global_ and \
                not caller.get("admin"):
            # skills resolve with the CALLER's secrets/knowledge, but a
            # private app's config (prompts, API endpoints) is still
            # the owner's — same guard as the HTTP app routes
            raise KeyError(app_id)
        assistant = app.config.assistants[0] if app.config.assistants \
            else None
        skills = []
        if assistant is not None:
            skills = list(self.agent_runner.build_skills(assistant, owner,
                                                         app_id))
        return skills + self._org_tools(owner)

    async def handle(self, app_id: str, owner: str, req: dict) -> Optional[dict]:
        """Handle one JSON-RPC request; returns a response dict (or None
        for notifications)."""
        rid = req.get("id")
        method = req.get("method", "")
        params = req.get("params") or {}

        def ok(result):
            return {"jsonrpc": "2.0", "id": rid, "result": result}

        def err(code, message):
            return {"jsonrpc": "2.0", "id": rid,
                    "error": {"code": code, "message": message}}

        try:
            if method == "initialize":
                return ok({
                    "protocolVersion": PROTOCOL_VERSION,
                    "capabilities": {"tools": {}},
                    "serverInfo": {"name": "helix_amd",
                                   "version": "0.1.0"},
                })
            if method == "notifications/initialized":
                return None
            if method == "ping":
                return ok({})
            if method == "tools/list":
                skills = await self._tools_for_app(app_id, owner)
                return ok({"tools": [
                    {"name": s.name, "description": s.description,
                     "inputSchema": s.parameters} for s in skills]})
            if method == "tools/call":
                name = params.get("name", "")
                args = params.get("arguments") or {}
                skills = await self._tools_for_app(app_id, owner)
                skill = next((s for s in skills if s.name == name), None)
                if skill is None:
                    return err(-32602, f"unknown tool: {name}")
                result = await skill.execute(args, {"owner": owner,
                                                    "app_id": app_id})
                return ok({"content": [{"type": "text", "text": result}],
                           "isError": False})
            return err(-32601, f"method not found: {method}")
        except KeyError as e:
            return err(-32602, f"app not found: {e}")
        except Exception as e:  # pragma: no cover
            log.exception("mcp error")
            return err(-32603, str(e))
