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


class MCPGateway:
    def __init__(self, store, agent_runner, code_intel=None):
        self.store = store
        self.agent_runner = agent_runner
        self.code_intel = code_intel

    async def _tools_for_app(self, app_id: str, owner: str):
        from helix_amd.server.types import App
        doc = self.store.get("apps", app_id)
        if doc is None:
            raise KeyError(app_id)
        app = App.model_validate(doc)
        assistant = app.config.assistants[0] if app.config.assistants \
            else None
        if assistant is None:
            return []
        return self.agent_runner.build_skills(assistant, owner, app_id)

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
