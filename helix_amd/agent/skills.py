"""Agent skills (parity with the reference's api/pkg/agent/skill/*:
calculator, knowledge/RAG, OpenAPI api-calling, web_search, memory).

Each skill surfaces as an OpenAI tool (reference agent.go:180
ConvertSkillsToTools) and executes server-side.
"""
from __future__ import annotations

import ast
import json
import logging
import operator as op
import time
from typing import Any, Dict, List, Optional

log = logging.getLogger("helix_amd.agent.skills")


class Skill:
    name: str = ""
    description: str = ""
    parameters: Dict[str, Any] = {"type": "object", "properties": {}}

    def to_tool(self) -> dict:
        return {"type": "function", "function": {
            "name": self.name, "description": self.description,
            "parameters": self.parameters}}

    async def execute(self, args: dict, ctx: dict) -> str:
        raise NotImplementedError


# ---------------------------------------------------------------------------
class CalculatorSkill(Skill):
    name = "calculator"
    description = "Evaluate an arithmetic expression (+,-,*,/,**,%,parens)."
    parameters = {"type": "object", "properties": {
        "expression": {"type": "string"}}, "required": ["expression"]}

    _OPS = {ast.Add: op.add, ast.Sub: op.sub, ast.Mult: op.mul,
            ast.Div: op.truediv, ast.Pow: op.pow, ast.Mod: op.mod,
            ast.USub: op.neg, ast.UAdd: op.pos, ast.FloorDiv: op.floordiv}

    def _eval(self, node):
        if isinstance(node, ast.Constant) and isinstance(node.value,
                                                         (int, float)):
            return node.value
        if isinstance(node, ast.BinOp) and type(node.op) in self._OPS:
            left = self._eval(node.left)
            right = self._eval(node.right)
            if isinstance(node.op, ast.Pow) and abs(right) > 1024:
                raise ValueError("exponent too large")
            return self._OPS[type(node.op)](left, right)
        if isinstance(node, ast.UnaryOp) and type(node.op) in self._OPS:
            return self._OPS[type(node.op)](self._eval(node.operand))
        raise ValueError(f"unsupported expression element: {node!r}")

    async def execute(self, args: dict, ctx: dict) -> str:
        try:
            tree = ast.parse(args.get("expression", ""), mode="eval")
            return str(self._eval(tree.body))
        except Exception as e:
            return f"calculator error: {e}"


# ---------------------------------------------------------------------------
class KnowledgeSkill(Skill):
    name = "search_knowledge"
    description = ("Search the attached knowledge base for passages "
                   "relevant to a query.")
    parameters = {"type": "object", "properties": {
        "query": {"type": "string"}}, "required": ["query"]}

    def __init__(self, rag, store, knowledge_names: List[str], owner: str):
        self.rag = rag
        self.store = store
        self.knowledge_names = knowledge_names
        self.owner = owner

    async def execute(self, args: dict, ctx: dict) -> str:
        if self.rag is None:
            return "knowledge search unavailable"
        query = args.get("query", "")
        results = []
        for name in self.knowledge_names:
            kn = self.store.find_one("knowledge", name=name,
                                     owner=self.owner)
            if kn is None or kn.get("state") != "ready":
                continue
            results.extend(await self.rag.query(kn["id"], query))
        if not results:
            return "no relevant passages found"
        results.sort(key=lambda r: -r["score"])
        return "\n\n".join(f"[{i+1}] {r['text']}"
                           for i, r in enumerate(results[:6]))


# ---------------------------------------------------------------------------
class APISkill(Skill):
    """REST tool from an OpenAPI spec (reference api/pkg/tools
    RunAPIAction): each operationId becomes callable; auth headers /
    query params injected."""

    def __init__(self, tool_cfg, http_client=None):
        import httpx
        self.cfg = tool_cfg
        self.base_url = tool_cfg.url.rstrip("/")
        self.headers = dict(tool_cfg.headers or {})
        self.query = dict(tool_cfg.query or {})
        self._http = http_client or httpx.AsyncClient(timeout=30)
        self.operations = self._parse_spec(tool_cfg.schema_)
        self.name = f"api_{tool_cfg.name}".replace(" ", "_").lower()
        self.description = (tool_cfg.description or
                            f"Call the {tool_cfg.name} API")
        ops = {oid: {"type": "object", "properties": {
            "parameters": {"type": "object"},
            "body": {"type": "object"}}}
            for oid in self.operations}
        self.parameters = {"type": "object", "properties": {
            "operation_id": {"type": "string",
                             "enum": list(self.operations.keys())},
            "parameters": {"type": "object",
                           "description": "path/query parameters"},
            "body": {"type": "object", "description": "JSON request body"},
        }, "required": ["operation_id"]}

    @staticmethod
    def _parse_spec(spec_text: str) -> Dict[str, dict]:
        if not spec_text:
            return {}
        try:
            spec = json.loads(spec_text)
        except json.JSONDecodeError:
            import yaml
            spec = yaml.safe_load(spec_text)
        ops = {}
        for path, methods in (spec.get("paths") or {}).items():
            for method, o in methods.items():
                if method.lower() not in ("get", "post", "put", "delete",
                                          "patch"):
                    continue
                oid = o.get("operationId") or f"{method}_{path}"
                ops[oid] = {"path": path, "method": method.upper(),
                            "summary": o.get("summary", ""),
                            "parameters": o.get("parameters", [])}
        return ops

    async def execute(self, args: dict, ctx: dict) -> str:
        oid = args.get("operation_id", "")
        if oid not in self.operations:
            return f"unknown operation: {oid}; known: {list(self.operations)}"
        o = self.operations[oid]
        params = args.get("parameters") or {}
        path = o["path"]
        for k, v in list(params.items()):
            if "{" + k + "}" in path:
                path = path.replace("{" + k + "}", str(v))
                params.pop(k)
        try:
            r = await self._http.request(
                o["method"], self.base_url + path,
                params={**self.query, **params},
                headers=self.headers,
                json=args.get("body") if o["method"] != "GET" else None)
            text = r.text[:4000]
            return f"HTTP {r.status_code}\n{text}"
        except Exception as e:
            return f"API call failed: {e}"


# ---------------------------------------------------------------------------
class WebSearchSkill(Skill):
    name = "web_search"
    description = "Search the web for current information."
    parameters = {"type": "object", "properties": {
        "query": {"type": "string"}}, "required": ["query"]}

    def __init__(self, searxng_url: str = "", http_client=None):
        import httpx
        self.url = searxng_url
        self._http = http_client or httpx.AsyncClient(timeout=20)

    async def execute(self, args: dict, ctx: dict) -> str:
        if not self.url:
            return "web search is not configured (no SEARXNG_URL)"
        try:
            r = await self._http.get(self.url + "/search", params={
                "q": args.get("query", ""), "format": "json"})
            results = r.json().get("results", [])[:5]
            return "\n".join(f"- {x.get('title')}: {x.get('content', '')} "
                             f"({x.get('url')})" for x in results) or \
                "no results"
        except Exception as e:
            return f"web search failed: {e}"


# ---------------------------------------------------------------------------
class MemorySkill(Skill):
    name = "memory"
    description = ("Store or recall durable user memories. "
                   "action=store saves `content`; action=recall lists "
                   "stored memories.")
    parameters = {"type": "object", "properties": {
        "action": {"type": "string", "enum": ["store", "recall"]},
        "content": {"type": "string"}}, "required": ["action"]}

    def __init__(self, store, owner: str, app_id: str = ""):
        self.store = store
        self.owner = owner
        self.app_id = app_id

    async def execute(self, args: dict, ctx: dict) -> str:
        from helix_amd.server.types import new_id
        if args.get("action") == "store":
            mid = new_id("mem")
            self.store.put("memories", mid,
                           {"id": mid, "content": args.get("content", ""),
                            "app_id": self.app_id, "ts": time.time()},
                           owner=self.owner, parent=self.app_id)
            return "memory stored"
        mems = self.store.list("memories", owner=self.owner,
                               parent=self.app_id or None, limit=50)
        if not mems:
            return "no memories stored"
        return "\n".join(f"- {m['content']}" for m in mems)
