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
        self.cfg = tool_cfg
        self.base_url = tool_cfg.url.rstrip("/")
        self.headers = dict(tool_cfg.headers or {})
        self.query = dict(tool_cfg.query or {})
        self._http = http_client   # lazy: skills are built per request
                                   # (incl. is_actionable) — eager
                                   # clients leak sockets under load

        self.operations = self._parse_spec(tool_cfg.schema_)
        self.name = f"api_{tool_cfg.name}".replace(" ", "_").lower()
        self.description = (tool_cfg.description or
                            f"Call the {tool_cfg.name} API")
        self.parameters = {"type": "object", "properties": {
            "operation_id": {"type": "string",
                             "enum": list(self.operations.keys())},
            "parameters": {"type": "object",
                           "description": "path/query parameters"},
            "body": {"type": "object", "description": "JSON request body"},
        }, "required": ["operation_id"]}

    def _client(self):
        if self._http is None:
            import httpx
            self._http = httpx.AsyncClient(timeout=30)
        return self._http

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
            r = await self._client().request(
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
        self.url = searxng_url
        self._http = http_client

    def _client(self):
        if self._http is None:
            import httpx
            self._http = httpx.AsyncClient(timeout=20)
        return self._http

    async def execute(self, args: dict, ctx: dict) -> str:
        if not self.url:
            return "web search is not configured (no SEARXNG_URL)"
        try:
            r = await self._client().get(self.url + "/search", params={
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


# ---------------------------------------------------------------------------
class BrowserSkill(Skill):
    """Fetch a URL, run readability extraction, and (optionally) distill
    the content with the small generation model (reference
    api/pkg/agent/skill/browser_skill.go:89-179: url+prompt parameters,
    NoBrowser HTTP fetch path, readability parse, processOutput LLM
    pass, per-URL cache)."""

    name = "browser"
    description = ("Open a URL in a browser, extract the readable "
                   "content, and answer a prompt about it.")
    parameters = {"type": "object", "properties": {
        "url": {"type": "string", "description": "The URL to visit"},
        "prompt": {"type": "string",
                   "description": "The prompt to run on the fetched "
                                  "content"}},
        "required": ["url", "prompt"]}

    def __init__(self, config: dict | None = None, llm=None,
                 http_client=None):
        self.config = config or {}
        self.llm = llm                  # async callable(messages) -> str
        self._http = http_client
        self._cache: Dict[str, str] = {}

    def _client(self):
        if self._http is None:
            import httpx
            self._http = httpx.AsyncClient(timeout=30,
                                           follow_redirects=True)
        return self._http

    async def execute(self, args: dict, ctx: dict) -> str:
        from helix_amd.server.extract import extract_html
        url = args.get("url", "")
        prompt = args.get("prompt", "")
        if not url:
            return "browser error: url is required"
        if not url.startswith(("http://", "https://")):
            return "browser error: only http(s) URLs are supported"
        text = self._cache.get(url) if self.config.get("cache", True) \
            else None
        if text is None:
            try:
                r = await self._client().get(url)
            except Exception as e:
                return f"browser error: {e}"
            if r.status_code >= 400:
                return f"browser error: HTTP {r.status_code} for {url}"
            page = extract_html(r.text)
            title = f"# {page['title']}\n\n" if page["title"] else ""
            text = (title + page["text"])[:16000]
            if self.config.get("cache", True):
                self._cache[url] = text
        if self.config.get("process_output", True) and \
                self.llm is not None and prompt:
            try:
                return await self.llm([
                    {"role": "system",
                     "content": "You extract the information requested "
                                "from fetched web content. Answer from "
                                "the content only."},
                    {"role": "user",
                     "content": f"Request: {prompt}\n\nContent:\n{text}"}])
            except Exception as e:
                log.warning("browser process_output failed: %s", e)
        return text


# ---------------------------------------------------------------------------
class EmailSkill(Skill):
    """Send an email to the requesting user (reference
    email_sending_skill.go:88-160: subject+message parameters, delivery
    to meta.UserEmail only — an agent cannot email arbitrary
    addresses)."""

    name = "send_email"
    description = "Send an email to the current user."
    parameters = {"type": "object", "properties": {
        "subject": {"type": "string"},
        "message": {"type": "string"}},
        "required": ["subject", "message"]}

    def __init__(self, notifications, user_email: str):
        self.notifications = notifications
        self.user_email = user_email

    async def execute(self, args: dict, ctx: dict) -> str:
        subject = args.get("subject", "")
        message = args.get("message", "")
        if not subject or not message:
            return "email error: subject and message are required"
        if not self.user_email:
            return "email error: the current user has no email address"
        email = getattr(self.notifications, "email", None)
        if email is None or not getattr(email, "host", ""):
            return "email error: no email provider configured"
        ok = email.send(self.user_email, subject, message)
        return "Email sent" if ok else "email error: delivery failed"


# ---------------------------------------------------------------------------
class MCPClient:
    """Minimal MCP client over streamable-HTTP JSON-RPC (reference
    api/pkg/agent/skill/mcp/mcp_client.go): initialize, tools/list,
    tools/call. Compatible with helix_amd's own MCP gateway."""

    def __init__(self, url: str, headers: dict | None = None,
                 http_client=None):
        import httpx
        self.url = url
        self.headers = headers or {}
        self._http = http_client or httpx.AsyncClient(timeout=30)
        self._id = 0
        self._initialized = False

    async def _rpc(self, method: str, params: dict | None = None):
        self._id += 1
        r = await self._http.post(self.url, headers=self.headers, json={
            "jsonrpc": "2.0", "id": self._id, "method": method,
            "params": params or {}})
        body = r.json()
        if "error" in body:
            raise RuntimeError(f"MCP {method}: {body['error'].get('message')}")
        return body.get("result", {})

    async def _ensure_init(self):
        if not self._initialized:
            await self._rpc("initialize", {
                "protocolVersion": "2024-11-05",
                "clientInfo": {"name": "helix_amd", "version": "1.0"},
                "capabilities": {}})
            self._initialized = True

    async def list_tools(self) -> List[dict]:
        await self._ensure_init()
        return (await self._rpc("tools/list")).get("tools", [])

    async def call_tool(self, name: str, arguments: dict) -> str:
        await self._ensure_init()
        result = await self._rpc("tools/call",
                                 {"name": name, "arguments": arguments})
        parts = []
        for c in result.get("content", []):
            if c.get("type") == "text":
                parts.append(c.get("text", ""))
            else:
                parts.append(json.dumps(c))
        out = "\n".join(parts)
        if result.get("isError"):
            return f"tool error: {out}"
        return out


def _sanitize_tool_name(name: str) -> str:
    import re as _re
    return _re.sub(r"[^A-Za-z0-9_\-]", "_", name)[:64]


class MCPToolSkill(Skill):
    """One remote MCP tool surfaced as an agent skill (reference
    mcp_skill.go:18-60: per-tool skills named mcp_<name>, inputSchema
    converted to OpenAI tool parameters)."""

    def __init__(self, client: MCPClient, tool: dict):
        self.client = client
        self.tool_name = tool.get("name", "")
        self.name = f"mcp_{_sanitize_tool_name(self.tool_name)}"
        self.description = tool.get("description", "") or \
            f"MCP tool {self.tool_name}"
        schema = tool.get("inputSchema") or {}
        self.parameters = {
            "type": "object",
            "properties": schema.get("properties", {}) or {},
            **({"required": schema["required"]}
               if schema.get("required") else {})}

    async def execute(self, args: dict, ctx: dict) -> str:
        try:
            return await self.client.call_tool(self.tool_name, args)
        except Exception as e:
            return f"mcp error: {e}"


async def build_mcp_skills(mcp_cfg: dict, http_client=None) -> List[Skill]:
    """Connect to one configured MCP server and surface each remote
    tool (config shape: {url, headers?} — reference ToolMCPClientConfig)."""
    client = MCPClient(mcp_cfg.get("url", ""),
                       mcp_cfg.get("headers") or {}, http_client)
    return [MCPToolSkill(client, t) for t in await client.list_tools()]


# ---------------------------------------------------------------------------
class RepositorySkill(Skill):
    """Repository tools over the in-platform git service (reference
    api/pkg/agent/skill/repository: list_files, find_files, get_file,
    grep — helix_repository_skill.go). One skill, action-dispatched,
    scoped to the owner's repositories."""

    name = "repository"
    description = ("Explore the user's git repositories: "
                   "action=list_repos|list_files|find_files|get_file|"
                   "grep.")
    parameters = {"type": "object", "properties": {
        "action": {"type": "string",
                   "enum": ["list_repos", "list_files", "find_files",
                            "get_file", "grep"]},
        "repo_id": {"type": "string"},
        "path": {"type": "string",
                 "description": "file path (get_file)"},
        "pattern": {"type": "string",
                    "description": "glob (find_files) or regex (grep)"},
        "ref": {"type": "string", "description": "git ref, default HEAD"}},
        "required": ["action"]}

    def __init__(self, git, owner: str):
        self.git = git
        self.owner = owner

    def _repo(self, repo_id: str):
        doc = self.git.get(repo_id)
        if doc is None or doc.get("owner") != self.owner:
            raise ValueError(f"repository not found: {repo_id}")
        return doc

    async def execute(self, args: dict, ctx: dict) -> str:
        import fnmatch
        import re as _re
        action = args.get("action", "")
        ref = args.get("ref") or "HEAD"
        if action == "list_repos":
            repos = self.git.list(self.owner)
            return "\n".join(f"{r['id']}  {r.get('name', '')}"
                             for r in repos) or "no repositories"
        rid = args.get("repo_id", "")
        self._repo(rid)
        if action == "list_files":
            return "\n".join(self.git.ls_tree(rid, ref)[:500])
        if action == "find_files":
            pat = args.get("pattern", "*")
            hits = [p for p in self.git.ls_tree(rid, ref)
                    if fnmatch.fnmatch(p, pat)]
            return "\n".join(hits[:200]) or "no matches"
        if action == "get_file":
            return self.git.read_file(rid, args.get("path", ""),
                                      ref)[:16000]
        if action == "grep":
            rx = _re.compile(args.get("pattern", ""))
            out = []
            for p in self.git.ls_tree(rid, ref)[:500]:
                try:
                    content = self.git.read_file(rid, p, ref)
                except Exception:
                    continue
                for i, line in enumerate(content.splitlines(), 1):
                    if rx.search(line):
                        out.append(f"{p}:{i}: {line.strip()[:200]}")
                        if len(out) >= 100:
                            return "\n".join(out)
            return "\n".join(out) or "no matches"
        return f"unknown action: {action}"


class ProjectSkill(Skill):
    """Spec-task project management (reference
    api/pkg/agent/skill/project: create/get/list/update/start spec
    tasks — the optimus PM agent's tool family)."""

    name = "project"
    description = ("Manage spec-driven tasks on the user's projects: "
                   "action=list_projects|list_tasks|get_task|"
                   "create_task|update_task|start_task.")
    parameters = {"type": "object", "properties": {
        "action": {"type": "string",
                   "enum": ["list_projects", "list_tasks", "get_task",
                            "create_task", "update_task", "start_task"]},
        "project_id": {"type": "string"},
        "task_id": {"type": "string"},
        "title": {"type": "string"},
        "description": {"type": "string"},
        "state": {"type": "string",
                  "description": "target state (update_task)"}},
        "required": ["action"]}

    def __init__(self, spec_tasks, owner: str):
        self.svc = spec_tasks
        self.owner = owner

    def _task(self, task_id: str):
        doc = self.svc.get_task(task_id)
        if doc is None or doc.get("owner") != self.owner:
            raise ValueError(f"task not found: {task_id}")
        return doc

    async def execute(self, args: dict, ctx: dict) -> str:
        action = args.get("action", "")
        if action == "list_projects":
            ps = self.svc.list_projects(self.owner)
            return "\n".join(f"{p['id']}  {p.get('name', '')}"
                             for p in ps) or "no projects"
        if action == "list_tasks":
            ts = self.svc.list_tasks(args.get("project_id", ""))
            return "\n".join(
                f"{t['id']}  [{t.get('state')}] {t.get('title', '')}"
                for t in ts if t.get("owner") == self.owner) or "no tasks"
        if action == "get_task":
            t = self._task(args.get("task_id", ""))
            return json.dumps({k: t.get(k) for k in
                               ("id", "title", "state", "description",
                                "spec", "branch", "verify")},
                              default=str)
        if action == "create_task":
            t = self.svc.create_task(self.owner,
                                     args.get("project_id", ""),
                                     args.get("title", ""),
                                     args.get("description", ""))
            return f"created {t['id']} in backlog"
        if action == "update_task":
            self._task(args.get("task_id", ""))
            t = self.svc.transition(args.get("task_id", ""),
                                    args.get("state", ""))
            return f"{t['id']} -> {t['state']}"
        if action == "start_task":
            self._task(args.get("task_id", ""))
            t = await self.svc.plan(args.get("task_id", ""))
            return f"{t['id']} planned -> {t['state']}"
        return f"unknown action: {action}"


class SandboxSkill(Skill):
    """Run shell commands in the session's sandbox workspace
    (reference: hydra dev-container exec surfaced to agents)."""

    name = "run_command"
    description = ("Run a shell command in your sandboxed workspace "
                   "and get stdout/stderr back.")
    parameters = {"type": "object", "properties": {
        "command": {"type": "string"},
        "timeout_s": {"type": "number"}},
        "required": ["command"]}

    def __init__(self, sandboxes, owner: str, session_id: str = ""):
        self.sandboxes = sandboxes
        self.owner = owner
        self.session_id = session_id
        self._sandbox_id: Optional[str] = None

    def _ensure(self) -> str:
        if self._sandbox_id:
            doc = self.sandboxes.get(self._sandbox_id)
            if doc is not None:
                return self._sandbox_id
        for doc in self.sandboxes.list(self.owner):
            if doc.get("session_id") == self.session_id:
                self._sandbox_id = doc["id"]
                return doc["id"]
        doc = self.sandboxes.create(self.owner, name="agent-workspace",
                                    session_id=self.session_id)
        self._sandbox_id = doc["id"]
        return doc["id"]

    async def execute(self, args: dict, ctx: dict) -> str:
        import asyncio as _aio
        sid = self._ensure()
        timeout = min(float(args.get("timeout_s") or 60), 300)
        r = await _aio.to_thread(self.sandboxes.exec, sid,
                                 args.get("command", ""), timeout)
        out = r["stdout"]
        if r["stderr"]:
            out += ("\n[stderr]\n" + r["stderr"])
        if r["timed_out"]:
            out += "\n[timed out]"
        return f"exit={r['exit_code']}\n{out}"[:8000]


# ---------------------------------------------------------------------------
class ZapierSkill(Skill):
    """Zapier NLA tool (reference api/pkg/tools/zapier.go: exposed
    actions listed and executed with natural-language instructions via
    the user's NLA API key)."""

    name = "zapier"
    description = ("Run the user's Zapier actions (email, sheets, CRM "
                   "...): action=list shows them, action=execute runs "
                   "one with natural-language instructions.")
    parameters = {"type": "object", "properties": {
        "action": {"type": "string", "enum": ["list", "execute"]},
        "action_id": {"type": "string"},
        "instructions": {"type": "string"}},
        "required": ["action"]}

    BASE = "https://nla.zapier.com/api/v1"

    def __init__(self, config: dict, http_client=None):
        self.api_key = (config or {}).get("api_key", "")
        self._http = http_client

    def _client(self):
        if self._http is None:
            import httpx
            self._http = httpx.AsyncClient(timeout=60)
        return self._http

    def _headers(self):
        return {"X-API-Key": self.api_key}

    async def execute(self, args: dict, ctx: dict) -> str:
        if not self.api_key:
            return "zapier error: no API key configured"
        try:
            if args.get("action") == "list":
                r = await self._client().get(self.BASE + "/exposed/",
                                             headers=self._headers())
                if r.status_code != 200:
                    return f"zapier error: HTTP {r.status_code}"
                rows = r.json().get("results", [])
                return "\n".join(f"{x.get('id')}: {x.get('description')}"
                                 for x in rows) or "no exposed actions"
            aid = args.get("action_id", "")
            if not aid:
                return "zapier error: action_id required for execute"
            r = await self._client().post(
                self.BASE + f"/exposed/{aid}/execute/",
                headers=self._headers(),
                json={"instructions": args.get("instructions", "")})
            if r.status_code != 200:
                return f"zapier error: HTTP {r.status_code}"
            out = r.json()
            if out.get("status") == "error":
                return f"zapier error: {out.get('error', 'unknown')}"
            return json.dumps(out.get("result",
                                      out))[:4000]
        except Exception as e:
            return f"zapier error: {e}"
