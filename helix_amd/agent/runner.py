"""Agent runtime — the skill-based agentic loop (parity with the
reference's api/pkg/agent Agent.Run, agent.go:374): iterate up to
max_iterations, decideNextAction via LLM with skills-as-tools, execute
tool calls, stream the final answer; step-info emitted per step
(observability.go:20 StepInfoEmitter -> pubsub + step_info table).
"""
from __future__ import annotations

import json
import logging
import time
from typing import Any, AsyncIterator, Dict, List, Optional

from helix_amd.agent.skills import (APISkill, BrowserSkill, CalculatorSkill,
                                    EmailSkill, KnowledgeSkill, MemorySkill,
                                    ProjectSkill, RepositorySkill,
                                    SandboxSkill, Skill, WebSearchSkill,
                                    ZapierSkill, build_mcp_skills)
from helix_amd.server import pubsub as ps
from helix_amd.server.types import AssistantConfig, new_id

log = logging.getLogger("helix_amd.agent")

_json_re = __import__("re").compile(r"\{.*\}", __import__("re").S)


class AgentRunner:
    def __init__(self, cfg, store, providers, pubsub, rag=None,
                 notifications=None):
        self.cfg = cfg
        self.store = store
        self.providers = providers
        self.pubsub = pubsub
        self.rag = rag
        self.notifications = notifications
        # late-bound platform services (set by create_app after the
        # full dependency graph exists)
        self.git = None
        self.spec_tasks = None
        self.sandboxes = None
        self.external_agents = None

    def _small_llm(self, assistant: AssistantConfig, owner: str):
        """Async messages->str callable on the small generation model
        slot (used by the browser skill's processOutput pass,
        browser_skill.go:333)."""
        provider, model = self._model_for(assistant,
                                          "small_generation_model")
        client = self.providers.get_client(provider, owner)

        async def call(messages):
            resp = await client.chat({"model": model,
                                      "messages": messages})
            return resp["choices"][0]["message"].get("content", "")
        return call

    # ------------------------------------------------------------------
    def _resolve_secrets(self, owner: str, mapping: dict) -> dict:
        """Interpolate ${secrets.NAME} in tool header/query values
        (reference tools secret injection), decrypting at-rest values."""
        import re as _re
        from helix_amd.server.crypto import decrypt_str, secrets_key
        key = secrets_key(self.cfg.web.admin_api_key)

        def sub(val: str) -> str:
            def repl(m):
                doc = self.store.get("secrets",
                                     f"{owner}:{m.group(1)}")
                if doc is None:
                    return m.group(0)
                try:
                    return decrypt_str(doc.get("value", ""), key)
                except ValueError:
                    return m.group(0)
            return _re.sub(r"\$\{secrets\.([A-Za-z0-9_\-]+)\}", repl,
                           val)

        return {k: sub(v) if isinstance(v, str) else v
                for k, v in (mapping or {}).items()}

    def build_skills(self, assistant: AssistantConfig, owner: str,
                     app_id: str = "") -> List[Skill]:
        skills: List[Skill] = []
        if assistant.calculator.get("enabled", bool(assistant.calculator)):
            skills.append(CalculatorSkill())
        if assistant.knowledge:
            skills.append(KnowledgeSkill(
                self.rag, self.store,
                [k.name for k in assistant.knowledge], owner))
        for api in assistant.apis:
            try:
                sk = APISkill(api)
                sk.headers = self._resolve_secrets(owner, sk.headers)
                sk.query = self._resolve_secrets(owner, sk.query)
                skills.append(sk)
            except Exception as e:
                log.warning("failed to build API skill %s: %s", api.name, e)
        if assistant.web_search.get("enabled",
                                    bool(assistant.web_search)):
            import os
            skills.append(WebSearchSkill(os.environ.get("SEARXNG_URL", "")))
        if assistant.memory.get("enabled", bool(assistant.memory)):
            skills.append(MemorySkill(self.store, owner, app_id))
        if assistant.browser.get("enabled", bool(assistant.browser)):
            skills.append(BrowserSkill(assistant.browser,
                                       llm=self._small_llm(assistant,
                                                           owner)))
        if assistant.email.get("enabled", bool(assistant.email)):
            user = self.store.get("users", owner) or {}
            skills.append(EmailSkill(self.notifications,
                                     user.get("email", "")))
        for zc in assistant.zapier:
            cfg2 = self._resolve_secrets(owner, dict(zc))
            skills.append(ZapierSkill(cfg2))
        if self.git is not None and \
                assistant.repository.get("enabled",
                                         bool(assistant.repository)):
            skills.append(RepositorySkill(self.git, owner))
        if self.spec_tasks is not None and \
                assistant.project.get("enabled", bool(assistant.project)):
            skills.append(ProjectSkill(self.spec_tasks, owner))
        if self.sandboxes is not None and \
                assistant.sandbox.get("enabled", bool(assistant.sandbox)):
            skills.append(SandboxSkill(self.sandboxes, owner,
                                       app_id))
        return skills

    async def resolve_skills(self, assistant: AssistantConfig, owner: str,
                             app_id: str = "") -> List[Skill]:
        """build_skills + remote MCP tool discovery (async because each
        MCP server is queried with tools/list, mcp_skill.go:18)."""
        skills = self.build_skills(assistant, owner, app_id)
        for mcp_cfg in assistant.mcps:
            try:
                skills.extend(await build_mcp_skills(mcp_cfg))
            except Exception as e:
                log.warning("MCP server %s unavailable: %s",
                            mcp_cfg.get("url", "?"), e)
        return skills

    # ------------------------------------------------------------------
    async def is_actionable(self, assistant: AssistantConfig,
                            messages: List[dict], owner: str,
                            ctx: dict) -> dict:
        """Classify whether the latest user input needs a tool call
        (reference api/pkg/tools/informative_or_actionable.go: JSON
        {needs_tool: yes|no, api, justification}, retried, over the
        tools' name+description list and truncated history)."""
        skills = self.build_skills(assistant, owner,
                                   ctx.get("app_id", ""))
        if not skills:
            return {"needs_tool": "no", "api": "",
                    "justification": "no tools available"}
        tool_lines = "\n".join(f"- {s.name}: {s.description}"
                               for s in skills)
        system = (
            "You decide whether the user's last input requires calling "
            "one of the available tools (actionable) or can be answered "
            "directly (informative).\nAvailable tools:\n" + tool_lines +
            "\nRespond with ONLY a JSON object: "
            '{"needs_tool": "yes"|"no", "api": "<tool name or empty>", '
            '"justification": "<why>"}')
        history = [m for m in messages if m.get("role") in
                   ("user", "assistant")][-6:]
        provider, model = self._model_for(assistant,
                                          "small_reasoning_model")
        client = self.providers.get_client(provider, owner)
        from helix_amd.server.providers import LoggingClient
        client = LoggingClient(client, self.store)
        req_messages = [{"role": "system", "content": system}] + history
        if req_messages[-1]["role"] == "user":
            req_messages[-1] = dict(
                req_messages[-1],
                content=(req_messages[-1].get("content") or "") +
                "\nReturn the corresponding json for the last user input")
        for attempt in range(3):
            resp = await client.chat({
                "model": model, "messages": req_messages,
                "temperature": 0.0,
                "_ctx": {**ctx, "step": "is_actionable"}})
            text = resp["choices"][0]["message"].get("content", "")
            m = _json_re.search(text)
            if m:
                try:
                    out = json.loads(m.group(0))
                    if str(out.get("needs_tool", "")).lower() in \
                            ("yes", "no"):
                        out["needs_tool"] = out["needs_tool"].lower()
                        return out
                except json.JSONDecodeError:
                    pass
            log.warning("is_actionable parse failure (attempt %d): %r",
                        attempt, text[:200])
        return {"needs_tool": "no", "api": "",
                "justification": "classification failed; informative"}

    def _model_for(self, assistant: AssistantConfig, slot: str) -> tuple:
        """4-slot model selection (reference llm_client.go:14-19)."""
        mc = getattr(assistant, slot, None)
        if mc is not None and mc.model:
            return (mc.provider or assistant.provider or
                    self.cfg.inference.default_provider, mc.model)
        return (assistant.provider or self.cfg.inference.default_provider,
                assistant.model or self.cfg.inference.default_model)

    def _emit_step(self, ctx: dict, step: dict):
        sid = new_id("step")
        doc = {"id": sid, "session_id": ctx.get("session_id", ""),
               "interaction_id": ctx.get("interaction_id", ""),
               "created": time.time(), **step}
        self.store.put("step_info", sid, doc, owner=ctx.get("owner", ""),
                       parent=ctx.get("session_id", ""), buffered=True)
        return doc

    async def _publish_step(self, ctx: dict, doc: dict):
        owner = ctx.get("owner", "")
        session_id = ctx.get("session_id", "")
        if owner and session_id:
            await self.pubsub.publish(ps.session_queue(owner, session_id),
                                      {"type": "step_info", "step": doc})

    # ------------------------------------------------------------------
    async def _loop(self, assistant: AssistantConfig, req: dict, owner: str,
                    ctx: dict) -> tuple:
        """Run the agentic loop; returns (messages, final_req) where
        final_req is what produces the user-facing answer."""
        app_id = ctx.get("app_id", "")
        skills = await self.resolve_skills(assistant, owner, app_id)
        tools = [s.to_tool() for s in skills]
        by_name = {s.name: s for s in skills}

        messages = [dict(m) for m in req.get("messages", [])]
        if assistant.system_prompt and not any(
                m.get("role") == "system" for m in messages):
            messages.insert(0, {"role": "system",
                                "content": assistant.system_prompt})
        provider, model = self._model_for(assistant, "generation_model")
        r_provider, r_model = self._model_for(assistant, "reasoning_model")
        client = self.providers.get_client(r_provider, owner)
        from helix_amd.server.providers import LoggingClient
        client = LoggingClient(client, self.store)

        max_iter = self.cfg.agent_max_iterations
        if not skills:
            # no-skill direct path (reference agent.go:290)
            return messages, {"model": model, "provider": provider}

        for it in range(max_iter):
            decide_req = {
                "model": r_model,
                "messages": messages,
                "tools": tools,
                "tool_choice": "auto",
                "temperature": (assistant.temperature
                                if assistant.temperature is not None else 0.1),
                "_ctx": {**ctx, "step": f"agent_decide_{it}"},
            }
            resp = await client.chat(decide_req)
            choice = resp["choices"][0]
            msg = choice.get("message", {})
            tool_calls = msg.get("tool_calls") or []
            if not tool_calls:
                # model answered directly — done
                messages.append({"role": "assistant",
                                 "content": msg.get("content", "")})
                return messages, None
            messages.append({"role": "assistant",
                             "content": msg.get("content") or "",
                             "tool_calls": tool_calls})
            for tc in tool_calls:
                fn = tc.get("function", {})
                name = fn.get("name", "")
                try:
                    args = json.loads(fn.get("arguments") or "{}")
                except json.JSONDecodeError:
                    args = {}
                skill = by_name.get(name)
                t0 = time.time()
                if skill is None:
                    result = f"unknown tool: {name}"
                else:
                    try:
                        result = await skill.execute(args, ctx)
                    except Exception as e:
                        # a failing tool is feedback for the model, not
                        # a dead agent turn (reference tool-error flow)
                        log.warning("tool %s failed: %s", name, e)
                        result = f"tool error: {e}"
                if not isinstance(result, str):
                    result = json.dumps(result, default=str)
                doc = self._emit_step(ctx, {
                    "step": name, "arguments": args,
                    "result": result[:2000],
                    "duration_ms": int((time.time() - t0) * 1000),
                    "iteration": it})
                await self._publish_step(ctx, doc)
                messages.append({"role": "tool",
                                 "tool_call_id": tc.get("id", ""),
                                 "name": name, "content": result})
        # iteration cap reached: ask for a final summary answer
        messages.append({
            "role": "user",
            "content": "Summarize your findings and answer the original "
                       "question now, without calling more tools."})
        return messages, {"model": model, "provider": provider}

    # ------------------------------------------------------------------
    def _external_key(self, ctx: dict) -> str:
        return ctx.get("session_id") or ctx.get("app_id") or "default"

    def _external_payload(self, assistant, req, ctx) -> dict:
        return {"messages": req.get("messages", []),
                "model": assistant.model or req.get("model", ""),
                "session_id": ctx.get("session_id", ""),
                "interaction_id": ctx.get("interaction_id", "")}

    async def run_blocking(self, assistant: AssistantConfig, req: dict,
                           owner: str, ctx: dict) -> dict:
        if assistant.agent_type == "zed_external" and \
                self.external_agents is not None:
            # turns execute on the externally connected agent
            # (reference controller_external_agent.go RunExternalAgent)
            from helix_amd.server.external_agent import ExternalAgentError
            try:
                text = await self.external_agents.run_turn_blocking(
                    self._external_key(ctx),
                    self._external_payload(assistant, req, ctx))
            except ExternalAgentError as e:
                text = f"[external agent error] {e}"
            return _completion_dict(req.get("model", ""), text)
        messages, final = await self._loop(assistant, req, owner, ctx)
        if final is None and messages and messages[-1]["role"] == "assistant":
            content = messages[-1].get("content", "")
            return _completion_dict(req.get("model", ""), content)
        provider, model = (final or {}).get("provider"), \
            (final or {}).get("model")
        client = self.providers.get_client(
            provider or self.cfg.inference.default_provider, owner)
        from helix_amd.server.providers import LoggingClient
        resp = await LoggingClient(client, self.store).chat({
            "model": model, "messages": messages,
            "temperature": assistant.temperature,
            "max_tokens": assistant.max_tokens,
            "_ctx": {**ctx, "step": "agent_final"}})
        return resp

    async def run_stream(self, assistant: AssistantConfig, req: dict,
                         owner: str, ctx: dict) -> AsyncIterator[dict]:
        if assistant.agent_type == "zed_external" and \
                self.external_agents is not None:
            from helix_amd.server.external_agent import ExternalAgentError
            base_id = new_id("chatcmpl")
            created = int(time.time())
            model = req.get("model", "")
            try:
                async for chunk in self.external_agents.run_turn(
                        self._external_key(ctx),
                        self._external_payload(assistant, req, ctx)):
                    yield {"id": base_id,
                           "object": "chat.completion.chunk",
                           "created": created, "model": model,
                           "choices": [{"index": 0, "delta": {
                               "content": chunk},
                               "finish_reason": None}]}
            except ExternalAgentError as e:
                yield {"id": base_id, "object": "chat.completion.chunk",
                       "created": created, "model": model,
                       "choices": [{"index": 0, "delta": {
                           "content": f"[external agent error] {e}"},
                           "finish_reason": None}]}
            yield {"id": base_id, "object": "chat.completion.chunk",
                   "created": created, "model": model,
                   "choices": [{"index": 0, "delta": {},
                                "finish_reason": "stop"}]}
            return
        messages, final = await self._loop(assistant, req, owner, ctx)
        if final is None and messages and messages[-1]["role"] == "assistant":
            content = messages[-1].get("content", "")
            base = _completion_dict(req.get("model", ""), content)
            yield {
                "id": base["id"], "object": "chat.completion.chunk",
                "created": base["created"], "model": base["model"],
                "choices": [{"index": 0, "delta": {
                    "role": "assistant", "content": content},
                    "finish_reason": None}]}
            yield {
                "id": base["id"], "object": "chat.completion.chunk",
                "created": base["created"], "model": base["model"],
                "choices": [{"index": 0, "delta": {},
                             "finish_reason": "stop"}]}
            return
        provider, model = (final or {}).get("provider"), \
            (final or {}).get("model")
        client = self.providers.get_client(
            provider or self.cfg.inference.default_provider, owner)
        from helix_amd.server.providers import LoggingClient
        async for chunk in LoggingClient(client, self.store).chat_stream({
                "model": model, "messages": messages, "stream": True,
                "temperature": assistant.temperature,
                "max_tokens": assistant.max_tokens,
                "_ctx": {**ctx, "step": "agent_final"}}):
            yield chunk


def _completion_dict(model: str, content: str) -> dict:
    return {
        "id": new_id("chatcmpl"),
        "object": "chat.completion",
        "created": int(time.time()),
        "model": model,
        "choices": [{"index": 0, "message": {
            "role": "assistant", "content": content},
            "finish_reason": "stop"}],
        "usage": {"prompt_tokens": 0, "completion_tokens": 0,
                  "total_tokens": 0},
    }
