"""Controller — the ChatCompletion pipeline and session lifecycle
(parity with api/pkg/controller: inference.go ChatCompletion /
ChatCompletionStream, loadAssistant, enrichPromptWithKnowledge,
sessions.go).
"""
from __future__ import annotations

import logging
import time
from typing import Any, AsyncIterator, Dict, List, Optional

from helix_amd.server import pubsub as ps
from helix_amd.server.providers import ProviderError, ProviderManager
from helix_amd.server.types import (App, AssistantConfig, Interaction,
                                    InteractionState, Session, new_id, now_ms)

log = logging.getLogger("helix_amd.controller")


class Controller:
    def __init__(self, cfg, store, providers: ProviderManager,
                 pubsub: ps.PubSub, rag=None, agent_runner=None, usage=None):
        self.cfg = cfg
        self.store = store
        self.providers = providers
        self.pubsub = pubsub
        self.rag = rag
        self.agent_runner = agent_runner
        self.usage = usage

    def _check_quota(self, owner: str, provider: str):
        if self.usage is None:
            return
        limit = getattr(self.cfg, "daily_token_limit", 0)
        if limit <= 0:
            return
        from helix_amd.server.usage import QuotaExceededError
        try:
            self.usage.check_quota(owner, provider, limit)
        except QuotaExceededError as e:
            raise ProviderError(str(e), 429)

    def _logging(self, client):
        from helix_amd.server.providers import LoggingClient
        logger = self.usage.log_call if self.usage is not None else None
        return LoggingClient(client, self.store, usage_logger=logger)

    # ------------------------------------------------------------------
    # Assistant / app resolution (reference inference.go:81 loadAssistant)
    # ------------------------------------------------------------------
    def load_assistant(self, app_id: str,
                       assistant_id: str = "") -> tuple[Optional[App],
                                                        Optional[AssistantConfig]]:
        if not app_id:
            return None, None
        doc = self.store.get("apps", app_id)
        if doc is None:
            raise ProviderError(f"app not found: {app_id}", 404)
        app = App.model_validate(doc)
        assistants = app.config.assistants
        if not assistants:
            return app, None
        if assistant_id:
            for a in assistants:
                if a.id == assistant_id or a.name == assistant_id:
                    return app, a
            raise ProviderError(f"assistant not found: {assistant_id}", 404)
        return app, assistants[0]

    # ------------------------------------------------------------------
    # Knowledge enrichment (reference inference.go:1097-1356)
    # ------------------------------------------------------------------
    async def enrich_with_knowledge(self, assistant: AssistantConfig,
                                    messages: List[dict],
                                    owner: str) -> List[dict]:
        if self.rag is None or not assistant or not assistant.knowledge:
            return messages
        last_user = next((m for m in reversed(messages)
                          if m.get("role") == "user"), None)
        if last_user is None:
            return messages
        query = last_user.get("content") or ""
        if not isinstance(query, str):
            return messages
        chunks = []
        for source in assistant.knowledge:
            kn = self.store.find_one("knowledge", name=source.name,
                                     owner=owner)
            if kn is None or kn.get("state") != "ready":
                continue
            results = await self.rag.query(kn["id"], query,
                                           self.cfg.rag.results_count)
            chunks.extend(results)
        if not chunks:
            return messages
        ctx = "\n\n".join(f"[{i+1}] {c['text']}" for i, c in enumerate(chunks))
        prompt = (
            "Use the following context to answer the question.\n"
            f"<context>\n{ctx}\n</context>\n\n"
            f"Question: {query}")
        out = [dict(m) for m in messages]
        for m in reversed(out):
            if m.get("role") == "user":
                m["content"] = prompt
                break
        return out

    # ------------------------------------------------------------------
    # ChatCompletion (reference inference.go:52 / :229)
    # ------------------------------------------------------------------
    def _apply_assistant(self, req: dict,
                         assistant: Optional[AssistantConfig]) -> dict:
        if assistant is None:
            return req
        out = dict(req)
        msgs = [dict(m) for m in out.get("messages", [])]
        if assistant.system_prompt and not any(
                m.get("role") == "system" for m in msgs):
            msgs.insert(0, {"role": "system",
                            "content": assistant.system_prompt})
        if assistant.context_limit and len(msgs) > assistant.context_limit:
            sys_msgs = [m for m in msgs if m.get("role") == "system"]
            rest = [m for m in msgs if m.get("role") != "system"]
            msgs = sys_msgs + rest[-assistant.context_limit:]
        out["messages"] = msgs
        if assistant.model and not out.get("model"):
            out["model"] = assistant.model
        for field in ("temperature", "top_p", "presence_penalty",
                      "frequency_penalty", "max_tokens"):
            v = getattr(assistant, field)
            if v is not None and out.get(field) is None:
                out[field] = v
        # reasoning-effort gating (reference llm_client.go:46: only pass
        # the field when the assistant configures it)
        if assistant.reasoning_effort and not out.get("reasoning_effort"):
            out["reasoning_effort"] = assistant.reasoning_effort
        return out

    def _resolve(self, req: dict, owner: str,
                 assistant: Optional[AssistantConfig]) -> tuple:
        provider = (assistant.provider if assistant and assistant.provider
                    else None)
        model = req.get("model") or ""
        if provider is None:
            provider, model = self.providers.resolve(
                model or self.cfg.inference.default_model,
                self.cfg.inference.default_provider, owner)
        if not model:
            model = self.cfg.inference.default_model
        return provider, model

    @staticmethod
    def _has_tools(assistant) -> bool:
        """Non-agent assistants with tools go through the actionable
        classification chain (reference ChainStrategy.IsActionable,
        tools/informative_or_actionable.go) before tool execution."""
        return bool(assistant.apis or assistant.mcps or
                    assistant.browser or assistant.email or
                    assistant.zapier)

    async def chat_completion(self, req: dict, owner: str,
                              app_id: str = "", assistant_id: str = "",
                              ctx: Optional[dict] = None) -> dict:
        app, assistant = self.load_assistant(app_id, assistant_id)
        if assistant is not None and (assistant.agent_mode or
                                      assistant.agent_type):
            if self.agent_runner is None:
                raise ProviderError("agent runtime unavailable", 500)
            return await self.agent_runner.run_blocking(
                assistant, req, owner, ctx or {})
        if assistant is not None and self.agent_runner is not None and \
                self._has_tools(assistant):
            cls = await self.agent_runner.is_actionable(
                assistant, req.get("messages", []), owner,
                {**(ctx or {}), "app_id": app.id if app else ""})
            if cls.get("needs_tool") == "yes":
                return await self.agent_runner.run_blocking(
                    assistant, req, owner, ctx or {})
        req = self._apply_assistant(req, assistant)
        if assistant is not None:
            req["messages"] = await self.enrich_with_knowledge(
                assistant, req["messages"], owner)
        provider, model = self._resolve(req, owner, assistant)
        self._check_quota(owner, provider)
        client = self.providers.get_client(provider, owner)
        call_req = {**req, "model": model, "_ctx": ctx or {"owner": owner}}
        return await self._logging(client).chat(call_req)

    async def chat_completion_stream(self, req: dict, owner: str,
                                     app_id: str = "", assistant_id: str = "",
                                     ctx: Optional[dict] = None
                                     ) -> AsyncIterator[dict]:
        app, assistant = self.load_assistant(app_id, assistant_id)
        if assistant is not None and (assistant.agent_mode or
                                      assistant.agent_type):
            if self.agent_runner is None:
                raise ProviderError("agent runtime unavailable", 500)
            async for chunk in self.agent_runner.run_stream(
                    assistant, req, owner, ctx or {}):
                yield chunk
            return
        if assistant is not None and self.agent_runner is not None and \
                self._has_tools(assistant):
            cls = await self.agent_runner.is_actionable(
                assistant, req.get("messages", []), owner,
                {**(ctx or {}), "app_id": app.id if app else ""})
            if cls.get("needs_tool") == "yes":
                async for chunk in self.agent_runner.run_stream(
                        assistant, req, owner, ctx or {}):
                    yield chunk
                return
        req = self._apply_assistant(req, assistant)
        if assistant is not None:
            req["messages"] = await self.enrich_with_knowledge(
                assistant, req["messages"], owner)
        provider, model = self._resolve(req, owner, assistant)
        self._check_quota(owner, provider)
        client = self.providers.get_client(provider, owner)
        call_req = {**req, "model": model, "stream": True,
                    "_ctx": ctx or {"owner": owner}}
        async for chunk in self._logging(client).chat_stream(call_req):
            yield chunk

    # ------------------------------------------------------------------
    # Sessions (reference session_handlers.go startChatSessionHandler)
    # ------------------------------------------------------------------
    def create_session(self, owner: str, model: str = "",
                       provider: str = "", app_id: str = "",
                       name: str = "") -> Session:
        s = Session(owner=owner, model_name=model, provider=provider,
                    parent_app=app_id, name=name or "New Session")
        self.store.put("sessions", s.id, s.model_dump(), owner=owner,
                       parent=app_id)
        return s

    def get_session(self, session_id: str) -> Optional[Session]:
        doc = self.store.get("sessions", session_id)
        return Session.model_validate(doc) if doc else None

    def list_sessions(self, owner: str) -> List[Session]:
        return [Session.model_validate(d)
                for d in self.store.list("sessions", owner=owner)]

    def delete_session(self, session_id: str) -> bool:
        for i in self.store.list("interactions", parent=session_id):
            self.store.delete("interactions", i["id"])
        return self.store.delete("sessions", session_id)

    def add_interaction(self, session: Session, prompt: str) -> Interaction:
        it = Interaction(session_id=session.id, prompt_message=prompt,
                         state=InteractionState.WAITING)
        self.store.put("interactions", it.id, it.model_dump(),
                       owner=session.owner, parent=session.id)
        return it

    def session_history(self, session_id: str) -> List[dict]:
        msgs = []
        for it in self.store.list("interactions", parent=session_id,
                                  desc=False):
            msgs.append({"role": "user", "content": it["prompt_message"]})
            if it.get("response_message"):
                msgs.append({"role": "assistant",
                             "content": it["response_message"]})
        return msgs

    async def run_session_turn(self, session: Session, interaction: Interaction,
                               stream_to_pubsub: bool = True
                               ) -> AsyncIterator[dict]:
        """Run one turn; yields chunks AND republishes to the session
        topic (the WS path the frontend watches, pubsub.go:74)."""
        messages = self.session_history(session.id)
        req = {
            "model": session.model_name or self.cfg.inference.default_model,
            "messages": messages,
            "stream": True,
        }
        ctx = {"owner": session.owner, "session_id": session.id,
               "interaction_id": interaction.id}
        t0 = time.monotonic()
        text = ""
        first_ms = 0
        last_persist = t0
        try:
            async for chunk in self.chat_completion_stream(
                    req, session.owner, app_id=session.parent_app, ctx=ctx):
                delta = ""
                if chunk.get("choices"):
                    delta = chunk["choices"][0].get("delta", {}).get(
                        "content") or ""
                if delta and not first_ms:
                    first_ms = int((time.monotonic() - t0) * 1000)
                text += delta
                # 200 ms partial-response write throttle (reference
                # wsprotocol accumulator, design/2026-02-25: O(N) deltas
                # on the wire, throttled DB writes — not per-token)
                now = time.monotonic()
                if delta and now - last_persist >= 0.2:
                    interaction.response_message = text
                    interaction.updated = now_ms()   # wedge-detector pulse
                    self.store.put("interactions", interaction.id,
                                   interaction.model_dump(),
                                   owner=session.owner, parent=session.id)
                    last_persist = now
                if stream_to_pubsub:
                    await self.pubsub.publish(
                        ps.session_queue(session.owner, session.id),
                        {"type": "chunk", "interaction_id": interaction.id,
                         "delta": delta, "chunk": chunk})
                yield chunk
            interaction.response_message = text
            interaction.state = InteractionState.COMPLETE
            # auto-title after the first completed turn (reference:
            # the controller names sessions with the small model)
            if session.name in ("", "New Session") or \
                    session.name == (interaction.prompt_message
                                     or "")[:40]:
                try:
                    await self._autotitle(session, interaction, text)
                except Exception:
                    log.debug("session auto-title failed", exc_info=True)
        except Exception as e:
            interaction.state = InteractionState.ERROR
            interaction.error = str(e)
            if stream_to_pubsub:
                await self.pubsub.publish(
                    ps.session_queue(session.owner, session.id),
                    {"type": "error", "interaction_id": interaction.id,
                     "error": str(e)})
            raise
        finally:
            interaction.ttft_ms = first_ms
            interaction.duration_ms = int((time.monotonic() - t0) * 1000)
            interaction.updated = now_ms()
            self.store.put("interactions", interaction.id,
                           interaction.model_dump(), owner=session.owner,
                           parent=session.id)
            if stream_to_pubsub:
                await self.pubsub.publish(
                    ps.session_queue(session.owner, session.id),
                    {"type": "done", "interaction_id": interaction.id,
                     "message": text, "state": interaction.state.value})

    async def _autotitle(self, session: Session,
                         interaction: Interaction, answer: str):
        """Name the session from its first exchange using the default
        provider (small-model role); falls back silently."""
        provider = self.cfg.inference.default_provider
        client = self.providers.get_client(provider, session.owner)
        resp = await client.chat({
            "model": session.model_name or
            self.cfg.inference.default_model,
            "max_tokens": 16,
            "messages": [
                {"role": "system",
                 "content": "Reply with a 3-6 word title for this "
                            "conversation. No quotes."},
                {"role": "user",
                 "content": f"Q: {interaction.prompt_message[:400]}\n"
                            f"A: {answer[:400]}"}]})
        title = resp["choices"][0]["message"].get("content", "").strip()
        title = title.strip('"' + "'").splitlines()[0][:60]
        if title:
            session.name = title
            session.updated = now_ms()
            self.store.put("sessions", session.id, session.model_dump(),
                           owner=session.owner,
                           parent=session.parent_app)
