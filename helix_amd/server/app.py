"""HelixAPIServer — the control-plane HTTP app (parity with
api/pkg/server registerRoutes, server.go:863): OpenAI-compatible surface,
sessions, apps/agents, runner plane, admin.
"""
from __future__ import annotations

import asyncio
import os
import json
import logging
from typing import Optional

from fastapi import Depends, FastAPI, HTTPException, Request, WebSocket
from fastapi.responses import JSONResponse, StreamingResponse

from helix_amd.server import pubsub as ps
from helix_amd.server.auth import (Authenticator, AuthUser, make_admin_dep,
                                   make_auth_dep, make_runner_dep)
from helix_amd.server.config import ServerConfig, load_config
from helix_amd.server.controller import Controller
from helix_amd.server.providers import (LocalRunnerClient, MockClient,
                                        OpenAIHTTPClient, ProviderError,
                                        ProviderManager, RetryableClient,
                                        RouterClient)
from helix_amd.server.router import InferenceRouter, NoRunnerError
from helix_amd.server.types import (App, AppHelixConfig, RunnerHeartbeat,
                                    new_id)
from helix_amd.store import Store

log = logging.getLogger("helix_amd.server")

auth_dep = make_auth_dep()
admin_dep = make_admin_dep()
runner_dep = make_runner_dep()


def create_app(cfg: Optional[ServerConfig] = None,
               store: Optional[Store] = None,
               providers: Optional[ProviderManager] = None,
               runner_service=None) -> FastAPI:
    cfg = cfg or load_config()
    store = store if store is not None else Store(cfg.store.path)
    app = FastAPI(title="helix_amd control plane", docs_url=None)
    auth = Authenticator(store, cfg.web.admin_api_key,
                         cfg.runner_plane.runner_token)
    router = InferenceRouter(cfg.runner_plane.dispatch_stale_s,
                             cfg.runner_plane.offline_after_s)
    pubsub = ps.PubSub()
    from helix_amd.server.tunnel import TunnelRegistry
    tunnels = TunnelRegistry()

    if providers is None:
        providers = ProviderManager(store)
        if cfg.providers.openai_api_key:
            providers.register("openai", RetryableClient(OpenAIHTTPClient(
                "openai", cfg.providers.openai_base_url,
                cfg.providers.openai_api_key)))
        if cfg.providers.together_api_key:
            providers.register("togetherai", RetryableClient(OpenAIHTTPClient(
                "togetherai", cfg.providers.together_base_url,
                cfg.providers.together_api_key)))
        if runner_service is not None:
            providers.register("helix", LocalRunnerClient(runner_service))
        elif cfg.runner_plane.local_runner:
            from helix_amd.runner.service import RunnerService
            runner_service = RunnerService(
                device=cfg.runner_plane.local_runner_device)
            providers.register("helix", LocalRunnerClient(runner_service))
        else:
            providers.register("helix", RouterClient(router,
                                                     tunnels=tunnels))

    from helix_amd.agent.runner import AgentRunner
    from helix_amd.rag.service import RAGService
    from helix_amd.server.notifications import NotificationService
    notifications = NotificationService(store=store)
    from helix_amd.server.filestore import FileStore
    from helix_amd.server.knowledge import KnowledgeReconciler
    from helix_amd.server.models_catalog import ModelCatalog
    from helix_amd.server.triggers import TriggerManager
    from helix_amd.server.usage import QuotaExceededError, UsageService
    rag = RAGService(cfg, store, providers)
    agent_runner = AgentRunner(cfg, store, providers, pubsub, rag=rag,
                               notifications=notifications)
    knowledge = KnowledgeReconciler(cfg, store, rag)
    catalog = ModelCatalog(store)
    from helix_amd.server.metrics import Metrics
    metrics = Metrics()
    app.state.metrics = metrics
    usage = UsageService(store, catalog, metrics=metrics)
    filestore = FileStore(cfg.filestore.path)

    from helix_amd.server.code_intel import CodeIntelService
    from helix_amd.server.git_service import GitService
    from helix_amd.server.oauth import OAuthManager
    from helix_amd.server.rbac import RBACService
    from helix_amd.server.runner_profiles import (ProfileService,
                                                  RunnerProfile)
    from helix_amd.server.spec_tasks import SpecTaskService
    controller = Controller(cfg, store, providers, pubsub, rag=rag,
                            agent_runner=agent_runner, usage=usage)
    triggers = TriggerManager(store, controller)
    rbac = RBACService(store)
    from helix_amd.server.org_runtime import OrgRuntime
    org_rt = OrgRuntime(store, controller)
    oauth = OAuthManager(store)
    profiles = ProfileService(store)
    git_svc = GitService(store, cfg.filestore.path)
    from helix_amd.server.sandbox import SandboxError, SandboxManager
    sandboxes = SandboxManager(
        store, os.path.join(cfg.filestore.path, "sandboxes"),
        golden_dir=os.environ.get("HELIX_SANDBOX_GOLDEN", ""))
    app.state.sandboxes = sandboxes
    spec_tasks = SpecTaskService(store, controller, git_svc,
                                 sandboxes=sandboxes)
    agent_runner.git = git_svc
    agent_runner.spec_tasks = spec_tasks
    agent_runner.sandboxes = sandboxes
    from helix_amd.server.external_agent import ExternalAgentRegistry
    external_agents = ExternalAgentRegistry()
    app.state.external_agents = external_agents
    agent_runner.external_agents = external_agents
    code_intel = CodeIntelService(rag, git_svc)
    from helix_amd.server.evaluations import EvaluationService
    evaluations = EvaluationService(store, controller, pubsub)
    app.state.evaluations = evaluations
    from helix_amd.server.mcp_gateway import MCPGateway
    mcp = MCPGateway(store, agent_runner, code_intel, org_rt=org_rt)
    app.state.mcp = mcp

    app.state.cfg = cfg
    app.state.store = store
    app.state.auth = auth
    app.state.router = router
    app.state.pubsub = pubsub
    app.state.providers = providers
    app.state.controller = controller
    app.state.runner_service = runner_service
    app.state.tunnels = tunnels
    app.state.rag = rag
    app.state.knowledge = knowledge
    app.state.catalog = catalog
    app.state.usage = usage
    app.state.filestore = filestore
    app.state.triggers = triggers
    app.state.rbac = rbac
    app.state.org_rt = org_rt
    app.state.oauth = oauth
    app.state.profiles = profiles
    app.state.git = git_svc
    app.state.spec_tasks = spec_tasks
    app.state.code_intel = code_intel
    app.state.notifications = notifications

    async def reap_wedged() -> int:
        """Stuck-interaction auto-wake (reference
        auto_wake_stuck_interactions.go): a turn whose partial-persist
        pulse went stale while "waiting" is wedged — flip it to error
        so clients and retries unblock. Returns #flipped."""
        import time as _t
        cutoff = int((_t.time() - cfg.web.wedge_timeout_s) * 1000)
        n = 0
        for it in store.list("interactions", limit=10000):
            if it.get("state") in ("waiting", "editing") and \
                    it.get("updated", 0) < cutoff:
                it["state"] = "error"
                it["error"] = "wedged: no generation progress; auto-errored"
                store.put("interactions", it["id"], it,
                          owner=it.get("owner", ""),
                          parent=it.get("session_id", ""))
                await pubsub.publish(
                    ps.session_queue(it.get("owner", ""),
                                     it.get("session_id", "")),
                    {"type": "error", "interaction_id": it["id"],
                     "error": it["error"]})
                n += 1
        return n

    app.state.reap_wedged = reap_wedged

    async def _reaper_loop():
        # offline-runner reaper (reference 5 m threshold) + wedge sweep
        while True:
            router.reap_offline()
            try:
                await reap_wedged()
            except Exception:
                log.exception("wedge reaper failed")
            await asyncio.sleep(30)

    @app.on_event("startup")
    async def _start_reconciler():
        # Startup recovery (reference serve.go:270-280
        # ResetRunningInteractions): interactions stuck mid-stream from a
        # previous process flip to error so clients never hang on them.
        for it in store.list("interactions", limit=100000):
            if it.get("state") in ("waiting", "editing"):
                it["state"] = "error"
                it["error"] = "server restarted mid-generation"
                store.put("interactions", it["id"], it,
                          owner=it.get("owner", ""),
                          parent=it.get("session_id", ""))
        app.state._reconciler_task = asyncio.create_task(knowledge.run())
        app.state._trigger_task = asyncio.create_task(triggers.run())

        app.state._reaper_task = asyncio.create_task(_reaper_loop())

    @app.on_event("shutdown")
    async def _stop_reconciler():
        for attr in ("_reconciler_task", "_trigger_task", "_reaper_task"):
            t = getattr(app.state, attr, None)
            if t:
                t.cancel()

    # error reporting (the reference wires Sentry; here unhandled
    # exceptions land in the error_events table with a fingerprint for
    # grouping, listed via /api/v1/admin/errors)
    def record_error(where: str, exc: BaseException):
        import hashlib as _h
        import time as _t
        import traceback
        tb = traceback.format_exc()[-4000:]
        fp = _h.sha1(f"{type(exc).__name__}:{where}".encode()
                     ).hexdigest()[:16]
        doc = store.get("error_events", fp) or {
            "id": fp, "type": type(exc).__name__, "where": where,
            "first_seen": _t.time(), "count": 0}
        doc["count"] += 1
        doc["last_seen"] = _t.time()
        doc["message"] = str(exc)[:500]
        doc["traceback"] = tb
        store.put("error_events", fp, doc, buffered=True)

    app.state.record_error = record_error

    @app.middleware("http")
    async def _prom_mw(request: Request, call_next):
        import time as _t
        t0 = _t.monotonic()
        try:
            resp = await call_next(request)
            status = resp.status_code
        except Exception as exc:
            status = 500
            record_error(request.url.path, exc)
            raise
        finally:
            route = request.scope.get("route")
            rname = getattr(route, "path", request.url.path)
            metrics.http_requests.labels(request.method, rname,
                                         str(status)).inc()
            metrics.http_latency.labels(rname).observe(_t.monotonic() - t0)
        return resp

    @app.get("/metrics")
    async def prometheus_metrics():
        from fastapi.responses import Response
        metrics.runners_online.set(len(router.runners()))
        metrics.models_loaded.set(
            len(runner_service.loaded_models())
            if runner_service is not None else 0)
        body, ctype = metrics.render()
        return Response(content=body, media_type=ctype)

    @app.exception_handler(ProviderError)
    async def _pe(request, exc: ProviderError):
        return JSONResponse({"error": {"message": str(exc)}},
                            status_code=exc.status)

    @app.exception_handler(PermissionError)
    async def _perm(request, exc):
        return JSONResponse({"error": {"message": str(exc)}},
                            status_code=400)

    @app.exception_handler(NoRunnerError)
    async def _nr(request, exc: NoRunnerError):
        # reference router.go:62: NoRunnerError -> 503 with available list
        return JSONResponse({"error": {"message": str(exc),
                                       "available": exc.available}},
                            status_code=503)

    # ------------------------------------------------------------------
    # OpenAI-compatible surface
    # ------------------------------------------------------------------
    async def _chat(request: Request, user: AuthUser, body: dict):
        app_id = body.pop("app_id", "") or \
            request.headers.get("X-Helix-App-Id", "") or \
            user.app_id
        assistant_id = body.pop("assistant_id", "")
        ctx = {"owner": user.id}
        if body.get("stream"):
            async def sse():
                try:
                    async for chunk in controller.chat_completion_stream(
                            body, user.id, app_id, assistant_id, ctx):
                        yield f"data: {json.dumps(chunk)}\n\n"
                except (ProviderError, NoRunnerError) as e:
                    yield ("data: " + json.dumps(
                        {"error": {"message": str(e)}}) + "\n\n")
                yield "data: [DONE]\n\n"
            return StreamingResponse(sse(), media_type="text/event-stream")
        return await controller.chat_completion(body, user.id, app_id,
                                                assistant_id, ctx)

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request,
                               user: AuthUser = Depends(auth_dep)):
        return await _chat(request, user, await request.json())

    @app.post("/openai/deployments/{model}/chat/completions")
    async def azure_chat(model: str, request: Request,
                         user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        body.setdefault("model", model)
        return await _chat(request, user, body)

    @app.post("/v1/completions")
    async def legacy_completions(request: Request,
                                 user: AuthUser = Depends(auth_dep)):
        """Legacy text-completions: translated onto the chat surface
        (prompt -> user message; text_completion response shape)."""
        body = await request.json()
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = prompt[0] if prompt else ""
        chat_body = {k: v for k, v in body.items()
                     if k not in ("prompt", "echo", "suffix")}
        chat_body["messages"] = [{"role": "user", "content": prompt}]
        if body.get("stream"):
            raise HTTPException(400, "streaming not supported on the "
                                     "legacy completions surface; use "
                                     "/v1/chat/completions")
        resp = await _chat(request, user, chat_body)
        if isinstance(resp, dict):
            choices = [{
                "index": c.get("index", i),
                "text": c.get("message", {}).get("content", ""),
                "finish_reason": c.get("finish_reason"),
                "logprobs": c.get("logprobs"),
            } for i, c in enumerate(resp.get("choices", []))]
            return {"id": resp.get("id"), "object": "text_completion",
                    "created": resp.get("created"),
                    "model": resp.get("model"), "choices": choices,
                    "usage": resp.get("usage")}
        return resp

    @app.post("/v1/embeddings")
    async def embeddings(request: Request,
                         user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        provider, model = providers.resolve(
            body.get("model", ""), cfg.rag.embeddings_provider, user.id)
        client = providers.get_client(provider, user.id)
        return await client.embeddings({**body, "model": model})

    @app.post("/v1/images/generations")
    async def images_generations(request: Request,
                                 user: AuthUser = Depends(auth_dep)):
        """OpenAI images surface (SURVEY §2.8 last row: the reference
        proxies this to a diffusers container; here it dispatches to a
        runner's native rectified-flow DiT engine)."""
        body = await request.json()
        provider, model = providers.resolve(
            body.get("model", ""), cfg.inference.default_provider,
            user.id)
        client = providers.get_client(provider, user.id)
        return await client.images({**body, "model": model})

    @app.get("/v1/models")
    async def models(request: Request, user: AuthUser = Depends(auth_dep)):
        # anthropic-version header => Anthropic-style model list
        # (reference openai_model_handlers.go:17-45)
        if request.headers.get("anthropic-version"):
            data = await providers.aggregate_models(user.id)
            return {"data": [{"id": m["id"], "type": "model",
                              "display_name": m["id"]} for m in data],
                    "has_more": False}
        data = await providers.aggregate_models(user.id)
        return {"object": "list", "data": data}

    # ------------------------------------------------------------------
    # Sessions API (reference server.go:1024-1064)
    # ------------------------------------------------------------------
    @app.post("/api/v1/sessions/chat")
    async def session_chat(request: Request,
                           user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        session_id = body.get("session_id", "")
        if session_id:
            session = controller.get_session(session_id)
            if session is None or session.owner != user.id:
                raise HTTPException(404, "session not found")
        else:
            session = controller.create_session(
                user.id, model=body.get("model", ""),
                provider=body.get("provider", ""),
                app_id=body.get("app_id", ""),
                name=(body.get("messages") or [{}])[-1].get(
                    "content", "")[:40] or "New Session")
        msgs = body.get("messages") or []
        prompt = msgs[-1].get("content", "") if msgs else body.get("text", "")
        interaction = controller.add_interaction(session, prompt)

        async def sse():
            yield ("data: " + json.dumps(
                {"type": "session", "session_id": session.id,
                 "interaction_id": interaction.id}) + "\n\n")
            try:
                async for chunk in controller.run_session_turn(session,
                                                               interaction):
                    yield f"data: {json.dumps(chunk)}\n\n"
            except (ProviderError, NoRunnerError) as e:
                yield ("data: " + json.dumps(
                    {"error": {"message": str(e)}}) + "\n\n")
            yield "data: [DONE]\n\n"
        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.get("/api/v1/sessions")
    async def list_sessions(user: AuthUser = Depends(auth_dep)):
        return [s.model_dump() for s in controller.list_sessions(user.id)]

    @app.get("/api/v1/sessions/{session_id}")
    async def get_session(session_id: str,
                          user: AuthUser = Depends(auth_dep)):
        s = controller.get_session(session_id)
        if s is None or (s.owner != user.id and not user.admin):
            raise HTTPException(404, "session not found")
        doc = s.model_dump()
        doc["interactions"] = store.list("interactions", parent=session_id,
                                         desc=False)
        return doc

    @app.post("/api/v1/sessions/{session_id}/resume")
    async def resume_session(session_id: str,
                             user: AuthUser = Depends(auth_dep)):
        """Re-run the last interaction if it errored mid-stream
        (reference session_handlers.go:2110)."""
        s = controller.get_session(session_id)
        if s is None or s.owner != user.id:
            raise HTTPException(404, "session not found")
        its = store.list("interactions", parent=session_id, desc=True,
                         limit=1)
        if not its:
            raise HTTPException(400, "nothing to resume")
        from helix_amd.server.types import Interaction
        last = Interaction.model_validate(its[0])
        if last.state == "complete":
            return {"ok": True, "note": "already complete"}

        async def sse():
            yield ("data: " + json.dumps(
                {"type": "session", "session_id": s.id,
                 "interaction_id": last.id}) + "\n\n")
            try:
                async for chunk in controller.run_session_turn(s, last):
                    yield f"data: {json.dumps(chunk)}\n\n"
            except (ProviderError, NoRunnerError) as e:
                yield ("data: " + json.dumps(
                    {"error": {"message": str(e)}}) + "\n\n")
            yield "data: [DONE]\n\n"
        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.put("/api/v1/sessions/{session_id}/agent")
    async def switch_agent(session_id: str, request: Request,
                           user: AuthUser = Depends(auth_dep)):
        """Switch the app/agent mid-session, history carried across
        (reference server.go:1042 switch-agent)."""
        s = controller.get_session(session_id)
        if s is None or s.owner != user.id:
            raise HTTPException(404, "session not found")
        body = await request.json()
        new_app = body.get("app_id", "")
        if new_app and store.get("apps", new_app) is None:
            raise HTTPException(404, "app not found")
        s.parent_app = new_app
        if body.get("model"):
            s.model_name = body["model"]
        store.put("sessions", s.id, s.model_dump(), owner=s.owner,
                  parent=new_app)
        return s.model_dump()

    @app.get("/api/v1/sessions/{session_id}/step-info")
    async def step_info(session_id: str,
                        user: AuthUser = Depends(auth_dep)):
        """Agent step trace for a session (reference server.go:1036)."""
        s = controller.get_session(session_id)
        if s is None or (s.owner != user.id and not user.admin):
            raise HTTPException(404, "session not found")
        return store.list("step_info", parent=session_id, desc=False)

    @app.post("/api/v1/sessions/{session_id}/fork")
    async def fork_session(session_id: str,
                           user: AuthUser = Depends(auth_dep)):
        """Duplicate a session with its history (reference fork route)."""
        s = controller.get_session(session_id)
        if s is None or s.owner != user.id:
            raise HTTPException(404, "session not found")
        ns = controller.create_session(user.id, model=s.model_name,
                                       provider=s.provider,
                                       app_id=s.parent_app,
                                       name=f"{s.name} (fork)")
        from helix_amd.server.types import Interaction
        for doc in store.list("interactions", parent=session_id,
                              desc=False):
            it = Interaction.model_validate(doc)
            it.id = f"int_fork_{it.id[-12:]}_{ns.id[-6:]}"
            it.session_id = ns.id
            store.put("interactions", it.id, it.model_dump(),
                      owner=user.id, parent=ns.id)
        return ns.model_dump()

    @app.delete("/api/v1/sessions/{session_id}")
    async def delete_session(session_id: str,
                             user: AuthUser = Depends(auth_dep)):
        s = controller.get_session(session_id)
        if s is None or (s.owner != user.id and not user.admin):
            raise HTTPException(404, "session not found")
        controller.delete_session(session_id)
        return {"ok": True}

    # -- interactions CRUD (reference sessions API, server.go:1024-1064:
    #    interaction listing/edit/delete alongside step-info) ----------
    def _owned_interaction(iid: str, user: AuthUser) -> dict:
        doc = store.get("interactions", iid)
        if doc is None:
            raise HTTPException(404, "interaction not found")
        s = controller.get_session(doc.get("session_id", ""))
        if s is None or (s.owner != user.id and not user.admin):
            raise HTTPException(404, "interaction not found")
        return doc

    @app.get("/api/v1/sessions/{session_id}/interactions")
    async def list_interactions(session_id: str, limit: int = 100,
                                offset: int = 0,
                                user: AuthUser = Depends(auth_dep)):
        s = controller.get_session(session_id)
        if s is None or (s.owner != user.id and not user.admin):
            raise HTTPException(404, "session not found")
        rows = store.list("interactions", parent=session_id,
                          desc=False, limit=limit, offset=offset)
        return rows

    @app.put("/api/v1/interactions/{iid}")
    async def update_interaction(iid: str, request: Request,
                                 user: AuthUser = Depends(auth_dep)):
        """Edit a turn (prompt correction before resume / response
        redaction)."""
        doc = _owned_interaction(iid, user)
        body = await request.json()
        for k in ("prompt_message", "response_message", "state"):
            if k in body:
                doc[k] = body[k]
        s2 = controller.get_session(doc.get("session_id", ""))
        store.put("interactions", iid, doc,
                  owner=s2.owner if s2 else "",
                  parent=doc.get("session_id", ""))
        return doc

    @app.delete("/api/v1/interactions/{iid}")
    async def delete_interaction(iid: str,
                                 user: AuthUser = Depends(auth_dep)):
        _owned_interaction(iid, user)
        return {"ok": store.delete("interactions", iid)}

    @app.post("/api/v1/tokenize")
    async def tokenize(request: Request,
                       user: AuthUser = Depends(auth_dep)):
        """Token counting for a prompt under a model's tokenizer +
        chat template (client-side budget planning)."""
        from helix_amd.utils.tokenizer import get_tokenizer
        body = await request.json()
        tok = get_tokenizer(body.get("model", ""))
        if body.get("messages"):
            ids = tok.apply_chat_template(
                body["messages"], template=body.get("template", ""))
        else:
            ids = tok.encode(body.get("text", ""))
        return {"count": len(ids),
                "tokens": ids[:int(body.get("return_tokens", 0) or 0)]}

    @app.post("/api/v1/auth/token")
    async def issue_token(request: Request,
                          user: AuthUser = Depends(auth_dep)):
        """Exchange an API key for a short-lived HS256 JWT (the
        reference's Keycloak-token flow, self-issued here)."""
        try:
            body = await request.json()
        except Exception:
            body = {}
        ttl = min(int(body.get("ttl_s", 3600)), 86400)
        return {"access_token": auth.issue_jwt(user, ttl),
                "token_type": "Bearer", "expires_in": ttl}

    # -- OIDC login mode (reference api/pkg/auth/oidc.go +
    #    session_manager.go: discovery, state+nonce, code exchange,
    #    ID-token verification, user mapping, session token) ----------
    app.state.oidc_client = None
    if cfg.oidc.enabled and cfg.oidc.issuer:
        from helix_amd.server.oidc import OIDCClient
        app.state.oidc_client = OIDCClient(
            cfg.oidc.issuer, cfg.oidc.client_id, cfg.oidc.client_secret,
            cfg.oidc.redirect_url,
            allowed_domains=cfg.oidc.allowed_domains)

    @app.get("/api/v1/auth/oidc/login")
    async def oidc_login():
        client = app.state.oidc_client
        if client is None:
            raise HTTPException(404, "OIDC is not configured")
        import time as _t
        state, nonce = new_id("oidcst"), new_id("nonce")
        store.put("oidc_states", state,
                  {"id": state, "nonce": nonce, "ts": _t.time()})
        return {"url": await client.get_auth_url(state, nonce),
                "state": state}

    @app.get("/api/v1/auth/oidc/callback")
    async def oidc_callback(code: str, state: str):
        from helix_amd.server.oidc import OIDCError
        client = app.state.oidc_client
        if client is None:
            raise HTTPException(404, "OIDC is not configured")
        import time as _t
        st = store.get("oidc_states", state)
        if st is None or _t.time() - st.get("ts", 0) > 600:
            raise HTTPException(400, "invalid or expired state")
        store.delete("oidc_states", state)
        try:
            tok = await client.exchange(code)
            claims = await client.verify_id_token(
                tok.get("id_token", ""), st.get("nonce", ""))
        except OIDCError as e:
            raise HTTPException(401, str(e))
        sub = claims.get("sub", "")
        email = claims.get("email", "")
        username = claims.get("preferred_username") or email or sub
        import hashlib as _h
        uid = f"user_{_h.sha1(('oidc:' + sub).encode()).hexdigest()[:16]}"
        doc = store.get("users", uid)
        if doc is None:
            doc = {"id": uid, "username": username, "admin": False,
                   "email": email, "oidc_sub": sub,
                   "name": claims.get("name", "")}
            store.put("users", uid, doc, owner=uid)
        elif email and doc.get("email") != email:
            doc["email"] = email
            store.put("users", uid, doc, owner=uid)
        jwt = auth.issue_jwt(AuthUser(id=uid, username=username,
                                      admin=doc.get("admin", False)),
                             ttl_s=86400)
        return {"access_token": jwt, "token_type": "Bearer",
                "expires_in": 86400,
                "user": {"id": uid, "username": username,
                         "email": email}}

    @app.websocket("/api/v1/ws/user")
    async def ws_user(ws: WebSocket):
        token = ws.query_params.get("access_token", "")
        user = auth.resolve(token)
        if user is None:
            await ws.close(code=4401)
            return
        await ws.accept()
        sub = await pubsub.subscribe(ps.session_queue(user.id, "*"))
        try:
            while True:
                topic, msg = await sub.get()
                await ws.send_json({"topic": topic, "payload": msg})
        except Exception:
            pass
        finally:
            await sub.close()

    # -- cross-process message bus (reference embedded NATS WS listener
    #    + JetStream, nats.go:119-163,608-699): external processes SUB/
    #    PUB ephemeral topics and FETCH/ACK durable streams over one WS --
    from helix_amd.server.pubsub import StreamBus
    bus = StreamBus(store, pubsub)
    app.state.bus = bus

    @app.websocket("/api/v1/external-agents/ws")
    async def external_agent_ws(ws: WebSocket):
        """External agent uplink (reference external-agent executor):
        the agent connects with its agent_id (session or app scope),
        receives chat commands, streams delta/done/error frames back."""
        token = ws.query_params.get("access_token", "")
        agent_id = ws.query_params.get("agent_id", "")
        user = auth.resolve(token)
        is_runner = token and token == cfg.runner_plane.runner_token
        if (user is None and not is_runner) or not agent_id:
            await ws.close(code=4401)
            return
        if not is_runner and not (user and user.admin):
            # the agent key is a session/app id: only its owner may
            # serve turns for it (otherwise any authenticated user
            # could hijack another tenant's zed_external sessions)
            sess = controller.get_session(agent_id)
            app_doc = store.get("apps", agent_id)
            owns = (sess is not None and sess.owner == user.id) or                 (app_doc is not None and
                 app_doc.get("owner") == user.id)
            if not owns:
                await ws.close(code=4403)
                return
        await ws.accept()
        outbox = external_agents.attach(agent_id)

        async def pump_out():
            try:
                while True:
                    frame = await outbox.get()
                    await ws.send_json(frame)
            except Exception:
                pass

        out_task = asyncio.ensure_future(pump_out())
        try:
            while True:
                frame = await ws.receive_json()
                if frame.get("type") == "ping":
                    await ws.send_json({"type": "pong"})
                    continue
                external_agents.deliver(frame)
        except Exception:
            pass
        finally:
            out_task.cancel()
            external_agents.detach(agent_id)

    @app.websocket("/api/v1/ws/bus")
    async def ws_bus(ws: WebSocket):
        token = ws.query_params.get("access_token", "")
        user = auth.resolve(token)
        is_runner = token and token == cfg.runner_plane.runner_token
        if user is None and not is_runner:
            await ws.close(code=4401)
            return
        privileged = is_runner or (user is not None and user.admin)

        def _topic_allowed(topic: str) -> bool:
            # non-privileged connections are scoped to their own
            # namespaces: session events and user topics they own
            # (otherwise any key could read every tenant's streams)
            if privileged:
                return True
            uid = user.id
            return topic.startswith((f"session.{uid}.", f"user.{uid}."))

        def _stream_allowed(stream: str) -> bool:
            if privileged:
                return True
            return stream.startswith(f"user-{user.id}-")

        await ws.accept()
        subs = {}
        forwarders = []

        async def forward(pattern, sub):
            try:
                while True:
                    topic, msg = await sub.get()
                    await ws.send_json({"op": "msg", "pattern": pattern,
                                        "topic": topic, "payload": msg})
            except Exception:
                pass

        try:
            while True:
                frame = await ws.receive_json()
                op = frame.get("op")
                if op == "sub":
                    pattern = frame.get("pattern", "")
                    if pattern and not _topic_allowed(pattern):
                        await ws.send_json({"op": "error",
                                            "message": "pattern not "
                                                       "allowed"})
                        continue
                    if pattern and pattern not in subs:
                        sub = await pubsub.subscribe(pattern)
                        subs[pattern] = sub
                        forwarders.append(
                            asyncio.ensure_future(forward(pattern, sub)))
                elif op == "unsub":
                    sub = subs.pop(frame.get("pattern", ""), None)
                    if sub:
                        await sub.close()
                elif op == "pub":
                    topic = frame.get("topic", "")
                    if not _topic_allowed(topic):
                        await ws.send_json({"op": "error",
                                            "message": "topic not "
                                                       "allowed"})
                        continue
                    await pubsub.publish(topic, frame.get("payload"))
                elif op == "stream_pub":
                    stream = frame.get("stream", "default")
                    if not _stream_allowed(stream):
                        await ws.send_json({"op": "error",
                                            "message": "stream not "
                                                       "allowed"})
                        continue
                    seq = await bus.publish_notify(
                        stream, frame.get("subject", ""),
                        frame.get("payload"))
                    await ws.send_json({"op": "pub_ack", "seq": seq})
                elif op == "fetch":
                    if not _stream_allowed(frame.get("stream",
                                                     "default")):
                        await ws.send_json({"op": "batch",
                                            "messages": [],
                                            "error": "stream not "
                                                     "allowed"})
                        continue
                    msgs = bus.fetch(frame.get("stream", "default"),
                                     frame.get("durable", "d"),
                                     int(frame.get("batch", 10)),
                                     frame.get("subject_filter", "*"))
                    await ws.send_json({"op": "batch",
                                        "messages": msgs})
                elif op == "ack":
                    if not _stream_allowed(frame.get("stream",
                                                     "default")):
                        continue
                    bus.ack(frame.get("stream", "default"),
                            frame.get("durable", "d"),
                            int(frame.get("seq", 0)))
                elif op == "ping":
                    await ws.send_json({"op": "pong"})
        except Exception:
            pass
        finally:
            for f in forwarders:
                f.cancel()
            for sub in subs.values():
                try:
                    await sub.close()
                except Exception:
                    pass

    # ------------------------------------------------------------------
    # Apps / agents (registered under BOTH aliases, agent_routes.go:12-50)
    # ------------------------------------------------------------------
    def _get_app(app_id: str, user: AuthUser) -> App:
        doc = store.get("apps", app_id)
        if doc is None:
            raise HTTPException(404, "app not found")
        a = App.model_validate(doc)
        if a.owner != user.id and not a.global_ and not user.admin:
            raise HTTPException(403, "forbidden")
        return a

    def _reconcile_app_triggers(a: App, owner: str):
        """helix.yaml `triggers:` -> TriggerManager rows (reference
        apply registers cron/slack/webhook triggers from the app
        config; stale rows for the app are replaced)."""
        for t in store.list("triggers", limit=10000):
            if t.get("app_id") == a.id and t.get("from_app_config"):
                store.delete("triggers", t["id"])
        made = []
        for spec in (a.config.triggers or []):
            kind = spec.get("kind") or spec.get("type") or ""
            config = dict(spec.get("config") or
                          {k: v for k, v in spec.items()
                           if k not in ("kind", "type")})
            if not kind:
                continue
            try:
                doc = triggers.create(owner, a.id, kind, config)
                doc["from_app_config"] = True
                store.put("triggers", doc["id"], doc, owner=owner,
                          parent=a.id)
                made.append(doc["id"])
            except ValueError as e:
                raise HTTPException(400, f"bad trigger {kind}: {e}")
        return made

    for prefix in ("/api/v1/apps", "/api/v1/agents"):
        def _bind(prefix=prefix):
            @app.post(prefix, name=f"create_app_{prefix}")
            async def create_app_h(request: Request,
                                   user: AuthUser = Depends(auth_dep)):
                body = await request.json()
                config = AppHelixConfig.model_validate(
                    body.get("config", body))
                a = App(owner=user.id, config=config,
                        global_=bool(body.get("global", False)))
                store.put("apps", a.id, a.model_dump(by_alias=True),
                          owner=user.id)
                _reconcile_app_triggers(a, user.id)
                return a.model_dump(by_alias=True)

            @app.get(prefix, name=f"list_apps_{prefix}")
            async def list_apps_h(user: AuthUser = Depends(auth_dep)):
                docs = store.list("apps", owner=user.id)
                return docs

            @app.get(prefix + "/{app_id}", name=f"get_app_{prefix}")
            async def get_app_h(app_id: str,
                                user: AuthUser = Depends(auth_dep)):
                return _get_app(app_id, user).model_dump(by_alias=True)

            @app.put(prefix + "/{app_id}", name=f"update_app_{prefix}")
            async def update_app_h(app_id: str, request: Request,
                                   user: AuthUser = Depends(auth_dep)):
                a = _get_app(app_id, user)
                body = await request.json()
                a.config = AppHelixConfig.model_validate(
                    body.get("config", body))
                from helix_amd.server.types import now_ms
                a.updated = now_ms()
                store.put("apps", a.id, a.model_dump(by_alias=True),
                          owner=a.owner)
                _reconcile_app_triggers(a, a.owner)
                return a.model_dump(by_alias=True)

            @app.delete(prefix + "/{app_id}", name=f"delete_app_{prefix}")
            async def delete_app_h(app_id: str,
                                   user: AuthUser = Depends(auth_dep)):
                _get_app(app_id, user)
                store.delete("apps", app_id)
                for t in store.list("triggers", limit=10000):
                    if t.get("app_id") == app_id and \
                            t.get("from_app_config"):
                        store.delete("triggers", t["id"])
                return {"ok": True}
        _bind()

    @app.post("/api/v1/apps/{app_id}/keys")
    async def create_app_key(app_id: str, request: Request,
                             user: AuthUser = Depends(auth_dep)):
        """App-scoped API key (reference GetAppAPIKeys, client/app.go:
        44): requests made with it default to this app."""
        a = _get_app(app_id, user)
        if a.owner != user.id and not user.admin:
            raise HTTPException(403, "only the owner can mint app keys")
        try:
            body = await request.json()
        except Exception:
            body = {}
        key = auth.create_api_key(a.owner,
                                  body.get("name", f"app-{app_id[:8]}"),
                                  app_id=app_id)
        return {"key": key, "app_id": app_id}

    @app.get("/api/v1/apps/{app_id}/keys")
    async def list_app_keys(app_id: str,
                            user: AuthUser = Depends(auth_dep)):
        a = _get_app(app_id, user)
        if a.owner != user.id and not user.admin:
            raise HTTPException(403, "forbidden")
        return [{"id": k["id"][:12] + "...", "name": k.get("name", "")}
                for k in store.list("api_keys", owner=a.owner)
                if k.get("app_id") == app_id]

    # ------------------------------------------------------------------
    # Local-model admin (reference local-models/load|unload,
    # server.go:1018-1020): proxied to the runner serving the model.
    # ------------------------------------------------------------------
    @app.post("/api/v1/local-models")
    async def register_local_model(request: Request,
                                   user: AuthUser = Depends(admin_dep)):
        """Register a model spec (preset + serving options: tp,
        quantization, context length) on the local runner."""
        if runner_service is None:
            raise HTTPException(400, "no local runner on this server")
        from helix_amd.runner.service import ModelSpec
        b = await request.json()
        try:
            spec = ModelSpec(
                name=b["name"], kind=b.get("kind", "llm"),
                preset=b.get("preset", b["name"]),
                max_model_len=int(b.get("max_model_len", 8192)),
                max_num_seqs=int(b.get("max_num_seqs", 64)),
                kv_cache_blocks=b.get("kv_cache_blocks"),
                tp=int(b.get("tp", 1)),
                quantization=b.get("quantization"),
                kv_cache_dtype=b.get("kv_cache_dtype", "bf16"))
            runner_service.register_spec(spec)
        except KeyError as e:
            raise HTTPException(400, f"missing field {e}")
        except ValueError as e:
            raise HTTPException(409, str(e))
        return {"ok": True, "model": spec.name}

    @app.post("/api/v1/local-models/{model}/load")
    async def load_local_model(model: str,
                               user: AuthUser = Depends(admin_dep)):
        if runner_service is not None:
            import asyncio as aio
            await aio.get_event_loop().run_in_executor(
                None, runner_service.ensure_loaded, model)
            return {"ok": True, "where": "local"}
        import httpx
        errors = {}
        for r in router.runners():
            try:
                async with httpx.AsyncClient(timeout=600) as http:
                    resp = await http.post(
                        f"{r.address}/api/v1/models/{model}/load")
                    if resp.status_code == 200:
                        return {"ok": True, "where": r.runner_id}
                    errors[r.runner_id] = resp.text[:200]
            except Exception as e:
                errors[r.runner_id] = str(e)
        raise HTTPException(503, f"no runner could load {model}: {errors}")

    @app.post("/api/v1/local-models/{model}/unload")
    async def unload_local_model(model: str,
                                 user: AuthUser = Depends(admin_dep)):
        if runner_service is not None:
            runner_service.unload(model)
            return {"ok": True}
        import httpx
        for r in router.runners():
            try:
                async with httpx.AsyncClient(timeout=60) as http:
                    await http.post(
                        f"{r.address}/api/v1/models/{model}/unload")
            except Exception:
                pass
        return {"ok": True}

    # ------------------------------------------------------------------
    # Knowledge API (reference server.go knowledge routes + reconciler)
    # ------------------------------------------------------------------
    @app.post("/api/v1/knowledge")
    async def create_knowledge(request: Request,
                               user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        try:
            doc = knowledge.create(user.id, body.get("name", ""),
                                   body.get("source", {}),
                                   body.get("app_id", ""),
                                   body.get("refresh_schedule", ""))
        except ValueError as e:
            raise HTTPException(400, str(e))
        return doc

    @app.get("/api/v1/knowledge")
    async def list_knowledge(user: AuthUser = Depends(auth_dep)):
        return knowledge.list(user.id)

    @app.get("/api/v1/knowledge/{kid}")
    async def get_knowledge(kid: str, user: AuthUser = Depends(auth_dep)):
        doc = knowledge.get(kid)
        if doc is None or (doc["owner"] != user.id and not user.admin):
            raise HTTPException(404, "knowledge not found")
        return doc

    @app.delete("/api/v1/knowledge/{kid}")
    async def delete_knowledge(kid: str,
                               user: AuthUser = Depends(auth_dep)):
        doc = knowledge.get(kid)
        if doc is None or (doc["owner"] != user.id and not user.admin):
            raise HTTPException(404, "knowledge not found")
        knowledge.delete(kid)
        return {"ok": True}

    @app.post("/api/v1/knowledge/{kid}/refresh")
    async def refresh_knowledge(kid: str,
                                user: AuthUser = Depends(auth_dep)):
        doc = knowledge.get(kid)
        if doc is None or (doc["owner"] != user.id and not user.admin):
            raise HTTPException(404, "knowledge not found")
        knowledge.request_refresh(kid)
        return {"ok": True}

    @app.post("/api/v1/knowledge/{kid}/query")
    async def query_knowledge(kid: str, request: Request,
                              user: AuthUser = Depends(auth_dep)):
        doc = knowledge.get(kid)
        if doc is None or (doc["owner"] != user.id and not user.admin):
            raise HTTPException(404, "knowledge not found")
        body = await request.json()
        return await rag.query(kid, body.get("query", ""),
                               body.get("k"))

    # ------------------------------------------------------------------
    # Runner plane (heartbeats -> router, reference sandbox_handlers.go:84)
    # ------------------------------------------------------------------
    @app.post("/api/v1/runner/heartbeat")
    async def runner_heartbeat(request: Request,
                               _=Depends(runner_dep)):
        hb = RunnerHeartbeat.model_validate(await request.json())
        router.on_heartbeat(hb)
        return {"ok": True}

    @app.get("/api/v1/runner/tunnel/{runner_id}")
    async def runner_tunnel(runner_id: str, _=Depends(runner_dep)):
        """Reverse-dial downlink: runner holds this SSE stream open and
        receives dispatched requests (RevDial parity)."""
        q = tunnels.connect(runner_id)

        async def sse():
            try:
                while True:
                    try:
                        msg = await asyncio.wait_for(q.get(), 15.0)
                    except asyncio.TimeoutError:
                        yield 'data: {"type": "ping"}\n\n'
                        continue
                    yield f"data: {json.dumps(msg)}\n\n"
            finally:
                tunnels.disconnect(runner_id, q)
        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.post("/api/v1/runner/tunnel/{runner_id}/reply")
    async def runner_tunnel_reply(runner_id: str, request: Request,
                                  _=Depends(runner_dep)):
        body = await request.json()
        ok = await tunnels.push_reply(body.get("id", ""),
                                      body.get("event") or {})
        return {"ok": ok}

    @app.get("/api/v1/events/{session_id}")
    async def session_events(session_id: str,
                             user: AuthUser = Depends(auth_dep)):
        """SSE session event stream (WS-free deployments)."""
        s_obj = controller.get_session(session_id)
        if s_obj is None or (s_obj.owner != user.id and not user.admin):
            raise HTTPException(404, "session not found")
        sub = await pubsub.subscribe(ps.session_queue(s_obj.owner,
                                                      session_id))

        async def sse():
            try:
                while True:
                    try:
                        topic, msg = await sub.get(timeout=30.0)
                    except asyncio.TimeoutError:
                        yield 'data: {"type": "ping"}\n\n'
                        continue
                    yield f"data: {json.dumps(msg)}\n\n"
            finally:
                await sub.close()
        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.get("/api/v1/admin/runners/{runner_id}/logs")
    async def runner_logs(runner_id: str, n: int = 200,
                          user: AuthUser = Depends(admin_dep)):
        import httpx
        state = next((r for r in router.runners()
                      if r.runner_id == runner_id), None)
        if state is None:
            raise HTTPException(404, "runner not found")
        async with httpx.AsyncClient(timeout=20) as http:
            r = await http.get(f"{state.address}/api/v1/logs",
                               params={"n": n})
            return r.json()

    @app.get("/api/v1/admin/runners")
    async def list_runners(user: AuthUser = Depends(admin_dep)):
        router.reap_offline()
        return [r.model_dump() for r in router.runners()]

    # ------------------------------------------------------------------
    # Runner profiles (reference server.go:1295-1320 + §3.5 flow)
    # ------------------------------------------------------------------
    @app.post("/api/v1/runner-profiles")
    async def create_profile(request: Request,
                             user: AuthUser = Depends(admin_dep)):
        prof = RunnerProfile.model_validate(await request.json())
        return profiles.create(prof).model_dump()

    @app.get("/api/v1/runner-profiles")
    async def list_profiles(user: AuthUser = Depends(admin_dep)):
        return [p.model_dump() for p in profiles.list()]

    @app.delete("/api/v1/runner-profiles/{pid}")
    async def delete_profile(pid: str,
                             user: AuthUser = Depends(admin_dep)):
        return {"ok": profiles.delete(pid)}

    @app.post("/api/v1/runners/{runner_id}/assign-profile")
    async def assign_profile(runner_id: str, request: Request,
                             user: AuthUser = Depends(admin_dep)):
        body = await request.json()
        state = next((r for r in router.runners()
                      if r.runner_id == runner_id), None)
        gpus = state.gpus if state else []
        ok, why = profiles.assign(runner_id, body.get("profile_id", ""),
                                  gpus)
        if not ok:
            raise HTTPException(409, why)
        return {"ok": True}

    @app.get("/api/v1/runners/{runner_id}/compatible-profiles")
    async def compatible_profiles(runner_id: str,
                                  user: AuthUser = Depends(admin_dep)):
        """Profiles whose GPU requirements this runner satisfies
        (reference gpucloud scenario 2 `compatibility_filter`)."""
        from helix_amd.server.runner_profiles import filter_compatible
        state = next((r for r in router.runners()
                      if r.runner_id == runner_id), None)
        if state is None:
            raise HTTPException(404, "runner not found")
        return [p.model_dump()
                for p in filter_compatible(profiles.list(), state.gpus)]

    @app.get("/api/v1/runner/{runner_id}/assignment")
    async def get_assignment(runner_id: str, _=Depends(runner_dep)):
        # runner polls its assigned profile (reference compose-manager
        # 15 s poll, §3.5)
        return profiles.assignment(runner_id) or {}

    @app.delete("/api/v1/runners/{runner_id}/assignment")
    async def clear_assignment(runner_id: str,
                               user: AuthUser = Depends(admin_dep)):
        return {"ok": profiles.clear_assignment(runner_id)}

    # ------------------------------------------------------------------
    # Orgs / teams / access grants (reference org routes + authz)
    # ------------------------------------------------------------------
    @app.post("/api/v1/organizations")
    async def create_org(request: Request,
                         user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        return rbac.create_org(user.id, body.get("name", ""))

    @app.get("/api/v1/organizations")
    async def list_orgs(user: AuthUser = Depends(auth_dep)):
        return rbac.list_orgs_for(user.id)

    @app.post("/api/v1/organizations/{oid}/members")
    async def add_org_member(oid: str, request: Request,
                             user: AuthUser = Depends(auth_dep)):
        if rbac.member_role(oid, user.id) not in ("owner", "admin") \
                and not user.admin:
            raise HTTPException(403, "org admin only")
        body = await request.json()
        return rbac.add_member(oid, body["user_id"],
                               body.get("role", "member"))

    @app.post("/api/v1/organizations/{oid}/teams")
    async def create_team(oid: str, request: Request,
                          user: AuthUser = Depends(auth_dep)):
        if rbac.member_role(oid, user.id) is None and not user.admin:
            raise HTTPException(403, "not a member")
        body = await request.json()
        return rbac.create_team(oid, body.get("name", ""))

    @app.get("/api/v1/organizations/{oid}/teams")
    async def list_org_teams(oid: str,
                             user: AuthUser = Depends(auth_dep)):
        if rbac.member_role(oid, user.id) is None and not user.admin:
            raise HTTPException(403, "not a member")
        return rbac.list_teams(oid)

    @app.post("/api/v1/teams/{tid}/members")
    async def add_team_member(tid: str, request: Request,
                              user: AuthUser = Depends(auth_dep)):
        team = store.get("teams", tid)
        if not team:
            raise HTTPException(404, "team not found")
        if rbac.member_role(team["org_id"], user.id) not in ("owner", "admin") \
                and not user.admin:
            raise HTTPException(403, "org admin only")
        body = await request.json()
        return rbac.add_team_member(tid, body["user_id"])

    # -- org runtime: positions / bots / streams (reference api/pkg/org)
    def _org_member(oid: str, user: AuthUser):
        if rbac.member_role(oid, user.id) is None and not user.admin:
            raise HTTPException(403, "not a member")

    @app.post("/api/v1/organizations/{oid}/positions")
    async def create_position(oid: str, request: Request,
                              user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        b = await request.json()
        return org_rt.create_position(
            oid, b.get("name", ""), b.get("role", "worker"),
            b.get("app_id", ""), b.get("system_prompt", ""),
            b.get("model", ""))

    @app.get("/api/v1/organizations/{oid}/positions")
    async def list_positions(oid: str,
                             user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        return org_rt.list_positions(oid)

    @app.post("/api/v1/organizations/{oid}/bots")
    async def create_bot(oid: str, request: Request,
                         user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        b = await request.json()
        try:
            return org_rt.create_bot(oid, b.get("name", ""),
                                     b.get("position_id", ""), user.id)
        except ValueError as e:
            raise HTTPException(400, str(e))

    @app.get("/api/v1/organizations/{oid}/bots")
    async def list_bots(oid: str, user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        return org_rt.list_bots(oid)

    @app.post("/api/v1/bots/{bid}/subscribe")
    async def bot_subscribe(bid: str, request: Request,
                            user: AuthUser = Depends(auth_dep)):
        b = await request.json()
        bot = org_rt.get_bot(bid)
        if bot is None:
            raise HTTPException(404, "bot not found")
        _org_member(bot["org_id"], user)
        try:
            return org_rt.subscribe(bid, b.get("stream_id", ""))
        except ValueError as e:
            raise HTTPException(400, str(e))

    @app.put("/api/v1/bots/{bid}/parents")
    async def set_bot_parents(bid: str, request: Request,
                              user: AuthUser = Depends(auth_dep)):
        """Reporting lines (reference org_reporting_lines: bot -> its
        managers, cycle-guarded DAG)."""
        b = await request.json()
        bot = org_rt.get_bot(bid)
        if bot is None:
            raise HTTPException(404, "bot not found")
        _org_member(bot["org_id"], user)
        try:
            return org_rt.set_reporting_lines(
                bid, list(b.get("parent_ids", [])))
        except ValueError as e:
            raise HTTPException(400, str(e))

    @app.delete("/api/v1/bots/{bid}")
    async def delete_bot(bid: str, user: AuthUser = Depends(auth_dep)):
        bot = org_rt.get_bot(bid)
        if bot is None:
            raise HTTPException(404, "bot not found")
        _org_member(bot["org_id"], user)
        return {"ok": org_rt.delete_bot_cascade(bid)}

    @app.get("/api/v1/organizations/{oid}/chart")
    async def org_chart(oid: str, user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        return org_rt.chart(oid)

    @app.post("/api/v1/bots/{bid}/escalate")
    async def bot_escalate(bid: str, request: Request,
                           user: AuthUser = Depends(auth_dep)):
        b = await request.json()
        bot = org_rt.get_bot(bid)
        if bot is None:
            raise HTTPException(404, "bot not found")
        _org_member(bot["org_id"], user)
        try:
            return await org_rt.escalate(bid, b.get("text", ""))
        except ValueError as e:
            raise HTTPException(400, str(e))

    @app.get("/api/v1/organizations/{oid}/audit")
    async def org_audit(oid: str, user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        store.flush("org_audit")
        return org_rt.audit_log(oid)

    @app.post("/api/v1/organizations/{oid}/streams")
    async def create_stream(oid: str, request: Request,
                            user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        b = await request.json()
        return org_rt.create_stream(oid, b.get("name", ""),
                                    b.get("topic", ""))

    @app.get("/api/v1/organizations/{oid}/streams")
    async def list_streams(oid: str, user: AuthUser = Depends(auth_dep)):
        _org_member(oid, user)
        return org_rt.list_streams(oid)

    @app.post("/api/v1/streams/{sid}/messages")
    async def post_stream_message(sid: str, request: Request,
                                  user: AuthUser = Depends(auth_dep)):
        b = await request.json()
        try:
            return await org_rt.post_message(sid, user.username or user.id,
                                             b.get("text", ""))
        except ValueError as e:
            raise HTTPException(404, str(e))

    @app.get("/api/v1/streams/{sid}/messages")
    async def get_stream_messages(sid: str,
                                  user: AuthUser = Depends(auth_dep)):
        return org_rt.stream_messages(sid)

    @app.post("/api/v1/access-grants")
    async def create_grant(request: Request,
                           user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        return rbac.grant(body.get("resource_type", ""),
                          body.get("resource_id", ""),
                          body.get("role", "viewer"),
                          user_id=body.get("user_id", ""),
                          team_id=body.get("team_id", ""),
                          org_id=body.get("org_id", ""))

    @app.get("/api/v1/access-grants/{resource_id}")
    async def list_grants(resource_id: str,
                          user: AuthUser = Depends(auth_dep)):
        return rbac.grants_for(resource_id)

    # ------------------------------------------------------------------
    # OAuth (reference api/pkg/oauth routes)
    # ------------------------------------------------------------------
    @app.post("/api/v1/oauth/providers")
    async def configure_oauth(request: Request,
                              user: AuthUser = Depends(admin_dep)):
        body = await request.json()
        return oauth.configure_provider(
            body["name"], body.get("client_id", ""),
            body.get("client_secret", ""), body.get("auth_url", ""),
            body.get("token_url", ""), body.get("scopes"))

    @app.get("/api/v1/oauth/{provider}/authorize-url")
    async def oauth_authorize(provider: str, redirect_uri: str,
                              user: AuthUser = Depends(auth_dep)):
        try:
            return {"url": oauth.authorize_url(provider, redirect_uri,
                                               state=user.id)}
        except KeyError as e:
            raise HTTPException(404, str(e))

    @app.post("/api/v1/oauth/{provider}/token")
    async def oauth_save_token(provider: str, request: Request,
                               user: AuthUser = Depends(auth_dep)):
        tok = await request.json()
        doc = oauth.save_token(user.id, provider, tok)
        return {"provider": provider,
                "expires_at": doc["expires_at"]}

    # ------------------------------------------------------------------
    # Provider endpoints (user/org-defined OpenAI-compatible providers)
    # ------------------------------------------------------------------
    @app.post("/api/v1/provider-endpoints")
    async def create_provider_endpoint(request: Request,
                                       user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        name = body.get("name", "")
        pid = f"{user.id}:{name}"
        store.put("provider_endpoints", pid,
                  {"id": pid, "name": name,
                   "base_url": body.get("base_url", ""),
                   "api_key": body.get("api_key", "")}, owner=user.id)
        return {"name": name, "ok": True}

    @app.get("/api/v1/provider-endpoints")
    async def list_provider_endpoints(user: AuthUser = Depends(auth_dep)):
        return [{"name": e["name"], "base_url": e["base_url"]}
                for e in store.list("provider_endpoints", owner=user.id)]

    # ------------------------------------------------------------------
    # Anthropic-compatible surface (reference api/pkg/anthropic proxy)
    # ------------------------------------------------------------------
    @app.post("/v1/messages")
    async def anthropic_messages(request: Request,
                                 user: AuthUser = Depends(auth_dep)):
        from helix_amd.server.anthropic_api import (AnthropicPassthrough,
                                                    anthropic_to_openai,
                                                    openai_to_anthropic,
                                                    stream_anthropic_events)
        areq = await request.json()
        # native passthrough when an Anthropic endpoint is configured:
        # raw body forwarded => prompt caching (cache_control) survives
        base = os.environ.get("HELIX_ANTHROPIC_BASE_URL", "")
        akey = os.environ.get("HELIX_ANTHROPIC_API_KEY", "")
        if not base:
            for ep in store.list("provider_endpoints", limit=1000):
                if ep.get("provider") == "anthropic" and \
                        ep.get("base_url"):
                    base = ep["base_url"]
                    akey = ep.get("api_key", "")
                    break
        if base:
            pt = getattr(app.state, "anthropic_passthrough", None)
            if pt is None or pt.base_url != base.rstrip("/"):
                pt = AnthropicPassthrough(base, akey)
                app.state.anthropic_passthrough = pt
            in_h = {k.lower(): v for k, v in request.headers.items()
                    if k.lower().startswith("anthropic-")}
            if areq.get("stream"):
                return StreamingResponse(
                    pt.forward_stream(areq, in_h),
                    media_type="text/event-stream")
            status, body = await pt.forward(areq, in_h)
            return JSONResponse(body, status_code=status)
        oreq = anthropic_to_openai(areq)
        ctx = {"owner": user.id}
        if oreq.get("stream"):
            chunks = controller.chat_completion_stream(oreq, user.id,
                                                       ctx=ctx)

            async def sse():
                async for ev in stream_anthropic_events(
                        chunks, oreq.get("model", "")):
                    yield ev
            return StreamingResponse(sse(), media_type="text/event-stream")
        resp = await controller.chat_completion(oreq, user.id, ctx=ctx)
        return openai_to_anthropic(resp)

    # ------------------------------------------------------------------
    # Model catalog (reference helix-models + model-info routes)
    # ------------------------------------------------------------------
    @app.get("/api/v1/helix-models")
    async def helix_models(user: AuthUser = Depends(auth_dep)):
        return catalog.list()

    @app.get("/api/v1/model-info/{model}")
    async def model_info(model: str, user: AuthUser = Depends(auth_dep)):
        info = catalog.get(model)
        if info is None:
            raise HTTPException(404, "unknown model")
        return info

    @app.put("/api/v1/model-info/{model}")
    async def set_model_info(model: str, request: Request,
                             user: AuthUser = Depends(admin_dep)):
        catalog.set_override(model, await request.json())
        return catalog.get(model)

    # ------------------------------------------------------------------
    # Usage / wallet (reference usage_metrics + wallets)
    # ------------------------------------------------------------------
    @app.get("/api/v1/apps/{app_id}/usage")
    async def app_usage(app_id: str, user: AuthUser = Depends(auth_dep),
                        day: str = ""):
        """Per-app daily usage (reference server.go:1021-1022): rolls up
        llm_calls whose session belongs to the app."""
        sessions = {s["id"] for s in store.list("sessions", parent=app_id,
                                                limit=100000)}
        out = {}
        for m in store.list("usage_metrics", owner=user.id, limit=100000):
            if m.get("session_id") not in sessions:
                continue
            if day and m.get("day") != day:
                continue
            k = (m["day"], m["model"])
            agg = out.setdefault(k, {"day": m["day"], "model": m["model"],
                                     "prompt_tokens": 0,
                                     "completion_tokens": 0, "calls": 0,
                                     "cost_usd": 0.0})
            agg["prompt_tokens"] += m["prompt_tokens"]
            agg["completion_tokens"] += m["completion_tokens"]
            agg["calls"] += 1
            agg["cost_usd"] += m.get("cost_usd", 0.0)
        return list(out.values())

    @app.get("/api/v1/usage")
    async def my_usage(user: AuthUser = Depends(auth_dep), day: str = ""):
        return usage.usage_for(user.id, day or None)

    @app.get("/api/v1/wallet")
    async def my_wallet(user: AuthUser = Depends(auth_dep)):
        return usage.wallet(user.id)

    @app.post("/api/v1/wallet/topup")
    async def wallet_topup(request: Request,
                           user: AuthUser = Depends(admin_dep)):
        body = await request.json()
        return usage.topup(body.get("owner", user.id),
                           float(body.get("amount_usd", 0)))

    # -- Stripe billing (reference api/pkg/stripe: checkout top-ups,
    #    webhook-driven wallet credit + subscription sync) -------------
    from helix_amd.server.billing import BillingService, WebhookError
    billing = BillingService(
        store, usage,
        webhook_secret=os.environ.get("HELIX_STRIPE_WEBHOOK_SECRET", ""))
    app.state.billing = billing

    @app.post("/api/v1/billing/topup-session")
    async def billing_topup_session(request: Request,
                                    user: AuthUser = Depends(auth_dep)):
        if not billing.enabled():
            raise HTTPException(
                501, "billing is not configured "
                "(HELIX_STRIPE_WEBHOOK_SECRET unset)")
        body = await request.json()
        udoc = store.get("users", user.id) or {"id": user.id}
        try:
            return await billing.create_topup_session(
                udoc, float(body.get("amount_usd", 0)))
        except ValueError as e:
            raise HTTPException(400, str(e))

    @app.post("/api/v1/stripe/webhook")
    async def stripe_webhook(request: Request):
        if not billing.enabled():
            raise HTTPException(501, "billing is not configured")
        payload = await request.body()
        if len(payload) > 65536:           # stripe.go:138 MaxBodyBytes
            raise HTTPException(413, "payload too large")
        try:
            return billing.process_webhook(
                payload, request.headers.get("Stripe-Signature", ""))
        except WebhookError as e:
            raise HTTPException(400, str(e))

    @app.get("/api/v1/billing")
    async def billing_state(user: AuthUser = Depends(auth_dep)):
        w = usage.wallet(user.id)
        invoices = [i for i in store.list("billing_invoices",
                                          owner=user.id, limit=100)]
        return {"wallet": w, "invoices": invoices,
                "enabled": billing.enabled()}

    # ------------------------------------------------------------------
    # Triggers (reference api/pkg/trigger: cron + webhook)
    # ------------------------------------------------------------------
    @app.post("/api/v1/triggers")
    async def create_trigger(request: Request,
                             user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        try:
            return triggers.create(user.id, body.get("app_id", ""),
                                   body.get("kind", "cron"),
                                   body.get("config", {}))
        except ValueError as e:
            raise HTTPException(400, str(e))

    @app.get("/api/v1/triggers")
    async def list_triggers(user: AuthUser = Depends(auth_dep)):
        return triggers.list(user.id)

    @app.delete("/api/v1/triggers/{tid}")
    async def delete_trigger(tid: str, user: AuthUser = Depends(auth_dep)):
        doc = store.get("triggers", tid)
        if doc is None or (doc["owner"] != user.id and not user.admin):
            raise HTTPException(404, "trigger not found")
        triggers.delete(tid)
        return {"ok": True}

    @app.post("/api/v1/webhooks/{tid}")
    async def fire_webhook(tid: str, request: Request):
        doc = store.get("triggers", tid)
        if doc is None or doc.get("kind") != "webhook":
            raise HTTPException(404, "webhook not found")
        try:
            payload = await request.json()
        except Exception:
            payload = {}
        return await triggers.fire(doc, payload)

    @app.post("/api/v1/discord/interactions/{tid}")
    async def discord_interactions(tid: str, request: Request):
        """Discord interactions inbound (Ed25519-signed; reference
        api/pkg/trigger/discord): PING->PONG + slash commands."""
        doc = store.get("triggers", tid)
        if doc is None or doc.get("kind") != "discord":
            raise HTTPException(404, "discord trigger not found")
        body = await request.body()
        try:
            return await triggers.handle_discord_event(
                doc, body,
                request.headers.get("X-Signature-Timestamp", ""),
                request.headers.get("X-Signature-Ed25519", ""))
        except PermissionError:
            raise HTTPException(401, "bad discord signature")

    @app.post("/api/v1/azure-devops/webhook/{tid}")
    async def azure_devops_webhook(tid: str, request: Request):
        """Azure DevOps service-hook inbound (basic-auth checked)."""
        doc = store.get("triggers", tid)
        if doc is None or doc.get("kind") != "azure_devops":
            raise HTTPException(404, "azure devops trigger not found")
        body = await request.body()
        try:
            return await triggers.handle_azure_devops_event(
                doc, body, request.headers.get("Authorization", ""))
        except PermissionError:
            raise HTTPException(401, "bad azure devops auth")

    @app.post("/api/v1/crisp/webhook/{tid}")
    async def crisp_webhook(tid: str, request: Request):
        """Crisp webhook inbound (HMAC-signed)."""
        doc = store.get("triggers", tid)
        if doc is None or doc.get("kind") != "crisp":
            raise HTTPException(404, "crisp trigger not found")
        body = await request.body()
        try:
            return await triggers.handle_crisp_event(
                doc, body,
                request.headers.get("X-Crisp-Request-Timestamp", ""),
                request.headers.get("X-Crisp-Signature", ""))
        except PermissionError:
            raise HTTPException(401, "bad crisp signature")

    @app.post("/api/v1/slack/events/{tid}")
    async def slack_events(tid: str, request: Request):
        """Slack Events API inbound (reference api/pkg/trigger slack):
        url_verification challenge + signature-verified message events."""
        doc = store.get("triggers", tid)
        if doc is None or doc.get("kind") != "slack":
            raise HTTPException(404, "slack trigger not found")
        body = await request.body()
        try:
            return await triggers.handle_slack_event(
                doc, body,
                request.headers.get("X-Slack-Request-Timestamp", ""),
                request.headers.get("X-Slack-Signature", ""))
        except PermissionError:
            raise HTTPException(401, "bad slack signature")

    @app.post("/api/v1/teams/webhook/{tid}")
    async def teams_webhook(tid: str, request: Request):
        """Teams outgoing-webhook inbound (HMAC of raw body)."""
        doc = store.get("triggers", tid)
        if doc is None or doc.get("kind") != "teams":
            raise HTTPException(404, "teams trigger not found")
        body = await request.body()
        try:
            return await triggers.handle_teams_event(
                doc, body, request.headers.get("Authorization", ""))
        except PermissionError:
            raise HTTPException(401, "bad teams hmac")

    # ------------------------------------------------------------------
    # Filestore (reference api/pkg/filestore routes)
    # ------------------------------------------------------------------
    @app.get("/api/v1/filestore/list")
    async def fs_list(user: AuthUser = Depends(auth_dep), path: str = ""):
        return filestore.list(user.id, path)

    @app.put("/api/v1/filestore/upload")
    async def fs_upload(request: Request,
                        user: AuthUser = Depends(auth_dep), path: str = ""):
        if not path:
            raise HTTPException(400, "path query param required")
        data = await request.body()
        return filestore.write(user.id, path, data)

    @app.get("/api/v1/filestore/download")
    async def fs_download(user: AuthUser = Depends(auth_dep),
                          path: str = ""):
        from fastapi.responses import Response
        try:
            return Response(filestore.read(user.id, path),
                            media_type="application/octet-stream")
        except FileNotFoundError:
            raise HTTPException(404, "not found")

    @app.delete("/api/v1/filestore")
    async def fs_delete(user: AuthUser = Depends(auth_dep), path: str = ""):
        return {"ok": filestore.delete(user.id, path)}

    # ------------------------------------------------------------------
    # Secrets (reference api/pkg/server secrets routes)
    # ------------------------------------------------------------------
    @app.post("/api/v1/secrets")
    async def set_secret(request: Request,
                         user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        name = body.get("name", "")
        if not name:
            raise HTTPException(400, "name required")
        sid = f"{user.id}:{name}"
        from helix_amd.server.crypto import encrypt_str, secrets_key
        store.put("secrets", sid, {"id": sid, "name": name,
                                   "value": encrypt_str(
                                       str(body.get("value", "")),
                                       secrets_key(cfg.web.admin_api_key))},
                  owner=user.id)
        return {"name": name, "ok": True}

    @app.get("/api/v1/secrets")
    async def list_secrets(user: AuthUser = Depends(auth_dep)):
        return [{"name": s["name"]}
                for s in store.list("secrets", owner=user.id)]

    @app.delete("/api/v1/secrets/{name}")
    async def delete_secret(name: str, user: AuthUser = Depends(auth_dep)):
        ok = store.delete("secrets", f"{user.id}:{name}")
        return {"ok": ok}

    # ------------------------------------------------------------------
    # Users / keys / admin
    # ------------------------------------------------------------------
    @app.get("/api/v1/users")
    async def list_users(user: AuthUser = Depends(admin_dep)):
        """Admin user listing (reference client.go:275 ListUsers)."""
        return [{k: u.get(k) for k in
                 ("id", "username", "admin", "email")}
                for u in store.list("users", limit=10000)]

    @app.post("/api/v1/users")
    async def create_user(request: Request,
                          user: AuthUser = Depends(admin_dep)):
        body = await request.json()
        from helix_amd.server.license import LicenseError
        try:
            app.state.licenses.check_seat()
        except LicenseError as e:
            raise HTTPException(402, str(e))
        u = auth.create_user(body["username"], bool(body.get("admin")))
        key = auth.create_api_key(u["id"])
        return {**u, "api_key": key}

    @app.post("/api/v1/api_keys")
    async def create_key(user: AuthUser = Depends(auth_dep)):
        return {"api_key": auth.create_api_key(user.id)}

    @app.get("/api/v1/llm_calls")
    async def llm_calls(user: AuthUser = Depends(admin_dep),
                        session_id: str = ""):
        if session_id:
            return store.list("llm_calls", parent=session_id)
        return store.list("llm_calls", limit=200)

    # ------------------------------------------------------------------
    # Projects / spec-tasks / git / code-intel (reference services)
    # ------------------------------------------------------------------
    def _owned(table: str, rid: str, user: AuthUser) -> dict:
        """Load an id-addressed doc and enforce owner-or-admin (the same
        guard sessions/knowledge/triggers use; reference authz.go
        authorizeUserToResource). 404 if missing, 403 if not yours."""
        doc = store.get(table, rid)
        if not doc:
            raise HTTPException(404, "not found")
        if doc.get("owner") != user.id and not user.admin:
            raise HTTPException(403, "forbidden")
        return doc

    @app.post("/api/v1/projects")
    async def create_project(request: Request,
                             user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        return spec_tasks.create_project(user.id, body.get("name", ""))

    @app.get("/api/v1/projects")
    async def list_projects(user: AuthUser = Depends(auth_dep)):
        return spec_tasks.list_projects(user.id)

    @app.post("/api/v1/projects/{pid}/tasks")
    async def create_task(pid: str, request: Request,
                          user: AuthUser = Depends(auth_dep)):
        _owned("projects", pid, user)
        body = await request.json()
        return spec_tasks.create_task(user.id, pid, body.get("title", ""),
                                      body.get("description", ""))

    @app.get("/api/v1/projects/{pid}/tasks")
    async def list_tasks(pid: str, user: AuthUser = Depends(auth_dep)):
        _owned("projects", pid, user)
        return spec_tasks.list_tasks(pid)

    @app.post("/api/v1/spec-tasks/{tid}/transition")
    async def transition_task(tid: str, request: Request,
                              user: AuthUser = Depends(auth_dep)):
        _owned("spec_tasks", tid, user)
        body = await request.json()
        try:
            return spec_tasks.transition(tid, body.get("state", ""))
        except (KeyError, ValueError) as e:
            raise HTTPException(400, str(e))

    @app.post("/api/v1/spec-tasks/{tid}/plan")
    async def plan_task(tid: str, user: AuthUser = Depends(auth_dep)):
        _owned("spec_tasks", tid, user)
        try:
            return await spec_tasks.plan(tid)
        except KeyError:
            raise HTTPException(404, "task not found")

    @app.post("/api/v1/spec-tasks/{tid}/implement")
    async def implement_task(tid: str,
                             user: AuthUser = Depends(auth_dep)):
        _owned("spec_tasks", tid, user)
        try:
            return await spec_tasks.implement(tid)
        except KeyError:
            raise HTTPException(404, "task not found")
        except (ValueError, RuntimeError) as e:
            raise HTTPException(400, str(e))

    @app.post("/api/v1/spec-tasks/{tid}/merge")
    async def merge_task(tid: str, user: AuthUser = Depends(auth_dep)):
        doc = spec_tasks.get_task(tid)
        if doc is None or (doc.get("owner") != user.id and
                           not user.admin):
            raise HTTPException(404, "task not found")
        try:
            return await asyncio.to_thread(spec_tasks.merge, tid)
        except ValueError as e:
            raise HTTPException(409, str(e))

    @app.post("/api/v1/spec-tasks/{tid}/comments")
    async def comment_task(tid: str, request: Request,
                           user: AuthUser = Depends(auth_dep)):
        _owned("spec_tasks", tid, user)
        body = await request.json()
        return spec_tasks.add_comment(tid, user.id, body.get("text", ""))

    # ------------------------------------------------------------------
    # MCP gateway (reference mcp gateway routes): JSON-RPC over POST
    # ------------------------------------------------------------------
    @app.post("/api/v1/mcp/{app_id}")
    async def mcp_endpoint(app_id: str, request: Request,
                           user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        if isinstance(body, list):  # JSON-RPC batch
            out = [await mcp.handle(app_id, user.id, r) for r in body]
            return [r for r in out if r is not None]
        resp = await mcp.handle(app_id, user.id, body)
        return resp if resp is not None else JSONResponse(status_code=202,
                                                          content=None)

    # ------------------------------------------------------------------
    # Evaluation suites / runs (reference agent_routes evaluation API)
    # ------------------------------------------------------------------
    @app.post("/api/v1/apps/{app_id}/evaluation-suites")
    async def create_suite(app_id: str, request: Request,
                           user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        return evaluations.create_suite(user.id, app_id,
                                        body.get("name", ""),
                                        body.get("tests", []))

    @app.get("/api/v1/evaluation-suites")
    async def list_suites(user: AuthUser = Depends(auth_dep)):
        return evaluations.list_suites(user.id)

    @app.post("/api/v1/evaluation-suites/{sid}/runs")
    async def run_suite(sid: str, user: AuthUser = Depends(auth_dep)):
        _owned("evaluation_runs", f"suite:{sid}", user)
        try:
            return await evaluations.run_suite(sid)
        except KeyError:
            raise HTTPException(404, "suite not found")

    @app.get("/api/v1/evaluation-runs/{rid}")
    async def get_run(rid: str, user: AuthUser = Depends(auth_dep)):
        run = evaluations.get_run(rid)
        if run is None:
            raise HTTPException(404, "run not found")
        if run.get("owner") != user.id and not user.admin:
            raise HTTPException(403, "forbidden")
        return run

    @app.get("/api/v1/git/repos")
    async def list_repos(user: AuthUser = Depends(auth_dep)):
        return git_svc.list(user.id)

    @app.get("/api/v1/git/repos/{rid}/log")
    async def repo_log(rid: str, user: AuthUser = Depends(auth_dep),
                       ref: str = "HEAD"):
        _owned("git_repositories", rid, user)
        return git_svc.log(rid, ref)

    @app.get("/api/v1/git/repos/{rid}/files")
    async def repo_files(rid: str, user: AuthUser = Depends(auth_dep),
                         ref: str = "HEAD"):
        _owned("git_repositories", rid, user)
        return git_svc.ls_tree(rid, ref)

    @app.post("/api/v1/git/repos/{rid}/index")
    async def index_repo(rid: str, user: AuthUser = Depends(auth_dep)):
        _owned("git_repositories", rid, user)
        n = await code_intel.index_repo(rid)
        return {"chunks": n}

    @app.post("/api/v1/git/repos/{rid}/search")
    async def search_repo(rid: str, request: Request,
                          user: AuthUser = Depends(auth_dep)):
        _owned("git_repositories", rid, user)
        body = await request.json()
        return await code_intel.query(rid, body.get("query", ""),
                                      body.get("k", 6))

    # ------------------------------------------------------------------
    # Sandboxes (reference api/pkg/hydra dev containers + api/pkg/sandbox:
    # create/list/get/delete, exec, file IO, PTY terminal)
    # ------------------------------------------------------------------
    def _owned_sandbox(sid: str, user: AuthUser) -> dict:
        doc = sandboxes.get(sid)
        if doc is None:
            raise HTTPException(404, "sandbox not found")
        if doc.get("owner") != user.id and not user.admin:
            raise HTTPException(403, "not your sandbox")
        return doc

    @app.post("/api/v1/sandboxes")
    async def create_sandbox(request: Request,
                             user: AuthUser = Depends(auth_dep)):
        body = await request.json()
        return sandboxes.create(user.id, body.get("name", ""),
                                body.get("session_id", ""))

    @app.get("/api/v1/sandboxes")
    async def list_sandboxes(user: AuthUser = Depends(auth_dep)):
        return sandboxes.list(user.id)

    @app.get("/api/v1/sandboxes/{sid}")
    async def get_sandbox(sid: str, user: AuthUser = Depends(auth_dep)):
        return _owned_sandbox(sid, user)

    @app.delete("/api/v1/sandboxes/{sid}")
    async def delete_sandbox(sid: str,
                             user: AuthUser = Depends(auth_dep)):
        _owned_sandbox(sid, user)
        return {"ok": sandboxes.delete(sid)}

    @app.post("/api/v1/sandboxes/{sid}/exec")
    async def exec_in_sandbox(sid: str, request: Request,
                              user: AuthUser = Depends(auth_dep)):
        _owned_sandbox(sid, user)
        body = await request.json()
        cmd = body.get("command", "")
        if not cmd:
            raise HTTPException(400, "command required")
        try:
            return await asyncio.to_thread(
                sandboxes.exec, sid, cmd,
                min(float(body.get("timeout_s", 60)), 300),
                body.get("cwd", ""), body.get("env"))
        except SandboxError as e:
            raise HTTPException(400, str(e))

    @app.get("/api/v1/sandboxes/{sid}/files")
    async def sandbox_files(sid: str, path: str = "",
                            user: AuthUser = Depends(auth_dep)):
        _owned_sandbox(sid, user)
        try:
            return sandboxes.list_files(sid, path)
        except (SandboxError, OSError) as e:
            raise HTTPException(400, str(e))

    @app.get("/api/v1/sandboxes/{sid}/file")
    async def sandbox_read_file(sid: str, path: str,
                                user: AuthUser = Depends(auth_dep)):
        _owned_sandbox(sid, user)
        try:
            data = sandboxes.read_file(sid, path)
        except (SandboxError, OSError) as e:
            raise HTTPException(400, str(e))
        return {"path": path,
                "content": data.decode("utf-8", errors="replace")}

    @app.put("/api/v1/sandboxes/{sid}/file")
    async def sandbox_write_file(sid: str, request: Request,
                                 user: AuthUser = Depends(auth_dep)):
        _owned_sandbox(sid, user)
        body = await request.json()
        try:
            sandboxes.write_file(sid, body.get("path", ""),
                                 body.get("content", "").encode())
        except (SandboxError, OSError) as e:
            raise HTTPException(400, str(e))
        return {"ok": True}

    @app.websocket("/api/v1/sandboxes/{sid}/terminal")
    async def sandbox_terminal(ws: WebSocket, sid: str):
        user = auth.resolve(ws.query_params.get("access_token", ""))
        if user is None:
            await ws.close(code=4401)
            return
        doc = sandboxes.get(sid)
        if doc is None or (doc.get("owner") != user.id and
                           not user.admin):
            await ws.close(code=4403)
            return
        await ws.accept()
        pid, master = sandboxes.open_terminal(sid)
        loop = asyncio.get_event_loop()

        async def pump_out():
            try:
                while True:
                    data = await loop.run_in_executor(
                        None, lambda: os.read(master, 4096))
                    if not data:
                        break
                    await ws.send_bytes(data)
            except (OSError, RuntimeError):
                pass

        out_task = asyncio.ensure_future(pump_out())
        try:
            while True:
                data = await ws.receive_bytes()
                os.write(master, data)
        except Exception:
            pass
        finally:
            out_task.cancel()
            try:
                os.kill(pid, 9)
                os.close(master)
            except OSError:
                pass

    # -- git smart-HTTP (reference git_http_server.go): real clone /
    #    push against platform repos via `git http-backend`, auth'd
    #    with the same bearer keys (git clients send them via
    #    http.extraHeader, or basic auth with the key as password) ----
    def _git_http_user(request: Request) -> Optional[AuthUser]:
        hdr = request.headers.get("Authorization", "")
        if hdr.startswith("Basic "):
            import base64 as _b64
            try:
                raw = _b64.b64decode(hdr[6:]).decode()
                _, _, pwd = raw.partition(":")
                return auth.resolve(pwd)
            except Exception:
                return None
        if hdr.startswith("Bearer "):
            return auth.resolve(hdr[7:])
        return None

    def _git_http_repo(rid: str, request: Request) -> dict:
        user = _git_http_user(request)
        if user is None:
            raise HTTPException(
                401, "authentication required",
                headers={"WWW-Authenticate": 'Basic realm="helix-git"'})
        doc = git_svc.get(rid)
        if doc is None or (doc.get("owner") != user.id and
                           not user.admin):
            raise HTTPException(404, "repository not found")
        return doc

    @app.get("/api/v1/git/repos/{rid}.git/info/refs")
    async def git_info_refs(rid: str, request: Request,
                            service: str = ""):
        from helix_amd.server.git_service import run_http_backend
        doc = _git_http_repo(rid, request)
        if service not in ("git-upload-pack", "git-receive-pack"):
            raise HTTPException(400, "dumb HTTP protocol not served")
        status, headers, payload = await asyncio.to_thread(
            run_http_backend, doc["path"], "GET",
            f"/{rid}.git/info/refs", f"service={service}", "", b"")
        from fastapi.responses import Response
        return Response(content=payload, status_code=status,
                        media_type=headers.get(
                            "Content-Type",
                            f"application/x-{service}-advertisement"))

    @app.post("/api/v1/git/repos/{rid}.git/{service}")
    async def git_service_rpc(rid: str, service: str, request: Request):
        from helix_amd.server.git_service import run_http_backend
        doc = _git_http_repo(rid, request)
        if service not in ("git-upload-pack", "git-receive-pack"):
            raise HTTPException(404, "unknown git service")
        body = await request.body()
        status, headers, payload = await asyncio.to_thread(
            run_http_backend, doc["path"], "POST",
            f"/{rid}.git/{service}", "",
            request.headers.get("Content-Type", ""), body)
        from fastapi.responses import Response
        return Response(content=payload, status_code=status,
                        media_type=headers.get(
                            "Content-Type",
                            f"application/x-{service}-result"))

    @app.get("/api/v1/config")
    async def get_config(user: AuthUser = Depends(auth_dep)):
        return {"version": "helix_amd-0.1.0",
                "default_model": cfg.inference.default_model,
                "default_provider": cfg.inference.default_provider}

    @app.get("/api/v1/search")
    async def global_search(q: str = "",
                            user: AuthUser = Depends(auth_dep)):
        """Owner-scoped search across sessions, apps, knowledge and
        spec tasks (the reference frontend's global search bar)."""
        needle = q.lower().strip()
        if not needle:
            return {"sessions": [], "apps": [], "knowledge": [],
                    "tasks": []}

        def match(doc, *fields):
            return any(needle in str(doc.get(f, "")).lower()
                       for f in fields)

        out = {
            "sessions": [
                {"id": d["id"], "name": d.get("name", "")}
                for d in store.list("sessions", owner=user.id,
                                    limit=2000)
                if match(d, "name")][:20],
            "apps": [
                {"id": d["id"],
                 "name": (d.get("config", {}) or {}).get("name", "")}
                for d in store.list("apps", owner=user.id, limit=2000)
                if needle in str((d.get("config", {}) or {}).get(
                    "name", "")).lower()][:20],
            "knowledge": [
                {"id": d["id"], "name": d.get("name", ""),
                 "state": d.get("state", "")}
                for d in store.list("knowledge", owner=user.id,
                                    limit=2000)
                if match(d, "name")][:20],
            "tasks": [
                {"id": d["id"], "title": d.get("title", ""),
                 "state": d.get("state", "")}
                for d in store.list("spec_tasks", owner=user.id,
                                    limit=2000)
                if match(d, "title", "description")][:20],
        }
        return out

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    @app.get("/debug/stats")
    async def debug_stats(user: AuthUser = Depends(admin_dep)):
        # pprof-equivalent quick stats (reference mounts net/http/pprof,
        # server.go:1473)
        import threading as _t

        import psutil
        p = psutil.Process()
        return {
            "rss_bytes": p.memory_info().rss,
            "num_threads": p.num_threads(),
            "cpu_percent": p.cpu_percent(interval=0.0),
            "open_files": len(p.open_files()),
            "threads": [t.name for t in _t.enumerate()][:50],
            "store_counts": {t: store.count(t) for t in
                             ("sessions", "interactions", "llm_calls",
                              "apps", "knowledge")},
        }

    @app.get("/debug/threads")
    async def debug_threads(user: AuthUser = Depends(admin_dep)):
        # python stack dump (pprof goroutine-profile equivalent)
        import sys
        import traceback
        import threading as _t
        names = {t.ident: t.name for t in _t.enumerate()}
        out = {}
        for tid, frame in sys._current_frames().items():
            out[f"{names.get(tid, '?')}-{tid}"] = \
                traceback.format_stack(frame)[-6:]
        return out

    @app.post("/api/v1/admin/backup")
    async def store_backup(request: Request,
                           user: AuthUser = Depends(admin_dep)):
        """Online store backup to a server-side path (reference
        Postgres dump role)."""
        try:
            body = await request.json()
        except Exception:
            body = {}
        import time as _t
        path = body.get("path") or os.path.join(
            cfg.filestore.path, "backups",
            _t.strftime("helix-%Y%m%d-%H%M%S.db"))
        os.makedirs(os.path.dirname(path), exist_ok=True)
        store.backup(path)
        return {"ok": True, "path": path,
                "bytes": os.path.getsize(path)}

    # -- license (reference api/pkg/license: signed envelope, expiry,
    #    revocation, seat limits; development mode without one) -------
    from helix_amd.server.license import LicenseError, LicenseManager
    _pub_hex = os.environ.get("HELIX_LICENSE_PUBKEY", "")
    licenses = LicenseManager(
        store, bytes.fromhex(_pub_hex) if _pub_hex else b"\x00" * 32)
    app.state.licenses = licenses

    @app.get("/api/v1/license")
    async def license_status(user: AuthUser = Depends(admin_dep)):
        licenses.revalidate()
        return licenses.status()

    @app.post("/api/v1/license")
    async def install_license(request: Request,
                              user: AuthUser = Depends(admin_dep)):
        body = await request.json()
        try:
            lic = licenses.install(body.get("envelope", ""))
        except LicenseError as e:
            raise HTTPException(400, str(e))
        return lic.to_dict()

    @app.get("/api/v1/admin/errors")
    async def list_errors(user: AuthUser = Depends(admin_dep)):
        store.flush("error_events")
        rows = store.list("error_events", limit=200)
        rows.sort(key=lambda r: -r.get("last_seen", 0))
        return rows

    @app.post("/api/v1/admin/janitor")
    async def run_janitor(request: Request,
                          user: AuthUser = Depends(admin_dep)):
        """Retention sweep (reference janitor/cleanup manager): prune
        llm_calls/usage_metrics/step_info older than retention_days and
        sessions idle past session_retention_days (0 = keep)."""
        try:
            body = await request.json()
        except Exception:
            body = {}
        import time as _t
        days = float(body.get("retention_days", 30))
        sdays = float(body.get("session_retention_days", 0))
        cutoff_ms = int((_t.time() - days * 86400) * 1000)
        cutoff_s = _t.time() - days * 86400
        pruned = {"llm_calls": 0, "usage_metrics": 0, "step_info": 0,
                  "sessions": 0, "interactions": 0}
        for table, ts_field, cut in (("llm_calls", "created", cutoff_ms),
                                     ("usage_metrics", "ts", cutoff_s),
                                     ("step_info", "created", cutoff_ms)):
            for doc in store.list(table, limit=100000, desc=False):
                v = doc.get(ts_field, 0)
                if v and v < cut:
                    store.delete(table, doc["id"])
                    pruned[table] += 1
        if sdays > 0:
            scut = (_t.time() - sdays * 86400) * 1000
            for doc in store.list("sessions", limit=100000, desc=False):
                if doc.get("updated", doc.get("created", 0)) < scut:
                    for it in store.list("interactions",
                                         parent=doc["id"]):
                        store.delete("interactions", it["id"])
                        pruned["interactions"] += 1
                    store.delete("sessions", doc["id"])
                    pruned["sessions"] += 1
        # bounded retention for bus streams and error events
        keep_bus = int(body.get("bus_keep_last", 10000))
        streams = {r["id"].split(":", 1)[0]
                   for r in store.list("bus_messages", limit=100000)}
        pruned["bus_messages"] = 0
        for stream in streams:
            before = len(store.list("bus_messages", parent=stream,
                                    limit=1000000))
            bus.purge(stream, keep_last=keep_bus)
            after = len(store.list("bus_messages", parent=stream,
                                   limit=1000000))
            pruned["bus_messages"] += before - after
        pruned["error_events"] = 0
        for doc in store.list("error_events", limit=100000):
            if doc.get("last_seen", 0) < cutoff_s:
                store.delete("error_events", doc["id"])
                pruned["error_events"] += 1
        # idle-sandbox GC (reference hydra handleGCReconcile)
        sbx_hours = float(body.get("sandbox_idle_hours", 24))
        pruned["sandboxes"] = 0
        if sbx_hours > 0:
            cut = _t.time() - sbx_hours * 3600
            for doc in store.list("sandboxes", limit=100000):
                last = max(doc.get("last_exec", 0),
                           doc.get("created", 0))
                if last < cut:
                    sandboxes.delete(doc["id"])
                    pruned["sandboxes"] += 1
        licenses.revalidate()
        return pruned

    @app.get("/")
    async def index():
        from fastapi.responses import HTMLResponse
        from helix_amd.server.webui import INDEX_HTML
        return HTMLResponse(INDEX_HTML)

    return app
