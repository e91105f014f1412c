"""Provider layer — parity with the reference's api/pkg/openai +
openai/manager (MultiClientManager): a Client interface multiplexed over
external OpenAI-compatible providers AND the internal "helix" provider
(the native MI355X runner), with retries, per-call logging and TTFT.
"""
from __future__ import annotations

import asyncio
import json
import logging
import random
import time
from typing import Any, AsyncIterator, Dict, List, Optional

import httpx

from helix_amd.server.types import LLMCall, new_id

log = logging.getLogger("helix_amd.providers")


class ProviderError(Exception):
    def __init__(self, message: str, status: int = 500):
        super().__init__(message)
        self.status = status


class Client:
    """Chat + embeddings client (reference openai_client.go Client iface)."""
    provider: str = ""

    async def chat(self, req: dict) -> dict:
        raise NotImplementedError

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        raise NotImplementedError

    async def embeddings(self, req: dict) -> dict:
        raise NotImplementedError

    async def images(self, req: dict) -> dict:
        raise ProviderError(
            f"provider {self.provider or '?'} does not support image "
            "generation", 501)

    async def list_models(self) -> List[str]:
        return []


class MockClient(Client):
    """Deterministic scripted client for tests (reference's gomock role)."""

    def __init__(self, provider: str = "mock", responses: Optional[List] = None):
        self.provider = provider
        self.responses = responses or ["mock response"]
        self.calls: List[dict] = []
        self._i = 0

    def _next(self, req) -> dict:
        self.calls.append(req)
        r = self.responses[min(self._i, len(self.responses) - 1)]
        self._i += 1
        if callable(r):
            r = r(req)
        if isinstance(r, dict):
            return r
        return {
            "id": new_id("chatcmpl"),
            "object": "chat.completion",
            "created": int(time.time()),
            "model": req.get("model", "mock-model"),
            "choices": [{"index": 0, "message": {
                "role": "assistant", "content": str(r)},
                "finish_reason": "stop"}],
            "usage": {"prompt_tokens": 7, "completion_tokens": 5,
                      "total_tokens": 12},
        }

    async def chat(self, req: dict) -> dict:
        return self._next(req)

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        full = self._next(req)
        msg = full["choices"][0]["message"]
        content = msg.get("content") or ""
        base = {"id": full["id"], "object": "chat.completion.chunk",
                "created": full["created"], "model": full["model"]}
        yield {**base, "choices": [{"index": 0, "delta": {
            "role": "assistant", "content": ""}, "finish_reason": None}]}
        for i in range(0, len(content), 8):
            yield {**base, "choices": [{"index": 0, "delta": {
                "content": content[i:i + 8]}, "finish_reason": None}]}
        last = {**base, "choices": [{"index": 0, "delta": {},
                                     "finish_reason": "stop"}]}
        if msg.get("tool_calls"):
            last["choices"][0]["delta"]["tool_calls"] = msg["tool_calls"]
        yield last

    async def embeddings(self, req: dict) -> dict:
        inputs = req.get("input")
        inputs = [inputs] if isinstance(inputs, str) else list(inputs)
        self.calls.append(req)
        data = []
        for i, text in enumerate(inputs):
            random.seed(hash(text) & 0xFFFF)
            data.append({"object": "embedding", "index": i,
                         "embedding": [random.random() for _ in range(8)]})
        return {"object": "list", "data": data, "model": req.get("model")}

    async def images(self, req: dict) -> dict:
        import base64
        import struct
        import zlib
        self.calls.append(req)
        # minimal valid 1x1 red PNG built by hand
        def chunk(tag, payload):
            return (struct.pack(">I", len(payload)) + tag + payload +
                    struct.pack(">I", zlib.crc32(tag + payload)))
        raw = zlib.compress(b"\x00\xff\x00\x00")
        png = (b"\x89PNG\r\n\x1a\n" +
               chunk(b"IHDR", struct.pack(">IIBBBBB", 1, 1, 8, 2,
                                          0, 0, 0)) +
               chunk(b"IDAT", raw) + chunk(b"IEND", b""))
        n = max(1, int(req.get("n", 1) or 1))
        return {"created": int(time.time()),
                "model": req.get("model", "mock-image"),
                "data": [{"b64_json": base64.b64encode(png).decode(),
                          "revised_prompt": None}] * n}

    async def list_models(self) -> List[str]:
        return ["mock-model"]


class OpenAIHTTPClient(Client):
    """External OpenAI-compatible provider over HTTP (openai.com,
    together.ai, any base_url)."""

    def __init__(self, provider: str, base_url: str, api_key: str,
                 timeout: float = 180.0, transport=None):
        self.provider = provider
        self.base_url = base_url.rstrip("/")
        self.api_key = api_key
        self._http = httpx.AsyncClient(
            timeout=timeout, transport=transport,
            headers={"Authorization": f"Bearer {api_key}"})

    async def chat(self, req: dict) -> dict:
        r = await self._http.post(f"{self.base_url}/chat/completions",
                                  json={**req, "stream": False})
        if r.status_code != 200:
            raise ProviderError(r.text, r.status_code)
        return r.json()

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        async with self._http.stream(
                "POST", f"{self.base_url}/chat/completions",
                json={**req, "stream": True}) as r:
            if r.status_code != 200:
                body = await r.aread()
                raise ProviderError(body.decode(), r.status_code)
            async for line in r.aiter_lines():
                if not line.startswith("data: "):
                    continue
                payload = line[6:]
                if payload.strip() == "[DONE]":
                    return
                yield json.loads(payload)

    async def embeddings(self, req: dict) -> dict:
        r = await self._http.post(f"{self.base_url}/embeddings", json=req)
        if r.status_code != 200:
            raise ProviderError(r.text, r.status_code)
        return r.json()

    async def images(self, req: dict) -> dict:
        r = await self._http.post(f"{self.base_url}/images/generations",
                                  json=req)
        if r.status_code != 200:
            raise ProviderError(r.text, r.status_code)
        return r.json()

    async def list_models(self) -> List[str]:
        try:
            r = await self._http.get(f"{self.base_url}/models")
            if r.status_code != 200:
                return []
            return [m["id"] for m in r.json().get("data", [])]
        except httpx.HTTPError:
            return []


class LocalRunnerClient(Client):
    """The internal "helix" provider: in-process RunnerService (local
    mode) — the seam the reference implements via pubsub+RevDial
    (helix_openai_server.go)."""

    def __init__(self, service):
        self.provider = "helix"
        self.service = service

    async def chat(self, req: dict) -> dict:
        from helix_amd.runner.openai_adapter import chat_completion
        return await chat_completion(self.service, {**req, "stream": False})

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        from helix_amd.runner.openai_adapter import chat_completion
        it = await chat_completion(self.service, {**req, "stream": True})
        async for chunk in it:
            yield chunk

    async def embeddings(self, req: dict) -> dict:
        from helix_amd.runner.openai_adapter import embeddings
        return await embeddings(self.service, req)

    async def images(self, req: dict) -> dict:
        from helix_amd.runner.openai_adapter import images_generations
        return await images_generations(self.service, req)

    async def list_models(self) -> List[str]:
        return list(self.service.specs.keys())


class RouterClient(Client):
    """The internal "helix" provider in control-plane mode: dispatches to
    remote runners through the inference router (reference
    helix_openai_server.go: PickRunner -> dispatchToSandbox). Runners
    reachable only via reverse tunnel advertise `tunnel:<id>` addresses
    and are dispatched through the TunnelRegistry (RevDial parity)."""

    def __init__(self, router, timeout: float = 300.0, transport=None,
                 tunnels=None):
        self.provider = "helix"
        self.router = router
        self.tunnels = tunnels
        self._http = httpx.AsyncClient(timeout=timeout, transport=transport)

    def _pick(self, model: str):
        addr = self.router.pick_runner(model)
        from helix_amd.server.tunnel import TUNNEL_ADDR_PREFIX, TunnelClient
        if addr.startswith(TUNNEL_ADDR_PREFIX):
            if self.tunnels is None:
                raise ProviderError("tunnel registry unavailable", 502)
            return TunnelClient(self.tunnels,
                                addr[len(TUNNEL_ADDR_PREFIX):])
        return addr

    async def chat(self, req: dict) -> dict:
        addr = self._pick(req.get("model", ""))
        if not isinstance(addr, str):
            return await addr.chat(req)
        r = await self._http.post(f"{addr}/v1/chat/completions",
                                  json={**req, "stream": False})
        if r.status_code != 200:
            raise ProviderError(r.text, r.status_code)
        return r.json()

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        addr = self._pick(req.get("model", ""))
        if not isinstance(addr, str):
            async for chunk in addr.chat_stream(req):
                yield chunk
            return
        async with self._http.stream(
                "POST", f"{addr}/v1/chat/completions",
                json={**req, "stream": True}) as r:
            if r.status_code != 200:
                body = await r.aread()
                raise ProviderError(body.decode(), r.status_code)
            async for line in r.aiter_lines():
                if not line.startswith("data: "):
                    continue
                if line[6:].strip() == "[DONE]":
                    return
                yield json.loads(line[6:])

    async def embeddings(self, req: dict) -> dict:
        addr = self._pick(req.get("model", ""))
        if not isinstance(addr, str):
            return await addr.embeddings(req)
        r = await self._http.post(f"{addr}/v1/embeddings", json=req)
        if r.status_code != 200:
            raise ProviderError(r.text, r.status_code)
        return r.json()

    async def images(self, req: dict) -> dict:
        addr = self._pick(req.get("model", ""))
        if not isinstance(addr, str):
            return await addr.images(req)
        r = await self._http.post(f"{addr}/v1/images/generations",
                                  json=req)
        if r.status_code != 200:
            raise ProviderError(r.text, r.status_code)
        return r.json()

    async def list_models(self) -> List[str]:
        return self.router.available_models()


class RetryableClient(Client):
    """Retry with backoff on 429/5xx (reference openai_client.go)."""

    def __init__(self, inner: Client, retries: int = 3, base_delay: float = 0.5):
        self.inner = inner
        self.provider = inner.provider
        self.retries = retries
        self.base_delay = base_delay

    async def chat(self, req: dict) -> dict:
        delay = self.base_delay
        for attempt in range(self.retries + 1):
            try:
                return await self.inner.chat(req)
            except ProviderError as e:
                if e.status not in (429, 500, 502, 503, 504) or \
                        attempt == self.retries:
                    raise
                await asyncio.sleep(delay)
                delay *= 2

    def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        return self.inner.chat_stream(req)

    async def embeddings(self, req: dict) -> dict:
        return await self.inner.embeddings(req)

    async def images(self, req: dict) -> dict:
        return await self.inner.images(req)

    async def list_models(self) -> List[str]:
        return await self.inner.list_models()


class LoggingClient(Client):
    """LLM-call logging middleware: request/response, duration, TTFT,
    usage -> llm_calls table (reference openai/logger/openai_logger.go:249
    firstTokenMs)."""

    def __init__(self, inner: Client, store, usage_logger=None):
        self.inner = inner
        self.provider = inner.provider
        self.store = store
        self.usage_logger = usage_logger

    def _log(self, call: LLMCall):
        if self.store is not None:
            self.store.put("llm_calls", call.id, call.model_dump(),
                           owner=call.owner, parent=call.session_id,
                           buffered=True)
        if self.usage_logger is not None:
            self.usage_logger(call)

    @staticmethod
    def _ctx(req):
        return req.pop("_ctx", {}) if isinstance(req, dict) else {}

    async def chat(self, req: dict) -> dict:
        ctx = self._ctx(req)
        t0 = time.monotonic()
        call = LLMCall(provider=self.provider, model=req.get("model", ""),
                       owner=ctx.get("owner", ""),
                       session_id=ctx.get("session_id", ""),
                       interaction_id=ctx.get("interaction_id", ""),
                       step=ctx.get("step", ""), request=_redact(req))
        try:
            resp = await self.inner.chat(req)
            call.duration_ms = int((time.monotonic() - t0) * 1000)
            call.first_token_ms = call.duration_ms
            usage = resp.get("usage") or {}
            call.prompt_tokens = usage.get("prompt_tokens", 0)
            call.completion_tokens = usage.get("completion_tokens", 0)
            call.response = {"id": resp.get("id"), "usage": usage}
            self._log(call)
            return resp
        except Exception as e:
            call.error = str(e)
            call.duration_ms = int((time.monotonic() - t0) * 1000)
            self._log(call)
            raise

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        ctx = self._ctx(req)
        t0 = time.monotonic()
        call = LLMCall(provider=self.provider, model=req.get("model", ""),
                       owner=ctx.get("owner", ""),
                       session_id=ctx.get("session_id", ""),
                       interaction_id=ctx.get("interaction_id", ""),
                       step=ctx.get("step", ""), request=_redact(req))
        first = True
        try:
            async for chunk in self.inner.chat_stream(req):
                if first:
                    call.first_token_ms = int((time.monotonic() - t0) * 1000)
                    first = False
                usage = chunk.get("usage")
                if usage:
                    call.prompt_tokens = usage.get("prompt_tokens", 0)
                    call.completion_tokens = usage.get("completion_tokens", 0)
                yield chunk
        except BaseException as e:
            # BaseException so a client disconnect (GeneratorExit) is
            # also recorded; the original exception propagates unchanged
            if isinstance(e, GeneratorExit):
                call.error = "client disconnected"
            else:
                call.error = str(e)
            raise
        finally:
            call.duration_ms = int((time.monotonic() - t0) * 1000)
            self._log(call)

    async def embeddings(self, req: dict) -> dict:
        return await self.inner.embeddings(req)

    async def images(self, req: dict) -> dict:
        return await self.inner.images(req)

    async def list_models(self) -> List[str]:
        return await self.inner.list_models()


def _redact(req: dict) -> dict:
    out = dict(req)
    out.pop("_ctx", None)
    return out


class ProviderManager:
    """Global + user provider endpoints; model-list aggregation with
    `provider/` prefixes (reference provider_manager.go +
    openai_model_handlers.go:17-45)."""

    def __init__(self, store=None):
        self.store = store
        self._global: Dict[str, Client] = {}
        self._model_cache: Dict[str, List[str]] = {}

    def register(self, name: str, client: Client):
        self._global[name] = client

    def providers(self, owner: str = "") -> List[str]:
        names = list(self._global.keys())
        if self.store is not None and owner:
            for ep in self.store.list("provider_endpoints", owner=owner):
                names.append(ep["name"])
        return names

    def get_client(self, provider: str, owner: str = "") -> Client:
        if provider in self._global:
            return self._global[provider]
        if self.store is not None and owner:
            for ep in self.store.list("provider_endpoints", owner=owner):
                if ep["name"] == provider:
                    return RetryableClient(OpenAIHTTPClient(
                        provider, ep["base_url"], ep.get("api_key", "")))
        raise ProviderError(f"unknown provider: {provider}", 400)

    async def refresh_models(self):
        for name, client in self._global.items():
            try:
                self._model_cache[name] = await client.list_models()
            except Exception as e:
                log.warning("model list failed for %s: %s", name, e)

    async def aggregate_models(self, owner: str = "") -> List[dict]:
        """All providers' models; non-default providers prefixed
        `provider/model` (reference model_handlers behavior)."""
        if not self._model_cache:
            await self.refresh_models()
        out = []
        for name, models in self._model_cache.items():
            for m in models:
                mid = m if name == "helix" else f"{name}/{m}"
                out.append({"id": mid, "object": "model", "owned_by": name})
        return out

    def resolve(self, model: str, default_provider: str,
                owner: str = "") -> tuple[str, str]:
        """Model name -> (provider, bare model). Order mirrors the
        reference (openai_chat_handlers.go:148-192): cached-list lookup,
        then prefix parse, then default provider."""
        for name, models in self._model_cache.items():
            if model in models:
                return name, model
        if "/" in model:
            prefix, rest = model.split("/", 1)
            if prefix in self._global or (
                    self.store is not None and owner and
                    any(e["name"] == prefix for e in
                        self.store.list("provider_endpoints", owner=owner))):
                return prefix, rest
        return default_provider, model
