"""Reverse tunnel — RevDial parity (reference api/pkg/revdial + connman):
NAT'd runners dial OUT once and the control plane dispatches inference
back through that connection, keyed by runner id, never by address
(design/2026-06-04 Bug B rationale).

Transport: a long-lived SSE stream (control plane -> runner carries
requests) + HTTP POST replies (runner -> control plane carries response
events). Pure httpx/uvicorn — no websocket client dependency needed.
"""
from __future__ import annotations

import asyncio
import json
import logging
import uuid
from typing import AsyncIterator, Dict, Optional

log = logging.getLogger("helix_amd.tunnel")

TUNNEL_ADDR_PREFIX = "tunnel:"


class TunnelRegistry:
    """Control-plane side: per-runner outbound queues + per-request
    reply queues."""

    def __init__(self):
        self._outbound: Dict[str, asyncio.Queue] = {}
        self._pending: Dict[str, asyncio.Queue] = {}

    def connect(self, runner_id: str) -> asyncio.Queue:
        q: asyncio.Queue = asyncio.Queue()
        self._outbound[runner_id] = q
        return q

    def disconnect(self, runner_id: str, q: asyncio.Queue):
        if self._outbound.get(runner_id) is q:
            self._outbound.pop(runner_id, None)

    def is_connected(self, runner_id: str) -> bool:
        return runner_id in self._outbound

    async def send_request(self, runner_id: str, path: str,
                           body: dict) -> str:
        q = self._outbound.get(runner_id)
        if q is None:
            raise ConnectionError(f"runner {runner_id} has no tunnel")
        rid = uuid.uuid4().hex
        self._pending[rid] = asyncio.Queue()
        try:
            await q.put({"id": rid, "path": path, "body": body})
        except BaseException:
            self._pending.pop(rid, None)
            raise
        return rid

    async def events(self, rid: str, timeout: float = 300.0
                     ) -> AsyncIterator[dict]:
        """Yield reply events for a request until 'end'/'error'."""
        q = self._pending.get(rid)
        if q is None:                      # already reaped / unknown id
            return
        try:
            while True:
                ev = await asyncio.wait_for(q.get(), timeout)
                yield ev
                if ev.get("type") in ("end", "error", "response"):
                    return
        finally:
            self._pending.pop(rid, None)

    async def push_reply(self, rid: str, ev: dict) -> bool:
        q = self._pending.get(rid)
        if q is None:
            return False
        await q.put(ev)
        return True


class TunnelClient:
    """The 'helix' provider transport for tunnel-connected runners
    (counterpart of the reference's dispatchToSandbox over RevDial,
    helix_openai_server.go:222-295)."""

    def __init__(self, registry: TunnelRegistry, runner_id: str,
                 timeout: float = 300.0):
        self.registry = registry
        self.runner_id = runner_id
        self.timeout = timeout

    async def chat(self, req: dict) -> dict:
        rid = await self.registry.send_request(
            self.runner_id, "/v1/chat/completions",
            {**req, "stream": False})
        async for ev in self.registry.events(rid, self.timeout):
            if ev["type"] == "response":
                return ev["data"]
            if ev["type"] == "error":
                from helix_amd.server.providers import ProviderError
                raise ProviderError(ev.get("message", "tunnel error"),
                                    ev.get("status", 502))
        from helix_amd.server.providers import ProviderError
        raise ProviderError("tunnel closed without response", 502)

    async def chat_stream(self, req: dict) -> AsyncIterator[dict]:
        rid = await self.registry.send_request(
            self.runner_id, "/v1/chat/completions", {**req, "stream": True})
        async for ev in self.registry.events(rid, self.timeout):
            if ev["type"] == "chunk":
                yield ev["data"]
            elif ev["type"] == "error":
                from helix_amd.server.providers import ProviderError
                raise ProviderError(ev.get("message", "tunnel error"),
                                    ev.get("status", 502))
            elif ev["type"] in ("end", "response"):
                return

    async def _unary(self, path: str, req: dict) -> dict:
        rid = await self.registry.send_request(self.runner_id, path, req)
        async for ev in self.registry.events(rid, self.timeout):
            if ev["type"] == "response":
                return ev["data"]
            if ev["type"] == "error":
                from helix_amd.server.providers import ProviderError
                raise ProviderError(ev.get("message", "tunnel error"),
                                    ev.get("status", 502))
        from helix_amd.server.providers import ProviderError
        raise ProviderError("tunnel closed without response", 502)

    async def embeddings(self, req: dict) -> dict:
        return await self._unary("/v1/embeddings", req)

    async def images(self, req: dict) -> dict:
        return await self._unary("/v1/images/generations", req)


# ---------------------------------------------------------------------------
# Runner side
# ---------------------------------------------------------------------------

async def tunnel_loop(api_url: str, runner_token: str, runner_id: str,
                      service, stop_event=None,
                      reconnect_delay: float = 3.0):
    """Dial out to the control plane and serve dispatched requests
    locally. Reconnects forever (the reference's redial semantics)."""
    import httpx
    headers = {"Authorization": f"Bearer {runner_token}"}
    while stop_event is None or not stop_event.is_set():
        try:
            async with httpx.AsyncClient(timeout=None) as http:
                async with http.stream(
                        "GET", f"{api_url}/api/v1/runner/tunnel/{runner_id}",
                        headers=headers) as resp:
                    if resp.status_code != 200:
                        raise ConnectionError(f"tunnel rejected: "
                                              f"{resp.status_code}")
                    log.info("tunnel established for %s", runner_id)
                    async for line in resp.aiter_lines():
                        if not line.startswith("data: "):
                            continue
                        msg = json.loads(line[6:])
                        if msg.get("type") == "ping":
                            continue
                        asyncio.ensure_future(_serve_one(
                            http, api_url, headers, runner_id, service, msg))
        except asyncio.CancelledError:
            return
        except Exception as e:
            log.warning("tunnel dropped (%s); redialing", e)
        if stop_event is not None and stop_event.is_set():
            return
        await asyncio.sleep(reconnect_delay)


async def _serve_one(http, api_url, headers, runner_id, service, msg):
    from helix_amd.runner.openai_adapter import (chat_completion,
                                                 embeddings,
                                                 images_generations)
    rid = msg["id"]
    reply_url = f"{api_url}/api/v1/runner/tunnel/{runner_id}/reply"

    async def reply(ev: dict):
        await http.post(reply_url, json={"id": rid, "event": ev},
                        headers=headers)

    try:
        body = msg.get("body") or {}
        if msg.get("path") == "/v1/embeddings":
            out = await embeddings(service, body)
            await reply({"type": "response", "data": out})
        elif msg.get("path") == "/v1/images/generations":
            out = await images_generations(service, body)
            await reply({"type": "response", "data": out})
        elif body.get("stream"):
            it = await chat_completion(service, body, request_id=rid)
            async for chunk in it:
                await reply({"type": "chunk", "data": chunk})
            await reply({"type": "end"})
        else:
            out = await chat_completion(service, body, request_id=rid)
            await reply({"type": "response", "data": out})
    except Exception as e:
        log.exception("tunnel request %s failed", rid)
        try:
            await reply({"type": "error", "message": str(e), "status": 500})
        except Exception:
            pass
