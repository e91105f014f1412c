"""External-agent bridge (parity with the reference's external-agent
plane, api/pkg/external-agent + controller_external_agent.go:16-80: an
assistant with agent_type "zed_external" has its turns executed by an
externally connected agent process instead of the in-process loop; the
agent connects over WebSocket, receives chat commands, and streams
chunks back).

Semantics carried over from the reference:
- idle-activity timeout, not absolute: every chunk resets the clock
  (controller_external_agent.go:16-24 — agent turns can run for hours;
  a fixed deadline marked productive turns as timeouts);
- a hard max-wait runaway guard on top;
- per-request response channels cleaned up when the turn ends.
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import AsyncIterator, Dict, Optional

from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.external_agent")


class ExternalAgentError(Exception):
    pass


class ExternalAgentRegistry:
    def __init__(self, idle_timeout_s: float = 7200.0,
                 max_wait_s: float = 86400.0):
        self.idle_timeout_s = idle_timeout_s
        self.max_wait_s = max_wait_s
        self._outboxes: Dict[str, asyncio.Queue] = {}
        self._responses: Dict[str, asyncio.Queue] = {}

    # -- connection lifecycle (the WS route drives these) -------------
    def attach(self, agent_id: str) -> asyncio.Queue:
        """Register a connected agent; returns its outbox (frames the
        WS pump must forward to the agent)."""
        q: asyncio.Queue = asyncio.Queue(maxsize=256)
        self._outboxes[agent_id] = q
        return q

    def detach(self, agent_id: str):
        self._outboxes.pop(agent_id, None)

    def connected(self, agent_id: str) -> bool:
        return agent_id in self._outboxes

    def list_agents(self):
        return sorted(self._outboxes)

    # -- inbound frames from the agent ---------------------------------
    def deliver(self, frame: dict):
        """Route an agent frame to the waiting turn by request_id."""
        rid = frame.get("request_id", "")
        q = self._responses.get(rid)
        if q is None:
            log.warning("frame for unknown request %s dropped", rid)
            return
        q.put_nowait(frame)

    # -- turn execution -------------------------------------------------
    async def run_turn(self, agent_id: str, payload: dict,
                       idle_timeout_s: Optional[float] = None
                       ) -> AsyncIterator[str]:
        """Send one chat command; yield content chunks until the agent
        signals done. Idle-reset timeout per chunk; hard max-wait cap."""
        outbox = self._outboxes.get(agent_id)
        if outbox is None:
            raise ExternalAgentError(
                f"external agent {agent_id!r} is not connected")
        rid = new_id("xreq")
        q: asyncio.Queue = asyncio.Queue()
        self._responses[rid] = q
        idle = idle_timeout_s or self.idle_timeout_s
        t0 = time.monotonic()
        try:
            await outbox.put({"type": "chat_message",
                              "request_id": rid, **payload})
            while True:
                if time.monotonic() - t0 > self.max_wait_s:
                    raise ExternalAgentError(
                        "external agent exceeded the hard max wait")
                try:
                    frame = await asyncio.wait_for(q.get(), idle)
                except asyncio.TimeoutError:
                    raise ExternalAgentError(
                        f"external agent idle for {idle:.0f}s")
                ftype = frame.get("type")
                if ftype == "delta":
                    chunk = frame.get("content", "")
                    if chunk:
                        yield chunk
                elif ftype == "done":
                    return
                elif ftype == "error":
                    raise ExternalAgentError(
                        frame.get("message", "external agent error"))
                # any other frame type (keepalive/status) resets idle
        finally:
            self._responses.pop(rid, None)

    async def run_turn_blocking(self, agent_id: str, payload: dict,
                                idle_timeout_s: Optional[float] = None
                                ) -> str:
        parts = []
        async for chunk in self.run_turn(agent_id, payload,
                                         idle_timeout_s):
            parts.append(chunk)
        return "".join(parts)
