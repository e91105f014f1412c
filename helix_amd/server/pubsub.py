"""In-process async pub/sub — the role of the reference's embedded NATS
(SURVEY.md §2.1 PubSub): session event streams, inference request/response
queues. Topic helpers mirror pubsub.go:74,91.
"""
from __future__ import annotations

import asyncio
import fnmatch
from collections import defaultdict
from typing import Any, AsyncIterator, Dict, List, Tuple


class PubSub:
    def __init__(self):
        self._subs: Dict[str, List[asyncio.Queue]] = defaultdict(list)
        self._lock = asyncio.Lock()

    async def publish(self, topic: str, msg: Any):
        for pattern, queues in list(self._subs.items()):
            if fnmatch.fnmatch(topic, pattern):
                for q in list(queues):
                    # slow-consumer policy (NATS-like): drop the OLDEST
                    # message rather than grow without bound or block
                    # every publisher behind one dead SSE client
                    while True:
                        try:
                            q.put_nowait((topic, msg))
                            break
                        except asyncio.QueueFull:
                            try:
                                q.get_nowait()
                            except asyncio.QueueEmpty:
                                break

    async def subscribe(self, pattern: str) -> "Subscription":
        q: asyncio.Queue = asyncio.Queue(maxsize=1024)
        async with self._lock:
            self._subs[pattern].append(q)
        return Subscription(self, pattern, q)

    async def _unsubscribe(self, pattern: str, q: asyncio.Queue):
        async with self._lock:
            if q in self._subs.get(pattern, []):
                self._subs[pattern].remove(q)
            if not self._subs.get(pattern):
                self._subs.pop(pattern, None)

    async def request(self, topic: str, msg: Any, timeout: float = 30.0) -> Any:
        """RPC over pub/sub: publish to `topic`, await one reply."""
        reply_topic = f"_reply.{id(msg)}.{asyncio.get_event_loop().time()}"
        sub = await self.subscribe(reply_topic)
        try:
            await self.publish(topic, {"reply_to": reply_topic, "data": msg})
            _, reply = await asyncio.wait_for(sub.get(), timeout)
            return reply
        finally:
            await sub.close()


class Subscription:
    def __init__(self, ps: PubSub, pattern: str, q: asyncio.Queue):
        self._ps = ps
        self.pattern = pattern
        self._q = q

    async def get(self, timeout: float | None = None) -> Tuple[str, Any]:
        if timeout is None:
            return await self._q.get()
        return await asyncio.wait_for(self._q.get(), timeout)

    async def stream(self) -> AsyncIterator[Tuple[str, Any]]:
        while True:
            yield await self._q.get()

    async def close(self):
        await self._ps._unsubscribe(self.pattern, self._q)


def session_queue(owner: str, session_id: str) -> str:
    """Session event stream topic (reference pubsub.go:74)."""
    return f"session.{owner}.{session_id}"


def runner_responses_queue(owner: str, request_id: str) -> str:
    """Inference response topic (reference pubsub.go:91)."""
    return f"runner.responses.{owner}.{request_id}"
