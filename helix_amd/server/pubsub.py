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


# ---------------------------------------------------------------------------
# Durable streams — the JetStream role of the reference's embedded NATS
# (nats.go:256-349: CreateOrUpdateStream + durable consumers with ack
# deadlines and redelivery backoff). Messages persist in the store, so
# consumers survive process restarts and delivery is at-least-once:
# fetched messages are leased for ack_wait_s, unacked leases expire and
# the message is redelivered.

import time as _time


class StreamBus:
    def __init__(self, store, pubsub: "PubSub" = None,
                 ack_wait_s: float = 30.0):
        import threading as _threading
        self.store = store
        self.pubsub = pubsub
        self.ack_wait_s = ack_wait_s
        self._seq_lock = _threading.Lock()

    # -- publish ------------------------------------------------------------
    def publish(self, stream: str, subject: str, payload: Any) -> int:
        """Append to the stream; returns the sequence number."""
        meta_id = f"{stream}:@meta"
        with self._seq_lock:
            meta = self.store.get("bus_consumers", meta_id) or \
                {"id": meta_id, "next_seq": 1}
            seq = meta["next_seq"]
            meta["next_seq"] = seq + 1
            self.store.put("bus_consumers", meta_id, meta)
        self.store.put("bus_messages", f"{stream}:{seq:012d}",
                       {"id": f"{stream}:{seq:012d}", "seq": seq,
                        "subject": subject, "payload": payload,
                        "ts": _time.time()}, parent=stream)
        return seq

    async def publish_notify(self, stream: str, subject: str,
                             payload: Any) -> int:
        """publish + wake in-proc subscribers of `stream.subject` so
        local consumers need not poll."""
        seq = self.publish(stream, subject, payload)
        if self.pubsub is not None:
            await self.pubsub.publish(f"{stream}.{subject}",
                                      {"seq": seq})
        return seq

    # -- durable consumers ----------------------------------------------------
    def _consumer_doc(self, stream: str, durable: str) -> dict:
        cid = f"{stream}:{durable}"
        return self.store.get("bus_consumers", cid) or \
            {"id": cid, "floor": 0, "acked": [], "inflight": {}}

    def fetch(self, stream: str, durable: str, batch: int = 10,
              subject_filter: str = "*") -> List[dict]:
        """Lease up to `batch` undelivered-or-expired messages."""
        doc = self._consumer_doc(stream, durable)
        now = _time.time()
        acked = set(doc.get("acked", []))
        floor = doc.get("floor", 0)
        inflight = {int(k): v for k, v in doc.get("inflight", {}).items()}
        out = []
        rows = self.store.list("bus_messages", parent=stream,
                               limit=100000, desc=False)
        for row in rows:
            if len(out) >= batch:
                break
            seq = row["seq"]
            if seq <= floor or seq in acked:
                continue
            if inflight.get(seq, 0) > now:
                continue                     # leased to another fetcher
            if not fnmatch.fnmatch(row.get("subject", ""),
                                   subject_filter):
                continue
            inflight[seq] = now + self.ack_wait_s
            out.append({"seq": seq, "subject": row["subject"],
                        "payload": row["payload"], "ts": row["ts"]})
        doc["inflight"] = {str(k): v for k, v in inflight.items()}
        self.store.put("bus_consumers", doc["id"], doc)
        return out

    def ack(self, stream: str, durable: str, seq: int):
        doc = self._consumer_doc(stream, durable)
        doc["inflight"].pop(str(seq), None)
        acked = set(doc.get("acked", []))
        acked.add(seq)
        floor = doc.get("floor", 0)
        # compact: advance the floor over contiguous acks
        while floor + 1 in acked:
            floor += 1
            acked.discard(floor)
        doc["floor"] = floor
        doc["acked"] = sorted(acked)
        self.store.put("bus_consumers", doc["id"], doc)

    def pending(self, stream: str, durable: str) -> int:
        doc = self._consumer_doc(stream, durable)
        acked = set(doc.get("acked", []))
        floor = doc.get("floor", 0)
        n = 0
        for row in self.store.list("bus_messages", parent=stream,
                                   limit=100000, desc=False):
            if row["seq"] > floor and row["seq"] not in acked:
                n += 1
        return n

    def purge(self, stream: str, keep_last: int = 10000):
        """Bound stream growth (JetStream retention role)."""
        rows = self.store.list("bus_messages", parent=stream,
                               limit=1000000, desc=False)
        for row in rows[:-keep_last] if keep_last else rows:
            self.store.delete("bus_messages", row["id"])
