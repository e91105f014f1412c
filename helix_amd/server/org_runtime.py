"""Org runtime: bots, positions, roles, streams (parity with the
reference's "helix-org" graph — api/pkg/org, DDD runtime of
Bots/Workers/Positions/Roles/Streams; re-based here as a single
message-driven service on the document store).

Model:
- A *position* is a named seat in an org (e.g. "support engineer")
  bound to an agent/app configuration and a role.
- A *bot* occupies a position: it is the runnable agent instance.
- A *stream* is a message channel in the org. Bots subscribe to
  streams; each message posted to a stream fans out a session turn to
  every subscribed bot (the reference's worker loop), and the bot's
  reply is appended back to the stream.
- Org primitives are also exposed as MCP tools (mcp_gateway) so agents
  can inspect and post to their org.
"""
from __future__ import annotations

import logging
import time
from typing import List, Optional

from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.org")


class OrgRuntime:
    def __init__(self, store, controller):
        self.store = store
        self.controller = controller

    # ---------------- positions --------------------------------------
    def create_position(self, org_id: str, name: str, role: str = "worker",
                        app_id: str = "", system_prompt: str = "",
                        model: str = "") -> dict:
        pid = new_id("pos")
        doc = {"id": pid, "org_id": org_id, "name": name, "role": role,
               "app_id": app_id, "system_prompt": system_prompt,
               "model": model, "created": time.time()}
        self.store.put("org_positions", pid, doc, parent=org_id)
        return doc

    def list_positions(self, org_id: str) -> List[dict]:
        return self.store.list("org_positions", parent=org_id)

    def delete_position(self, pid: str) -> bool:
        return self.store.delete("org_positions", pid)

    # ---------------- bots -------------------------------------------
    def create_bot(self, org_id: str, name: str, position_id: str,
                   owner: str) -> dict:
        pos = self.store.get("org_positions", position_id)
        if pos is None or pos["org_id"] != org_id:
            raise ValueError("position not found in org")
        bid = new_id("bot")
        doc = {"id": bid, "org_id": org_id, "name": name, "owner": owner,
               "position_id": position_id, "state": "idle",
               "streams": [], "turns": 0, "created": time.time()}
        self.store.put("org_bots", bid, doc, owner=owner, parent=org_id)
        return doc

    def list_bots(self, org_id: str) -> List[dict]:
        return self.store.list("org_bots", parent=org_id)

    def get_bot(self, bot_id: str) -> Optional[dict]:
        return self.store.get("org_bots", bot_id)

    def delete_bot(self, bot_id: str) -> bool:
        return self.store.delete("org_bots", bot_id)

    def subscribe(self, bot_id: str, stream_id: str) -> dict:
        bot = self.store.get("org_bots", bot_id)
        if bot is None:
            raise ValueError("bot not found")
        if self.store.get("org_streams", stream_id) is None:
            raise ValueError("stream not found")
        if stream_id not in bot["streams"]:
            bot["streams"].append(stream_id)
        self.store.put("org_bots", bot_id, bot, owner=bot.get("owner"),
                       parent=bot["org_id"])
        return bot

    # ---------------- streams ----------------------------------------
    def create_stream(self, org_id: str, name: str,
                      topic: str = "") -> dict:
        sid = new_id("strm")
        doc = {"id": sid, "org_id": org_id, "name": name, "topic": topic,
               "created": time.time()}
        self.store.put("org_streams", sid, doc, parent=org_id)
        return doc

    def list_streams(self, org_id: str) -> List[dict]:
        return self.store.list("org_streams", parent=org_id)

    def stream_messages(self, stream_id: str, limit: int = 200) -> List[dict]:
        msgs = self.store.list("org_messages", parent=stream_id,
                               limit=limit)
        return sorted(msgs, key=lambda m: m.get("ts", 0))

    async def post_message(self, stream_id: str, sender: str, text: str,
                           fan_out: bool = True) -> dict:
        """Post to a stream; fan a session turn out to every subscribed
        bot (the reference's org worker loop) and append their replies."""
        stream = self.store.get("org_streams", stream_id)
        if stream is None:
            raise ValueError("stream not found")
        mid = new_id("msg")
        msg = {"id": mid, "stream_id": stream_id, "sender": sender,
               "text": text, "ts": time.time(), "replies": []}
        self.store.put("org_messages", mid, msg, parent=stream_id)
        if not fan_out:
            return msg
        for bot in self.list_bots(stream["org_id"]):
            if stream_id not in bot.get("streams", []):
                continue
            if bot.get("name") == sender:
                continue          # no self-replies (loop prevention)
            try:
                reply = await self._bot_turn(bot, stream, text, sender)
                msg["replies"].append({"bot": bot["name"],
                                       "text": reply})
            except Exception as e:
                log.exception("bot %s turn failed", bot["id"])
                msg["replies"].append({"bot": bot["name"],
                                       "text": f"error: {e}"})
        self.store.put("org_messages", mid, msg, parent=stream_id)
        return msg

    async def _bot_turn(self, bot: dict, stream: dict, text: str,
                        sender: str) -> str:
        t0 = time.time()
        try:
            reply = await self._bot_turn_inner(bot, stream, text, sender)
            self._audit(bot, stream.get("name", ""), sender,
                        int((time.time() - t0) * 1000), True)
            return reply
        except Exception as e:
            self._audit(bot, stream.get("name", ""), sender,
                        int((time.time() - t0) * 1000), False, str(e))
            raise

    async def _bot_turn_inner(self, bot: dict, stream: dict, text: str,
                              sender: str) -> str:
        pos = self.store.get("org_positions", bot["position_id"]) or {}
        prompt = (f"[stream {stream['name']}] message from {sender}: "
                  f"{text}")
        if pos.get("system_prompt"):
            prompt = f"{pos['system_prompt']}\n\n{prompt}"
        session = self.controller.create_session(
            bot.get("owner", ""), app_id=pos.get("app_id", ""),
            model=pos.get("model", ""),
            name=f"bot {bot['name']} @ {stream['name']}")
        interaction = self.controller.add_interaction(session, prompt)
        async for _ in self.controller.run_session_turn(
                session, interaction, stream_to_pubsub=False):
            pass
        bot["turns"] = bot.get("turns", 0) + 1
        bot["state"] = "idle"
        self.store.put("org_bots", bot["id"], bot,
                       owner=bot.get("owner"), parent=bot["org_id"])
        # store refetch: run_session_turn persists the interaction
        doc = self.store.get("interactions", interaction.id)
        return (doc or {}).get("response_message", "")


# ---------------------------------------------------------------------------
# Org-graph mechanics (reference QA.md mental model: Bots form a
# cycle-guarded DAG of reporting lines; activations are audited;
# messages can be escalated up the line).

class OrgGraphMixin:
    def set_reporting_lines(self, bot_id: str,
                            parent_ids: List[str]) -> dict:
        """Replace a bot's managers (org_reporting_lines role). The
        graph must stay a DAG — adding a line that closes a cycle is
        rejected (reference: cycle-guarded chart)."""
        bot = self.store.get("org_bots", bot_id)
        if bot is None:
            raise ValueError("bot not found")
        org = bot["org_id"]
        for pid in parent_ids:
            p = self.store.get("org_bots", pid)
            if p is None or p["org_id"] != org:
                raise ValueError(f"parent bot not in org: {pid}")
            if pid == bot_id:
                raise ValueError("a bot cannot report to itself")
        # cycle guard: walk up from each proposed parent; reaching
        # bot_id again means the new lines close a loop
        def ancestors(start: str, seen: set):
            if start in seen:
                return
            seen.add(start)
            doc = self.store.get("org_bots", start) or {}
            for pp in doc.get("parent_ids", []):
                ancestors(pp, seen)
        for pid in parent_ids:
            seen: set = set()
            ancestors(pid, seen)
            if bot_id in seen:
                raise ValueError(
                    f"reporting line {pid} -> {bot_id} closes a cycle")
        bot["parent_ids"] = list(dict.fromkeys(parent_ids))
        self.store.put("org_bots", bot_id, bot,
                       owner=bot.get("owner"), parent=org)
        return bot

    def chart(self, org_id: str) -> dict:
        """Chart-tab data: bots as nodes, reporting lines as edges."""
        bots = self.list_bots(org_id)
        ids = {b["id"] for b in bots}
        edges = []
        for b in bots:
            for pid in b.get("parent_ids", []):
                if pid in ids:
                    edges.append({"manager": pid, "report": b["id"]})
        return {"nodes": [{"id": b["id"], "name": b["name"],
                           "state": b.get("state", "idle"),
                           "turns": b.get("turns", 0)} for b in bots],
                "edges": edges}

    def delete_bot_cascade(self, bot_id: str) -> bool:
        """Bot deletion drops every reporting line referencing it
        (reference ON DELETE CASCADE)."""
        bot = self.store.get("org_bots", bot_id)
        if bot is None:
            return False
        for other in self.list_bots(bot["org_id"]):
            if bot_id in other.get("parent_ids", []):
                other["parent_ids"] = [p for p in other["parent_ids"]
                                       if p != bot_id]
                self.store.put("org_bots", other["id"], other,
                               owner=other.get("owner"),
                               parent=other["org_id"])
        return self.store.delete("org_bots", bot_id)

    async def escalate(self, bot_id: str, text: str) -> List[dict]:
        """Send a message up the reporting line: each manager gets an
        activation and the replies come back (reference escalation
        path in the worker loop)."""
        bot = self.store.get("org_bots", bot_id)
        if bot is None:
            raise ValueError("bot not found")
        out = []
        for pid in bot.get("parent_ids", []):
            mgr = self.store.get("org_bots", pid)
            if mgr is None:
                continue
            pseudo_stream = {"org_id": bot["org_id"],
                             "name": f"escalation from {bot['name']}"}
            try:
                reply = await self._bot_turn(mgr, pseudo_stream, text,
                                             bot["name"])
            except Exception as e:
                reply = f"error: {e}"
            out.append({"manager": mgr["name"], "reply": reply})
        return out

    def audit_log(self, org_id: str, limit: int = 100) -> List[dict]:
        rows = self.store.list("org_audit", parent=org_id, limit=limit)
        return sorted(rows, key=lambda r: -r.get("ts", 0))

    def _audit(self, bot: dict, trigger: str, sender: str,
               duration_ms: int, ok: bool, detail: str = ""):
        aid = new_id("orgaud")
        self.store.put("org_audit", aid, {
            "id": aid, "org_id": bot["org_id"], "bot_id": bot["id"],
            "bot_name": bot.get("name", ""), "trigger": trigger,
            "sender": sender, "duration_ms": duration_ms, "ok": ok,
            "detail": detail[:500], "ts": time.time()},
            parent=bot["org_id"], buffered=True)


for _name in ("set_reporting_lines", "chart", "delete_bot_cascade",
              "escalate", "audit_log", "_audit"):
    setattr(OrgRuntime, _name, getattr(OrgGraphMixin, _name))
