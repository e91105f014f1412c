"""Usage metering, quotas and wallets (parity with the reference's
UsageLogger -> usage_metrics, api/pkg/quota token quotas and
wallet/transactions, controller/balance_check.go)."""
from __future__ import annotations

import time
from typing import List, Optional

from helix_amd.server.types import LLMCall, new_id


class QuotaExceededError(Exception):
    pass


class UsageService:
    def __init__(self, store, catalog=None, metrics=None):
        self.store = store
        self.catalog = catalog
        self.metrics = metrics

    # -- metering ----------------------------------------------------------
    def log_call(self, call: LLMCall):
        """Per-call metric row + daily rollup per (owner, provider, model)."""
        if self.metrics is not None:
            self.metrics.observe_call(call)
        day = time.strftime("%Y-%m-%d", time.gmtime())
        mid = new_id("use")
        cost = 0.0
        if self.catalog is not None:
            cost = self.catalog.cost_usd(call.model, call.prompt_tokens,
                                         call.completion_tokens)
        self.store.put("usage_metrics", mid, {
            "id": mid, "owner": call.owner, "provider": call.provider,
            "model": call.model, "session_id": call.session_id,
            "prompt_tokens": call.prompt_tokens,
            "completion_tokens": call.completion_tokens,
            "duration_ms": call.duration_ms, "cost_usd": cost, "day": day,
            "ts": time.time()}, owner=call.owner, parent=call.session_id,
            buffered=True)
        rid = f"{call.owner}:{call.provider}:{call.model}:{day}"
        roll = self.store.get("usage_rollups", rid) or {
            "id": rid, "owner": call.owner, "provider": call.provider,
            "model": call.model, "day": day, "prompt_tokens": 0,
            "completion_tokens": 0, "calls": 0, "cost_usd": 0.0}
        roll["prompt_tokens"] += call.prompt_tokens
        roll["completion_tokens"] += call.completion_tokens
        roll["calls"] += 1
        roll["cost_usd"] += cost
        self.store.put("usage_rollups", rid, roll, owner=call.owner)
        if cost > 0:
            self.debit(call.owner, cost, f"llm:{call.model}")

    def usage_for(self, owner: str, day: Optional[str] = None) -> List[dict]:
        rows = self.store.list("usage_rollups", owner=owner, limit=10000)
        if day:
            rows = [r for r in rows if r.get("day") == day]
        return rows

    # -- quotas (per-provider daily token caps) ----------------------------
    def check_quota(self, owner: str, provider: str,
                    daily_token_limit: int = 0):
        if daily_token_limit <= 0:
            return
        day = time.strftime("%Y-%m-%d", time.gmtime())
        used = sum(r["prompt_tokens"] + r["completion_tokens"]
                   for r in self.usage_for(owner, day)
                   if r.get("provider") == provider)
        if used >= daily_token_limit:
            raise QuotaExceededError(
                f"daily token quota exceeded for {provider}: "
                f"{used}/{daily_token_limit}")

    # -- wallets -----------------------------------------------------------
    def wallet(self, owner: str) -> dict:
        w = self.store.get("wallets", owner)
        if w is None:
            w = {"id": owner, "balance_usd": 0.0}
            self.store.put("wallets", owner, w, owner=owner)
        return w

    def topup(self, owner: str, amount_usd: float, ref: str = "manual"):
        w = self.wallet(owner)
        w["balance_usd"] += amount_usd
        self.store.put("wallets", owner, w, owner=owner)
        tid = new_id("txn")
        self.store.put("transactions", tid, {
            "id": tid, "owner": owner, "amount_usd": amount_usd,
            "kind": "topup", "ref": ref, "ts": time.time()}, owner=owner)
        return w

    def debit(self, owner: str, amount_usd: float, ref: str):
        w = self.wallet(owner)
        w["balance_usd"] -= amount_usd
        self.store.put("wallets", owner, w, owner=owner)
        tid = new_id("txn")
        self.store.put("transactions", tid, {
            "id": tid, "owner": owner, "amount_usd": -amount_usd,
            "kind": "usage", "ref": ref, "ts": time.time()}, owner=owner)
        return w
