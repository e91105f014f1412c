"""Inference router — parity with api/pkg/inferencerouter (router.go):
in-memory model->runner table fed by heartbeats, per-model round-robin,
NoRunnerError carrying the available-model list (surfaced as 503).
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional

from helix_amd.server.types import RunnerHeartbeat, RunnerState


class NoRunnerError(Exception):
    def __init__(self, model: str, available: List[str]):
        super().__init__(
            f"no runner is serving model '{model}'; available: {available}")
        self.model = model
        self.available = available


class InferenceRouter:
    def __init__(self, dispatch_stale_s: float = 90.0,
                 offline_after_s: float = 300.0):
        self._runners: Dict[str, RunnerState] = {}
        self._rr: Dict[str, int] = {}
        self._lock = threading.RLock()
        self.dispatch_stale_s = dispatch_stale_s
        self.offline_after_s = offline_after_s

    # -- heartbeat ingestion (reference runner_assignment_handlers.go:28) --
    def on_heartbeat(self, hb: RunnerHeartbeat):
        with self._lock:
            self._runners[hb.runner_id] = RunnerState(
                runner_id=hb.runner_id, address=hb.address, status="online",
                gpus=hb.gpus, models=hb.models, last_seen=time.time())

    def remove_runner(self, runner_id: str):
        with self._lock:
            self._runners.pop(runner_id, None)

    def reap_offline(self):
        """Flip stale runners offline / drop them (reference reaper)."""
        now = time.time()
        with self._lock:
            for rid in list(self._runners):
                if now - self._runners[rid].last_seen > self.offline_after_s:
                    del self._runners[rid]

    # -- queries ----------------------------------------------------------
    def _fresh(self) -> List[RunnerState]:
        now = time.time()
        return [r for r in self._runners.values()
                if now - r.last_seen <= self.dispatch_stale_s]

    def runners(self) -> List[RunnerState]:
        with self._lock:
            return list(self._runners.values())

    def available_models(self) -> List[str]:
        with self._lock:
            models = set()
            for r in self._fresh():
                for m in r.models:
                    if m.state == "ready":
                        models.add(m.model_id)
            return sorted(models)

    def pick_runner(self, model: str) -> str:
        """Per-model round-robin over fresh runners serving it
        (reference router.go:168-198). Returns the runner address."""
        with self._lock:
            cands = [r for r in self._fresh()
                     if any(m.model_id == model and m.state == "ready"
                            for m in r.models)]
            if not cands:
                raise NoRunnerError(model, self.available_models())
            cands.sort(key=lambda r: r.runner_id)
            i = self._rr.get(model, 0) % len(cands)
            self._rr[model] = i + 1
            return cands[i].address
