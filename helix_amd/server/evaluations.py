"""Evaluation suites/runs (parity with the reference's app evaluation
routes, agent_routes.go: suites CRUD, runs with LLM-judged steps, run
streaming): each suite holds helix.yaml-style multi-turn tests; a run
executes them against the app's assistant and judges with an LLM."""
from __future__ import annotations

import logging
import time
from typing import List, Optional

from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.evaluations")


class EvaluationService:
    def __init__(self, store, controller, pubsub=None):
        self.store = store
        self.controller = controller
        self.pubsub = pubsub

    # -- suites -----------------------------------------------------------
    def create_suite(self, owner: str, app_id: str, name: str,
                     tests: List[dict]) -> dict:
        sid = new_id("evs")
        doc = {"id": sid, "owner": owner, "app_id": app_id, "name": name,
               "tests": tests, "created": time.time()}
        self.store.put("evaluation_runs", f"suite:{sid}", doc, owner=owner,
                       parent=app_id)
        return doc

    def get_suite(self, sid: str) -> Optional[dict]:
        return self.store.get("evaluation_runs", f"suite:{sid}")

    def list_suites(self, owner: str) -> List[dict]:
        return [d for d in self.store.list("evaluation_runs", owner=owner,
                                           limit=10000)
                if d["id"].startswith("evs")]

    # -- runs -------------------------------------------------------------
    async def run_suite(self, sid: str, judge_model: str = "") -> dict:
        suite = self.get_suite(sid)
        if suite is None:
            raise KeyError(sid)
        rid = new_id("evr")
        run = {"id": rid, "suite_id": sid, "owner": suite["owner"],
               "app_id": suite["app_id"], "state": "running",
               "results": [], "started": time.time()}
        self._save_run(run)
        passed = 0
        for t in suite["tests"]:
            for step in t.get("steps", []):
                prompt = step.get("prompt", "")
                expected = step.get("expected_output", "")
                try:
                    resp = await self.controller.chat_completion(
                        {"messages": [{"role": "user", "content": prompt}]},
                        suite["owner"], app_id=suite["app_id"],
                        ctx={"owner": suite["owner"], "step": "eval"})
                    answer = resp["choices"][0]["message"]["content"]
                    verdict = await self._judge(prompt, expected, answer,
                                                suite, judge_model)
                except Exception as e:
                    answer, verdict = "", f"error: {e}"
                ok = verdict.strip().upper().startswith("YES")
                passed += 1 if ok else 0
                run["results"].append({
                    "test": t.get("name", ""), "prompt": prompt,
                    "expected": expected, "answer": answer,
                    "verdict": verdict.strip()[:200], "passed": ok})
                self._save_run(run)
                await self._publish(run)
        total = len(run["results"])
        run["state"] = "complete"
        run["passed"] = passed
        run["total"] = total
        run["finished"] = time.time()
        self._save_run(run)
        await self._publish(run)
        return run

    async def _judge(self, prompt, expected, answer, suite,
                     judge_model) -> str:
        resp = await self.controller.chat_completion(
            {"model": judge_model or None,
             "messages": [{
                 "role": "user",
                 "content": ("You are a test judge. Question: "
                             f"{prompt}\nExpectation: {expected}\n"
                             f"Answer: {answer}\nDoes the answer satisfy "
                             "the expectation? Reply YES or NO with a "
                             "short reason.")}]},
            suite["owner"], ctx={"owner": suite["owner"],
                                 "step": "eval_judge"})
        return resp["choices"][0]["message"]["content"] or ""

    def _save_run(self, run: dict):
        self.store.put("evaluation_runs", run["id"], run,
                       owner=run["owner"], parent=run["suite_id"])

    async def _publish(self, run: dict):
        if self.pubsub is None:
            return
        from helix_amd.server import pubsub as ps
        await self.pubsub.publish(
            ps.session_queue(run["owner"], f"eval-{run['id']}"),
            {"type": "eval_run", "run": {k: run[k] for k in
                                         ("id", "state", "results")}})

    def get_run(self, rid: str) -> Optional[dict]:
        return self.store.get("evaluation_runs", rid)

    def list_runs(self, suite_id: str) -> List[dict]:
        return [d for d in self.store.list("evaluation_runs",
                                           parent=suite_id, limit=1000)]
