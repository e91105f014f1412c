"""Runner heartbeat loop — pushes GPU + model status to the control
plane every interval (parity with api/cmd/sandbox-heartbeat: 30 s beat
with gpudetect inventory)."""
from __future__ import annotations

import asyncio
import logging

import httpx

from helix_amd.runner import gpudetect

log = logging.getLogger("helix_amd.runner.heartbeat")


async def heartbeat_loop(api_url: str, runner_token: str, runner_id: str,
                         advertise_addr: str, service,
                         interval: float = 30.0, stop_event=None):
    async with httpx.AsyncClient(timeout=15) as http:
        while stop_event is None or not stop_event.is_set():
            try:
                payload = {
                    "runner_id": runner_id,
                    "address": advertise_addr,
                    "gpus": [g.model_dump() for g in gpudetect.detect()],
                    "models": [
                        {"model_id": m["model_id"], "state": m["state"],
                         "memory_bytes": m["memory_bytes"],
                         "last_used": m["last_used"]}
                        for m in service.status()],
                }
                r = await http.post(
                    f"{api_url}/api/v1/runner/heartbeat", json=payload,
                    headers={"Authorization": f"Bearer {runner_token}"})
                if r.status_code != 200:
                    log.warning("heartbeat rejected: %s %s", r.status_code,
                                r.text[:200])
            except Exception as e:
                log.warning("heartbeat failed: %s", e)
            try:
                if stop_event is not None:
                    await asyncio.wait_for(stop_event.wait(), interval)
                else:
                    await asyncio.sleep(interval)
            except asyncio.TimeoutError:
                pass
