"""Runner profile-assignment poller (parity with api/cmd/compose-manager:
poll GET /api/v1/runner/{id}/assignment every 15 s and reconcile loaded
models — here the 'profile' is a model manifest, not compose YAML)."""
from __future__ import annotations

import asyncio
import logging

import httpx

from helix_amd.runner.service import ModelSpec, RunnerService

log = logging.getLogger("helix_amd.runner.assignment")


def apply_profile(service: RunnerService, profile: dict) -> dict:
    """Reconcile loaded models with the assigned profile's manifest."""
    wanted = {}
    for m in profile.get("models", []):
        spec = ModelSpec(
            name=m.get("name", ""),
            kind=m.get("kind", "llm"),
            preset=m.get("preset", m.get("name", "")),
            max_model_len=int(m.get("max_model_len", 8192)),
            max_num_seqs=int(m.get("max_num_seqs", 64)),
            kv_cache_blocks=m.get("kv_cache_blocks"),
            tp=int(m.get("tp", 1)),
            quantization=m.get("quantization"),
            kv_cache_dtype=m.get("kv_cache_dtype", "bf16"),
        )
        wanted[spec.name] = spec
        service.specs[spec.name] = spec
    loaded = set(service.loaded_models())
    result = {"loaded": [], "unloaded": [], "failed": {}}
    # unload models not in the profile
    for name in loaded - set(wanted):
        service.unload(name)
        result["unloaded"].append(name)
    # load missing ones
    for name in wanted:
        if name in loaded:
            continue
        try:
            service.ensure_loaded(name)
            result["loaded"].append(name)
        except Exception as e:
            log.warning("failed to load %s: %s", name, e)
            result["failed"][name] = str(e)
    return result


async def assignment_loop(api_url: str, runner_token: str, runner_id: str,
                          service: RunnerService, interval: float = 15.0,
                          stop_event=None):
    last_profile_id = None
    async with httpx.AsyncClient(timeout=30) as http:
        while stop_event is None or not stop_event.is_set():
            try:
                r = await http.get(
                    f"{api_url}/api/v1/runner/{runner_id}/assignment",
                    headers={"Authorization": f"Bearer {runner_token}"})
                if r.status_code == 200:
                    profile = r.json()
                    pid = profile.get("id")
                    if pid != last_profile_id:
                        log.info("applying profile %s", pid)
                        res = apply_profile(service, profile or {})
                        log.info("profile applied: %s", res)
                        last_profile_id = pid
            except Exception as e:
                log.warning("assignment poll failed: %s", e)
            try:
                if stop_event is not None:
                    await asyncio.wait_for(stop_event.wait(), interval)
                else:
                    await asyncio.sleep(interval)
            except asyncio.TimeoutError:
                pass
