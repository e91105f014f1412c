#!/usr/bin/env python3
"""GPU validation harness — the reference's gpucloud integration
scenarios (integration-test/gpucloud/README.md:49-56) adapted to this
stack: boots a real control plane + a real GPU runner as subprocesses
on localhost and drives the seven scenarios end-to-end over HTTP.

    python scripts/gpu_validate.py [--device cuda:0] [--small]

Prints one JSON line per scenario and a final summary; exit code 0 only
if every scenario passed. Runs on an MI355X box (gpurun) in ~2-4 min;
--small uses the tiny test preset so a smoke pass also works on slower
boxes.
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import time

import httpx

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
API = "http://127.0.0.1:18080"
ADMIN = {"Authorization": "Bearer admin-key"}
RUNNER_ID = "gpuval-runner"

RESULTS = []


def scenario(name):
    def deco(fn):
        def wrapper(*a, **kw):
            t0 = time.time()
            try:
                fn(*a, **kw)
                ok, err = True, ""
            except Exception as e:
                ok, err = False, f"{type(e).__name__}: {e}"
            rec = {"scenario": name, "ok": ok,
                   "seconds": round(time.time() - t0, 1), "error": err}
            RESULTS.append(rec)
            print(json.dumps(rec), flush=True)
            return ok
        return wrapper
    return deco


def wait_for(pred, timeout, what, interval=1.0):
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            if pred():
                return
        except Exception:
            pass
        time.sleep(interval)
    raise TimeoutError(f"timed out waiting for {what} ({timeout}s)")


def runner_state(http):
    r = http.get(API + "/api/v1/admin/runners", headers=ADMIN)
    r.raise_for_status()
    for it in r.json():
        if it.get("runner_id") == RUNNER_ID:
            return it
    return None


# -- scenarios (gpucloud README:49-56) ---------------------------------------

@scenario("boot_smoke")
def s1_boot_smoke(http):
    """Runner connects, heartbeat lands, GPU inventory matches."""
    wait_for(lambda: runner_state(http) is not None, 90,
             "runner heartbeat")
    st = runner_state(http)
    gpus = st.get("gpus", [])
    assert gpus, "no GPUs in heartbeat"
    assert gpus[0].get("vendor") == "amd", gpus[0]
    assert gpus[0].get("arch") == "cdna4", gpus[0]
    assert gpus[0].get("total_memory", 0) > 200e9, \
        f"expected ~288GB HBM, got {gpus[0].get('total_memory')}"


@scenario("compatibility_filter")
def s2_compat(http, profile_id):
    r = http.get(API + f"/api/v1/runners/{RUNNER_ID}/compatible-profiles",
                 headers=ADMIN)
    r.raise_for_status()
    ids = [p["id"] for p in r.json()]
    assert profile_id in ids, f"{profile_id} not in {ids}"


@scenario("assignment_apply")
def s3_assign(http, profile_id, model):
    r = http.post(API + f"/api/v1/runners/{RUNNER_ID}/assign-profile",
                  headers=ADMIN, json={"profile_id": profile_id})
    r.raise_for_status()

    def model_ready():
        st = runner_state(http)
        return st and any(m.get("model_id") == model and
                          m.get("state") == "ready"
                          for m in st.get("models", []))
    wait_for(model_ready, 240, f"model {model} ready")


@scenario("inference_roundtrip")
def s4_inference(http, model):
    r = http.post(API + "/v1/chat/completions", headers=ADMIN, json={
        "model": model,
        "messages": [{"role": "user", "content": "Say hello."}],
        "max_tokens": 16}, timeout=180)
    r.raise_for_status()
    out = r.json()
    # random-init weights at full vocab can sample ids the byte-level
    # tokenizer decodes to '' — assert the serving contract (structure
    # + generated tokens), not linguistic output
    msg = out["choices"][0]["message"]
    assert msg["role"] == "assistant" and isinstance(
        msg.get("content", ""), str)
    assert out.get("usage", {}).get("completion_tokens", 0) > 0, out
    r = http.post(API + "/v1/embeddings", headers=ADMIN, json={
        "model": "bge-base", "input": ["hello world"]}, timeout=180)
    r.raise_for_status()
    emb = r.json()["data"][0]["embedding"]
    assert len(emb) >= 128


@scenario("profile_switch")
def s5_switch(http, profile2_id, model2):
    r = http.post(API + f"/api/v1/runners/{RUNNER_ID}/assign-profile",
                  headers=ADMIN, json={"profile_id": profile2_id})
    r.raise_for_status()

    def swapped():
        st = runner_state(http)
        models = {m.get("model_id"): m.get("state")
                  for m in (st or {}).get("models", [])}
        return models.get(model2) == "ready"
    wait_for(swapped, 240, f"swap to {model2}")


@scenario("clear_profile")
def s6_clear(http, old_model):
    r = http.delete(API + f"/api/v1/runners/{RUNNER_ID}/assignment",
                    headers=ADMIN)
    r.raise_for_status()

    def idle():
        st = runner_state(http)
        return all(m.get("model_id") != old_model
                   for m in (st or {}).get("models", []))
    wait_for(idle, 120, "old model unloaded")


@scenario("incompatible_rejection")
def s7_incompatible(http):
    r = http.post(API + "/api/v1/runner-profiles", headers=ADMIN, json={
        "name": "cuda-only",
        "models": [{"name": "llama3-8b"}],
        "gpu_requirement": {"vendor": "nvidia",
                            "architectures": ["hopper"]}})
    r.raise_for_status()
    pid = r.json()["id"]
    r = http.post(API + f"/api/v1/runners/{RUNNER_ID}/assign-profile",
                  headers=ADMIN, json={"profile_id": pid})
    assert r.status_code in (409, 422), \
        f"expected rejection, got {r.status_code}"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--small", action="store_true",
                    help="tiny presets for smoke runs")
    args = ap.parse_args()

    model1 = "tiny" if args.small else "llama3-8b"
    model2 = "tiny-gqa" if args.small else "mistral-7b"

    env = dict(os.environ,
               HELIX_STORE_PATH="/tmp/gpuval.db",
               HELIX_FILESTORE_PATH="/tmp/gpuval-fs",
               SERVER_PORT="18080",
               PYTHONPATH=ROOT)
    for f in ("/tmp/gpuval.db", "/tmp/gpuval.db-wal",
              "/tmp/gpuval.db-shm"):
        if os.path.exists(f):
            os.unlink(f)
    serve = subprocess.Popen(
        [sys.executable, "-m", "helix_amd.cli", "serve"],
        env=env, cwd=ROOT, start_new_session=True)
    runner = None
    try:
        http = httpx.Client(timeout=60)
        wait_for(lambda: http.get(API + "/healthz").status_code == 200,
                 60, "control plane")
        runner = subprocess.Popen(
            [sys.executable, "-m", "helix_amd.cli", "runner",
             "--api-url", API, "--runner-id", RUNNER_ID,
             "--port", "18090", "--device", args.device,
             "--tunnel"],
            env=env, cwd=ROOT, start_new_session=True)

        # two compatible profiles + the scenarios
        def mk_profile(name, model):
            r = http.post(API + "/api/v1/runner-profiles",
                          headers=ADMIN, json={
                              "name": name,
                              "models": [{"name": model,
                                          "max_num_seqs": 16,
                                          "max_model_len": 2048},
                                         {"name": "bge-base",
                                          "kind": "embedding"}],
                              "gpu_requirement": {
                                  "vendor": "amd",
                                  "architectures": ["cdna4"]}})
            r.raise_for_status()
            return r.json()["id"]

        p1 = mk_profile("amd-primary", model1)
        p2 = mk_profile("amd-secondary", model2)

        s1_boot_smoke(http)
        s2_compat(http, p1)
        s3_assign(http, p1, model1)
        s4_inference(http, model1)
        s5_switch(http, p2, model2)
        s6_clear(http, model1)
        s7_incompatible(http)
    finally:
        for proc in (runner, serve):
            if proc is not None:
                try:
                    os.killpg(proc.pid, signal.SIGTERM)
                except (ProcessLookupError, PermissionError):
                    pass
        time.sleep(2)
        for proc in (runner, serve):
            if proc is not None and proc.poll() is None:
                try:
                    os.killpg(proc.pid, signal.SIGKILL)
                except (ProcessLookupError, PermissionError):
                    pass

    passed = sum(1 for r in RESULTS if r["ok"])
    summary = {"summary": True, "passed": passed,
               "total": len(RESULTS),
               "ok": passed == len(RESULTS) and len(RESULTS) == 7}
    print(json.dumps(summary), flush=True)
    sys.exit(0 if summary["ok"] else 1)


if __name__ == "__main__":
    main()
