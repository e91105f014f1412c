"""End-to-end smoke over real sockets: control plane (uvicorn) + runner
(tiny CPU engine) + heartbeat -> router -> dispatch -> SSE back — the
reference's gpucloud inference_roundtrip / boot_smoke scenarios at unit
scale, plus the split-plane topology."""
import json
import socket
import threading
import time

import httpx
import pytest
import uvicorn

from helix_amd.runner.http import create_runner_app
from helix_amd.runner.service import RunnerService
from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.providers import ProviderManager, RouterClient
from helix_amd.store import Store


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


class ServerThread:
    def __init__(self, app, port):
        self.config = uvicorn.Config(app, host="127.0.0.1", port=port,
                                     log_level="error")
        self.server = uvicorn.Server(self.config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    def start(self):
        self.thread.start()
        for _ in range(100):
            if self.server.started:
                return
            time.sleep(0.05)
        raise RuntimeError("server failed to start")

    def stop(self):
        self.server.should_exit = True
        self.thread.join(timeout=10)


@pytest.fixture(scope="module")
def e2e():
    cp_port, rn_port = free_port(), free_port()
    cfg = ServerConfig()
    cfg.inference.default_provider = "helix"
    cfg.inference.default_model = "tiny"
    store = Store(":memory:")
    cp_app = create_app(cfg, store=store)
    svc = RunnerService(device="cpu")
    rn_app = create_runner_app(svc, "runner-e2e")
    cp = ServerThread(cp_app, cp_port)
    rn = ServerThread(rn_app, rn_port)
    cp.start()
    rn.start()
    base = f"http://127.0.0.1:{cp_port}"
    # runner heartbeat announcing the tiny model (pre-loaded)
    svc.ensure_loaded("tiny")
    hb = {"runner_id": "runner-e2e",
          "address": f"http://127.0.0.1:{rn_port}",
          "gpus": [], "models": [{"model_id": "tiny", "state": "ready"}]}
    httpx.post(f"{base}/api/v1/runner/heartbeat", json=hb,
               headers={"Authorization": "Bearer runner-token"})
    r = httpx.post(f"{base}/api/v1/users", json={"username": "e2e"},
                   headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    yield base, key, svc
    cp.stop()
    rn.stop()
    svc.shutdown()


def test_e2e_chat_roundtrip(e2e):
    base, key, _ = e2e
    r = httpx.post(f"{base}/v1/chat/completions", json={
        "model": "tiny",
        "messages": [{"role": "user", "content": "hello e2e"}],
        "max_tokens": 6, "temperature": 0}, timeout=120,
        headers={"Authorization": f"Bearer {key}"})
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["usage"]["completion_tokens"] >= 1
    assert body["choices"][0]["message"]["role"] == "assistant"


def test_e2e_streaming(e2e):
    base, key, _ = e2e
    with httpx.stream("POST", f"{base}/v1/chat/completions", json={
        "model": "tiny", "stream": True, "max_tokens": 5, "temperature": 0,
        "messages": [{"role": "user", "content": "stream me"}]},
            timeout=120,
            headers={"Authorization": f"Bearer {key}"}) as r:
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    assert len(lines) >= 3


def test_e2e_embeddings_via_router(e2e):
    base, key, svc = e2e
    svc.ensure_loaded("tiny-bert")
    hb = {"runner_id": "runner-e2e",
          "address": svc and f"http://127.0.0.1:{0}",  # placeholder
          "gpus": [], "models": [
              {"model_id": "tiny", "state": "ready"},
              {"model_id": "tiny-bert", "state": "ready"}]}
    # re-send heartbeat with the right address (reuse stored runner addr)
    r0 = httpx.get(f"{base}/api/v1/admin/runners",
                   headers={"Authorization": "Bearer admin-key"}).json()
    hb["address"] = r0[0]["address"]
    httpx.post(f"{base}/api/v1/runner/heartbeat", json=hb,
               headers={"Authorization": "Bearer runner-token"})
    r = httpx.post(f"{base}/v1/embeddings", json={
        "model": "tiny-bert", "input": "embed this"}, timeout=120,
        headers={"Authorization": f"Bearer {key}"})
    assert r.status_code == 200, r.text
    assert len(r.json()["data"][0]["embedding"]) == 128


def test_e2e_images_via_router(e2e):
    base, key, svc = e2e
    svc.ensure_loaded("tiny-dit")
    r0 = httpx.get(f"{base}/api/v1/admin/runners",
                   headers={"Authorization": "Bearer admin-key"}).json()
    hb = {"runner_id": "runner-e2e", "address": r0[0]["address"],
          "gpus": [], "models": [
              {"model_id": "tiny", "state": "ready"},
              {"model_id": "tiny-dit", "state": "ready"}]}
    httpx.post(f"{base}/api/v1/runner/heartbeat", json=hb,
               headers={"Authorization": "Bearer runner-token"})
    r = httpx.post(f"{base}/v1/images/generations", json={
        "model": "tiny-dit", "prompt": "a blue circle", "steps": 3,
        "seed": 5}, timeout=120,
        headers={"Authorization": f"Bearer {key}"})
    assert r.status_code == 200, r.text
    import base64
    import io

    from PIL import Image
    img = Image.open(io.BytesIO(
        base64.b64decode(r.json()["data"][0]["b64_json"])))
    assert img.size == (32, 32)


def test_e2e_models_endpoint(e2e):
    base, key, _ = e2e
    r = httpx.get(f"{base}/v1/models",
                  headers={"Authorization": f"Bearer {key}"})
    ids = [m["id"] for m in r.json()["data"]]
    assert "tiny" in ids


def test_e2e_cli_test_verb(e2e, tmp_path):
    """`helix-amd test -f helix.yaml` runs assistant tests against the
    live stack with the LLM judge (reference `helix test`)."""
    import os
    import subprocess
    import sys
    base, key, _ = e2e
    yaml_path = tmp_path / "app.yaml"
    yaml_path.write_text("""
name: cli-test-app
assistants:
  - name: default
    model: tiny
    tests:
      - name: replies
        steps:
          - prompt: "say anything"
            expected_output: "any reply at all is accepted by the judge"
""")
    env = dict(os.environ, HELIX_URL=base, HELIX_API_KEY=key)
    r = subprocess.run(
        [sys.executable, "-m", "helix_amd.cli", "test", "-f",
         str(yaml_path), "--url", base],
        capture_output=True, text=True, timeout=300, env=env)
    # tiny random-init model judges arbitrarily: PASS or FAIL both prove
    # the loop ran end-to-end; crash/timeout would be a real failure
    assert r.returncode in (0, 1), r.stderr
    assert "replies" in r.stdout
    assert ("PASS" in r.stdout) or ("FAIL" in r.stdout)
