"""Reverse-tunnel e2e over real sockets: a runner with NO inbound port
dials out, the control plane dispatches chat through the tunnel
(RevDial parity, reference revdial.go + helix_openai_server.go:279)."""
import asyncio
import threading

import httpx
import pytest

from helix_amd.runner.service import RunnerService
from helix_amd.server.app import create_app
from helix_amd.server.config import ServerConfig
from helix_amd.server.tunnel import tunnel_loop
from helix_amd.store import Store
from tests.test_e2e import ServerThread, free_port


@pytest.fixture(scope="module")
def tunnel_stack():
    cp_port = free_port()
    cfg = ServerConfig()
    cfg.inference.default_provider = "helix"
    cfg.inference.default_model = "tiny"
    store = Store(":memory:")
    cp_app = create_app(cfg, store=store)
    cp = ServerThread(cp_app, cp_port)
    cp.start()
    base = f"http://127.0.0.1:{cp_port}"

    svc = RunnerService(device="cpu")
    svc.ensure_loaded("tiny")
    stop = None
    loop_holder = {}

    def run_tunnel():
        loop = asyncio.new_event_loop()
        loop_holder["loop"] = loop
        loop.run_until_complete(tunnel_loop(
            base, "runner-token", "nat-runner", svc,
            reconnect_delay=0.5))

    t = threading.Thread(target=run_tunnel, daemon=True)
    t.start()
    # announce via heartbeat with a tunnel address
    import time
    for _ in range(50):
        r = httpx.post(f"{base}/api/v1/runner/heartbeat", json={
            "runner_id": "nat-runner", "address": "tunnel:nat-runner",
            "gpus": [], "models": [{"model_id": "tiny", "state": "ready"}]},
            headers={"Authorization": "Bearer runner-token"})
        if r.status_code == 200 and cp_app.state.tunnels.is_connected(
                "nat-runner"):
            break
        time.sleep(0.1)
    r = httpx.post(f"{base}/api/v1/users", json={"username": "t"},
                   headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    yield base, key, cp_app
    loop_holder["loop"].call_soon_threadsafe(
        lambda: [task.cancel()
                 for task in asyncio.all_tasks(loop_holder["loop"])])
    cp.stop()
    svc.shutdown()


def test_tunnel_chat_roundtrip(tunnel_stack):
    base, key, app = tunnel_stack
    assert app.state.tunnels.is_connected("nat-runner")
    r = httpx.post(f"{base}/v1/chat/completions", json={
        "model": "tiny", "max_tokens": 6, "temperature": 0,
        "messages": [{"role": "user", "content": "through the tunnel"}]},
        timeout=120, headers={"Authorization": f"Bearer {key}"})
    assert r.status_code == 200, r.text
    assert r.json()["usage"]["completion_tokens"] >= 1


def test_tunnel_streaming(tunnel_stack):
    base, key, _ = tunnel_stack
    with httpx.stream("POST", f"{base}/v1/chat/completions", json={
        "model": "tiny", "max_tokens": 5, "temperature": 0, "stream": True,
        "messages": [{"role": "user", "content": "stream"}]},
            timeout=120,
            headers={"Authorization": f"Bearer {key}"}) as r:
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    assert len(lines) >= 3


def test_tunnel_embeddings(tunnel_stack):
    base, key, app = tunnel_stack
    app.state.runner_service = None  # force router path
    svc_models = [{"model_id": "tiny", "state": "ready"},
                  {"model_id": "tiny-bert", "state": "ready"}]
    httpx.post(f"{base}/api/v1/runner/heartbeat", json={
        "runner_id": "nat-runner", "address": "tunnel:nat-runner",
        "gpus": [], "models": svc_models},
        headers={"Authorization": "Bearer runner-token"})
    r = httpx.post(f"{base}/v1/embeddings", json={
        "model": "tiny-bert", "input": "hello"}, timeout=120,
        headers={"Authorization": f"Bearer {key}"})
    assert r.status_code == 200, r.text
    assert len(r.json()["data"][0]["embedding"]) == 128


def test_tunnel_images(tunnel_stack):
    """images/generations over the reverse tunnel (server/tunnel.py
    `_unary` + `_serve_one` image branch)."""
    base, key, app = tunnel_stack
    app.state.runner_service = None  # force router -> tunnel path
    httpx.post(f"{base}/api/v1/runner/heartbeat", json={
        "runner_id": "nat-runner", "address": "tunnel:nat-runner",
        "gpus": [], "models": [{"model_id": "tiny", "state": "ready"},
                               {"model_id": "tiny-dit",
                                "state": "ready"}]},
        headers={"Authorization": "Bearer runner-token"})
    r = httpx.post(f"{base}/v1/images/generations", json={
        "model": "tiny-dit", "prompt": "tunnel image", "steps": 3,
        "seed": 2}, timeout=120,
        headers={"Authorization": f"Bearer {key}"})
    assert r.status_code == 200, r.text
    import base64
    import io

    from PIL import Image
    img = Image.open(io.BytesIO(
        base64.b64decode(r.json()["data"][0]["b64_json"])))
    assert img.size == (32, 32)
