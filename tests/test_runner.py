"""Runner service + OpenAI adapter tests on CPU (tiny models).

Mirrors the reference's inferenceproxy/proxy_test.go + gpucloud
inference_roundtrip scenario at unit scale.
"""
import pytest
import torch
from fastapi.testclient import TestClient

from helix_amd.runner.http import create_runner_app
from helix_amd.runner.service import (DEFAULT_SPECS, ModelSpec,
                                      NoCapacityError, RunnerService,
                                      estimate_model_bytes)


@pytest.fixture()
def service():
    svc = RunnerService(device="cpu", memory_budget=64 << 30)
    yield svc
    svc.shutdown()


@pytest.fixture()
def client(service):
    app = create_runner_app(service)
    with TestClient(app) as c:
        yield c


def test_estimate_llama8b_size():
    est = estimate_model_bytes(DEFAULT_SPECS["llama3-8b"])
    # 8B params bf16 ~16 GB + KV + headroom
    assert 16 << 30 < est < 120 << 30


def test_chat_completion_roundtrip(client):
    r = client.post("/v1/chat/completions", json={
        "model": "tiny",
        "messages": [{"role": "user", "content": "hello"}],
        "max_tokens": 8, "temperature": 0,
    })
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["object"] == "chat.completion"
    assert body["choices"][0]["message"]["role"] == "assistant"
    assert body["usage"]["completion_tokens"] >= 1


def test_chat_completion_stream(client):
    with client.stream("POST", "/v1/chat/completions", json={
        "model": "tiny",
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 6, "temperature": 0, "stream": True,
    }) as r:
        assert r.status_code == 200
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    import json
    chunks = [json.loads(l[6:]) for l in lines[:-1]]
    assert chunks[0]["choices"][0]["delta"].get("role") == "assistant"
    assert chunks[-1]["choices"][0]["finish_reason"] is not None
    assert chunks[-1]["usage"]["completion_tokens"] >= 1


def test_model_not_found(client):
    r = client.post("/v1/chat/completions", json={
        "model": "nope", "messages": [{"role": "user", "content": "x"}]})
    assert r.status_code == 404


def test_embeddings(client):
    r = client.post("/v1/embeddings", json={
        "model": "tiny-bert", "input": ["hello world", "second text"]})
    assert r.status_code == 200
    data = r.json()["data"]
    assert len(data) == 2
    v = torch.tensor(data[0]["embedding"])
    assert abs(float(v.norm()) - 1.0) < 1e-3  # L2 normalized


def test_models_list(client):
    r = client.get("/v1/models")
    ids = [m["id"] for m in r.json()["data"]]
    assert "tiny" in ids and "llama3-8b" in ids


def test_scheduler_evicts_lru(service):
    # budget fits ~1 tiny model at a time (estimate ~1.1 GiB incl. the
    # 1 GiB activation headroom)
    svc = RunnerService(device="cpu", memory_budget=int(1.5e9))
    specs = {
        "a": ModelSpec("a", "llm", "tiny", max_model_len=256,
                       kv_cache_blocks=64),
        "b": ModelSpec("b", "llm", "tiny", max_model_len=256,
                       kv_cache_blocks=64),
    }
    svc.specs.update(specs)
    try:
        svc.ensure_loaded("a")
        assert "a" in svc.loaded_models()
        svc.ensure_loaded("b")   # must evict a
        assert "b" in svc.loaded_models()
        assert "a" not in svc.loaded_models()
    finally:
        svc.shutdown()


def test_scheduler_no_capacity():
    svc = RunnerService(device="cpu", memory_budget=1000)
    try:
        with pytest.raises(NoCapacityError):
            svc.ensure_loaded("tiny")
    finally:
        svc.shutdown()


def test_status_endpoint(client):
    client.post("/api/v1/models/tiny/load")
    r = client.get("/api/v1/status")
    body = r.json()
    assert any(m["model_id"] == "tiny" for m in body["models"])


def test_apply_profile_reconciles(service):
    from helix_amd.runner.assignment import apply_profile
    res = apply_profile(service, {"id": "p1", "models": [
        {"name": "tiny", "preset": "tiny", "max_model_len": 256,
         "kv_cache_blocks": 64},
        {"name": "tiny-bert", "preset": "tiny-bert", "kind": "embedding"},
    ]})
    assert set(res["loaded"]) == {"tiny", "tiny-bert"}
    assert set(service.loaded_models()) == {"tiny", "tiny-bert"}
    # switching profile unloads removed models (profile_switch scenario)
    res = apply_profile(service, {"id": "p2", "models": [
        {"name": "tiny-gqa", "preset": "tiny-gqa", "max_model_len": 256,
         "kv_cache_blocks": 64}]})
    assert set(res["unloaded"]) == {"tiny", "tiny-bert"}
    assert service.loaded_models() == ["tiny-gqa"]


def test_stream_disconnect_cancels_sequence(service):
    """Dropping the SSE stream mid-generation must cancel the engine
    sequence (no orphan decode burning KV/compute)."""
    import asyncio
    from helix_amd.runner.openai_adapter import chat_completion

    async def run():
        it = await chat_completion(service, {
            "model": "tiny", "stream": True, "temperature": 0,
            "max_tokens": 200,
            "messages": [{"role": "user", "content": "go"}]})
        count = 0
        async for _ in it:
            count += 1
            if count >= 3:
                await it.aclose()      # simulate client disconnect
                break
        return count

    asyncio.run(run())
    inst = service.instances["tiny"]
    # wait for the engine loop to drain the cancelled sequence
    import time
    for _ in range(100):
        if inst.in_flight == 0:
            break
        time.sleep(0.05)
    assert inst.in_flight == 0
    seqs = inst.engine.seqs
    assert any(s.status.value == "cancelled" for s in seqs.values())


def test_register_spec_and_load(tmp_path):
    """Dynamic model registration: admin registers an fp8 tiny spec and
    loads it through the admin API."""
    from fastapi.testclient import TestClient
    from helix_amd.runner.service import RunnerService
    from helix_amd.server.app import create_app
    from helix_amd.server.config import ServerConfig
    from helix_amd.store import Store
    cfg = ServerConfig()
    cfg.filestore.path = str(tmp_path / "fs")
    svc = RunnerService(device="cpu")
    app = create_app(cfg, store=Store(":memory:"), runner_service=svc)
    client = TestClient(app)
    H = {"Authorization": "Bearer admin-key"}
    r = client.post("/api/v1/local-models", json={
        "name": "tiny-fp8", "preset": "tiny", "max_model_len": 256,
        "kv_cache_blocks": 64, "quantization": "fp8"}, headers=H)
    assert r.status_code == 200, r.text
    r = client.post("/api/v1/local-models/tiny-fp8/load", headers=H)
    assert r.status_code == 200
    assert "tiny-fp8" in svc.loaded_models()
    from helix_amd.models.quant import FP8Linear
    inst = svc.instances["tiny-fp8"]
    assert isinstance(inst.engine.model.layers[0].attn.qkv_proj, FP8Linear)
    r = client.post("/api/v1/local-models", json={
        "name": "tiny-fp8", "preset": "tiny"}, headers=H)
    assert r.status_code == 409        # loaded: must unload first
    svc.shutdown()


def test_openai_logprobs(tmp_path):
    """OpenAI logprobs field on the native runner surface."""
    import asyncio
    from helix_amd.runner.openai_adapter import chat_completion
    from helix_amd.runner.service import RunnerService
    svc = RunnerService(device="cpu")
    try:
        resp = asyncio.run(chat_completion(svc, {
            "model": "tiny", "messages": [{"role": "user",
                                           "content": "hi"}],
            "max_tokens": 4, "temperature": 0, "logprobs": True,
            "top_logprobs": 2}))
        lp = resp["choices"][0]["logprobs"]["content"]
        assert len(lp) == 4
        assert len(lp[0]["top_logprobs"]) == 2
        assert lp[0]["logprob"] <= 0.0
        assert isinstance(lp[0]["token"], str)
        # absent unless requested
        resp2 = asyncio.run(chat_completion(svc, {
            "model": "tiny", "messages": [{"role": "user",
                                           "content": "hi"}],
            "max_tokens": 2, "temperature": 0}))
        assert "logprobs" not in resp2["choices"][0]
    finally:
        svc.shutdown()


def test_graceful_drain():
    """drain(): in-flight requests finish; new submissions are rejected
    with an error finish."""
    import threading
    from helix_amd.engine.sampling_params import SamplingParams
    from helix_amd.runner.service import RunnerService
    svc = RunnerService(device="cpu")
    try:
        inst = svc.ensure_loaded("tiny")
        done = threading.Event()
        toks = []

        def cb(seq, tok, fin):
            toks.append(tok)
            if fin:
                done.set()
        inst.submit("d0", [1, 2, 3],
                    SamplingParams(temperature=0.0, max_tokens=20,
                                   ignore_eos=True), cb)
        assert svc.drain(timeout=60) is True
        # in_flight hits 0 just before the final callback fires — allow
        # the callback thread a moment
        assert done.wait(timeout=10)
        assert len(toks) == 20           # in-flight ran to completion
        # post-drain submissions are rejected immediately
        rejected = threading.Event()
        reasons = []

        def cb2(seq, tok, fin):
            reasons.append(seq.finish_reason)
            rejected.set()
        inst.submit("d1", [4, 5], SamplingParams(max_tokens=4), cb2)
        assert rejected.wait(timeout=5)
        assert reasons[0].startswith("error")
    finally:
        svc.shutdown()


def test_runner_metrics_endpoint():
    from fastapi.testclient import TestClient
    from helix_amd.runner.http import create_runner_app
    from helix_amd.runner.service import RunnerService
    svc = RunnerService(device="cpu")
    try:
        svc.ensure_loaded("tiny")
        client = TestClient(create_runner_app(svc, "r-test"))
        r = client.get("/metrics")
        assert r.status_code == 200
        assert "helix_runner_models_loaded 1" in r.text
        assert 'helix_runner_in_flight{model="tiny"}' in r.text
    finally:
        svc.shutdown()


def test_vision_embeddings_instance():
    """SigLIP2-role image embedder (reference kodit vision path):
    base64/data-URL images embed to stable normalized vectors through
    the runner's OpenAI embeddings surface."""
    import asyncio
    import base64
    import io

    import numpy as np
    from PIL import Image

    from helix_amd.runner.openai_adapter import embeddings
    from helix_amd.runner.service import RunnerService

    svc = RunnerService(device="cpu")

    def png(color):
        img = Image.new("RGB", (48, 48), color)
        buf = io.BytesIO()
        img.save(buf, "PNG")
        return base64.b64encode(buf.getvalue()).decode()

    red, blue = png((255, 0, 0)), png((0, 0, 255))
    out = asyncio.run(embeddings(svc, {
        "model": "tiny-vit",
        "input": [{"image": red}, {"image": blue},
                  "data:image/png;base64," + red]}))
    vecs = [np.array(d["embedding"]) for d in out["data"]]
    assert len(vecs) == 3 and len(vecs[0]) == 128
    assert abs(vecs[0] @ vecs[2] - 1.0) < 1e-3   # same image, same vec
    assert vecs[0] @ vecs[1] < 0.999             # different images
    assert abs(np.linalg.norm(vecs[0]) - 1.0) < 1e-3
