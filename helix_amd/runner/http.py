"""Runner HTTP app: OpenAI-compatible inference surface + admin.

Replaces the reference's in-sandbox inference-proxy (:8090 model->port
routing, inferenceproxy/proxy.go) — here the models are in-process
engines, so routing is a dict lookup. Also serves the heartbeat the
sandbox-heartbeat binary used to push (the control plane can poll or the
runner pushes, see heartbeat.py).
"""
from __future__ import annotations

import collections
import json
import logging
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from helix_amd.runner import gpudetect
from helix_amd.runner.openai_adapter import (chat_completion, embeddings,
                                             images_generations)
from helix_amd.runner.service import (ModelNotFoundError, NoCapacityError,
                                      RunnerService)

log = logging.getLogger("helix_amd.runner.http")


class RingBufferHandler(logging.Handler):
    """In-memory log ring (parity with hydra's logbuf + the admin
    runner-logs surface, reference design/2026-05-29)."""

    def __init__(self, capacity: int = 2000):
        super().__init__()
        self.buf = collections.deque(maxlen=capacity)
        self.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(name)s %(message)s"))

    def emit(self, record):
        try:
            self.buf.append(self.format(record))
        except Exception:
            pass


def create_runner_app(service: RunnerService,
                      runner_id: str = "runner-0") -> FastAPI:
    app = FastAPI(title="helix_amd runner", docs_url=None)
    app.state.service = service
    app.state.runner_id = runner_id
    logbuf = RingBufferHandler()
    logging.getLogger("helix_amd").addHandler(logbuf)
    app.state.logbuf = logbuf

    @app.exception_handler(ModelNotFoundError)
    async def _nf(request, exc):
        return JSONResponse({"error": {"message": f"model not found: {exc}",
                                       "type": "invalid_request_error"}},
                            status_code=404)

    @app.exception_handler(NoCapacityError)
    async def _cap(request, exc):
        return JSONResponse({"error": {"message": str(exc),
                                       "type": "no_capacity"}},
                            status_code=503)

    @app.post("/v1/chat/completions")
    @app.post("/v1/completions")
    async def chat(request: Request):
        req = await request.json()
        result = await chat_completion(service, req)
        if req.get("stream"):
            async def sse():
                async for chunk in result:
                    yield f"data: {json.dumps(chunk)}\n\n"
                yield "data: [DONE]\n\n"
            return StreamingResponse(sse(), media_type="text/event-stream")
        return result

    @app.post("/v1/embeddings")
    async def embed(request: Request):
        return await embeddings(service, await request.json())

    @app.post("/v1/images/generations")
    async def images(request: Request):
        return await images_generations(service, await request.json())

    @app.get("/v1/models")
    async def models():
        loaded = set(service.loaded_models())
        return {"object": "list", "data": [
            {"id": name, "object": "model", "owned_by": "helix_amd",
             "loaded": name in loaded}
            for name in service.specs.keys()]}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    @app.get("/metrics")
    async def prometheus():
        # minimal runner-side Prometheus text (control plane has the
        # full registry; this covers per-node scraping)
        lines = [
            "# TYPE helix_runner_models_loaded gauge",
            f"helix_runner_models_loaded {len(service.loaded_models())}",
        ]
        for m in service.status():
            mid = m["model_id"]
            lines.append(f'helix_runner_in_flight{{model="{mid}"}} '
                         f'{m["in_flight"]}')
            lines.append(f'helix_runner_model_bytes{{model="{mid}"}} '
                         f'{m["memory_bytes"]}')
        try:
            import torch
            if torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info()
                lines.append(f"helix_runner_hbm_free_bytes {free}")
                lines.append(f"helix_runner_hbm_total_bytes {total}")
        except Exception:
            pass
        from fastapi.responses import PlainTextResponse
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.get("/api/v1/status")
    async def status():
        return {
            "runner_id": runner_id,
            "gpus": [g.model_dump() for g in gpudetect.detect()],
            "models": service.status(),
        }

    @app.get("/api/v1/logs")
    async def logs(n: int = 200):
        return {"lines": list(logbuf.buf)[-n:]}

    @app.post("/api/v1/models/{model}/load")
    async def load(model: str):
        import asyncio
        await asyncio.get_event_loop().run_in_executor(
            None, service.ensure_loaded, model)
        return {"ok": True}

    @app.post("/api/v1/models/{model}/unload")
    async def unload(model: str):
        service.unload(model)
        return {"ok": True}

    return app
