"""OpenAI-compatible request handling over RunnerService.

Used by the runner's HTTP app and by the control plane's in-process
"helix" provider (the seam the reference keeps between
helix_openai_server.go and the sandbox vLLM containers).
"""
from __future__ import annotations

import asyncio
import time
import uuid
from typing import AsyncIterator, Dict, List, Optional

from helix_amd.engine.sampling_params import SamplingParams
from helix_amd.runner.service import (LLMInstance, ModelNotFoundError,
                                      RunnerService)
from helix_amd.utils.chat_templates import (parse_tool_calls,
                                             template_for_model)
from helix_amd.utils.tokenizer import get_tokenizer


class StreamDetokenizer:
    """Incremental byte-level detokenizer (UTF-8 partials buffered)."""

    def __init__(self, tokenizer):
        self.tok = tokenizer
        self.ids: List[int] = []
        self.emitted = ""

    def push(self, token_id: int) -> str:
        self.ids.append(token_id)
        full = self.tok.decode(self.ids)
        # strip a trailing replacement char from an incomplete sequence
        while full.endswith("�"):
            full = full[:-1]
        delta = full[len(self.emitted):]
        self.emitted = full
        return delta


def _params_from_request(req: dict, max_model_len: int) -> SamplingParams:
    temp = req.get("temperature")
    if temp is None:
        temp = 1.0
    stop = req.get("stop")
    return SamplingParams(
        temperature=float(temp),
        top_p=float(req.get("top_p") or 1.0),
        top_k=int(req.get("top_k") or 0),
        max_tokens=int(req.get("max_tokens") or 256),
        presence_penalty=float(req.get("presence_penalty") or 0.0),
        frequency_penalty=float(req.get("frequency_penalty") or 0.0),
        seed=req.get("seed"),
        logprobs=(int(req.get("top_logprobs") or 1)
                  if req.get("logprobs") else None),
        json_mode=(req.get("response_format") or {}).get(
            "type") == "json_object",
        logit_bias={int(k): float(v) for k, v in
                    (req.get("logit_bias") or {}).items()} or None,
    ), ([stop] if isinstance(stop, str) else list(stop or []))


class TokenStream:
    """Bridges the engine's callback thread into an asyncio queue."""

    def __init__(self, loop: asyncio.AbstractEventLoop):
        self.loop = loop
        self.q: asyncio.Queue = asyncio.Queue()

    def on_token(self, seq, token_id: int, finished: bool):
        lp = None
        lps = getattr(seq, "logprobs", None)
        if lps:
            lp = lps[-1]       # engine appended this token's entry
        try:
            self.loop.call_soon_threadsafe(
                self.q.put_nowait,
                (token_id, finished, seq.finish_reason, lp))
        except RuntimeError:
            # client's event loop is gone (disconnect) — tokens go nowhere
            pass

    async def __aiter__(self):
        while True:
            tok, fin, reason, lp = await self.q.get()
            yield tok, fin, reason, lp
            if fin:
                return


async def chat_completion(service: RunnerService, req: dict,
                          request_id: Optional[str] = None):
    """Returns a response dict, or an async iterator of chunk dicts when
    req['stream'] is true."""
    model = req.get("model", "")
    loop = asyncio.get_event_loop()
    inst = await loop.run_in_executor(None, service.ensure_loaded, model)
    assert isinstance(inst, LLMInstance), f"{model} is not an LLM"
    tok = get_tokenizer(model)
    tools = req.get("tools")
    if "messages" in req and req["messages"] is not None:
        prompt_ids = tok.apply_chat_template(
            req["messages"], template=template_for_model(model),
            tools=tools)
    else:
        prompt_ids = tok.encode(str(req.get("prompt", "")), add_bos=True)
    max_ctx = inst.spec.max_model_len
    if len(prompt_ids) >= max_ctx - 8:
        prompt_ids = prompt_ids[-(max_ctx - 8):]
    params, stop_strs = _params_from_request(req, max_ctx)
    params.stop_token_ids = [tok.eos_token_id]
    rid = request_id or f"chatcmpl-{uuid.uuid4().hex[:24]}"
    created = int(time.time())

    if req.get("stream"):
        return _stream(service, inst, model, rid, created, prompt_ids,
                       params, stop_strs, tok, tools)

    n = int(req.get("n") or 1)
    choices = []
    total_completion = 0
    for i in range(n):
        text, finish_reason, ntok, lp_content = await _generate_one(
            inst, f"{rid}-{i}", prompt_ids, params, stop_strs, tok, loop)
        total_completion += ntok
        message = {"role": "assistant", "content": text}
        if tools:
            content, calls = parse_tool_calls(text)
            if calls:
                message = {"role": "assistant",
                           "content": content or None,
                           "tool_calls": calls}
                finish_reason = "tool_calls"
        choice = {
            "index": i,
            "message": message,
            "finish_reason": finish_reason,
        }
        if params.logprobs:
            choice["logprobs"] = {"content": lp_content}
        choices.append(choice)
    return {
        "id": rid,
        "object": "chat.completion",
        "created": created,
        "model": model,
        "choices": choices,
        "usage": {
            "prompt_tokens": len(prompt_ids),
            "completion_tokens": total_completion,
            "total_tokens": len(prompt_ids) + total_completion,
        },
    }


async def _generate_one(inst, seq_id, prompt_ids, params, stop_strs, tok,
                        loop):
    ts = TokenStream(loop)
    inst.submit(seq_id, prompt_ids, params, ts.on_token)
    detok = StreamDetokenizer(tok)
    text = ""
    finish_reason = "stop"
    ntok = 0
    lp_content = []
    async for token_id, fin, reason, lp in ts.__aiter__():
        ntok += 1
        text += detok.push(token_id)
        if lp is not None:
            lp_content.append(_lp_entry(lp, tok))
        if fin:
            finish_reason = reason or "stop"
            break
        hit = _find_stop(text, stop_strs)
        if hit is not None:
            inst.cancel(seq_id)
            text = text[:hit]
            finish_reason = "stop"
            break
    return text, finish_reason, ntok, lp_content


def _lp_entry(lp: dict, tok) -> dict:
    """Engine logprob record -> OpenAI chat logprobs content entry."""
    return {
        "token": tok.decode([lp["token"]]),
        "logprob": lp["logprob"],
        "top_logprobs": [{"token": tok.decode([t["token"]]),
                          "logprob": t["logprob"]}
                         for t in lp.get("top_logprobs", [])],
    }


def _find_stop(text: str, stop_strs: List[str]) -> Optional[int]:
    best = None
    for s in stop_strs:
        if s and s in text:
            i = text.index(s)
            best = i if best is None else min(best, i)
    return best


async def _stream(service, inst, model, rid, created, prompt_ids, params,
                  stop_strs, tok, tools=None) -> AsyncIterator[dict]:
    loop = asyncio.get_event_loop()
    ts = TokenStream(loop)
    inst.submit(rid, prompt_ids, params, ts.on_token)
    detok = StreamDetokenizer(tok)
    ntok = 0
    finished = False

    def chunk(delta: dict, finish: Optional[str] = None, usage=None):
        c = {
            "id": rid,
            "object": "chat.completion.chunk",
            "created": created,
            "model": model,
            "choices": [{"index": 0, "delta": delta,
                         "finish_reason": finish}],
        }
        if usage:
            c["usage"] = usage
        return c

    try:
        yield chunk({"role": "assistant", "content": ""})
        emitted = 0
        finish_reason = "stop"
        muted = False   # inside a <tool_call> block: buffer, don't emit
        async for token_id, fin, reason, _lp in ts.__aiter__():
            ntok += 1
            delta = detok.push(token_id)
            hit = _find_stop(detok.emitted, stop_strs)
            if hit is not None:
                inst.cancel(rid)
                keep = max(0, hit - emitted)
                if keep and not muted:
                    yield chunk({"content": delta[:keep]})
                finish_reason = "stop"
                finished = True
                break
            if tools and not muted and "<tool_call>" in detok.emitted:
                # stop emitting content once a tool call begins; the
                # parsed calls are delivered in the final delta
                muted = True
                pre = detok.emitted.index("<tool_call>")
                keep = max(0, pre - emitted)
                if keep:
                    yield chunk({"content": delta[:keep]})
                emitted = len(detok.emitted)
            if delta and not muted:
                emitted += len(delta)
                yield chunk({"content": delta})
            if fin:
                finish_reason = reason or "stop"
                finished = True
                break
        if tools:
            _, calls = parse_tool_calls(detok.emitted)
            if calls:
                finish_reason = "tool_calls"
                yield chunk({"tool_calls": [
                    {"index": i, **c} for i, c in enumerate(calls)]})
        yield chunk({}, finish=finish_reason, usage={
            "prompt_tokens": len(prompt_ids),
            "completion_tokens": ntok,
            "total_tokens": len(prompt_ids) + ntok,
        })
    finally:
        # Client disconnect / generator close mid-stream: stop decoding
        # (the reference's cancellation-correctness requirement —
        # helix_openai_server.go:285-293 conn hard-close semantics).
        if not finished:
            inst.cancel(rid)


async def images_generations(service: RunnerService, req: dict) -> dict:
    """OpenAI images surface (reference delegates to a diffusers
    container via `/v1/images/generations`, inferenceproxy/proxy.go:113).
    Always returns b64_json (no public URL storage on a runner)."""
    import base64
    model = req.get("model", "flux-lite")
    spec = service.specs.get(model)
    if spec is not None and spec.kind != "image":
        # refuse before admission — don't load an LLM just to 404
        raise ModelNotFoundError(f"{model} is not an image model")
    loop = asyncio.get_event_loop()
    inst = await loop.run_in_executor(None, service.ensure_loaded, model)
    if not hasattr(inst, "generate"):
        raise ModelNotFoundError(f"{model} is not an image model")
    n = max(1, min(int(req.get("n", 1) or 1), 8))
    steps = max(1, min(int(req.get("steps", 8) or 8), 64))
    seed = req.get("seed")
    pngs = await loop.run_in_executor(
        None, lambda: inst.generate(
            str(req.get("prompt", "")), n=n, steps=steps,
            seed=int(seed) if seed is not None else None,
            size=req.get("size")))
    return {
        "created": int(time.time()),
        "model": model,
        "data": [{"b64_json": base64.b64encode(p).decode(),
                  "revised_prompt": None} for p in pngs],
    }


async def embeddings(service: RunnerService, req: dict) -> dict:
    model = req.get("model", "")
    loop = asyncio.get_event_loop()
    inst = await loop.run_in_executor(None, service.ensure_loaded, model)
    inputs = req.get("input", "")
    if isinstance(inputs, str):
        inputs = [inputs]
    vecs = await loop.run_in_executor(None, inst.embed, inputs)
    return {
        "object": "list",
        "model": model,
        "data": [{"object": "embedding", "index": i, "embedding": v}
                 for i, v in enumerate(vecs)],
        "usage": {"prompt_tokens": sum(
                      len(t) if isinstance(t, str) else 1
                      for t in inputs),
                  "total_tokens": sum(
                      len(t) if isinstance(t, str) else 1
                      for t in inputs)},
    }
