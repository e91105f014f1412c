"""Anthropic-compatible /v1/messages surface (parity with
api/pkg/anthropic: native passthrough when an Anthropic provider is
configured; otherwise translated onto any chat provider — including the
local MI355X runner — so Anthropic SDK clients work against helix_amd).
"""
from __future__ import annotations

import json
import uuid
from typing import AsyncIterator


def anthropic_to_openai(req: dict) -> dict:
    messages = []
    system = req.get("system")
    if system:
        if isinstance(system, list):
            system = " ".join(b.get("text", "") for b in system)
        messages.append({"role": "system", "content": system})
    for m in req.get("messages", []):
        content = m.get("content")
        if isinstance(content, list):
            content = " ".join(b.get("text", "") for b in content
                               if isinstance(b, dict) and
                               b.get("type") == "text")
        messages.append({"role": m.get("role", "user"), "content": content})
    out = {
        "model": req.get("model", ""),
        "messages": messages,
        "max_tokens": req.get("max_tokens", 256),
        "stream": bool(req.get("stream")),
    }
    for k in ("temperature", "top_p"):
        if req.get(k) is not None:
            out[k] = req[k]
    if req.get("stop_sequences"):
        out["stop"] = req["stop_sequences"]
    return out


def openai_to_anthropic(resp: dict) -> dict:
    choice = resp.get("choices", [{}])[0]
    text = choice.get("message", {}).get("content", "") or ""
    usage = resp.get("usage", {})
    stop_reason = {"stop": "end_turn", "length": "max_tokens"}.get(
        choice.get("finish_reason", "stop"), "end_turn")
    return {
        "id": f"msg_{uuid.uuid4().hex[:24]}",
        "type": "message",
        "role": "assistant",
        "model": resp.get("model", ""),
        "content": [{"type": "text", "text": text}],
        "stop_reason": stop_reason,
        "stop_sequence": None,
        "usage": {"input_tokens": usage.get("prompt_tokens", 0),
                  "output_tokens": usage.get("completion_tokens", 0)},
    }


async def stream_anthropic_events(chunks: AsyncIterator[dict],
                                  model: str) -> AsyncIterator[str]:
    """Convert an OpenAI chunk stream into Anthropic SSE events."""
    mid = f"msg_{uuid.uuid4().hex[:24]}"

    def ev(name: str, data: dict) -> str:
        return f"event: {name}\ndata: {json.dumps(data)}\n\n"

    yield ev("message_start", {"type": "message_start", "message": {
        "id": mid, "type": "message", "role": "assistant", "model": model,
        "content": [], "stop_reason": None,
        "usage": {"input_tokens": 0, "output_tokens": 0}}})
    yield ev("content_block_start", {
        "type": "content_block_start", "index": 0,
        "content_block": {"type": "text", "text": ""}})
    out_tokens = 0
    finish = "end_turn"
    async for chunk in chunks:
        if not chunk.get("choices"):
            continue
        c = chunk["choices"][0]
        delta = c.get("delta", {}).get("content") or ""
        if delta:
            out_tokens += 1
            yield ev("content_block_delta", {
                "type": "content_block_delta", "index": 0,
                "delta": {"type": "text_delta", "text": delta}})
        if c.get("finish_reason"):
            finish = {"stop": "end_turn", "length": "max_tokens"}.get(
                c["finish_reason"], "end_turn")
        usage = chunk.get("usage")
        if usage:
            out_tokens = usage.get("completion_tokens", out_tokens)
    yield ev("content_block_stop", {"type": "content_block_stop", "index": 0})
    yield ev("message_delta", {
        "type": "message_delta",
        "delta": {"stop_reason": finish, "stop_sequence": None},
        "usage": {"output_tokens": out_tokens}})
    yield ev("message_stop", {"type": "message_stop"})
