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
        role = m.get("role", "user")
        if isinstance(content, list):
            texts = []
            tool_calls = []
            tool_results = []
            for b in content:
                if not isinstance(b, dict):
                    continue
                if b.get("type") == "text":
                    texts.append(b.get("text", ""))
                elif b.get("type") == "tool_use":
                    tool_calls.append({
                        "id": b.get("id", ""), "type": "function",
                        "function": {
                            "name": b.get("name", ""),
                            "arguments": json.dumps(b.get("input", {}))}})
                elif b.get("type") == "tool_result":
                    rc = b.get("content", "")
                    if isinstance(rc, list):
                        rc = " ".join(x.get("text", "") for x in rc
                                      if isinstance(x, dict))
                    tool_results.append({
                        "role": "tool",
                        "tool_call_id": b.get("tool_use_id", ""),
                        "content": rc})
            msg = {"role": role, "content": " ".join(texts)}
            if tool_calls:
                msg["tool_calls"] = tool_calls
            if texts or tool_calls:
                messages.append(msg)
            messages.extend(tool_results)
        else:
            messages.append({"role": role, "content": content})
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
    # tool definitions (Anthropic input_schema -> OpenAI parameters)
    if req.get("tools"):
        out["tools"] = [{
            "type": "function",
            "function": {"name": t.get("name", ""),
                         "description": t.get("description", ""),
                         "parameters": t.get("input_schema", {})}}
            for t in req["tools"]]
    tc = req.get("tool_choice")
    if isinstance(tc, dict):
        if tc.get("type") == "tool":
            out["tool_choice"] = {"type": "function", "function":
                                  {"name": tc.get("name", "")}}
        elif tc.get("type") in ("any", "auto"):
            out["tool_choice"] = ("required" if tc["type"] == "any"
                                  else "auto")
    return out


def openai_to_anthropic(resp: dict) -> dict:
    choice = resp.get("choices", [{}])[0]
    msg = choice.get("message", {})
    text = msg.get("content", "") or ""
    usage = resp.get("usage", {})
    stop_reason = {"stop": "end_turn", "length": "max_tokens",
                   "tool_calls": "tool_use"}.get(
        choice.get("finish_reason", "stop"), "end_turn")
    content = []
    if text:
        content.append({"type": "text", "text": text})
    for tc in msg.get("tool_calls") or []:
        fn = tc.get("function", {})
        try:
            args = json.loads(fn.get("arguments") or "{}")
        except Exception:
            args = {"_raw": fn.get("arguments", "")}
        content.append({"type": "tool_use",
                        "id": tc.get("id", f"toolu_{uuid.uuid4().hex[:16]}"),
                        "name": fn.get("name", ""), "input": args})
        stop_reason = "tool_use"
    if not content:
        content = [{"type": "text", "text": ""}]
    return {
        "id": f"msg_{uuid.uuid4().hex[:24]}",
        "type": "message",
        "role": "assistant",
        "model": resp.get("model", ""),
        "content": content,
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


# ---------------------------------------------------------------------------
# Native passthrough (reference api/pkg/anthropic/anthropic_proxy.go:
# reverse proxy that forwards the RAW /v1/messages body to the upstream
# Anthropic-compatible endpoint — preserving cache_control blocks and
# thus prompt caching — swaps the caller's helix token for the provider
# key (anthropic_proxy.go:158,212), and retries the thinking.type
# adaptive<->enabled mismatch (anthropic_proxy.go:108-111)).

class AnthropicPassthrough:
    def __init__(self, base_url: str, api_key: str, http_client=None,
                 version: str = "2023-06-01"):
        self.base_url = base_url.rstrip("/")
        self.api_key = api_key
        self.version = version
        self._http = http_client

    def _client(self):
        if self._http is None:
            import httpx
            self._http = httpx.AsyncClient(timeout=600)
        return self._http

    def _headers(self, incoming: dict | None = None) -> dict:
        h = {"content-type": "application/json",
             "x-api-key": self.api_key,
             "anthropic-version":
                 (incoming or {}).get("anthropic-version", self.version)}
        beta = (incoming or {}).get("anthropic-beta")
        if beta:
            h["anthropic-beta"] = beta
        return h

    @staticmethod
    def _flip_thinking(body: dict) -> dict | None:
        th = body.get("thinking")
        if not isinstance(th, dict) or "type" not in th:
            return None
        flipped = dict(body)
        t = th.get("type")
        alt = {"adaptive": "enabled", "enabled": "adaptive"}.get(t)
        if alt is None:
            return None
        flipped["thinking"] = dict(th, type=alt)
        return flipped

    async def forward(self, body: dict, incoming_headers: dict
                      | None = None):
        """Non-streaming: returns (status_code, response_json). Retries
        once with the alternate thinking.type on a 400 naming it."""
        url = self.base_url + "/v1/messages"
        r = await self._client().post(url, json=body,
                                      headers=self._headers(
                                          incoming_headers))
        if r.status_code == 400:
            try:
                err = r.json()
            except Exception:
                err = {}
            if "thinking" in json.dumps(err):
                flipped = self._flip_thinking(body)
                if flipped is not None:
                    r = await self._client().post(
                        url, json=flipped,
                        headers=self._headers(incoming_headers))
        try:
            return r.status_code, r.json()
        except Exception:
            return r.status_code, {"type": "error", "error": {
                "type": "api_error", "message": r.text[:500]}}

    async def forward_stream(self, body: dict,
                             incoming_headers: dict | None = None):
        """Streaming: yields raw SSE bytes verbatim from upstream."""
        url = self.base_url + "/v1/messages"
        client = self._client()
        async with client.stream("POST", url, json=body,
                                 headers=self._headers(
                                     incoming_headers)) as r:
            if r.status_code >= 400:
                raw = await r.aread()
                yield (b"event: error\ndata: " + raw[:2000] + b"\n\n")
                return
            async for chunk in r.aiter_bytes():
                yield chunk
