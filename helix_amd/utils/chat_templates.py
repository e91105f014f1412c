"""Per-model chat templates + hermes tool-call parsing.

The reference serves real models through vLLM with
`--enable-auto-tool-choice --tool-call-parser hermes`
(design/2026-04-28-cloud-gpu-smoke-results.md:58-60); parity serving
needs the same surface: render OpenAI `messages` (+`tools`) into each
model family's prompt format, and parse `<tool_call>{json}</tool_call>`
blocks out of the generation into OpenAI `tool_calls`.

Templates render to TEXT; the tokenizer encodes the result (real
tokenizer.json files carry the special strings as added tokens, so they
map to the correct single ids).
"""
from __future__ import annotations

import json
import re
import uuid
from typing import List, Optional, Tuple


def _content_str(m: dict) -> str:
    c = m.get("content") or ""
    if isinstance(c, str):
        return c
    return " ".join(p.get("text", "") for p in c if isinstance(p, dict))


def _tools_system_suffix(tools: Optional[List[dict]]) -> str:
    """Hermes-style function-calling preamble appended to the system
    prompt when the request carries tools."""
    if not tools:
        return ""
    sigs = []
    for t in tools:
        fn = t.get("function", t)
        sigs.append(json.dumps({
            "name": fn.get("name", ""),
            "description": fn.get("description", ""),
            "parameters": fn.get("parameters", {})}, sort_keys=True))
    return (
        "\n\nYou have access to the following functions. To call a "
        "function, respond with a <tool_call>{\"name\": ..., "
        "\"arguments\": {...}}</tool_call> block.\n<tools>\n"
        + "\n".join(sigs) + "\n</tools>")


def _merge_system(messages: List[dict], tools) -> Tuple[str, List[dict]]:
    system = ""
    rest = []
    for m in messages:
        if m.get("role") == "system" and not rest:
            system += (("\n" if system else "") + _content_str(m))
        else:
            rest.append(m)
    system += _tools_system_suffix(tools)
    return system, rest


def render_llama3(messages: List[dict], add_generation_prompt: bool = True,
                  tools=None) -> str:
    """Llama-3 instruct format (header tokens are added tokens in the
    real tokenizer.json)."""
    system, rest = _merge_system(messages, tools)
    out = "<|begin_of_text|>"
    if system:
        out += ("<|start_header_id|>system<|end_header_id|>\n\n"
                f"{system}<|eot_id|>")
    for m in rest:
        role = m.get("role", "user")
        if role == "tool":
            out += ("<|start_header_id|>ipython<|end_header_id|>\n\n"
                    f"{_content_str(m)}<|eot_id|>")
            continue
        body = _content_str(m)
        if role == "assistant" and m.get("tool_calls"):
            body += "".join(
                "<tool_call>" + json.dumps(
                    {"name": tc["function"]["name"],
                     "arguments": json.loads(tc["function"]["arguments"])
                     if isinstance(tc["function"].get("arguments"), str)
                     else tc["function"].get("arguments", {})})
                + "</tool_call>" for tc in m["tool_calls"])
        out += (f"<|start_header_id|>{role}<|end_header_id|>\n\n"
                f"{body}<|eot_id|>")
    if add_generation_prompt:
        out += "<|start_header_id|>assistant<|end_header_id|>\n\n"
    return out


def render_mistral(messages: List[dict], add_generation_prompt: bool = True,
                   tools=None) -> str:
    """Mistral [INST] format; system prompt folded into the first user
    turn (upstream convention)."""
    system, rest = _merge_system(messages, tools)
    out = "<s>"
    pending_sys = system
    for m in rest:
        role = m.get("role", "user")
        body = _content_str(m)
        if role in ("user", "tool"):
            if pending_sys:
                body = pending_sys + "\n\n" + body
                pending_sys = ""
            out += f"[INST] {body} [/INST]"
        else:
            out += f" {body}</s>"
    return out


def render_chatml(messages: List[dict], add_generation_prompt: bool = True,
                  tools=None) -> str:
    """ChatML (Qwen2 family)."""
    system, rest = _merge_system(messages, tools)
    out = ""
    if system:
        out += f"<|im_start|>system\n{system}<|im_end|>\n"
    for m in rest:
        role = m.get("role", "user")
        out += f"<|im_start|>{role}\n{_content_str(m)}<|im_end|>\n"
    if add_generation_prompt:
        out += "<|im_start|>assistant\n"
    return out


TEMPLATES = {
    "llama3": render_llama3,
    "mistral": render_mistral,
    "chatml": render_chatml,
}

# model-name prefix -> template family
_MODEL_TEMPLATE = [
    ("llama3", "llama3"),
    ("llama-3", "llama3"),
    ("mistral", "mistral"),
    ("qwen", "chatml"),
]


def template_for_model(model: str) -> str:
    low = (model or "").lower()
    for prefix, tpl in _MODEL_TEMPLATE:
        if low.startswith(prefix):
            return tpl
    return "llama3"


_TOOL_CALL_RE = re.compile(r"<tool_call>\s*(\{.*?\})\s*</tool_call>",
                           re.DOTALL)


def parse_tool_calls(text: str) -> Tuple[str, List[dict]]:
    """Hermes parser: extract <tool_call>{json}</tool_call> blocks into
    OpenAI tool_calls; returns (content_without_blocks, tool_calls).
    Malformed JSON blocks are left in the content untouched (the
    reference's parser behavior: only well-formed calls become calls)."""
    calls: List[dict] = []

    def _sub(match: "re.Match") -> str:
        try:
            obj = json.loads(match.group(1))
            name = obj["name"]
        except Exception:
            return match.group(0)        # leave malformed block in text
        args = obj.get("arguments", {})
        calls.append({
            "id": f"call_{uuid.uuid4().hex[:24]}",
            "type": "function",
            "function": {"name": name,
                         "arguments": json.dumps(args)},
        })
        return ""

    content = _TOOL_CALL_RE.sub(_sub, text).strip()
    return content, calls
