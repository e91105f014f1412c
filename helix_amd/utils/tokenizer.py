"""Byte-level tokenizer + chat template.

No network => no real tokenizer files; this deterministic byte-level
tokenizer (256 byte tokens + specials) makes the text->tokens->text path
fully functional end-to-end with random-init weights ("data: synthetic"
per BASELINE.json). Vocabulary ids stay < 300 so any model preset works.
"""
from __future__ import annotations

from typing import List

PAD = 0
BOS = 1
EOS = 2
ROLE_SYSTEM = 3
ROLE_USER = 4
ROLE_ASSISTANT = 5
ROLE_TOOL = 6
N_SPECIAL = 8

_ROLE_TOKENS = {
    "system": ROLE_SYSTEM,
    "user": ROLE_USER,
    "assistant": ROLE_ASSISTANT,
    "tool": ROLE_TOOL,
}


class ByteTokenizer:
    vocab_size = N_SPECIAL + 256
    bos_token_id = BOS
    eos_token_id = EOS

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = [BOS] if add_bos else []
        ids.extend(b + N_SPECIAL for b in text.encode("utf-8"))
        return ids

    def decode(self, ids: List[int]) -> str:
        bs = bytes(i - N_SPECIAL for i in ids
                   if N_SPECIAL <= i < N_SPECIAL + 256)
        return bs.decode("utf-8", errors="replace")

    def apply_chat_template(self, messages: List[dict],
                            add_generation_prompt: bool = True,
                            template: str = "", tools=None) -> List[int]:
        from helix_amd.utils.chat_templates import _tools_system_suffix
        ids = [BOS]
        suffix = _tools_system_suffix(tools)
        saw_system = any(m.get("role") == "system" for m in messages)
        if suffix and not saw_system:
            ids.append(ROLE_SYSTEM)
            ids.extend(self.encode(suffix.strip()))
            ids.append(EOS)
        for m in messages:
            role = m.get("role", "user")
            content = m.get("content") or ""
            if not isinstance(content, str):
                # OpenAI content-parts form
                content = " ".join(p.get("text", "") for p in content
                                   if isinstance(p, dict))
            if role == "system" and suffix:
                content += suffix
                suffix = ""
            ids.append(_ROLE_TOKENS.get(role, ROLE_USER))
            ids.extend(self.encode(content))
            ids.append(EOS)
        if add_generation_prompt:
            ids.append(ROLE_ASSISTANT)
        return ids


_DEFAULT = ByteTokenizer()
_CACHE = {}


def _jinja_raise(msg):
    raise ValueError(msg)


def _token_str(tok):
    """HF tokenizer_config token fields are strings or AddedToken
    dicts."""
    if isinstance(tok, dict):
        return tok.get("content", "")
    return tok or ""


class HFTokenizer:
    """Real tokenizer files (tokenizer.json) when a deployment provides
    them via HELIX_TOKENIZER_DIR/<model>/tokenizer.json — same interface
    as ByteTokenizer."""

    def __init__(self, path: str):
        import json as _json
        import os as _os

        from tokenizers import Tokenizer
        self._tok = Tokenizer.from_file(path)
        # the model's own jinja chat template, when the deployment
        # ships tokenizer_config.json next to tokenizer.json (HF
        # layout) — takes precedence over the built-in family
        # templates
        self.chat_template = None
        cfg_path = _os.path.join(_os.path.dirname(path),
                                 "tokenizer_config.json")
        if _os.path.exists(cfg_path):
            try:
                with open(cfg_path) as fh:
                    tc = _json.load(fh)
                tpl = tc.get("chat_template")
                if isinstance(tpl, list):      # HF multi-template form
                    tpl = next((t.get("template") for t in tpl
                                if t.get("name") == "default"),
                               tpl[0].get("template")
                               if tpl else None)
                if tpl:
                    import jinja2
                    env = jinja2.Environment(
                        trim_blocks=True, lstrip_blocks=True,
                        undefined=jinja2.ChainableUndefined)
                    env.globals["raise_exception"] = _jinja_raise
                    env.filters["tojson"] = _json.dumps
                    self.chat_template = env.from_string(tpl)
                self._special = {
                    "bos_token": _token_str(tc.get("bos_token")),
                    "eos_token": _token_str(tc.get("eos_token")),
                }
            except Exception:
                self.chat_template = None
        self.vocab_size = self._tok.get_vocab_size()
        self.bos_token_id = self._tok.token_to_id("<|begin_of_text|>") or             self._tok.token_to_id("<s>") or 1
        self.eos_token_id = self._tok.token_to_id("<|end_of_text|>") or             self._tok.token_to_id("</s>") or 2

    def encode(self, text: str, add_bos: bool = False):
        ids = self._tok.encode(text, add_special_tokens=False).ids
        return ([self.bos_token_id] + ids) if add_bos else ids

    def decode(self, ids):
        return self._tok.decode(list(ids), skip_special_tokens=True)

    def apply_chat_template(self, messages, add_generation_prompt=True,
                            template: str = "", tools=None):
        """Render with the model family's real prompt format (llama3
        header tokens / mistral [INST] / qwen ChatML) and encode; the
        special strings are added tokens in real tokenizer.json files."""
        if self.chat_template is not None:
            # the model's own jinja template (exact HF semantics for
            # the fields real templates use)
            text = self.chat_template.render(
                messages=messages, tools=tools,
                add_generation_prompt=add_generation_prompt,
                **getattr(self, "_special", {}))
            return self.encode(text, add_bos=False)
        from helix_amd.utils.chat_templates import TEMPLATES
        render = TEMPLATES.get(template or "llama3", TEMPLATES["llama3"])
        text = render(messages, add_generation_prompt, tools=tools)
        # templates carry their own BOS markers (<|begin_of_text|>/<s>)
        return self.encode(text, add_bos=False)


def get_tokenizer(model: str = ""):
    import os
    base = os.environ.get("HELIX_TOKENIZER_DIR", "")
    if base and model:
        path = os.path.join(base, model, "tokenizer.json")
        if os.path.exists(path):
            if path not in _CACHE:
                try:
                    _CACHE[path] = HFTokenizer(path)
                except Exception:
                    _CACHE[path] = _DEFAULT
            return _CACHE[path]
    return _DEFAULT
