"""JSON-constrained decoding (OpenAI `response_format:
{"type": "json_object"}`; the reference serves this through its vLLM
backend's guided decoding).

A byte-level pushdown automaton over RFC 8259 JSON: given the bytes
emitted so far, `allowed_bytes()` is the exact set of next bytes that
keep the output a prefix of a valid JSON value, and `complete` flips
when a full top-level value has been consumed (the engine then forces
EOS). With the byte-level tokenizer every vocab token is one byte, so
masking logits to the allowed set makes invalid JSON unrepresentable —
grammar enforcement, not prompting. (Multi-byte BPE tokenizers fall
back to instruction-nudged JSON in the adapter.)
"""
from __future__ import annotations

from typing import List, Optional, Set

WS = frozenset(b" \t\n\r")
DIGITS = frozenset(b"0123456789")
HEX = frozenset(b"0123456789abcdefABCDEF")
VALUE_START = frozenset(b"{[\"-0123456789tfn")
ESCAPABLE = frozenset(b'"\\/bfnrtu')
# printable string contents (no control chars, quote/backslash handled)
STR_CHARS = frozenset(range(0x20, 0x100)) - frozenset(b'"\\')


class JSONByteMask:
    """Incremental validator. States are kept on an explicit stack:
    'obj_key'   — inside object, before/at a key string
    'obj_colon' — after key, expecting ':'
    'obj_value' — expecting a value inside an object
    'obj_next'  — after a value, expecting ',' or '}'
    'arr_value' — expecting a value or ']' (only right after '[')
    'arr_next'  — after a value, expecting ',' or ']'
    plus scalar micro-states: in_string/escape/unicode, in_number,
    literal progress. Top-level expects exactly one value (an object,
    per json_object semantics when strict=True)."""

    def __init__(self, strict_object: bool = True):
        self.stack: List[str] = []
        self.state = "start"          # start | value | done
        self.strict_object = strict_object
        # scalar micro-state
        self.str_ctx: Optional[str] = None   # None|'chars'|'escape'|'u0'..'u3'
        self.num_buf = b""
        self.lit_target: bytes = b""
        self.lit_pos = 0
        self.complete = False

    # -- helpers -----------------------------------------------------------
    def _enter_value(self, b: int) -> bool:
        """Consume the first byte of a value; push container state."""
        c = bytes([b])
        if c == b"{":
            self.stack.append("obj_first")
            return True
        if c == b"[":
            self.stack.append("arr_first")
            return True
        if c == b'"':
            self.str_ctx = "chars"
            self.stack.append("in_string_value")
            return True
        if b in DIGITS or c == b"-":
            self.num_buf = c
            self.stack.append("in_number")
            return True
        for lit in (b"true", b"false", b"null"):
            if lit[0] == b:
                self.lit_target = lit
                self.lit_pos = 1
                self.stack.append("in_literal")
                return True
        return False

    def _pop_value_done(self):
        """A complete value just finished; unwind to the container."""
        if not self.stack:
            self.complete = True
            self.state = "done"
            return
        top = self.stack[-1]
        if top in ("obj_value",):
            self.stack[-1] = "obj_next"
        elif top in ("arr_first", "arr_value"):
            self.stack[-1] = "arr_next"
        else:
            self.complete = not self.stack
            self.state = "done" if self.complete else self.state

    # -- the transition function ------------------------------------------
    def push_byte(self, b: int) -> bool:
        """Advance on byte b; returns False if b is not allowed."""
        if b not in self.allowed_bytes():
            return False
        c = bytes([b])
        top = self.stack[-1] if self.stack else None

        if top == "in_string_value" or top == "in_string_key":
            if self.str_ctx == "chars":
                if c == b'"':
                    kind = top
                    self.stack.pop()
                    self.str_ctx = None
                    if kind == "in_string_key":
                        self.stack[-1] = "obj_colon"
                    else:
                        self._pop_value_done()
                elif c == b"\\":
                    self.str_ctx = "escape"
            elif self.str_ctx == "escape":
                self.str_ctx = "u0" if c == b"u" else "chars"
            elif self.str_ctx.startswith("u"):
                n = int(self.str_ctx[1])
                self.str_ctx = f"u{n + 1}" if n < 3 else "chars"
            return True

        if top == "in_number":
            if b in self.allowed_bytes():
                if b in DIGITS or c in (b".", b"e", b"E", b"+", b"-"):
                    self.num_buf += c
                    return True
                # delimiter byte ends the number: pop then re-dispatch
                self.stack.pop()
                self.num_buf = b""
                self._pop_value_done()
                return self.push_byte(b)
            return False

        if top == "in_literal":
            self.lit_pos += 1
            if self.lit_pos == len(self.lit_target):
                self.stack.pop()
                self._pop_value_done()
            return True

        if b in WS:
            return True

        if self.state == "start":
            ok = self._enter_value(b)
            if ok:
                self.state = "value"
            return ok

        if top == "obj_first":
            if c == b"}":
                self.stack.pop()
                self._pop_value_done()
            else:                              # must be a key quote
                self.str_ctx = "chars"
                self.stack[-1] = "obj_key_open"
                self.stack.append("in_string_key")
            return True
        if top == "obj_key_open":
            # the opening quote of the next key (WS already passed)
            self.str_ctx = "chars"
            self.stack.append("in_string_key")
            return True
        if top == "obj_colon":
            self.stack[-1] = "obj_value"
            return True
        if top == "obj_value":
            return self._enter_value(b)
        if top == "obj_next":
            if c == b"}":
                self.stack.pop()
                self._pop_value_done()
            else:                              # ','
                self.str_ctx = None
                self.stack[-1] = "obj_key_open"
            return True
        if top == "arr_first":
            if c == b"]":
                self.stack.pop()
                self._pop_value_done()
                return True
            self.stack[-1] = "arr_value"
            return self._enter_value(b)
        if top == "arr_value":
            return self._enter_value(b)
        if top == "arr_next":
            if c == b"]":
                self.stack.pop()
                self._pop_value_done()
            else:                              # ','
                self.stack[-1] = "arr_value"
            return True
        return False

    # -- what may come next -------------------------------------------------
    def allowed_bytes(self) -> Set[int]:
        if self.complete:
            return set()
        top = self.stack[-1] if self.stack else None

        if top in ("in_string_value", "in_string_key"):
            if self.str_ctx == "chars":
                return set(STR_CHARS) | set(b'"\\')
            if self.str_ctx == "escape":
                return set(ESCAPABLE)
            return set(HEX)                   # \uXXXX digits

        if top == "in_number":
            allowed = set(DIGITS)
            nb = self.num_buf
            if nb in (b"-",):
                return set(DIGITS)
            if b"." not in nb and b"e" not in nb and b"E" not in nb:
                allowed |= set(b".eE")
            elif (b"e" not in nb and b"E" not in nb):
                allowed |= set(b"eE")
            if nb[-1:] in (b"e", b"E"):
                return set(DIGITS) | set(b"+-")
            if nb[-1:] in (b"+", b"-") or nb[-1:] == b".":
                return set(DIGITS)
            # a number can be terminated by its container's delimiter
            allowed |= self._delimiters_after_value()
            return allowed

        if top == "in_literal":
            return {self.lit_target[self.lit_pos]}

        ws = set(WS)
        if self.state == "start":
            if self.strict_object:
                return ws | set(b"{")
            return ws | set(VALUE_START)
        if top == "obj_first":
            return ws | set(b'"}')
        if top == "obj_key_open":
            return set(WS) | set(b'"')
        if top == "obj_colon":
            return ws | set(b":")
        if top == "obj_value":
            return ws | set(VALUE_START)
        if top == "obj_next":
            return ws | set(b",}")
        if top == "arr_first":
            return ws | set(VALUE_START) | set(b"]")
        if top == "arr_value":
            return ws | set(VALUE_START)
        if top == "arr_next":
            return ws | set(b",]")
        return set()

    def _delimiters_after_value(self) -> Set[int]:
        """Bytes that may legally follow a just-finished number, given
        the enclosing container (they terminate the number)."""
        if len(self.stack) < 2:
            return set(WS)
        outer = self.stack[-2]
        if outer == "obj_value":
            return set(WS) | set(b",}")
        if outer in ("arr_first", "arr_value"):
            return set(WS) | set(b",]")
        return set(WS)
