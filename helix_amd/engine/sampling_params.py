"""Sampling parameters — mirrors the request fields the reference forwards
from AssistantConfig to vLLM (reference types.go:1636-1661: temperature,
top_p, penalties, max_tokens)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class SamplingParams:
    temperature: float = 0.0           # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0                     # 0 => disabled
    max_tokens: int = 256
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    stop_token_ids: List[int] = field(default_factory=list)
    ignore_eos: bool = False
    seed: Optional[int] = None
    logprobs: Optional[int] = None     # top-k logprobs per emitted token
    json_mode: bool = False            # grammar-constrain output to JSON
    logit_bias: Optional[dict] = None  # token id -> additive bias

    @property
    def needs_logit_processing(self) -> bool:
        return (self.top_p < 1.0 or self.top_k > 0
                or self.presence_penalty != 0.0
                or self.frequency_penalty != 0.0
                or self.repetition_penalty != 1.0)
