"""Model catalog (parity with api/pkg/model: model_info.json static
price/context catalog + DynamicModelInfoProvider DB overrides,
ProcessModelName default resolution)."""
from __future__ import annotations

from typing import Dict, List, Optional

# Static catalog (reference model_info.json role). Prices are
# per-million-token placeholders for local models (cost accounting works;
# external providers carry their real prices via overrides).
STATIC_MODEL_INFO: Dict[str, dict] = {
    "llama3-8b": {"context_length": 8192, "prompt_price_per_m": 0.05,
                  "completion_price_per_m": 0.10, "family": "llama",
                  "runtime": "helix_amd", "kind": "chat"},
    "llama3-70b": {"context_length": 8192, "prompt_price_per_m": 0.6,
                   "completion_price_per_m": 0.8, "family": "llama",
                   "runtime": "helix_amd", "kind": "chat"},
    "llama3.1-8b": {"context_length": 32768, "prompt_price_per_m": 0.05,
                    "completion_price_per_m": 0.10, "family": "llama",
                    "runtime": "helix_amd", "kind": "chat"},
    "mistral-7b": {"context_length": 8192, "prompt_price_per_m": 0.05,
                   "completion_price_per_m": 0.10, "family": "mistral",
                   "runtime": "helix_amd", "kind": "chat"},
    "qwen2-7b": {"context_length": 32768, "prompt_price_per_m": 0.05,
                 "completion_price_per_m": 0.10, "family": "qwen",
                 "runtime": "helix_amd", "kind": "chat"},
    "bge-base": {"context_length": 512, "prompt_price_per_m": 0.005,
                 "completion_price_per_m": 0.0, "family": "bge",
                 "runtime": "helix_amd", "kind": "embedding"},
    "bge-large": {"context_length": 512, "prompt_price_per_m": 0.01,
                  "completion_price_per_m": 0.0, "family": "bge",
                  "runtime": "helix_amd", "kind": "embedding"},
    "siglip-base": {"context_length": 0, "prompt_price_per_m": 0.01,
                    "completion_price_per_m": 0.0, "family": "siglip",
                    "runtime": "helix_amd", "kind": "vision-embedding"},
    "flux-lite": {"context_length": 32, "prompt_price_per_m": 0.0,
                  "completion_price_per_m": 0.0, "family": "flux",
                  "runtime": "helix_amd", "kind": "image",
                  "price_per_image": 0.002},
    "gpt-4o": {"context_length": 128000, "prompt_price_per_m": 2.5,
               "completion_price_per_m": 10.0, "family": "openai",
               "runtime": "external", "kind": "chat"},
    "claude-sonnet-4-5": {"context_length": 200000,
                          "prompt_price_per_m": 3.0,
                          "completion_price_per_m": 15.0,
                          "family": "anthropic", "runtime": "external",
                          "kind": "chat"},
}


class ModelCatalog:
    def __init__(self, store):
        self.store = store

    def get(self, model: str) -> Optional[dict]:
        """Dynamic DB override wins over the static catalog."""
        dyn = self.store.get("models", model)
        if dyn:
            base = dict(STATIC_MODEL_INFO.get(model, {}))
            base.update(dyn)
            return base
        info = STATIC_MODEL_INFO.get(model)
        return dict(info) if info else None

    def set_override(self, model: str, info: dict):
        info = {**info, "id": model}
        self.store.put("models", model, info)

    def list(self) -> List[dict]:
        out = {}
        for mid, info in STATIC_MODEL_INFO.items():
            out[mid] = {"id": mid, **info}
        for dyn in self.store.list("models", limit=10000):
            mid = dyn["id"]
            out[mid] = {**out.get(mid, {}), **dyn}
        return list(out.values())

    def context_length(self, model: str, default: int = 8192) -> int:
        info = self.get(model)
        return int(info["context_length"]) if info else default

    def cost_usd(self, model: str, prompt_tokens: int,
                 completion_tokens: int) -> float:
        info = self.get(model) or {}
        return (prompt_tokens * info.get("prompt_price_per_m", 0.0) +
                completion_tokens * info.get("completion_price_per_m", 0.0)
                ) / 1e6
