"""RunnerService — the native MI355X model runner.

Replaces the reference's entire delegated GPU plane (compose-manager +
inference-proxy + vLLM/Ollama containers) with in-process
engines on hand-written CDNA4 kernels, plus the HBM-aware multi-model
scheduler the reference deleted (SURVEY.md §2.8 "Model pack/evict":
estimate-based admission driven by live free HBM, LRU eviction that
coordinates with in-flight batches).
"""
from __future__ import annotations

import asyncio
import logging
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from helix_amd.engine.engine import EngineConfig, LLMEngine, Sequence
from helix_amd.engine.sampling_params import SamplingParams
from helix_amd.models.bert import BERT_PRESETS, BertEmbeddingModel
from helix_amd.models.llama import PRESETS as LLAMA_PRESETS
from helix_amd.utils.tokenizer import get_tokenizer

log = logging.getLogger("helix_amd.runner")


@dataclass
class ModelSpec:
    """Manifest entry (replaces compose-profile YAML model entries)."""
    name: str
    kind: str = "llm"                  # llm | embedding
    preset: str = "llama3-8b"
    max_model_len: int = 8192
    max_num_seqs: int = 64
    kv_cache_blocks: Optional[int] = None
    kv_memory_fraction: float = 0.30   # of total HBM, for auto-sizing
    tp: int = 1                        # tensor-parallel degree (xGMI group)
    quantization: Optional[str] = None  # None | "fp8"
    kv_cache_dtype: str = "bf16"


DEFAULT_SPECS = {
    "llama3-8b": ModelSpec("llama3-8b", "llm", "llama3-8b"),
    "llama3-70b": ModelSpec("llama3-70b", "llm", "llama3-70b",
                            max_num_seqs=48, kv_cache_blocks=16384),
    "mistral-7b": ModelSpec("mistral-7b", "llm", "mistral-7b"),
    "qwen2-7b": ModelSpec("qwen2-7b", "llm", "qwen2-7b",
                          max_model_len=32768),
    "bge-base": ModelSpec("bge-base", "embedding", "bge-base"),
    "siglip-base": ModelSpec("siglip-base", "vision", "siglip-base"),
    "tiny-vit": ModelSpec("tiny-vit", "vision", "tiny-vit"),
    "bge-large": ModelSpec("bge-large", "embedding", "bge-large"),
    "flux-lite": ModelSpec("flux-lite", "image", "flux-lite"),
    "tiny-dit": ModelSpec("tiny-dit", "image", "tiny-dit"),
    # tiny models for CPU tests
    "tiny": ModelSpec("tiny", "llm", "tiny", max_model_len=256,
                      kv_cache_blocks=256),
    "tiny-gqa": ModelSpec("tiny-gqa", "llm", "tiny-gqa", max_model_len=512,
                          kv_cache_blocks=256),
    "tiny-bert": ModelSpec("tiny-bert", "embedding", "tiny-bert"),
}


def estimate_model_bytes(spec: ModelSpec, block_size: int = 16) -> int:
    """Admission estimate: weights + KV budget + workspace headroom.
    Descendant of the reference's GGUF estimator (api/pkg/memory/estimate.go)
    re-based on bf16 dense checkpoints."""
    if spec.kind == "image":
        from helix_amd.models.dit import DIT_PRESETS
        dcfg = DIT_PRESETS[spec.preset]
        h = dcfg.hidden
        per_layer = 18 * h * h          # qkv+o+mlp(8h^2)+adaLN mod(6h^2)
        n = dcfg.vocab_size * h + dcfg.seq_len * h + \
            dcfg.depth * per_layer + 40 * dcfg.vae_ch ** 2 * 9
        return int(n * 2 * 1.3) + (1 << 30)
    if spec.kind == "vision":
        from helix_amd.models.vit import VIT_PRESETS
        vcfg = VIT_PRESETS[spec.preset]
        n = vcfg.patch_dim * vcfg.hidden_size + \
            vcfg.num_patches * vcfg.hidden_size
        per_layer = 4 * vcfg.hidden_size ** 2 + \
            2 * vcfg.hidden_size * vcfg.intermediate_size
        return int((n + vcfg.num_layers * per_layer) * 2 * 1.3) + (512 << 20)
    if spec.kind == "embedding":
        cfg = BERT_PRESETS[spec.preset]
        n = cfg.vocab_size * cfg.hidden_size + cfg.max_position * cfg.hidden_size
        per_layer = 4 * cfg.hidden_size ** 2 + \
            2 * cfg.hidden_size * cfg.intermediate_size
        return 2 * (n + cfg.num_layers * per_layer) + (64 << 20)
    cfg = LLAMA_PRESETS[spec.preset]
    emb = cfg.vocab_size * cfg.hidden_size * (1 if cfg.tie_embeddings else 2)
    per_layer = (cfg.hidden_size * (cfg.q_size + 2 * cfg.kv_size)  # qkv
                 + cfg.q_size * cfg.hidden_size                     # o
                 + 3 * cfg.hidden_size * cfg.intermediate_size)     # mlp
    wbytes = 1 if spec.quantization == "fp8" else 2
    # fp8 halves the projection weights (embeddings/lm_head stay bf16)
    weights = 2 * emb + wbytes * cfg.num_layers * per_layer
    kv_block = 2 * cfg.num_layers * cfg.num_kv_heads * block_size * \
        cfg.head_dim * 2
    if spec.kv_cache_blocks:
        kv = kv_block * spec.kv_cache_blocks
    else:
        kv = kv_block * spec.max_num_seqs * \
            (spec.max_model_len // block_size + 1)
    return weights + kv + (1 << 30)    # +1 GiB activations/graphs headroom


class LLMInstance:
    """A loaded LLM: engine + dedicated step-loop thread."""

    def __init__(self, spec: ModelSpec, device: str):
        self.spec = spec
        self.device = device
        kv_blocks = spec.kv_cache_blocks
        if kv_blocks is None and torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info(torch.device(device))
            from helix_amd.engine.kv_cache import KVCache
            cfg = LLAMA_PRESETS[spec.preset]
            budget = int(total * spec.kv_memory_fraction)
            kv_blocks = KVCache.blocks_for_bytes(
                min(budget, max(free - (2 << 30), 1 << 28)),
                cfg.num_layers, cfg.num_kv_heads, cfg.head_dim, 16)
        self.engine = LLMEngine(
            EngineConfig(model=spec.preset, max_model_len=spec.max_model_len,
                         max_num_seqs=spec.max_num_seqs,
                         kv_cache_blocks=kv_blocks,
                         quantization=spec.quantization,
                         kv_cache_dtype=spec.kv_cache_dtype),
            device=device)
        if self.engine.graph_runner is not None:
            # eager hipGraph capture for all batch buckets (serving never
            # pays capture latency mid-request)
            t0 = time.time()
            self.engine.graph_runner.warmup()
            log.info("captured decode graphs in %.1fs", time.time() - t0)
        # Engine state is touched ONLY by the engine thread; submissions
        # and cancellations flow through an intake queue. (Holding a lock
        # across step() starved submitters: the engine thread re-acquired
        # it instantly between steps, serializing all requests.)
        self.intake: queue.Queue = queue.Queue()
        self.wake = threading.Event()
        self.stop = False
        self.draining = False
        self.last_used = time.time()
        self.thread = threading.Thread(target=self._loop, daemon=True,
                                       name=f"engine-{spec.name}")
        self.thread.start()

    @property
    def in_flight(self) -> int:
        # approximate (racy read is fine for scheduling/stats)
        return (len(self.engine.waiting) + len(self.engine.running)
                + self.intake.qsize())

    def _drain_intake(self):
        while True:
            try:
                kind, payload = self.intake.get_nowait()
            except queue.Empty:
                return
            try:
                if kind == "submit":
                    seq_id, prompt_ids, params, on_token = payload
                    self.engine.add_request(seq_id, prompt_ids, params,
                                            on_token=on_token)
                elif kind == "cancel":
                    self.engine.cancel(payload)
            except Exception as e:
                log.warning("intake %s failed: %s", kind, e)
                if kind == "submit" and payload[3] is not None:
                    # surface the rejection to the waiting caller
                    class _F:
                        seq_id = payload[0]
                        finish_reason = f"error: {e}"
                    try:
                        payload[3](_F(), 0, True)
                    except Exception:
                        pass

    def _loop(self):
        while not self.stop:
            self._drain_intake()
            if not self.engine.has_work:
                self.wake.wait(timeout=0.05)
                self.wake.clear()
                continue
            try:
                self.engine.step()
            except Exception:
                # a step failure (bad callback, transient OOM) must not
                # kill the serving loop; affected sequences error out,
                # the rest keep going
                log.exception("engine step failed (%s)", self.spec.name)
                time.sleep(0.01)

    def submit(self, seq_id: str, prompt_ids: List[int],
               params: SamplingParams, on_token) -> None:
        if self.draining:
            if on_token is not None:
                class _F:
                    pass
                f = _F()
                f.seq_id = seq_id
                f.finish_reason = "error: draining for shutdown"
                on_token(f, 0, True)
            return
        self.last_used = time.time()
        self.intake.put(("submit", (seq_id, prompt_ids, params, on_token)))
        self.wake.set()

    def cancel(self, seq_id: str):
        self.intake.put(("cancel", seq_id))
        self.wake.set()

    def drain(self, timeout: float = 30.0) -> bool:
        """Stop accepting new work and wait for in-flight sequences to
        finish (rolling-restart support). Returns True if fully drained
        before the deadline."""
        self.draining = True
        deadline = time.time() + timeout
        while time.time() < deadline:
            if self.in_flight == 0:
                return True
            time.sleep(0.05)
        return self.in_flight == 0

    def shutdown(self):
        self.stop = True
        self.wake.set()
        self.thread.join(timeout=10)
        del self.engine
        if torch.cuda.is_available():
            torch.cuda.empty_cache()


class EmbeddingInstance:
    def __init__(self, spec: ModelSpec, device: str):
        self.spec = spec
        self.device = torch.device(device)
        cfg = BERT_PRESETS[spec.preset]
        dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.model = BertEmbeddingModel(cfg).to(dtype).to(self.device)
        self.model.init_random(0)
        self.tokenizer = get_tokenizer(spec.name)
        self.lock = threading.Lock()
        self.last_used = time.time()

    @property
    def in_flight(self) -> int:
        return 0

    def embed(self, texts: List[str]) -> List[List[float]]:
        self.last_used = time.time()
        cfg = self.model.cfg
        ids_list = [self.tokenizer.encode(t)[: cfg.max_position - 1] or [1]
                    for t in texts]
        flat, cu = [], [0]
        for ids in ids_list:
            flat.extend(ids)
            cu.append(cu[-1] + len(ids))
        with self.lock:
            ids_t = torch.tensor(flat, dtype=torch.int64, device=self.device)
            cu_t = torch.tensor(cu, dtype=torch.int32, device=self.device)
            out = self.model(ids_t, cu_t, max(len(i) for i in ids_list))
        return out.cpu().tolist()

    def shutdown(self):
        del self.model
        if torch.cuda.is_available():
            torch.cuda.empty_cache()


class VisionEmbeddingInstance:
    """SigLIP2-role image embedder (reference kodit vision path):
    base64/bytes images -> normalized vectors in the retrieval space."""

    def __init__(self, spec: ModelSpec, device: str):
        from helix_amd.models.vit import VIT_PRESETS, ViTEmbeddingModel
        self.spec = spec
        self.device = torch.device(device)
        cfg = VIT_PRESETS[spec.preset]
        dtype = torch.bfloat16 if self.device.type == "cuda"             else torch.float32
        self.model = ViTEmbeddingModel(cfg).to(dtype).to(self.device)
        self.model.init_random(0)
        self.lock = threading.Lock()
        self.last_used = time.time()

    @property
    def in_flight(self) -> int:
        return 0

    def embed(self, inputs: List) -> List[List[float]]:
        """inputs: base64 strings, data URLs, or raw bytes."""
        import base64 as _b64
        self.last_used = time.time()
        blobs = []
        for item in inputs:
            if isinstance(item, dict):
                item = item.get("image", "")
            if isinstance(item, bytes):
                blobs.append(item)
                continue
            s = str(item)
            if s.startswith("data:"):
                s = s.split(",", 1)[-1]
            blobs.append(_b64.b64decode(s))
        with self.lock:
            out = self.model.embed_images(blobs)
        return out.cpu().tolist()

    def shutdown(self):
        del self.model
        if torch.cuda.is_available():
            torch.cuda.empty_cache()


class ImageGenInstance:
    """Diffusion image generator (the diffusers-container role of the
    reference, SURVEY §2.8 last row): prompt -> PNG bytes via the
    rectified-flow DiT pipeline (models/dit.py)."""

    def __init__(self, spec: ModelSpec, device: str):
        from helix_amd.models.dit import DIT_PRESETS, DiffusionImageModel
        self.spec = spec
        self.device = torch.device(device)
        self.cfg = DIT_PRESETS[spec.preset]
        dtype = torch.bfloat16 if self.device.type == "cuda" \
            else torch.float32
        self.model = DiffusionImageModel(self.cfg).to(dtype).to(
            self.device)
        self.model.init_random(0)
        self.tokenizer = get_tokenizer(spec.name)
        self.lock = threading.Lock()
        self.last_used = time.time()

    @property
    def in_flight(self) -> int:
        return 0

    def generate(self, prompt: str, n: int = 1, steps: int = 8,
                 seed: Optional[int] = None,
                 size: Optional[str] = None) -> List[bytes]:
        """-> PNG blobs. `size` ("WxH") resizes the preset's native
        output; seed defaults to a fresh draw per request."""
        import io as _io

        from PIL import Image
        self.last_used = time.time()
        if seed is None:
            seed = int.from_bytes(__import__("os").urandom(4), "little")
        ids = self.tokenizer.encode(prompt)
        with self.lock:
            # distinct images per sample: consecutive seeds
            imgs = [self.model.generate([ids], steps=steps,
                                        seed=seed + i)[0]
                    for i in range(max(1, int(n)))]
        out = []
        for img in imgs:
            pil = Image.fromarray(
                img.permute(1, 2, 0).cpu().numpy(), mode="RGB")
            if size:
                try:
                    w, h = (int(v) for v in size.lower().split("x"))
                    pil = pil.resize((w, h), Image.BILINEAR)
                except ValueError:
                    pass
            buf = _io.BytesIO()
            pil.save(buf, format="PNG")
            out.append(buf.getvalue())
        return out

    def shutdown(self):
        del self.model
        if torch.cuda.is_available():
            torch.cuda.empty_cache()


class RunnerService:
    """Owns model instances on one GPU (or CPU for tests); packs/evicts
    under the HBM budget."""

    def __init__(self, device: str = "cuda:0",
                 specs: Optional[Dict[str, ModelSpec]] = None,
                 memory_budget: Optional[int] = None):
        self.device = device
        self.specs = dict(DEFAULT_SPECS)
        if specs:
            self.specs.update(specs)
        self.instances: Dict[str, object] = {}
        self._lock = threading.Lock()
        if memory_budget is None:
            if torch.cuda.is_available() and device.startswith("cuda"):
                _, total = torch.cuda.mem_get_info(torch.device(device))
                memory_budget = int(total * 0.92)
            else:
                memory_budget = 64 << 30
        self.memory_budget = memory_budget

    # ---------------- scheduler -------------------------------------------
    def free_hbm(self) -> int:
        if torch.cuda.is_available() and self.device.startswith("cuda"):
            free, _ = torch.cuda.mem_get_info(torch.device(self.device))
            return free
        used = sum(estimate_model_bytes(i.spec)
                   for i in self.instances.values())
        return self.memory_budget - used

    def loaded_models(self) -> List[str]:
        return list(self.instances.keys())

    def register_spec(self, spec: ModelSpec):
        """Register/override a model manifest entry at runtime (the
        reference's dynamic model config; pairs with local-models
        load)."""
        with self._lock:
            if spec.name in self.instances:
                raise ValueError(f"{spec.name} is loaded; unload first")
            self.specs[spec.name] = spec

    def ensure_loaded(self, model: str):
        """Admission control: load `model`, LRU-evicting idle models if the
        estimate does not fit in free HBM. Raises NoCapacityError if it
        cannot fit even after eviction (control plane surfaces 503,
        matching the reference's NoRunnerError behavior, router.go:62)."""
        with self._lock:
            if model in self.instances:
                return self.instances[model]
            spec = self.specs.get(model)
            if spec is None:
                raise ModelNotFoundError(model)
            need = estimate_model_bytes(spec)
            # Evict LRU idle models until it fits.
            while self.free_hbm() < need:
                victim = self._pick_victim()
                if victim is None:
                    raise NoCapacityError(
                        f"model {model} needs {need >> 30} GiB; "
                        f"free {self.free_hbm() >> 30} GiB and no evictable "
                        f"model")
                log.info("evicting %s to fit %s", victim, model)
                self._unload(victim)
            log.info("loading %s (%d GiB est.)", model, need >> 30)
            t0 = time.time()
            if spec.kind == "embedding":
                inst = EmbeddingInstance(spec, self.device)
            elif spec.kind == "vision":
                inst = VisionEmbeddingInstance(spec, self.device)
            elif spec.kind == "image":
                inst = ImageGenInstance(spec, self.device)
            elif spec.tp > 1:
                from helix_amd.runner.tp_instance import TPLLMInstance
                inst = TPLLMInstance(spec, spec.tp)
            else:
                inst = LLMInstance(spec, self.device)
            log.info("loaded %s in %.1fs", model, time.time() - t0)
            self.instances[model] = inst
            return inst

    def _pick_victim(self) -> Optional[str]:
        idle = [(inst.last_used, name)
                for name, inst in self.instances.items()
                if inst.in_flight == 0]
        if not idle:
            return None
        return min(idle)[1]

    def _unload(self, model: str):
        inst = self.instances.pop(model)
        inst.shutdown()

    def unload(self, model: str):
        with self._lock:
            if model in self.instances:
                self._unload(model)

    def status(self) -> List[dict]:
        out = []
        for name, inst in self.instances.items():
            out.append({
                "model_id": name,
                "state": "ready",
                "memory_bytes": estimate_model_bytes(inst.spec),
                "in_flight": inst.in_flight,
                "last_used": inst.last_used,
            })
        return out

    def drain(self, timeout: float = 30.0) -> bool:
        """Drain every instance (new submissions rejected; in-flight
        finishes) within the shared deadline."""
        deadline = time.time() + timeout
        ok = True
        for inst in list(self.instances.values()):
            if hasattr(inst, "drain"):
                ok &= inst.drain(max(0.1, deadline - time.time()))
        return ok

    def shutdown(self):
        with self._lock:
            for name in list(self.instances):
                self._unload(name)


class ModelNotFoundError(Exception):
    pass


class NoCapacityError(Exception):
    pass
