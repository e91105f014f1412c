"""Weight IO: safetensors load/save with HF-checkpoint name mapping and
pinned-host + side-stream async H2D staging (SURVEY.md §2.8 "Weight
load": replaces vLLM's HF download/load with a native path that does not
stall serving streams).
"""
from __future__ import annotations

import glob
import json
import logging
import os
from typing import Dict, Iterator, Tuple

import torch

log = logging.getLogger("helix_amd.weights")


def save_sharded(model: torch.nn.Module, out_dir: str,
                 shard_bytes: int = 4 << 30):
    """Save a model as safetensors shards (our native checkpoint format)."""
    from safetensors.torch import save_file
    os.makedirs(out_dir, exist_ok=True)
    shard: Dict[str, torch.Tensor] = {}
    size = 0
    idx = 0
    index = {}
    sd = model.state_dict()
    for name, t in sd.items():
        tb = t.detach().cpu().contiguous()
        shard[name] = tb
        size += tb.numel() * tb.element_size()
        if size >= shard_bytes:
            fn = f"model-{idx:05d}.safetensors"
            save_file(shard, os.path.join(out_dir, fn))
            for n in shard:
                index[n] = fn
            shard, size = {}, 0
            idx += 1
    if shard:
        fn = f"model-{idx:05d}.safetensors"
        save_file(shard, os.path.join(out_dir, fn))
        for n in shard:
            index[n] = fn
    with open(os.path.join(out_dir, "model.safetensors.index.json"),
              "w") as f:
        json.dump({"weight_map": index}, f)


def iter_safetensors(ckpt_dir: str) -> Iterator[Tuple[str, torch.Tensor]]:
    from safetensors import safe_open
    files = sorted(glob.glob(os.path.join(ckpt_dir, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {ckpt_dir}")
    for path in files:
        with safe_open(path, framework="pt", device="cpu") as f:
            for name in f.keys():
                yield name, f.get_tensor(name)


# HF Llama checkpoint name -> (our param, role). Roles q/k/v and
# gate/up are packed into the fused projections.
def _map_hf_name(name: str):
    name = name.replace("model.", "")
    if name == "embed_tokens.weight":
        return "embed_tokens.weight", None
    if name == "norm.weight":
        return "final_norm_w", None
    if name == "lm_head.weight":
        return "lm_head.weight", None
    if name.startswith("layers."):
        parts = name.split(".")
        i = parts[1]
        rest = ".".join(parts[2:])
        m = {
            "input_layernorm.weight": (f"layers.{i}.input_norm_w", None),
            "post_attention_layernorm.weight":
                (f"layers.{i}.post_norm_w", None),
            "self_attn.q_proj.weight":
                (f"layers.{i}.attn.qkv_proj.weight", "q"),
            "self_attn.k_proj.weight":
                (f"layers.{i}.attn.qkv_proj.weight", "k"),
            "self_attn.v_proj.weight":
                (f"layers.{i}.attn.qkv_proj.weight", "v"),
            "self_attn.q_proj.bias":
                (f"layers.{i}.attn.qkv_proj.bias", "q"),
            "self_attn.k_proj.bias":
                (f"layers.{i}.attn.qkv_proj.bias", "k"),
            "self_attn.v_proj.bias":
                (f"layers.{i}.attn.qkv_proj.bias", "v"),
            "self_attn.o_proj.weight":
                (f"layers.{i}.attn.o_proj.weight", None),
            "mlp.gate_proj.weight":
                (f"layers.{i}.mlp.gate_up_proj.weight", "gate"),
            "mlp.up_proj.weight":
                (f"layers.{i}.mlp.gate_up_proj.weight", "up"),
            "mlp.down_proj.weight":
                (f"layers.{i}.mlp.down_proj.weight", None),
        }
        if rest in m:
            return m[rest]
    return name, None  # our native names pass through


@torch.inference_mode()
def load_llama_weights(model, ckpt_dir: str, use_async: bool = True):
    """Load a Llama checkpoint (HF layout or our native layout) into the
    model. On GPU, tensors are staged through a pinned-host buffer and
    copied on a side stream (hipMemcpyAsync) so decode streams on the
    default stream are not serialized behind weight traffic."""
    # GGUF checkpoints (llama.cpp ecosystem) route through the native
    # GGUF loader (engine/gguf.py): file path or a dir holding one.
    gguf_path = None
    if ckpt_dir.endswith(".gguf") and os.path.isfile(ckpt_dir):
        gguf_path = ckpt_dir
    elif os.path.isdir(ckpt_dir):
        ggufs = sorted(glob.glob(os.path.join(ckpt_dir, "*.gguf")))
        if ggufs and not glob.glob(os.path.join(ckpt_dir, "*.safetensors")):
            gguf_path = ggufs[0]
    if gguf_path is not None:
        from . import gguf as _gguf
        n = _gguf.load_gguf_weights(model, gguf_path)
        log.info("loaded %d tensors from GGUF %s", n, gguf_path)
        return n

    params = dict(model.named_parameters())
    cfg = model.cfg
    q, kv = cfg.q_size, cfg.kv_size
    inter = cfg.intermediate_size
    device = next(model.parameters()).device
    on_gpu = device.type == "cuda"
    side = torch.cuda.Stream(device) if (on_gpu and use_async) else None
    # Double-buffered pinned staging: wait on a buffer's event before
    # refilling it so the async copy out of it has completed.
    pinned = [None, None]
    events = [None, None]
    cur = 0

    def stage_copy(dst_view: torch.Tensor, src: torch.Tensor):
        nonlocal cur
        src = src.to(dst_view.dtype)
        if side is None:
            dst_view.copy_(src)
            return
        n = src.numel() * src.element_size()
        if pinned[cur] is None or pinned[cur].numel() < n:
            pinned[cur] = torch.empty(max(n, 64 << 20), dtype=torch.uint8,
                                      pin_memory=True)
            events[cur] = torch.cuda.Event()
        else:
            events[cur].synchronize()
        flat = pinned[cur][:n]
        flat.copy_(src.contiguous().view(-1).view(torch.uint8))
        with torch.cuda.stream(side):
            dst_view.view(-1).view(torch.uint8).copy_(flat,
                                                      non_blocking=True)
            events[cur].record(side)
        cur ^= 1

    loaded = 0
    for name, tensor in iter_safetensors(ckpt_dir):
        our, role = _map_hf_name(name)
        if our not in params:
            log.debug("skipping unknown weight %s", name)
            continue
        p = params[our]
        if role == "q":
            stage_copy(p.data[:q], tensor)
        elif role == "k":
            stage_copy(p.data[q:q + kv], tensor)
        elif role == "v":
            stage_copy(p.data[q + kv:], tensor)
        elif role == "gate":
            stage_copy(p.data[:inter], tensor)
        elif role == "up":
            stage_copy(p.data[inter:], tensor)
        else:
            stage_copy(p.data, tensor)
        loaded += 1
    if side is not None:
        torch.cuda.current_stream(device).wait_stream(side)
        torch.cuda.synchronize(device)
    log.info("loaded %d tensors from %s", loaded, ckpt_dir)
    return loaded
