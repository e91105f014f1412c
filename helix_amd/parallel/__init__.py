"""Tensor parallelism over RCCL/xGMI.

Design: one process per GPU, torch.distributed with the "nccl" backend
(RCCL on ROCm). Llama layers are head-sharded (column-parallel QKV and
gate_up, row-parallel o_proj / down_proj) so each decoder layer needs
exactly two all-reduces — sized by xGMI's per-link ring bound
(7 links x ~153 GB/s), not NVSwitch assumptions.

CPU tests run the same code over gloo (world_size 2).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_GROUP = None
_TP_RANK = 0
_TP_SIZE = 1


def init_distributed(backend: Optional[str] = None) -> tuple[int, int]:
    """Initialize torch.distributed from torchrun env. Returns (rank, world)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend)
    return dist.get_rank(), dist.get_world_size()


def init_tp(tp_size: int, backend: Optional[str] = None):
    """Set up the tensor-parallel group (currently world == TP group)."""
    global _TP_GROUP, _TP_RANK, _TP_SIZE
    rank, world = init_distributed(backend)
    if tp_size <= 1:
        _TP_GROUP, _TP_RANK, _TP_SIZE = None, 0, 1
        return 0, 1
    assert world == tp_size, f"world {world} != tp_size {tp_size}"
    _TP_GROUP = dist.group.WORLD
    _TP_RANK, _TP_SIZE = rank, tp_size
    return rank, tp_size


def tp_rank() -> int:
    return _TP_RANK


def tp_size() -> int:
    return _TP_SIZE


_CUSTOM_AR = False        # one-shot xGMI allreduce initialized
_CUSTOM_AR_CAP = 0        # its data capacity (bytes)


def init_custom_allreduce(capacity_bytes: int = 32 * 1024 * 1024) -> bool:
    """Set up the one-shot xGMI allreduce (ops/hip/allreduce.hip) for the
    TP group: allocate this rank's mailbox, exchange hipIpc handles over
    the existing process group, map the peers. Decode-sized messages
    (<= capacity) then bypass RCCL's per-link-bound ring entirely —
    pulled point-to-point over each GPU pair's own xGMI link — and the
    kernel is hipGraph-capturable (per-block device counters, no resets).

    Returns True when active. Safe no-op on CPU / TP=1 / missing IPC.
    """
    global _CUSTOM_AR, _CUSTOM_AR_CAP
    if _TP_SIZE <= 1 or not torch.cuda.is_available():
        return False
    if _CUSTOM_AR:
        return True
    from helix_amd import ops
    if not ops.have_native():
        return False
    try:
        handle = ops._native().ar_create(_TP_SIZE, _TP_RANK, capacity_bytes)
        gathered: list = [None] * _TP_SIZE
        dist.all_gather_object(gathered, handle.numpy().tobytes(),
                               group=_TP_GROUP)
        handles = [torch.frombuffer(bytearray(b), dtype=torch.uint8)
                   for b in gathered]
        ops._native().ar_open(handles)
        _CUSTOM_AR, _CUSTOM_AR_CAP = True, capacity_bytes
        return True
    except Exception:
        import logging
        logging.getLogger("helix_amd.parallel").exception(
            "custom allreduce init failed; falling back to RCCL")
        try:
            ops._native().ar_destroy()
        except Exception:
            pass
        return False


def destroy_custom_allreduce():
    global _CUSTOM_AR, _CUSTOM_AR_CAP
    if _CUSTOM_AR:
        from helix_amd import ops
        ops._native().ar_destroy()
        _CUSTOM_AR, _CUSTOM_AR_CAP = False, 0


def tp_all_reduce(x: torch.Tensor) -> torch.Tensor:
    """Sum partial activations across the TP group (row-parallel output).

    Dispatch order: one-shot xGMI kernel (decode-sized bf16 messages) →
    RCCL ring (large prefill messages) → CPU round-trip (gloo PG with a
    CUDA tensor: tests running two ranks on one GPU, where RCCL cannot).
    """
    if _TP_SIZE <= 1:
        return x
    if _CUSTOM_AR and x.dtype == torch.bfloat16 and x.is_cuda \
            and x.numel() * 2 <= _CUSTOM_AR_CAP:
        from helix_amd import ops
        xc = x if x.is_contiguous() else x.contiguous()
        ops._native().ar_allreduce(xc, xc)
        if xc is not x:
            x.copy_(xc)
        return x
    if x.is_cuda and dist.get_backend(_TP_GROUP) == "gloo":
        # gloo cannot reduce CUDA bf16 tensors; round-trip through host
        # fp32 (slow, correctness path for single-GPU multi-process tests)
        h = x.detach().float().cpu()
        dist.all_reduce(h, group=_TP_GROUP)
        x.copy_(h.to(x.dtype))
        return x
    dist.all_reduce(x, group=_TP_GROUP)
    return x


def shard_llama_state_dict(sd: dict, cfg, tp: int, rank: int) -> dict:
    """Slice a full Llama state dict into rank `rank`'s TP shard.

    Column-parallel: qkv_proj (by head), gate_up_proj (gate and up halves
    separately). Row-parallel: o_proj, down_proj (input dim). Norms,
    embeddings and lm_head are replicated.
    """
    out = {}
    q = cfg.num_heads * cfg.head_dim
    kv = cfg.num_kv_heads * cfg.head_dim
    ql, kvl = q // tp, kv // tp
    il = cfg.intermediate_size // tp
    for name, w in sd.items():
        if name.endswith("qkv_proj.bias"):
            qs = w[rank * ql:(rank + 1) * ql]
            ks = w[q + rank * kvl: q + (rank + 1) * kvl]
            vs = w[q + kv + rank * kvl: q + kv + (rank + 1) * kvl]
            out[name] = torch.cat([qs, ks, vs], dim=0).contiguous()
        elif name.endswith("qkv_proj.weight"):
            qs = w[rank * ql:(rank + 1) * ql]
            ks = w[q + rank * kvl: q + (rank + 1) * kvl]
            vs = w[q + kv + rank * kvl: q + kv + (rank + 1) * kvl]
            out[name] = torch.cat([qs, ks, vs], dim=0).contiguous()
        elif name.endswith("o_proj.weight"):
            out[name] = w[:, rank * ql:(rank + 1) * ql].contiguous()
        elif name.endswith("gate_up_proj.weight"):
            i = cfg.intermediate_size
            g = w[rank * il:(rank + 1) * il]
            u = w[i + rank * il: i + (rank + 1) * il]
            out[name] = torch.cat([g, u], dim=0).contiguous()
        elif name.endswith("down_proj.weight"):
            out[name] = w[:, rank * il:(rank + 1) * il].contiguous()
        else:
            out[name] = w
    return out
