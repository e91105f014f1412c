#!/usr/bin/env python3
"""Flagship serving benchmark: Llama-3-8B bf16 local-runner decode throughput.

Measures the BASELINE.json headline metric ("tokens/sec + p50 TTFT,
Llama-3-8B local runner") on synthetic data / random-init weights.

Single GPU:   python bench.py --gpus 1 --steps 32 --warmup 8
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...
Parallelism: data-parallel replicas (one engine per GPU, the reference's
round-robin-across-runners layout) => weak scaling.

One timed "step" = one continuous-batching decode iteration over the
steady-state batch (BATCH sequences), i.e. BATCH new tokens per GPU.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--model", type=str, default="llama3-8b")
    ap.add_argument("--parallelism", type=str, default="dp",
                    choices=["dp", "tp"],
                    help="dp: one engine per GPU (weak scaling); "
                         "tp: one tensor-parallel engine over all GPUs")
    ap.add_argument("--quant", type=str, default=None,
                    choices=[None, "fp8"],
                    help="fp8: e4m3 W8A8 projections (opt-in)")
    ap.add_argument("--kv-dtype", type=str, default="bf16",
                    choices=["bf16", "fp8"],
                    help="fp8: e4m3 KV cache (opt-in)")
    args = ap.parse_args()

    import torch.distributed as dist
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams

    distributed = args.gpus > 1 or "RANK" in os.environ
    rank, world = 0, 1
    if distributed:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group("nccl")
        rank = dist.get_rank()
        world = dist.get_world_size()
        torch.cuda.set_device(rank % torch.cuda.device_count())
    device = f"cuda:{rank % max(1, torch.cuda.device_count())}"
    tp = args.parallelism == "tp" and world > 1
    if tp:
        from helix_amd import parallel
        parallel.init_tp(world)

    max_len = args.prompt_len + args.warmup + args.steps + 64
    cfg = EngineConfig(
        model=args.model,
        max_num_seqs=args.batch,
        max_model_len=max_len,
        # chunked prefill admission (16k-token waves): earlier requests
        # start decoding sooner => TTFT p50 ~ halves vs one giant prefill
        max_prefill_tokens=16384,
        eos_token_id=-1,
        seed=rank,
        quantization=args.quant,
        kv_cache_dtype=args.kv_dtype,
    )
    if tp:
        # SPMD tensor parallelism: every rank runs the identical engine
        # loop; the model's row-parallel all-reduces (RCCL over xGMI) are
        # the only cross-rank communication. Identical seeds keep
        # scheduling and sampling converged across ranks.
        cfg.seed = 0
        cfg.enforce_eager = True
        eng = LLMEngine(cfg, device=device, tp_size=world, tp_rank=rank)
    else:
        eng = LLMEngine(cfg, device=device)

    torch.manual_seed(1234 + (0 if tp else rank))
    vocab = eng.model_cfg.vocab_size
    prompts = torch.randint(3, vocab - 1,
                            (args.batch, args.prompt_len)).tolist()
    sp = SamplingParams(temperature=0.0, max_tokens=10 ** 9, ignore_eos=True)
    for i, p in enumerate(prompts):
        eng.add_request(f"bench-{i}", p, sp)

    # Prefill (untimed) — also yields TTFT for every request.
    t0 = time.monotonic()
    while eng.waiting:
        eng.step()
    torch.cuda.synchronize()
    prefill_s = time.monotonic() - t0
    ttfts = sorted((s.first_token_time - s.arrival_time) * 1000
                   for s in eng.seqs.values())
    ttft_p50 = statistics.median(ttfts)

    # Warmup decode steps (untimed).
    for _ in range(args.warmup):
        eng.step()
    torch.cuda.synchronize()
    if distributed:
        dist.barrier()

    # Timed region: exactly --steps decode iterations.
    t0 = time.monotonic()
    for _ in range(args.steps):
        eng.step()
    torch.cuda.synchronize()
    elapsed = time.monotonic() - t0
    if distributed:
        dist.barrier()
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens = args.batch * args.steps * (1 if tp else world)
    value = tokens / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "strong" if tp else "weak",
            "vs_baseline": None,
            "dtype": ("fp8-w8a8" if args.quant == "fp8" else "bf16"),
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * world,
                "seq_len": args.prompt_len,
                "parallelism": (f"tp{world}" if tp else f"dp{world}"),
                "ttft_p50_ms": round(ttft_p50, 2),
                "prefill_s": round(prefill_s, 3),
                "kv_cache_dtype": args.kv_dtype,
            },
        }))
    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
