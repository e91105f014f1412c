#!/usr/bin/env python3
"""Flagship serving benchmark: Llama-3-8B bf16 local-runner decode throughput.

Measures the BASELINE.json headline metric ("tokens/sec + p50 TTFT,
Llama-3-8B local runner; models co-resident per GPU") on synthetic data /
random-init weights.

Single GPU:   python bench.py --gpus 1 --steps 200 --warmup 8
Co-resident:  python bench.py --coresident   (BASELINE config #4:
              Llama-3-8B + Mistral-7B decoding while bge-base embeds on a
              side HIP stream, all resident in one GPU's HBM)
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...
Parallelism: data-parallel replicas (one engine per GPU, the reference's
round-robin-across-runners layout) => weak scaling;
--parallelism tp shards one engine over all GPUs (RCCL + one-shot xGMI
allreduce, hipGraph-captured decode) => strong scaling.

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


def run_coresident(args) -> None:
    """BASELINE config #4: three models co-resident on one GPU. The two
    LLM engines decode every timed step; bge-base embeds a 16-text batch
    each step on its own HIP stream (overlapped, the MI355X-native way to
    serve mixed traffic). Reported value = total LLM tokens/s."""
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    from helix_amd.runner.service import DEFAULT_SPECS, EmbeddingInstance

    device = "cuda:0"
    b1, b2 = args.batch, max(32, args.batch // 4)
    max_len = args.prompt_len + args.warmup + args.steps + 64
    blocks = lambda b: b * (max_len // 16 + 2)  # noqa: E731

    engines = []
    for model, bsz in (("llama3-8b", b1), ("mistral-7b", b2)):
        cfg = EngineConfig(model=model, max_num_seqs=bsz,
                           max_model_len=max_len, max_prefill_tokens=16384,
                           eos_token_id=-1, seed=0,
                           kv_cache_blocks=blocks(bsz),
                           kv_cache_dtype=args.kv_dtype)
        engines.append(LLMEngine(cfg, device=device))
    bge = EmbeddingInstance(DEFAULT_SPECS["bge-base"], device)
    emb_texts = [f"document {i} about GPUs, HBM and xGMI topology" * 4
                 for i in range(16)]
    # Pre-tokenize once; the timed loop issues the raw encoder forward on
    # a side stream (EmbeddingInstance.embed fetches to host, which would
    # serialize the overlap we are measuring).
    ids_list = [bge.tokenizer.encode(t)[:bge.model.cfg.max_position - 1]
                for t in emb_texts]
    flat, cu = [], [0]
    for ids in ids_list:
        flat.extend(ids)
        cu.append(cu[-1] + len(ids))
    emb_ids = torch.tensor(flat, dtype=torch.int64, device=device)
    emb_cu = torch.tensor(cu, dtype=torch.int32, device=device)
    emb_maxlen = max(len(i) for i in ids_list)
    emb_stream = torch.cuda.Stream()

    free, total = torch.cuda.mem_get_info()
    resident_gb = round((total - free) / (1 << 30), 1)

    torch.manual_seed(1234)
    sp = SamplingParams(temperature=0.0, max_tokens=10 ** 9, ignore_eos=True)
    for eng, bsz in zip(engines, (b1, b2)):
        vocab = eng.model_cfg.vocab_size
        prompts = torch.randint(3, vocab - 1,
                                (bsz, args.prompt_len)).tolist()
        for i, p in enumerate(prompts):
            eng.add_request(f"bench-{eng.cfg.model}-{i}", p, sp)

    t0 = time.monotonic()
    for eng in engines:
        while eng.waiting:
            eng.step()
    torch.cuda.synchronize()
    prefill_s = time.monotonic() - t0
    ttfts = sorted((s.first_token_time - s.arrival_time) * 1000
                   for s in engines[0].seqs.values())
    ttft_p50 = statistics.median(ttfts)

    def one_step():
        with torch.cuda.stream(emb_stream):
            bge.model(emb_ids, emb_cu, emb_maxlen)
        for eng in engines:
            eng.step()

    for _ in range(args.warmup):
        one_step()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    elapsed = time.monotonic() - t0

    tokens = (b1 + b2) * args.steps
    print(json.dumps({
        "metric": "tokens/sec (co-resident)",
        "value": round(tokens / elapsed, 2),
        "unit": "tokens/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": "llama3-8b+mistral-7b+bge-base",
            "global_batch": b1 + b2,
            "seq_len": args.prompt_len,
            "parallelism": "coresident1",
            "ttft_p50_ms": round(ttft_p50, 2),
            "prefill_s": round(prefill_s, 3),
            "resident_hbm_gb": resident_gb,
            "embeds_per_step": len(emb_texts),
            "embed_texts_per_s": round(len(emb_texts) * args.steps /
                                       elapsed, 1),
            "per_model_batch": {"llama3-8b": b1, "mistral-7b": b2},
        },
    }))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--model", type=str, default="llama3-8b")
    ap.add_argument("--parallelism", type=str, default="dp",
                    choices=["dp", "tp"],
                    help="dp: one engine per GPU (weak scaling); "
                         "tp: one tensor-parallel engine over all GPUs")
    ap.add_argument("--coresident", action="store_true",
                    help="BASELINE config #4: 8B + 7B + bge-base on one GPU")
    ap.add_argument("--quant", type=str, default=None,
                    choices=[None, "fp8"],
                    help="fp8: e4m3 W8A8 projections (opt-in)")
    ap.add_argument("--kv-dtype", type=str, default="bf16",
                    choices=["bf16", "fp8"],
                    help="fp8: e4m3 KV cache (opt-in)")
    ap.add_argument("--temperature", type=float, default=0.0)
    ap.add_argument("--top-p", type=float, default=1.0)
    args = ap.parse_args()

    if args.coresident:
        run_coresident(args)
        return

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
        # loop; the model's row-parallel all-reduces are the only
        # cross-rank communication — one-shot xGMI kernel for the
        # decode-sized messages (hipGraph-capturable, so decode stays
        # graph-launched), RCCL rings for the large prefill messages.
        from helix_amd import parallel
        parallel.init_tp(world)
        cfg.seed = 0
        from helix_amd.models.llama import PRESETS
        hidden = PRESETS[args.model].hidden_size
        ar_ok = parallel.init_custom_allreduce(
            max(8 << 20, args.batch * hidden * 2))
        cfg.enforce_eager = not ar_ok
        eng = LLMEngine(cfg, device=device, tp_size=world, tp_rank=rank)
    else:
        eng = LLMEngine(cfg, device=device)

    torch.manual_seed(1234 + (0 if tp else rank))
    vocab = eng.model_cfg.vocab_size
    prompts = torch.randint(3, vocab - 1,
                            (args.batch, args.prompt_len)).tolist()
    sp = SamplingParams(temperature=args.temperature, top_p=args.top_p,
                        max_tokens=10 ** 9, ignore_eos=True)
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
                "temperature": args.temperature,
                "top_p": args.top_p,
            },
        }))
    if distributed:
        if tp:
            from helix_amd import parallel
            parallel.destroy_custom_allreduce()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
