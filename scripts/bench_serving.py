#!/usr/bin/env python3
"""Serving-QoS benchmark: Poisson request arrivals against the engine,
reporting TTFT and inter-token (TPOT) percentiles plus throughput.

python scripts/bench_serving.py --rate 8 --requests 64 [--model llama3-8b]
"""
import argparse
import os
import random
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def pct(v, p):
    v = sorted(v)
    return v[min(len(v) - 1, int(len(v) * p))]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b",
                    help="comma-separated for co-resident multi-model")
    ap.add_argument("--rate", type=float, default=8.0, help="req/s")
    ap.add_argument("--requests", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--max-tokens", type=int, default=128)
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--shared-prefix", type=int, default=0,
                    help="tokens of a common system prompt shared by all "
                         "requests (exercises the prefix cache)")
    args = ap.parse_args()

    from helix_amd.runner.service import RunnerService
    from helix_amd.engine.sampling_params import SamplingParams
    svc = RunnerService(device=args.device)
    models = [m.strip() for m in args.model.split(",")]
    insts = [svc.ensure_loaded(m) for m in models]

    stats = {}
    lock = threading.Lock()
    done = threading.Event()
    finished = [0]

    def on_token_factory(rid, t_submit):
        token_times = []

        def cb(seq, tok, fin):
            token_times.append(time.monotonic())
            if fin:
                with lock:
                    stats[rid] = (t_submit, token_times)
                    finished[0] += 1
                    if finished[0] == args.requests:
                        done.set()
        return cb

    random.seed(0)
    torch.manual_seed(0)
    shared = {}
    t0 = time.monotonic()
    for i in range(args.requests):
        inst = insts[i % len(insts)]
        vocab = inst.engine.model_cfg.vocab_size
        if args.shared_prefix and id(inst) not in shared:
            shared[id(inst)] = [random.randrange(3, vocab - 1)
                                for _ in range(args.shared_prefix)]
        prefix = shared.get(id(inst), [])
        prompt = prefix + [random.randrange(3, vocab - 1)
                           for _ in range(args.prompt_len
                                          - len(prefix))]
        inst.submit(f"q{i}", prompt,
                    SamplingParams(temperature=0.0,
                                   max_tokens=args.max_tokens,
                                   ignore_eos=True),
                    on_token_factory(f"q{i}", time.monotonic()))
        time.sleep(random.expovariate(args.rate))
    done.wait(timeout=600)
    wall = time.monotonic() - t0

    ttfts, tpots = [], []
    total_tokens = 0
    for t_submit, times in stats.values():
        ttfts.append((times[0] - t_submit) * 1000)
        total_tokens += len(times)
        for a, b in zip(times, times[1:]):
            tpots.append((b - a) * 1000)
    import json
    print(json.dumps({
        "models": models, "shared_prefix": args.shared_prefix,
        "requests": len(stats), "rate_rps": args.rate,
        "prompt_len": args.prompt_len, "max_tokens": args.max_tokens,
        "wall_s": round(wall, 2),
        "throughput_tok_s": round(total_tokens / wall, 1),
        "ttft_ms": {"p50": round(pct(ttfts, 0.5), 1),
                    "p90": round(pct(ttfts, 0.9), 1),
                    "p99": round(pct(ttfts, 0.99), 1)},
        "tpot_ms": {"p50": round(pct(tpots, 0.5), 2),
                    "p90": round(pct(tpots, 0.9), 2),
                    "p99": round(pct(tpots, 0.99), 2)},
    }))
    svc.shutdown()


if __name__ == "__main__":
    main()
