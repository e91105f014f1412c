#!/usr/bin/env python3
"""Measure the image-generation engine on MI355X: flux-lite (298M DiT,
256px) images/s at a few batch sizes + tiny-dit sanity. Evidence for
profiles/ (the §2.8 image row); NOT the headline bench (bench.py owns
that contract).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch

from helix_amd.models.dit import DIT_PRESETS, DiffusionImageModel


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--preset", default="flux-lite")
    ap.add_argument("--steps", type=int, default=28)
    ap.add_argument("--batches", default="1,4,8")
    ap.add_argument("--reps", type=int, default=3)
    args = ap.parse_args()
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev.startswith("cuda") else torch.float32
    cfg = DIT_PRESETS[args.preset]
    m = DiffusionImageModel(cfg).to(dtype).to(dev).init_random(0)
    prompt = list(range(7, 7 + cfg.text_len))
    out = {"preset": args.preset, "image_size": cfg.image_size,
           "ode_steps": args.steps, "device": dev, "dtype": str(dtype),
           "params_m": round(sum(p.numel() for p in
                                 m.parameters()) / 1e6, 1),
           "results": []}
    for b in [int(x) for x in args.batches.split(",")]:
        prompts = [prompt] * b
        m.generate(prompts, steps=args.steps, seed=0)       # warmup
        if dev.startswith("cuda"):
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for r in range(args.reps):
            m.generate(prompts, steps=args.steps, seed=r + 1)
        if dev.startswith("cuda"):
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.reps
        out["results"].append({
            "batch": b, "s_per_batch": round(dt, 4),
            "images_per_s": round(b / dt, 2),
            "ms_per_image": round(1000 * dt / b, 1)})
        print(json.dumps(out["results"][-1]))
    print(json.dumps(out))
    return out


if __name__ == "__main__":
    main()
