#!/usr/bin/env python3
"""Quantization accuracy harness: bf16 vs fp8 weights vs fp8 KV.

Measures, over N random prompts on the same random-init model:
  - mean KL(bf16 logits || variant logits) at each decode step
  - greedy top-1 agreement rate
  - max logit divergence

python scripts/eval_quant.py [--model tiny-gqa] [--prompts 16]
                             [--steps 24] [--device cpu|cuda:0]
(8B on a GPU box: --model llama3-8b --device cuda:0)
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from helix_amd.engine.engine import EngineConfig, LLMEngine  # noqa: E402
from helix_amd.engine.sampling_params import SamplingParams  # noqa: E402


def run_variant(name, base_cfg, device, prompts, steps, **over):
    eng = LLMEngine(EngineConfig(**base_cfg, **over), device=device)
    sp = SamplingParams(temperature=0.0, max_tokens=steps, ignore_eos=True)
    outs = eng.generate([list(p) for p in prompts], sp)
    # re-run prefill+decode capturing logits via compute_logits hook is
    # invasive; instead compare token streams + a one-shot logit probe
    # on the first prompt's prefill.
    del eng
    return outs


def logit_probe(base_cfg, device, prompt, **over):
    """Prefill logits for one prompt under a variant config."""
    eng = LLMEngine(EngineConfig(**base_cfg, **over), device=device)
    from helix_amd.models.llama import PrefillMeta
    ids = torch.tensor(prompt, dtype=torch.int64, device=device)
    T = len(prompt)
    meta = PrefillMeta(
        cu_seqlens=torch.tensor([0, T], dtype=torch.int32, device=device),
        max_seqlen=T,
        slot_mapping=torch.full((T,), -1, dtype=torch.int64, device=device),
        positions=torch.arange(T, device=device))
    with torch.inference_mode():
        hidden = eng.model(ids, None, meta)
        logits = eng.model.compute_logits(hidden).float().cpu()
    del eng
    return logits


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="tiny-gqa")
    ap.add_argument("--prompts", type=int, default=16)
    ap.add_argument("--steps", type=int, default=24)
    ap.add_argument("--prompt-len", type=int, default=32)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    from helix_amd.models.llama import PRESETS
    vocab = PRESETS[args.model].vocab_size
    g = torch.Generator().manual_seed(123)
    prompts = [torch.randint(1, vocab - 1, (args.prompt_len,),
                             generator=g).tolist()
               for _ in range(args.prompts)]
    base = dict(model=args.model, max_model_len=args.prompt_len +
                args.steps + 8, max_num_seqs=max(4, args.prompts),
                kv_cache_blocks=None if args.device.startswith("cuda")
                else 2048, eos_token_id=-1, seed=args.seed,
                enforce_eager=True)

    variants = {
        "bf16": {},
        "fp8_weights": {"quantization": "fp8"},
        "fp8_kv": {"kv_cache_dtype": "fp8"},
        "fp8_both": {"quantization": "fp8", "kv_cache_dtype": "fp8"},
    }
    ref_tokens = None
    ref_logits = logit_probe(base, args.device, prompts[0])
    report = {}
    for name, over in variants.items():
        outs = run_variant(name, base, args.device, prompts, args.steps,
                           **over)
        if name == "bf16":
            ref_tokens = outs
            report[name] = {"top1_agreement": 1.0, "mean_kl": 0.0}
            continue
        agree = total = 0
        for o, r in zip(outs, ref_tokens):
            for a, b in zip(o, r):
                agree += int(a == b)
                total += 1
        lg = logit_probe(base, args.device, prompts[0], **over)
        p = torch.log_softmax(ref_logits, -1)
        q = torch.log_softmax(lg, -1)
        kl = torch.sum(p.exp() * (p - q), dim=-1)
        report[name] = {
            "top1_agreement": round(agree / max(1, total), 4),
            "mean_kl": round(float(kl.mean()), 6),
            "max_kl": round(float(kl.max()), 6),
            "max_logit_absdiff": round(float(
                (ref_logits - lg).abs().max()), 4),
        }
    out = {"model": args.model, "device": args.device,
           "prompts": args.prompts, "steps": args.steps,
           "variants": report,
           "notes": [
               "random-init weights make top1_agreement pessimistic: "
               "near-uniform logits flip argmax under tiny perturbation; "
               "KL is the comparable signal (trained checkpoints have "
               "peaked distributions and far higher agreement)",
               "the logit probe is a cache-less prefill, so fp8_kv shows "
               "KL 0 there by construction — its effect is in the "
               "decode token stream",
           ]}
    print(json.dumps(out, indent=2))


if __name__ == "__main__":
    main()
