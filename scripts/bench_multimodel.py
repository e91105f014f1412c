#!/usr/bin/env python3
"""Multi-model co-residency demo (BASELINE config #4): Llama-3-8B +
Mistral-7B + bge-base co-resident on one MI355X under the HBM-aware
scheduler, then an oversize load (llama3-70b) forcing LRU eviction.
Prints one JSON line per phase with free-HBM telemetry."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from helix_amd.engine.sampling_params import SamplingParams
from helix_amd.runner.service import RunnerService
from helix_amd.utils.tokenizer import get_tokenizer


def free_gb():
    free, _ = torch.cuda.mem_get_info()
    return round(free / (1 << 30), 1)


def main():
    svc = RunnerService(device="cuda:0")
    tok = get_tokenizer()
    out = []

    for model in ["llama3-8b", "mistral-7b", "bge-base"]:
        t0 = time.time()
        inst = svc.ensure_loaded(model)
        load_s = time.time() - t0
        if model == "bge-base":
            texts = [f"document number {i} about GPUs and memory systems "
                     * 8 for i in range(64)]
            inst.embed(texts[:2])  # warmup
            t0 = time.time()
            vecs = inst.embed(texts)
            infer_s = time.time() - t0
            extra = {"embedding_dim": len(vecs[0]),
                     "texts_per_s": round(64 / infer_s, 1)}
        else:
            t0 = time.time()
            import threading
            done = threading.Event()
            toks = []
            inst.submit(f"demo-{model}",
                        tok.encode("Tell me about GPUs", add_bos=True),
                        SamplingParams(temperature=0.0, max_tokens=16,
                                       ignore_eos=True),
                        lambda s, t, f: (toks.append(t),
                                         done.set() if f else None))
            done.wait(timeout=120)
            infer_s = time.time() - t0
            extra = {"tokens": len(toks)}
        out.append({"phase": f"load+infer {model}", "load_s": round(load_s, 1),
                    "infer_s": round(infer_s, 2), "free_hbm_gb": free_gb(),
                    "loaded": svc.loaded_models(), **extra})
        print(json.dumps(out[-1]), flush=True)

    # Oversize load forces eviction of idle LRU models.
    t0 = time.time()
    svc.ensure_loaded("llama3-70b")
    out.append({"phase": "load llama3-70b (forces eviction)",
                "load_s": round(time.time() - t0, 1),
                "free_hbm_gb": free_gb(), "loaded": svc.loaded_models()})
    print(json.dumps(out[-1]), flush=True)
    svc.shutdown()


if __name__ == "__main__":
    main()
