"""fp8 weight quantization (W8A8 e4m3 with epilogue dequant), CPU."""
import pytest
import torch

import helix_amd.ops as ops
from helix_amd.engine.engine import EngineConfig, LLMEngine
from helix_amd.engine.sampling_params import SamplingParams
from helix_amd.models.llama import LlamaForCausalLM, PRESETS
from helix_amd.models.quant import FP8Linear, quantize_model_fp8


def test_quantize_fp8_roundtrip():
    torch.manual_seed(0)
    w = torch.randn(16, 128) * 0.1
    w8, ws = ops.quantize_fp8(w)
    back = w8.view(torch.float8_e4m3fn).float() * ws[:, None]
    rel = (back - w).norm() / w.norm()
    assert rel < 0.04                    # e4m3 has ~2 mantissa-ish digits
    # scales are per-row: scaling one row scales only its quantized row
    w2 = w.clone()
    w2[3] *= 100
    _, ws2 = ops.quantize_fp8(w2)
    assert ws2[3] > ws[3] * 50
    assert torch.allclose(ws2[0], ws[0])


def test_fp8_linear_matches_reference():
    torch.manual_seed(1)
    lin = torch.nn.Linear(256, 64, bias=False)
    q = FP8Linear.from_linear(lin)
    x = torch.randn(8, 256)
    got = q(x).float()
    want = lin(x).float()
    rel = (got - want).norm() / want.norm()
    assert rel < 0.05


def test_quantize_model_swaps_eligible_linears():
    model = LlamaForCausalLM(PRESETS["tiny"])
    n = quantize_model_fp8(model)
    # per layer: qkv, o, gate_up, down = 4; lm_head skipped
    assert n == 4 * PRESETS["tiny"].num_layers
    assert isinstance(model.layers[0].attn.qkv_proj, FP8Linear)
    assert isinstance(model.lm_head, torch.nn.Linear)


def test_engine_generates_quantized():
    """End-to-end: the fp8-quantized engine generates coherent greedy
    output whose tokens mostly agree with the bf16 engine (weights are
    identical up to e4m3 rounding)."""
    cfg = dict(model="tiny-gqa", max_model_len=256, max_num_seqs=4,
               kv_cache_blocks=128, eos_token_id=-1, seed=5)
    sp = SamplingParams(temperature=0.0, max_tokens=12, ignore_eos=True)
    prompts = [[1, 2, 3, 4, 5]]
    full = LLMEngine(EngineConfig(**cfg), device="cpu")
    want = full.generate(prompts, sp)[0]
    quant = LLMEngine(EngineConfig(**cfg, quantization="fp8"), device="cpu")
    got = quant.generate(prompts, sp)[0]
    assert len(got) == 12
    agree = sum(a == b for a, b in zip(got, want))
    assert agree >= 6, f"fp8 diverged early: {got} vs {want}"


def test_estimate_halves_fp8_weights():
    from helix_amd.runner.service import ModelSpec, estimate_model_bytes
    bf16 = estimate_model_bytes(ModelSpec("m", "llm", "llama3-8b"))
    fp8 = estimate_model_bytes(ModelSpec("m", "llm", "llama3-8b",
                                         quantization="fp8"))
    assert fp8 < bf16
    # projection weights are ~13.4 GiB of the 8B model; halving saves ~6.7
    assert (bf16 - fp8) > 5 << 30


def test_eval_quant_harness_runs(tmp_path):
    """The accuracy harness produces the expected report shape."""
    import json
    import subprocess
    import sys
    r = subprocess.run(
        [sys.executable, "scripts/eval_quant.py", "--prompts", "3",
         "--steps", "6", "--model", "tiny-gqa"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    rep = json.loads(r.stdout)
    assert set(rep["variants"]) == {"bf16", "fp8_weights", "fp8_kv",
                                    "fp8_both"}
    assert rep["variants"]["fp8_weights"]["mean_kl"] < 0.05
