"""Weights-quantized serving (fp8 e4m3, W8A8 with per-channel/per-token
scales).

The reference serves quantized checkpoints through vLLM/Ollama
quantization support; here quantization is a
post-load model pass: every eligible nn.Linear is swapped for an
FP8Linear holding e4m3 bytes + per-output-channel scales, and the
forward runs the hand-written 16x16x128 scaled-MFMA GEMM
(ops/hip/gemm_fp8.hip) with per-token activation quantization.

Why epilogue dequant instead of HW MX block scales: unity e8m0 scales
keep full per-channel accuracy without depending on the scale-lane
mapping (validated by tests/test_ops_gpu.py::test_mfma_probe_fp8_*),
and halve weight HBM (llama3-70b: 141 -> 71 GB resident).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from helix_amd import ops


class FP8Linear(nn.Module):
    """Drop-in nn.Linear replacement: e4m3 weights + fp32 out-channel
    scales; activations quantized per token at call time."""

    def __init__(self, weight8: torch.Tensor, w_scale: torch.Tensor):
        super().__init__()
        self.register_buffer("weight8", weight8, persistent=True)
        self.register_buffer("w_scale", w_scale, persistent=True)
        self.out_features, self.in_features = weight8.shape

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "FP8Linear":
        w8, ws = ops.quantize_fp8(lin.weight.data)
        return cls(w8.contiguous(), ws.contiguous())

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape[:-1]
        x2 = x.reshape(-1, self.in_features)
        x8, xs = ops.quantize_fp8(x2)
        out = ops.gemm_fp8(x8.contiguous(), self.weight8, xs.contiguous(),
                           self.w_scale)
        return out.reshape(*shape, self.out_features)


def quantize_model_fp8(model: nn.Module,
                       skip: tuple = ("lm_head",)) -> int:
    """Swap eligible Linears (in_features % 128 == 0, not in `skip`) for
    FP8Linear. Returns the number of modules converted."""
    converted = 0
    for name, mod in list(model.named_modules()):
        for child_name, child in list(mod.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if not isinstance(child, nn.Linear):
                continue
            if any(s in full for s in skip):
                continue
            if child.in_features % 128 != 0 or child.bias is not None:
                continue
            setattr(mod, child_name, FP8Linear.from_linear(child))
            converted += 1
    return converted
