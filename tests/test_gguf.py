"""GGUF reader/writer/loader tests (reference api/pkg/memory estimate.go
role + llama.cpp checkpoint interop), all CPU."""
import struct

import numpy as np
import pytest
import torch

from helix_amd.engine import gguf
from helix_amd.models.llama import LlamaForCausalLM, PRESETS


def _permute(w: torch.Tensor, n_head: int) -> torch.Tensor:
    """llama.cpp convert_hf_to_gguf.py Q/K permutation (forward)."""
    out, inp = w.shape
    return (w.reshape(n_head, 2, out // n_head // 2, inp)
            .swapaxes(1, 2).reshape(out, inp))


def test_roundtrip_metadata_and_tensors(tmp_path):
    path = str(tmp_path / "t.gguf")
    tensors = {
        "a.weight": torch.randn(4, 8, dtype=torch.float32),
        "b.weight": torch.randn(3, 5, dtype=torch.bfloat16),
        "c.weight": torch.randn(16, dtype=torch.float16),
    }
    meta = {"general.architecture": "llama", "llama.block_count": 2,
            "general.name": "tiny", "x.flag": True, "x.scale": 0.5}
    gguf.write_gguf(path, meta, tensors)
    g = gguf.GGUFFile(path)
    assert g.version == 3
    assert g.metadata["general.architecture"] == "llama"
    assert g.metadata["llama.block_count"] == 2
    assert g.metadata["x.flag"] is True
    assert abs(g.metadata["x.scale"] - 0.5) < 1e-6
    for name, t in tensors.items():
        assert g.tensors[name].shape == tuple(t.shape)
        back = g.load_tensor(name)
        assert torch.allclose(back.float(), t.float(), atol=1e-3)


def test_q8_0_dequant(tmp_path):
    """Hand-build a Q8_0 tensor and check dequantization math."""
    nblk = 4
    d = np.array([0.5, 1.0, 2.0, 0.25], dtype="<f2")
    qs = np.arange(-64, 64, dtype=np.int8).reshape(nblk, 32)
    raw = b"".join(d[i].tobytes() + qs[i].tobytes() for i in range(nblk))
    info = gguf.GGUFTensorInfo("x", (nblk * 32,), 8, 0)
    out = gguf._dequantize(raw, info)
    want = (d.astype(np.float32)[:, None] * qs.astype(np.float32)).reshape(-1)
    assert torch.allclose(out, torch.from_numpy(want))


def test_q4_0_dequant():
    d = np.array([2.0], dtype="<f2")
    nib = np.zeros(16, dtype=np.uint8)
    nib[0] = (9 << 4) | 1   # elem0 = 1-8 = -7, elem16 = 9-8 = 1
    raw = d.tobytes() + nib.tobytes()
    info = gguf.GGUFTensorInfo("x", (32,), 2, 0)
    out = gguf._dequantize(raw, info)
    assert out[0].item() == -14.0      # 2.0 * (1-8)
    assert out[16].item() == 2.0       # 2.0 * (9-8)
    assert out[1].item() == -16.0      # 2.0 * (0-8)


def test_estimate_gguf_bytes(tmp_path):
    path = str(tmp_path / "m.gguf")
    tensors = {"w": torch.randn(64, 64, dtype=torch.bfloat16)}
    meta = {"general.architecture": "llama", "llama.block_count": 4,
            "llama.embedding_length": 64,
            "llama.attention.head_count": 4,
            "llama.attention.head_count_kv": 2}
    gguf.write_gguf(path, meta, tensors)
    est = gguf.estimate_gguf_bytes(path, kv_tokens=100)
    assert est["weights"] == 64 * 64 * 2
    # 2 (K+V) * layers * kv_heads * head_dim * tokens * 2B
    assert est["kv"] == 2 * 4 * 2 * 16 * 100 * 2
    assert est["total"] == est["weights"] + est["kv"]


def test_load_gguf_llama_matches_native(tmp_path):
    """A converter-style GGUF (llama.cpp names + Q/K permute) loaded into
    our model must produce identical logits to the source weights."""
    torch.manual_seed(0)
    cfg = PRESETS["tiny"]
    src = LlamaForCausalLM(cfg)
    # export with llama.cpp naming and the converter's Q/K permutation
    q, kv = cfg.q_size, cfg.kv_size
    inter = cfg.intermediate_size
    tensors = {}
    sd = dict(src.named_parameters())
    tensors["token_embd.weight"] = sd["embed_tokens.weight"].data
    tensors["output_norm.weight"] = sd["final_norm_w"].data
    tensors["output.weight"] = sd["lm_head.weight"].data
    for i in range(cfg.num_layers):
        qkv = sd[f"layers.{i}.attn.qkv_proj.weight"].data
        gu = sd[f"layers.{i}.mlp.gate_up_proj.weight"].data
        tensors[f"blk.{i}.attn_q.weight"] = _permute(
            qkv[:q], cfg.num_heads)
        tensors[f"blk.{i}.attn_k.weight"] = _permute(
            qkv[q:q + kv], cfg.num_kv_heads)
        tensors[f"blk.{i}.attn_v.weight"] = qkv[q + kv:]
        tensors[f"blk.{i}.attn_output.weight"] = \
            sd[f"layers.{i}.attn.o_proj.weight"].data
        tensors[f"blk.{i}.ffn_gate.weight"] = gu[:inter]
        tensors[f"blk.{i}.ffn_up.weight"] = gu[inter:]
        tensors[f"blk.{i}.ffn_down.weight"] = \
            sd[f"layers.{i}.mlp.down_proj.weight"].data
        tensors[f"blk.{i}.attn_norm.weight"] = \
            sd[f"layers.{i}.input_norm_w"].data
        tensors[f"blk.{i}.ffn_norm.weight"] = \
            sd[f"layers.{i}.post_norm_w"].data
    path = str(tmp_path / "tiny.gguf")
    gguf.write_gguf(path, {"general.architecture": "llama"}, tensors)

    dst = LlamaForCausalLM(cfg)
    n = gguf.load_gguf_weights(dst, path)
    assert n == len(tensors)
    for name, p in src.named_parameters():
        assert torch.equal(p.data, dict(dst.named_parameters())[name].data), name


def test_bad_magic(tmp_path):
    p = tmp_path / "x.gguf"
    p.write_bytes(b"NOPE" + struct.pack("<I", 3))
    with pytest.raises(ValueError):
        gguf.GGUFFile(str(p))


def test_weights_loader_dispatches_gguf(tmp_path):
    """load_llama_weights auto-detects a GGUF checkpoint dir."""
    import subprocess
    import sys
    out = str(tmp_path / "tiny.gguf")
    r = subprocess.run(
        [sys.executable, "-m", "helix_amd.cli", "export-gguf",
         "--preset", "tiny", "-o", out, "--seed", "3"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    from helix_amd.engine.weights import load_llama_weights
    torch.manual_seed(0)
    src = LlamaForCausalLM(PRESETS["tiny"])
    src.init_random(3)
    dst = LlamaForCausalLM(PRESETS["tiny"])
    n = load_llama_weights(dst, out)
    assert n > 0
    for name, p in src.named_parameters():
        assert torch.equal(p.data, dict(dst.named_parameters())[name].data), name
