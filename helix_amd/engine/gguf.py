"""GGUF v3 reader/writer: model loading and memory estimation.

The reference estimates GPU memory from GGUF tensor metadata
(api/pkg/memory/estimate.go) and serves GGUF checkpoints through
llama.cpp containers; here GGUF is a first-class *input format* for the
native engine: we parse the header, map llama.cpp tensor names onto our
fused-projection layout (models/llama.py), un-permute the Q/K rope
reordering applied by llama.cpp's HF converter, and dequantize
F32/F16/BF16/Q8_0/Q4_0 tensors straight into bf16 device weights.

Quantized *serving* (keeping Q-weights resident and dequantizing in the
GEMM) is a round-2 item; this module makes GGUF checkpoints loadable and
sizeable today.
"""
from __future__ import annotations

import struct
from typing import Any, BinaryIO, Dict, List, NamedTuple, Optional, Tuple

import numpy as np
import torch

GGUF_MAGIC = b"GGUF"

# -- metadata value types (gguf spec) ----------------------------------
_T_U8, _T_I8, _T_U16, _T_I16, _T_U32, _T_I32 = 0, 1, 2, 3, 4, 5
_T_F32, _T_BOOL, _T_STRING, _T_ARRAY, _T_U64, _T_I64, _T_F64 = \
    6, 7, 8, 9, 10, 11, 12
_SCALAR_FMT = {
    _T_U8: "<B", _T_I8: "<b", _T_U16: "<H", _T_I16: "<h",
    _T_U32: "<I", _T_I32: "<i", _T_F32: "<f", _T_U64: "<Q",
    _T_I64: "<q", _T_F64: "<d",
}

# -- ggml tensor types: id -> (name, elems/block, bytes/block) ---------
GGML_TYPES: Dict[int, Tuple[str, int, int]] = {
    0: ("F32", 1, 4),
    1: ("F16", 1, 2),
    2: ("Q4_0", 32, 18),
    3: ("Q4_1", 32, 20),
    6: ("Q5_0", 32, 22),
    7: ("Q5_1", 32, 24),
    8: ("Q8_0", 32, 34),
    9: ("Q8_1", 32, 36),
    10: ("Q2_K", 256, 84),
    11: ("Q3_K", 256, 110),
    12: ("Q4_K", 256, 144),
    13: ("Q5_K", 256, 176),
    14: ("Q6_K", 256, 210),
    15: ("Q8_K", 256, 292),
    24: ("I8", 1, 1),
    25: ("I16", 1, 2),
    26: ("I32", 1, 4),
    27: ("I64", 1, 8),
    28: ("F64", 1, 8),
    30: ("BF16", 1, 2),
}
_NAME_TO_GGML = {v[0]: k for k, v in GGML_TYPES.items()}


class GGUFTensorInfo(NamedTuple):
    name: str
    shape: Tuple[int, ...]   # torch order ([out, in]); gguf stores reversed
    ggml_type: int
    offset: int              # relative to the aligned data section

    @property
    def numel(self) -> int:
        n = 1
        for d in self.shape:
            n *= d
        return n

    @property
    def nbytes(self) -> int:
        _, blk, bsz = GGML_TYPES[self.ggml_type]
        return (self.numel // blk) * bsz

    @property
    def type_name(self) -> str:
        return GGML_TYPES[self.ggml_type][0]


def _read_str(f: BinaryIO) -> str:
    (n,) = struct.unpack("<Q", f.read(8))
    return f.read(n).decode("utf-8")


def _read_value(f: BinaryIO, vtype: int) -> Any:
    if vtype in _SCALAR_FMT:
        fmt = _SCALAR_FMT[vtype]
        (v,) = struct.unpack(fmt, f.read(struct.calcsize(fmt)))
        return v
    if vtype == _T_BOOL:
        return f.read(1) != b"\x00"
    if vtype == _T_STRING:
        return _read_str(f)
    if vtype == _T_ARRAY:
        (etype,) = struct.unpack("<I", f.read(4))
        (count,) = struct.unpack("<Q", f.read(8))
        return [_read_value(f, etype) for _ in range(count)]
    raise ValueError(f"unknown gguf metadata type {vtype}")


class GGUFFile:
    """Parsed GGUF container: metadata dict + tensor directory.

    Tensor data is read lazily per-tensor (mmap-free sequential reads),
    so estimation never touches the multi-GB payload.
    """

    def __init__(self, path: str):
        self.path = path
        self.metadata: Dict[str, Any] = {}
        self.tensors: Dict[str, GGUFTensorInfo] = {}
        with open(path, "rb") as f:
            if f.read(4) != GGUF_MAGIC:
                raise ValueError(f"{path}: not a GGUF file")
            (self.version,) = struct.unpack("<I", f.read(4))
            if self.version < 2:
                raise ValueError(f"GGUF v{self.version} unsupported (<2)")
            n_tensors, n_kv = struct.unpack("<QQ", f.read(16))
            for _ in range(n_kv):
                key = _read_str(f)
                (vtype,) = struct.unpack("<I", f.read(4))
                self.metadata[key] = _read_value(f, vtype)
            for _ in range(n_tensors):
                name = _read_str(f)
                (nd,) = struct.unpack("<I", f.read(4))
                dims = struct.unpack(f"<{nd}Q", f.read(8 * nd))
                gtype, = struct.unpack("<I", f.read(4))
                (off,) = struct.unpack("<Q", f.read(8))
                if gtype not in GGML_TYPES:
                    raise ValueError(f"{name}: unknown ggml type {gtype}")
                # gguf dims are fastest-first; torch shape is the reverse
                self.tensors[name] = GGUFTensorInfo(
                    name, tuple(reversed(dims)), gtype, off)
            align = int(self.metadata.get("general.alignment", 32))
            pos = f.tell()
            self.data_start = (pos + align - 1) // align * align

    # -- sizing ---------------------------------------------------------
    def tensor_bytes(self) -> int:
        return sum(t.nbytes for t in self.tensors.values())

    def arch(self) -> Optional[str]:
        return self.metadata.get("general.architecture")

    # -- data -----------------------------------------------------------
    def read_raw(self, name: str) -> bytes:
        info = self.tensors[name]
        with open(self.path, "rb") as f:
            f.seek(self.data_start + info.offset)
            return f.read(info.nbytes)

    def load_tensor(self, name: str) -> torch.Tensor:
        """Dequantize one tensor to a torch tensor (fp32/bf16 source
        dtype preserved; Q-types dequantized to fp32)."""
        info = self.tensors[name]
        raw = self.read_raw(name)
        t = _dequantize(raw, info)
        return t.reshape(info.shape)


def _dequantize(raw: bytes, info: GGUFTensorInfo) -> torch.Tensor:
    tname = info.type_name
    if tname == "F32":
        return torch.from_numpy(
            np.frombuffer(raw, dtype="<f4").copy())
    if tname == "F16":
        return torch.from_numpy(
            np.frombuffer(raw, dtype="<f2").copy()).float()
    if tname == "BF16":
        u16 = np.frombuffer(raw, dtype="<u2").copy()
        return torch.from_numpy(u16).view(torch.bfloat16)
    if tname == "Q8_0":
        # block = f16 scale d + 32 int8 q; x = d * q
        blk = np.frombuffer(raw, dtype=np.uint8).reshape(-1, 34)
        d = blk[:, :2].copy().view("<f2").astype(np.float32)
        qs = blk[:, 2:].copy().view(np.int8).astype(np.float32)
        return torch.from_numpy((d * qs).reshape(-1))
    if tname == "Q4_0":
        # block = f16 scale d + 16 nibble bytes; elem j in low nibbles,
        # elem j+16 in high nibbles; x = d * (nib - 8)
        blk = np.frombuffer(raw, dtype=np.uint8).reshape(-1, 18)
        d = blk[:, :2].copy().view("<f2").astype(np.float32)
        qs = blk[:, 2:]
        lo = (qs & 0x0F).astype(np.float32) - 8.0
        hi = (qs >> 4).astype(np.float32) - 8.0
        out = np.concatenate([lo, hi], axis=1) * d
        return torch.from_numpy(out.reshape(-1))
    raise NotImplementedError(
        f"dequantize {tname}: only F32/F16/BF16/Q8_0/Q4_0 supported "
        f"(estimation supports all types)")


# -- llama.cpp -> helix_amd name mapping -------------------------------

def map_gguf_name(name: str):
    """llama.cpp tensor name -> (our param name, fuse-role)."""
    fixed = {
        "token_embd.weight": ("embed_tokens.weight", None),
        "output_norm.weight": ("final_norm_w", None),
        "output.weight": ("lm_head.weight", None),
    }
    if name in fixed:
        return fixed[name]
    if name.startswith("blk."):
        parts = name.split(".")
        i = parts[1]
        rest = ".".join(parts[2:])
        m = {
            "attn_norm.weight": (f"layers.{i}.input_norm_w", None),
            "ffn_norm.weight": (f"layers.{i}.post_norm_w", None),
            "attn_q.weight": (f"layers.{i}.attn.qkv_proj.weight", "q"),
            "attn_k.weight": (f"layers.{i}.attn.qkv_proj.weight", "k"),
            "attn_v.weight": (f"layers.{i}.attn.qkv_proj.weight", "v"),
            "attn_output.weight": (f"layers.{i}.attn.o_proj.weight", None),
            "ffn_gate.weight": (f"layers.{i}.mlp.gate_up_proj.weight",
                                "gate"),
            "ffn_up.weight": (f"layers.{i}.mlp.gate_up_proj.weight", "up"),
            "ffn_down.weight": (f"layers.{i}.mlp.down_proj.weight", None),
        }
        if rest in m:
            return m[rest]
    return name, None


def unpermute_rope(w: torch.Tensor, n_head: int) -> torch.Tensor:
    """Invert the Q/K row permutation llama.cpp's HF converter applies.

    The converter reorders each head's rows (n_head, 2, dh/2, in) ->
    swapaxes(1,2) so GGML's interleaved rope matches HF's half-rotation.
    Our rope is HF half-rotation (ops/hip/rope.hip), so loading GGUF we
    apply the inverse: (n_head, dh/2, 2, in) -> swapaxes(1,2).
    """
    out, inp = w.shape
    dh = out // n_head
    return (w.reshape(n_head, dh // 2, 2, inp)
            .swapaxes(1, 2).reshape(out, inp))


# -- estimation (reference estimate.go role) ---------------------------

def estimate_gguf_bytes(path: str, kv_tokens: int = 0,
                        kv_dtype_bytes: int = 2) -> Dict[str, int]:
    """Memory footprint of serving a GGUF model: weights as stored
    (quantized sizes honored, like the reference's GGUF estimator) plus
    optional KV for `kv_tokens` cached tokens."""
    g = GGUFFile(path)
    weights = g.tensor_bytes()
    arch = g.arch() or "llama"
    n_layer = int(g.metadata.get(f"{arch}.block_count", 0))
    n_kv_head = int(g.metadata.get(
        f"{arch}.attention.head_count_kv",
        g.metadata.get(f"{arch}.attention.head_count", 0)))
    n_embd = int(g.metadata.get(f"{arch}.embedding_length", 0))
    n_head = int(g.metadata.get(f"{arch}.attention.head_count", 1))
    head_dim = n_embd // max(1, n_head)
    kv = 2 * n_layer * n_kv_head * head_dim * kv_tokens * kv_dtype_bytes
    return {"weights": weights, "kv": kv, "total": weights + kv}


# -- writer (tests + checkpoint export) --------------------------------

def _write_str(f: BinaryIO, s: str):
    b = s.encode("utf-8")
    f.write(struct.pack("<Q", len(b)))
    f.write(b)


def _write_value(f: BinaryIO, v: Any):
    if isinstance(v, bool):
        f.write(struct.pack("<I", _T_BOOL))
        f.write(b"\x01" if v else b"\x00")
    elif isinstance(v, int):
        f.write(struct.pack("<I", _T_U32 if 0 <= v < 2**32 else _T_I64))
        f.write(struct.pack("<I" if 0 <= v < 2**32 else "<q", v))
    elif isinstance(v, float):
        f.write(struct.pack("<I", _T_F32))
        f.write(struct.pack("<f", v))
    elif isinstance(v, str):
        f.write(struct.pack("<I", _T_STRING))
        _write_str(f, v)
    else:
        raise TypeError(f"unsupported metadata value {type(v)}")


def write_gguf(path: str, metadata: Dict[str, Any],
               tensors: Dict[str, torch.Tensor], align: int = 32):
    """Write a GGUF v3 file (F32/F16/BF16 payloads). Used by tests and
    `helix export-gguf`; quantized writing is out of scope."""
    infos: List[Tuple[str, torch.Tensor, int, int]] = []
    off = 0
    for name, t in tensors.items():
        t = t.detach().cpu().contiguous()
        if t.dtype == torch.float32:
            gt = _NAME_TO_GGML["F32"]
        elif t.dtype == torch.float16:
            gt = _NAME_TO_GGML["F16"]
        elif t.dtype == torch.bfloat16:
            gt = _NAME_TO_GGML["BF16"]
        else:
            raise TypeError(f"{name}: dtype {t.dtype} not writable")
        nbytes = t.numel() * t.element_size()
        infos.append((name, t, gt, off))
        off += (nbytes + align - 1) // align * align
    with open(path, "wb") as f:
        f.write(GGUF_MAGIC)
        f.write(struct.pack("<I", 3))
        meta = dict(metadata)
        meta.setdefault("general.alignment", align)
        f.write(struct.pack("<QQ", len(infos), len(meta)))
        for k, v in meta.items():
            _write_str(f, k)
            _write_value(f, v)
        for name, t, gt, o in infos:
            _write_str(f, name)
            dims = tuple(reversed(t.shape)) if t.dim() else (1,)
            f.write(struct.pack("<I", len(dims)))
            f.write(struct.pack(f"<{len(dims)}Q", *dims))
            f.write(struct.pack("<I", gt))
            f.write(struct.pack("<Q", o))
        pos = f.tell()
        f.write(b"\x00" * ((pos + align - 1) // align * align - pos))
        for name, t, gt, o in infos:
            if t.dtype == torch.bfloat16:
                buf = t.view(torch.uint16).numpy().tobytes()
            else:
                buf = t.numpy().tobytes()
            f.write(buf)
            pad = (len(buf) + align - 1) // align * align - len(buf)
            f.write(b"\x00" * pad)


@torch.inference_mode()
def load_gguf_weights(model, path: str) -> int:
    """Load a GGUF llama checkpoint into a helix_amd Llama model
    (dequantizing to the model dtype; Q/K rope rows un-permuted)."""
    g = GGUFFile(path)
    params = dict(model.named_parameters())
    cfg = model.cfg
    q, kv = cfg.q_size, cfg.kv_size
    inter = cfg.intermediate_size
    loaded = 0
    for name in g.tensors:
        our, role = map_gguf_name(name)
        if our not in params:
            continue
        t = g.load_tensor(name)
        if role == "q":
            t = unpermute_rope(t, cfg.num_heads)
        elif role == "k":
            t = unpermute_rope(t, cfg.num_kv_heads)
        p = params[our]
        t = t.to(p.dtype)
        if role == "q":
            p.data[:q].copy_(t)
        elif role == "k":
            p.data[q:q + kv].copy_(t)
        elif role == "v":
            p.data[q + kv:].copy_(t)
        elif role == "gate":
            p.data[:inter].copy_(t)
        elif role == "up":
            p.data[inter:].copy_(t)
        else:
            p.data.copy_(t)
        loaded += 1
    return loaded
