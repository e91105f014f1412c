#include "hip/hip_runtime.h"
// Flash-attention prefill (varlen, causal, GQA) for CDNA4 / gfx950.
//
// Replaces the prefill attention the reference delegates to vLLM
// (SURVEY.md §2.8 "Prefill attention"). MFMA f32_16x16x32_bf16 tiles,
// LDS-staged K/V with XOR swizzle (guide §6 G4), online softmax in
// registers, fp32 accumulation.
//
// Geometry: 256-thread block = 4 waves. Q-tile 64 rows (16/wave),
// KV-tile 32 tokens. grid = (cdiv(max_len, 64), batch, Hq).
//
// MFMA fragment layouts (gfx950, f32_16x16x32_bf16):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + i]      (bf16x8)
//   B: lane l holds B[k = (l>>4)*8 + i][col = l&15]      (bf16x8)
//   C/D: lane l holds C[row = (l>>4)*4 + r][col = l&15]  (floatx4)
// Verified on hardware by tests/test_ops_gpu.py::test_mfma_probe with
// asymmetric operands (guide §3 "Always A=I-check with ASYMMETRIC B").
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int QTILE = 64;     // q rows per block
constexpr int KTILE = 32;     // kv tokens per tile
constexpr int DMAX = 128;
constexpr int P_PITCH = 40;   // p_lds row pitch (elements) — bank spread

__device__ __forceinline__ int swz(int tok, int byte_in_row) {
  // XOR-swizzle within a 256B row: spread rows across 16B slots.
  return tok * (DMAX * 2) + (byte_in_row ^ ((tok & 7) << 4));
}

__global__ __launch_bounds__(256) void attn_prefill_kernel(
    uint16_t* __restrict__ out,        // [T, Hq, D]
    const uint16_t* __restrict__ q,    // [T, Hq, D]
    const uint16_t* __restrict__ k,    // [T, Hkv, D]
    const uint16_t* __restrict__ v,    // [T, Hkv, D]
    const int* __restrict__ cu_seqlens,  // [B+1]
    float scale, int Hq, int Hkv, int D, int causal) {
  const int qtile = blockIdx.x;
  const int seq = blockIdx.y;
  const int hq = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int seq_start = cu_seqlens[seq];
  const int len = cu_seqlens[seq + 1] - seq_start;
  const int qbase = qtile * QTILE;
  if (qbase >= len) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int lane_hi = lane >> 4;   // 0..3
  const int lane_lo = lane & 15;   // 0..15

  __shared__ uint16_t k_lds[KTILE * DMAX];
  __shared__ uint16_t v_lds[KTILE * DMAX];
  __shared__ uint16_t p_lds[4][16 * P_PITCH];

  // ---- Load Q fragments for this wave's 16 rows (A-operand layout) ----
  const int my_qrow = qbase + wid * 16 + lane_lo;
  const int nkt = D / 32;  // K-dim subtiles for QK^T (D=128 -> 4)
  bf16x8 q_frag[DMAX / 32];
#pragma unroll
  for (int kt = 0; kt < DMAX / 32; ++kt) {
    if (kt < nkt && my_qrow < len) {
      const uint16_t* src =
          q + ((int64_t)(seq_start + my_qrow) * Hq + hq) * D + kt * 32 +
          lane_hi * 8;
      u16x8 raw = *reinterpret_cast<const u16x8*>(src);
      q_frag[kt] = *reinterpret_cast<bf16x8*>(&raw);
    } else {
      q_frag[kt] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // Softmax state per lane: 4 rows (r = 0..3 -> qrow = qbase+wid*16+lane_hi*4+r)
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.f;
  }
  floatx4 o_acc[DMAX / 16];
#pragma unroll
  for (int c = 0; c < DMAX / 16; ++c) o_acc[c] = floatx4{0, 0, 0, 0};
  const int nc = D / 16;

  const int kv_max = causal ? min(len, qbase + QTILE) : len;
  const int ntiles = (kv_max + KTILE - 1) / KTILE;

  for (int t = 0; t < ntiles; ++t) {
    const int kv_base = t * KTILE;
    __syncthreads();  // previous tile's compute done before overwrite
    // ---- Stage K and V tiles into LDS (swizzled) ----
    for (int c = tid; c < KTILE * (D / 8); c += 256) {
      const int tok = c / (D / 8);
      const int dgrp = c % (D / 8);
      const int gtok = kv_base + tok;
      u16x8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
      if (gtok < len) {
        const int64_t off =
            ((int64_t)(seq_start + gtok) * Hkv + hkv) * D + dgrp * 8;
        kv = *reinterpret_cast<const u16x8*>(k + off);
        vv = *reinterpret_cast<const u16x8*>(v + off);
      }
      *reinterpret_cast<u16x8*>((char*)k_lds + swz(tok, dgrp * 16)) = kv;
      *reinterpret_cast<u16x8*>((char*)v_lds + swz(tok, dgrp * 16)) = vv;
    }
    __syncthreads();

    // ---- QK^T for the two 16-token halves ----
    floatx4 s_frag[2];
    float p_val[2][4];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      s_frag[h] = floatx4{0, 0, 0, 0};
      const int tok = h * 16 + lane_lo;
      for (int kt = 0; kt < nkt; ++kt) {
        const int byte = (kt * 32 + lane_hi * 8) * 2;
        u16x8 raw =
            *reinterpret_cast<const u16x8*>((char*)k_lds + swz(tok, byte));
        s_frag[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kt], *reinterpret_cast<bf16x8*>(&raw), s_frag[h], 0, 0, 0);
      }
    }

    // ---- Online softmax ----
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qpos = qbase + wid * 16 + lane_hi * 4 + r;
      float s0 = s_frag[0][r] * scale;
      float s1 = s_frag[1][r] * scale;
      const int t0 = kv_base + lane_lo, t1 = kv_base + 16 + lane_lo;
      if ((causal && t0 > qpos) || t0 >= len) s0 = -1e30f;
      if ((causal && t1 > qpos) || t1 >= len) s1 = -1e30f;
      float m_tile = fmaxf(s0, s1);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        m_tile = fmaxf(m_tile, __shfl_xor(m_tile, off, WAVE));
      const float m_new = fmaxf(m_run[r], m_tile);
      corr[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
      const float p0 = __expf(s0 - m_new);
      const float p1 = __expf(s1 - m_new);
      float l_tile = p0 + p1;
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        l_tile += __shfl_xor(l_tile, off, WAVE);
      l_run[r] = l_run[r] * corr[r] + l_tile;
      m_run[r] = m_new;
      p_val[0][r] = p0;
      p_val[1][r] = p1;
    }

    // ---- P -> LDS (per-wave buffer), then PV ----
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds[wid][(lane_hi * 4 + r) * P_PITCH + h * 16 + lane_lo] =
            f32_to_bf16(p_val[h][r]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    bf16x8 p_frag;
    {
      u16x8 raw = *reinterpret_cast<const u16x8*>(
          &p_lds[wid][lane_lo * P_PITCH + lane_hi * 8]);
      p_frag = *reinterpret_cast<bf16x8*>(&raw);
    }
    for (int c = 0; c < nc; ++c) {
      // B-operand: V[tok = lane_hi*8 + i][dim = c*16 + lane_lo]
      u16x8 vraw;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int tok = lane_hi * 8 + i;
        vraw[i] = *reinterpret_cast<const uint16_t*>(
            (char*)v_lds + swz(tok, (c * 16 + lane_lo) * 2));
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[c][r] *= corr[r];
      o_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          p_frag, *reinterpret_cast<bf16x8*>(&vraw), o_acc[c], 0, 0, 0);
    }
  }

  // ---- Epilogue: normalize and write ----
  for (int c = 0; c < nc; ++c) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qbase + wid * 16 + lane_hi * 4 + r;
      if (qrow < len) {
        out[((int64_t)(seq_start + qrow) * Hq + hq) * D + c * 16 + lane_lo] =
            f32_to_bf16(o_acc[c][r] / fmaxf(l_run[r], 1e-20f));
      }
    }
  }
}

}  // namespace

void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, torch::Tensor cu_seqlens,
                  int64_t max_seqlen, double scale, bool causal) {
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k.size(1);
  const int B = cu_seqlens.size(0) - 1;
  TORCH_CHECK(D % 32 == 0 && D <= DMAX, "head_dim must be mult of 32, <=128");
  TORCH_CHECK(cu_seqlens.scalar_type() == torch::kInt32);
  auto stream = at::hip::getCurrentHIPStream();
  const int n_qtiles = cdiv((int)max_seqlen, QTILE);
  hipLaunchKernelGGL(attn_prefill_kernel, dim3(n_qtiles, B, Hq), dim3(256), 0,
                     stream, (uint16_t*)out.data_ptr(),
                     (const uint16_t*)q.data_ptr(),
                     (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(),
                     cu_seqlens.data_ptr<int>(), (float)scale, Hq, Hkv, D,
                     causal ? 1 : 0);
}
