// Flash-attention prefill (varlen, causal/bidirectional, GQA) for CDNA4.
//
// Replaces the prefill attention the reference delegates to vLLM
// (SURVEY.md §2.8 "Prefill attention"). v2 design:
//  - MFMA f32_16x16x32_bf16 tiles; 256-thread block = 4 waves; Q-tile 64
//    rows (16/wave); KV-tile 64 tokens.
//  - Async-stage split (guide §6 G15): tile t+1's K/V global loads issue
//    into registers BEFORE tile t's compute so HBM latency hides under
//    the MFMAs; registers are written to LDS after the barrier.
//  - K in LDS row-major with XOR row-swizzle (conflict-free b128 reads);
//    V stored TRANSPOSED [dim][tok] (scalar writes once per tile) so the
//    PV B-fragment is a contiguous swizzled ds_read_b128 (v1 used 8
//    scalar reads per fragment and reached only ~10 TF).
//  - Online softmax in registers, fp32 accumulation.
//
// MFMA fragment layouts (gfx950, f32_16x16x32_bf16) — hardware-verified
// by tests/test_ops_gpu.py::test_mfma_probe_layout:
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + i]      (bf16x8)
//   B: lane l holds B[k = (l>>4)*8 + i][col = l&15]      (bf16x8)
//   C/D: lane l holds C[row = (l>>4)*4 + r][col = l&15]  (floatx4)
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int QTILE = 64;     // q rows per block
constexpr int KTILE = 64;     // kv tokens per tile
constexpr int DMAX = 128;
constexpr int P_PITCH = 72;   // p_lds row pitch (elements): 144B = 9 slots

// K tile: row-major [tok][D], 256B rows, XOR swizzle spreads rows over
// 16B slots.
__device__ __forceinline__ int k_swz(int tok, int byte_in_row) {
  return tok * (DMAX * 2) + (byte_in_row ^ ((tok & 7) << 4));
}

// V tile: TRANSPOSED [dim][KTILE], 128B rows; swizzle on slot bits.
// The swizzle mixes dim>>3 so lanes of ONE scalar-write instruction
// (which share dim&7) still land in different banks — with the plain
// (dim&7) swizzle the XOR was constant per instruction and writes
// collapsed to the 4 tok values per wave (16-way conflicts; PMC showed
// 5.7e8 conflicts/dispatch, profiles/r01_decode_profile_v1.md).
__device__ __forceinline__ int v_swz(int dim, int byte_in_row) {
  const int s = ((dim >> 3) ^ dim) & 7;
  return dim * (KTILE * 2) + (byte_in_row ^ (s << 4));
}

template <int CAUSAL, int D>
__global__ __launch_bounds__(256, 2) void attn_prefill_kernel(
    uint16_t* __restrict__ out,        // [T, Hq, D]
    const uint16_t* __restrict__ q,    // [T, Hq, D]
    const uint16_t* __restrict__ k,    // [T, Hkv, D]
    const uint16_t* __restrict__ v,    // [T, Hkv, D]
    const int* __restrict__ cu_seqlens_q,  // [B+1] query rows
    const int* __restrict__ cu_seqlens_k,  // [B+1] kv rows (>= q rows)
    float scale, int Hq, int Hkv, int window) {
  const int qtile = blockIdx.x;
  const int seq = blockIdx.y;
  const int hq = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int q_start = cu_seqlens_q[seq];
  const int qlen = cu_seqlens_q[seq + 1] - q_start;
  const int k_start = cu_seqlens_k[seq];
  const int klen = cu_seqlens_k[seq + 1] - k_start;
  const int ctx = klen - qlen;   // cached-prefix offset (0 = plain prefill)
  const int qbase = qtile * QTILE;
  if (qbase >= qlen) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int lane_hi = lane >> 4;   // 0..3
  const int lane_lo = lane & 15;   // 0..15

  // double-buffered K/V so the main loop needs ONE barrier per tile
  // instead of two (store for tile t+1 overlaps nothing that reads it)
  __shared__ uint16_t k_lds[2][KTILE * DMAX];
  __shared__ uint16_t v_lds[2][DMAX * KTILE];       // transposed
  __shared__ uint16_t p_lds[4][16 * P_PITCH];

  // ---- Q fragments (A-operand layout), 16 rows per wave ----
  const int my_qrow = qbase + wid * 16 + lane_lo;
  constexpr int nkt = D / 32;
  bf16x8 q_frag[nkt];
#pragma unroll
  for (int kt = 0; kt < nkt; ++kt) {
    if (my_qrow < qlen) {
      const uint16_t* src =
          q + ((int64_t)(q_start + my_qrow) * Hq + hq) * D + kt * 32 +
          lane_hi * 8;
      u16x8 raw = *reinterpret_cast<const u16x8*>(src);
      q_frag[kt] = *reinterpret_cast<bf16x8*>(&raw);
    } else {
      q_frag[kt] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.f;
  }
  constexpr int nc = D / 16;
  floatx4 o_acc[nc];
#pragma unroll
  for (int c = 0; c < nc; ++c) o_acc[c] = floatx4{0, 0, 0, 0};

  const int kv_max = CAUSAL ? min(klen, ctx + qbase + QTILE) : klen;
  // sliding window: rows in this q-tile never look below kv_lo
  const int kv_lo = (CAUSAL && window > 0)
                        ? max(0, ctx + qbase - window + 1) : 0;
  const int t_first = kv_lo / KTILE;
  const int ntiles = (kv_max + KTILE - 1) / KTILE;

  // Per-thread staging assignment: KTILE*D/8 16B-chunks over 256 threads.
  constexpr int nchunk = KTILE * D / 8 / 256;  // 4 for D=128, 2 for D=64
  u16x8 k_reg[nchunk], v_reg[nchunk];

  auto load_tile = [&](int t) {
#pragma unroll
    for (int cch = 0; cch < nchunk; ++cch) {
      const int e = (cch * 256 + tid) * 8;
      const int tok = e / D;
      const int d0 = e % D;
      const int gtok = t * KTILE + tok;
      if (gtok < klen) {
        const int64_t off =
            ((int64_t)(k_start + gtok) * Hkv + hkv) * D + d0;
        k_reg[cch] = *reinterpret_cast<const u16x8*>(k + off);
        v_reg[cch] = *reinterpret_cast<const u16x8*>(v + off);
      } else {
        k_reg[cch] = u16x8{0, 0, 0, 0, 0, 0, 0, 0};
        v_reg[cch] = u16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };
  auto store_tile = [&](int buf) {
#pragma unroll
    for (int cch = 0; cch < nchunk; ++cch) {
      const int e = (cch * 256 + tid) * 8;
      const int tok = e / D;
      const int d0 = e % D;
      *reinterpret_cast<u16x8*>((char*)k_lds[buf] + k_swz(tok, d0 * 2)) =
          k_reg[cch];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        *reinterpret_cast<uint16_t*>(
            (char*)v_lds[buf] + v_swz(d0 + i, tok * 2)) = v_reg[cch][i];
      }
    }
  };

  load_tile(t_first);
  store_tile(0);
  if (t_first + 1 < ntiles) load_tile(t_first + 1);
  for (int t = t_first; t < ntiles; ++t) {
    const int buf = (t - t_first) & 1;
    __syncthreads();            // buf's staging (and last compute) done

    const int kv_base = t * KTILE;
    // ---- QK^T: 4 halves of 16 tokens ----
    floatx4 s_frag[KTILE / 16];
    float p_val[KTILE / 16][4];
#pragma unroll
    for (int h = 0; h < KTILE / 16; ++h) {
      s_frag[h] = floatx4{0, 0, 0, 0};
      const int tok = h * 16 + lane_lo;
#pragma unroll
      for (int kt = 0; kt < nkt; ++kt) {
        const int byte = (kt * 32 + lane_hi * 8) * 2;
        u16x8 raw = *reinterpret_cast<const u16x8*>(
            (char*)k_lds[buf] + k_swz(tok, byte));
        s_frag[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kt], *reinterpret_cast<bf16x8*>(&raw), s_frag[h], 0, 0, 0);
      }
    }

    // ---- Online softmax over the 4 halves ----
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qpos = ctx + qbase + wid * 16 + lane_hi * 4 + r;
      float s[KTILE / 16];
      float m_tile = -INFINITY;
#pragma unroll
      for (int h = 0; h < KTILE / 16; ++h) {
        s[h] = s_frag[h][r] * scale;
        const int tp = kv_base + h * 16 + lane_lo;
        if ((CAUSAL && (tp > qpos ||
                        (window > 0 && qpos - tp >= window))) ||
            tp >= klen) s[h] = -1e30f;
        m_tile = fmaxf(m_tile, s[h]);
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        m_tile = fmaxf(m_tile, __shfl_xor(m_tile, off, WAVE));
      const float m_new = fmaxf(m_run[r], m_tile);
      corr[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
      float l_tile = 0.f;
#pragma unroll
      for (int h = 0; h < KTILE / 16; ++h) {
        p_val[h][r] = __expf(s[h] - m_new);
        l_tile += p_val[h][r];
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        l_tile += __shfl_xor(l_tile, off, WAVE);
      l_run[r] = l_run[r] * corr[r] + l_tile;
      m_run[r] = m_new;
    }

    // ---- P -> per-wave LDS, then PV ----
#pragma unroll
    for (int h = 0; h < KTILE / 16; ++h) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds[wid][(lane_hi * 4 + r) * P_PITCH + h * 16 + lane_lo] =
            f32_to_bf16(p_val[h][r]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    bf16x8 p_frag[KTILE / 32];
#pragma unroll
    for (int kk = 0; kk < KTILE / 32; ++kk) {
      u16x8 raw = *reinterpret_cast<const u16x8*>(
          &p_lds[wid][lane_lo * P_PITCH + kk * 32 + lane_hi * 8]);
      p_frag[kk] = *reinterpret_cast<bf16x8*>(&raw);
    }
#pragma unroll
    for (int c = 0; c < nc; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[c][r] *= corr[r];
#pragma unroll
      for (int kk = 0; kk < KTILE / 32; ++kk) {
        // B-operand from transposed V: row dim = c*16+lane_lo, toks
        // kk*32 + lane_hi*8 .. +7 — one contiguous swizzled 16B read.
        u16x8 raw = *reinterpret_cast<const u16x8*>(
            (char*)v_lds[buf] +
            v_swz(c * 16 + lane_lo, (kk * 32 + lane_hi * 8) * 2));
        o_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag[kk], *reinterpret_cast<bf16x8*>(&raw), o_acc[c], 0, 0, 0);
      }
    }
    // stage tile t+1 into the other buffer (nobody reads it this iter);
    // prefetch tile t+2 loads afterwards so they fly during the barrier
    if (t + 1 < ntiles) {
      store_tile(buf ^ 1);
      if (t + 2 < ntiles) load_tile(t + 2);
    }
  }

  // ---- Epilogue ----
#pragma unroll
  for (int c = 0; c < nc; ++c) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qbase + wid * 16 + lane_hi * 4 + r;
      if (qrow < qlen) {
        out[((int64_t)(q_start + qrow) * Hq + hq) * D + c * 16 + lane_lo] =
            f32_to_bf16(o_acc[c][r] / fmaxf(l_run[r], 1e-20f));
      }
    }
  }
}

}  // namespace

void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, torch::Tensor cu_seqlens,
                  torch::Tensor cu_seqlens_k, int64_t max_seqlen,
                  double scale, bool causal, int64_t window) {
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k.size(1);
  const int B = cu_seqlens.size(0) - 1;
  TORCH_CHECK(D % 32 == 0 && D <= DMAX, "head_dim must be mult of 32, <=128");
  TORCH_CHECK(cu_seqlens.scalar_type() == torch::kInt32);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  auto stream = at::hip::getCurrentHIPStream();
  const int n_qtiles = cdiv((int)max_seqlen, QTILE);
#define LAUNCH(C, DH)                                                        \
  hipLaunchKernelGGL((attn_prefill_kernel<C, DH>), dim3(n_qtiles, B, Hq),    \
                     dim3(256), 0, stream, (uint16_t*)out.data_ptr(),        \
                     (const uint16_t*)q.data_ptr(),                          \
                     (const uint16_t*)k.data_ptr(),                          \
                     (const uint16_t*)v.data_ptr(),                          \
                     cu_seqlens.data_ptr<int>(),                             \
                     cu_seqlens_k.data_ptr<int>(), (float)scale, Hq, Hkv,    \
                     (int)window)
  if (causal) {
    if (D == 128) LAUNCH(1, 128); else LAUNCH(1, 64);
  } else {
    if (D == 128) LAUNCH(0, 128); else LAUNCH(0, 64);
  }
#undef LAUNCH
}
