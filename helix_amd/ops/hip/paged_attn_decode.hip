// Paged-attention decode (single query token per sequence) for CDNA4.
//
// Replaces the decode attention the reference delegates to vLLM
// (SURVEY.md §2.8 "Decode attention (paged KV)").
//
// Design (MI355X-first, v3 — templated + load-batched):
//  - KV-bandwidth-bound: each KV byte is read once per kv-head and
//    amortized across the G = Hq/Hkv GQA query heads.
//  - grid = (num_seqs, Hkv, num_partitions): flash-decoding split-K, host
//    sizes partitions to fill the 256 CUs (>= ~1024 workgroups).
//  - DHEAD and G are template parameters so every inner loop fully
//    unrolls; K rows and V tiles are loaded into registers in batches
//    BEFORE any math so the wave keeps many VMEM ops in flight (v2's
//    runtime-D loops serialized load->use and reached only 0.5 TB/s —
//    profiles/r01_decode_profile_v1.md).
//  - 128-thread blocks (2 waves) walk the partition in 128-token chunks:
//      A: thread t loads K row of token t (DHEAD/8 x 16B batched), dots
//         against the G query vectors in LDS (broadcast reads).
//      B: per-head online softmax (running m, l), one wave per head.
//      C: V accumulation, thread owns a dim pair, 8-token load batches.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int NTHREADS = 128;
constexpr int CHUNK = 128;
constexpr int MAX_G = 8;
constexpr int PART_QUANT = 128;

// MFMA_A: phase A computes scores with f32_16x16x32_bf16 matrix ops
// (16 tokens x G queries per instruction) instead of per-thread VALU
// dots — motivated by the B=512 PMC: 37 VALU per vector load and
// wait:busy 14:1 (profiles/r01_pmc_decode_b512.txt).
// KV8: the paged cache holds OCP e4m3 bytes (opt-in
// kv_cache_dtype="fp8"): halves the KV traffic this kernel is
// latency/BW-bound on; dequant is one v_cvt per element in the math
// loops. MFMA_A and KV8 are mutually exclusive instantiations.
// VPS: phase-C V rows staged RAW (u16x8) per batch — the round-2
// deep-pipelining lever (PMC wait:busy 14:1, latency-bound): more V
// bytes in flight per wave before any conversion VALU touches them.
// OCC: launch_bounds waves/SIMD floor override (0 = round-1 default).
// KPRE: inter-chunk K prefetch. The r2 ISA audit (profiles/
// r02_decode_kpre.md) showed the compiler sinking phase A's 16-deep raw
// K staging into the consume loop — steady state s_waitcnt vmcnt(1),
// i.e. only ~2 VMEM loads in flight per wave, matching the 14:1
// wait:busy PMC. KPRE hoists the next chunk's 16 b128 K loads to just
// after the current chunk's scores are consumed, so they land during
// phases B+C (softmax + V) instead of head-of-line blocking phase A.
template <int DHEAD, int G, bool MFMA_A, bool KV8, int VPS = 4, int OCC = 0,
          bool KPRE = false>
__global__ __launch_bounds__(NTHREADS, (OCC > 0 ? OCC : (G <= 2 ? 4 : 3))) void paged_attn_decode_kernel(
    uint16_t* __restrict__ out,          // [B, Hq, D] (used when nparts==1)
    float* __restrict__ tmp_out,         // [B, Hq, maxP, D]
    float* __restrict__ tmp_ml,          // [B, Hq, maxP, 2]
    const uint16_t* __restrict__ q,      // [B, Hq, D]
    const uint16_t* __restrict__ k_cache,// [nblocks, Hkv, bs, D]
    const uint16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,// [B, max_blocks]
    const int* __restrict__ seq_lens,    // [B]
    float scale, int Hkv, int block_size, int max_blocks,
    int partition_size, int max_parts, int window) {
  const int seq = blockIdx.x;
  const int hkv = blockIdx.y;
  const int part = blockIdx.z;
  const int nparts = gridDim.z;
  const int Hq = Hkv * G;
  const int len = seq_lens[seq];
  const int kv_lo = (window > 0) ? max(0, len - window) : 0;
  int p_start = part * partition_size;
  const int p_hi = min(len, p_start + partition_size);
  if (p_start < kv_lo) p_start = kv_lo;
  if (p_start >= len || p_hi <= kv_lo) {
    if (nparts > 1 && threadIdx.x == 0) {
#pragma unroll
      for (int g = 0; g < G; ++g) {
        const int hq = hkv * G + g;
        float* ml = tmp_ml + (((int64_t)seq * Hq + hq) * max_parts + part) * 2;
        ml[0] = -INFINITY;
        ml[1] = 0.f;
      }
    }
    return;
  }
  const int p_end = p_hi;

  __shared__ float q_lds[G][DHEAD];
  __shared__ float s_lds[G][CHUNK];
  __shared__ float head_m[MAX_G], head_l[MAX_G], head_corr[MAX_G];

  for (int idx = threadIdx.x; idx < G * DHEAD; idx += NTHREADS) {
    const int g = idx / DHEAD, d = idx % DHEAD;
    q_lds[g][d] =
        bf16_to_f32(q[((int64_t)seq * Hq + hkv * G + g) * DHEAD + d]) * scale;
  }
  if (threadIdx.x < MAX_G) {
    head_m[threadIdx.x] = -INFINITY;
    head_l[threadIdx.x] = 0.f;
  }
  __syncthreads();

  // Q fragments for the MFMA path: A[row=query g][k=dim], rows >= G
  // zero.  Lane holds A[lane&15][(lane>>4)*8+i] per 32-dim step.
  constexpr int NKK = DHEAD / 32;
  bf16x8 qfrag[MFMA_A ? NKK : 1];
  if constexpr (MFMA_A) {
    const int l_lo = threadIdx.x & 15;
    const int l_hi = (threadIdx.x & (WAVE - 1)) >> 4;
#pragma unroll
    for (int kk = 0; kk < NKK; ++kk) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float qv = (l_lo < G)
            ? q_lds[l_lo][kk * 32 + l_hi * 8 + i] : 0.f;
        reinterpret_cast<uint16_t*>(&qfrag[kk])[i] = f32_to_bf16(qv);
      }
    }
  }

  // Phase-C ownership: a thread owns 8 consecutive dims (one 16B load
  // per token) — the ablation probe showed b32 V loads were 64% of the
  // kernel (8x more VMEM instructions per byte than phase A's b128s).
  constexpr int NGRP = DHEAD / 8;            // dim-groups (16 for D=128)
  constexpr int C_PAR = NTHREADS / NGRP;     // token parities (8 / 16)
  const int d8 = (threadIdx.x % NGRP) * 8;   // my dims [d8, d8+8)
  const int cpar = threadIdx.x / NGRP;
  float acc[G][8];
#pragma unroll
  for (int g = 0; g < G; ++g)
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[g][i] = 0.f;

  const int* btable = block_tables + (int64_t)seq * max_blocks;
  const int nwaves = NTHREADS / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  // KPRE: raw K rows for the chunk about to be scored, loaded one full
  // chunk ahead (16 b128 loads live across phases B+C of the previous
  // chunk — the latency they need to hide).
  constexpr bool USE_KPRE = KPRE && !MFMA_A && !KV8;
  constexpr bool USE_KPRE8 = KPRE && !MFMA_A && KV8;
  u16x8 kpre[USE_KPRE ? DHEAD / 8 : 1];
  u8x16 kpre8[USE_KPRE8 ? DHEAD / 16 : 1];
  auto issue_kpre = [&](int cbase) {
    const int tok = min(cbase + (int)threadIdx.x, p_end - 1);
    const int64_t roff =
        (((int64_t)btable[tok / block_size] * Hkv + hkv) *
             (int64_t)block_size + tok % block_size) * DHEAD;
    if constexpr (USE_KPRE8) {
      const uint8_t* krow = (const uint8_t*)k_cache + roff;
#pragma unroll
      for (int j = 0; j < DHEAD / 16; ++j)
        kpre8[j] = *reinterpret_cast<const u8x16*>(krow + j * 16);
    } else {
      const uint16_t* krow = k_cache + roff;
#pragma unroll
      for (int j = 0; j < (USE_KPRE ? DHEAD / 8 : 1); ++j)
        kpre[j] = *reinterpret_cast<const u16x8*>(krow + j * 8);
    }
  };
  if constexpr (USE_KPRE || USE_KPRE8) issue_kpre(p_start);

  for (int base = p_start; base < p_end; base += CHUNK) {
    const int chunk_n = min(CHUNK, p_end - base);

    // --- Phase A: scores -------------------------------------------------
    if constexpr (MFMA_A) {
      // 16-token MFMA tiles; each wave owns alternating groups of 4.
      const int ntiles = (chunk_n + 15) / 16;
      const int l = threadIdx.x & (WAVE - 1);
      const int l_lo = l & 15, l_hi = l >> 4;
      for (int t0 = wid * 4; t0 < ntiles; t0 += nwaves * 4) {
        const int nt = min(4, ntiles - t0);
        u16x8 bfr[4][NKK];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (u >= nt) continue;
          const int tok = min(base + (t0 + u) * 16 + l_lo, p_end - 1);
          const int64_t blk = btable[tok / block_size];
          const uint16_t* krow =
              k_cache + (((blk * Hkv + hkv) * (int64_t)block_size +
                          tok % block_size)) * DHEAD;
#pragma unroll
          for (int kk = 0; kk < NKK; ++kk)
            bfr[u][kk] = *reinterpret_cast<const u16x8*>(
                krow + kk * 32 + l_hi * 8);
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (u >= nt) continue;
          floatx4 sacc = floatx4{0, 0, 0, 0};
#pragma unroll
          for (int kk = 0; kk < NKK; ++kk)
            sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                qfrag[kk], *reinterpret_cast<bf16x8*>(&bfr[u][kk]), sacc,
                0, 0, 0);
          // C[row=l_hi*4+r][col=l_lo]: row = query g, col = token
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int g = l_hi * 4 + r;
            const int ci = (t0 + u) * 16 + l_lo;
            if (g < G && ci < chunk_n) s_lds[g][ci] = sacc[r];
          }
        }
      }
    } else if constexpr (USE_KPRE) {
      if ((int)threadIdx.x < chunk_n) {
        float s[G];
#pragma unroll
        for (int g = 0; g < G; ++g) s[g] = 0.f;
#pragma unroll
        for (int j = 0; j < DHEAD / 8; ++j) {
          float kv[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) kv[i] = bf16_to_f32(kpre[j][i]);
#pragma unroll
          for (int g = 0; g < G; ++g) {
#pragma unroll
            for (int i = 0; i < 8; ++i)
              s[g] += q_lds[g][j * 8 + i] * kv[i];
          }
        }
#pragma unroll
        for (int g = 0; g < G; ++g) s_lds[g][threadIdx.x] = s[g];
      }
      // scores consumed: issue the NEXT chunk's K rows now so they
      // fly during softmax + V accumulation
      if (base + CHUNK < p_end) issue_kpre(base + CHUNK);
    } else if constexpr (USE_KPRE8) {
      if ((int)threadIdx.x < chunk_n) {
        float s[G];
#pragma unroll
        for (int g = 0; g < G; ++g) s[g] = 0.f;
#pragma unroll
        for (int j = 0; j < DHEAD / 16; ++j) {
          float kv[16];
#pragma unroll
          for (int i = 0; i < 16; ++i) kv[i] = fp8_to_f32(kpre8[j][i]);
#pragma unroll
          for (int g = 0; g < G; ++g) {
#pragma unroll
            for (int i = 0; i < 16; ++i)
              s[g] += q_lds[g][j * 16 + i] * kv[i];
          }
        }
#pragma unroll
        for (int g = 0; g < G; ++g) s_lds[g][threadIdx.x] = s[g];
      }
      if (base + CHUNK < p_end) issue_kpre(base + CHUNK);
    } else if ((int)threadIdx.x < chunk_n) {
      const int tok = base + threadIdx.x;
      const int64_t blk = btable[tok / block_size];
      const int64_t roff =
          (((blk * Hkv + hkv) * (int64_t)block_size + tok % block_size)) *
          DHEAD;
      float s[G];
#pragma unroll
      for (int g = 0; g < G; ++g) s[g] = 0.f;
      if constexpr (KV8) {
        const uint8_t* krow = (const uint8_t*)k_cache + roff;
        u8x16 kraw[DHEAD / 16];
#pragma unroll
        for (int j = 0; j < DHEAD / 16; ++j)
          kraw[j] = *reinterpret_cast<const u8x16*>(krow + j * 16);
#pragma unroll
        for (int j = 0; j < DHEAD / 16; ++j) {
          float kv[16];
#pragma unroll
          for (int i = 0; i < 16; ++i) kv[i] = fp8_to_f32(kraw[j][i]);
#pragma unroll
          for (int g = 0; g < G; ++g) {
#pragma unroll
            for (int i = 0; i < 16; ++i)
              s[g] += q_lds[g][j * 16 + i] * kv[i];
          }
        }
      } else {
        const uint16_t* krow = k_cache + roff;
        u16x8 kraw[DHEAD / 8];
#pragma unroll
        for (int j = 0; j < DHEAD / 8; ++j)
          kraw[j] = *reinterpret_cast<const u16x8*>(krow + j * 8);
#pragma unroll
        for (int j = 0; j < DHEAD / 8; ++j) {
          float kv[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) kv[i] = bf16_to_f32(kraw[j][i]);
#pragma unroll
          for (int g = 0; g < G; ++g) {
#pragma unroll
            for (int i = 0; i < 8; ++i)
              s[g] += q_lds[g][j * 8 + i] * kv[i];
          }
        }
      }
#pragma unroll
      for (int g = 0; g < G; ++g) s_lds[g][threadIdx.x] = s[g];
    }
    __syncthreads();

    // --- Phase B: online softmax per head --------------------------------
    for (int g = wid; g < G; g += nwaves) {
      float m_chunk = -INFINITY;
      for (int i = lane; i < chunk_n; i += WAVE)
        m_chunk = fmaxf(m_chunk, s_lds[g][i]);
      m_chunk = wave_reduce_max(m_chunk);
      const float m_old = head_m[g];
      const float m_new = fmaxf(m_old, m_chunk);
      const float corr = (m_old == -INFINITY) ? 0.f : __expf(m_old - m_new);
      float l_add = 0.f;
      for (int i = lane; i < chunk_n; i += WAVE) {
        const float p = __expf(s_lds[g][i] - m_new);
        s_lds[g][i] = p;
        l_add += p;
      }
      l_add = wave_reduce_sum(l_add);
      if (lane == 0) {
        head_l[g] = head_l[g] * corr + l_add;
        head_m[g] = m_new;
        head_corr[g] = corr;
      }
    }
    __syncthreads();

    // --- Phase C: V accumulation (16B lane loads, 4-pass batches) --------
    {
#pragma unroll
      for (int g = 0; g < G; ++g)
#pragma unroll
        for (int i = 0; i < 8; ++i) acc[g][i] *= head_corr[g];
      const int npass = (chunk_n - cpar + C_PAR - 1) / C_PAR;  // my tokens
      auto vrow_off = [&](int tok) {
        const int64_t blk = btable[tok / block_size];
        return (((blk * Hkv + hkv) * (int64_t)block_size +
                 tok % block_size)) * DHEAD + d8;
      };
      auto load_v = [&](int64_t off, float v[8]) {
        if constexpr (KV8) {
          const u8x8 raw = *reinterpret_cast<const u8x8*>(
              (const uint8_t*)v_cache + off);
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = fp8_to_f32(raw[i]);
        } else {
          const u16x8 raw = *reinterpret_cast<const u16x8*>(v_cache + off);
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = bf16_to_f32(raw[i]);
        }
      };
      int ps = 0;
      if constexpr (!KV8) {
        // raw-staged batches: VPS x 16B loads issue back-to-back (pure,
        // no dependent converts between them); conversion + FMA consume
        // them in order while later loads are still in flight
        for (; ps + VPS <= npass; ps += VPS) {
          u16x8 vraw[VPS];
#pragma unroll
          for (int u = 0; u < VPS; ++u)
            vraw[u] = *reinterpret_cast<const u16x8*>(
                v_cache + vrow_off(base + (ps + u) * C_PAR + cpar));
#pragma unroll
          for (int u = 0; u < VPS; ++u) {
            const int tok_i = (ps + u) * C_PAR + cpar;
            float v[8];
#pragma unroll
            for (int i = 0; i < 8; ++i) v[i] = bf16_to_f32(vraw[u][i]);
#pragma unroll
            for (int g = 0; g < G; ++g) {
              const float pv = s_lds[g][tok_i];
#pragma unroll
              for (int i = 0; i < 8; ++i) acc[g][i] += pv * v[i];
            }
          }
        }
      } else {
        for (; ps + 4 <= npass; ps += 4) {
          int64_t voff[4];
#pragma unroll
          for (int u = 0; u < 4; ++u)
            voff[u] = vrow_off(base + (ps + u) * C_PAR + cpar);
          float v4[4][8];
#pragma unroll
          for (int u = 0; u < 4; ++u) load_v(voff[u], v4[u]);
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            const int tok_i = (ps + u) * C_PAR + cpar;
#pragma unroll
            for (int g = 0; g < G; ++g) {
              const float pv = s_lds[g][tok_i];
#pragma unroll
              for (int i = 0; i < 8; ++i) acc[g][i] += pv * v4[u][i];
            }
          }
        }
      }
      for (; ps < npass; ++ps) {
        const int tok_i = ps * C_PAR + cpar;
        float v[8];
        load_v(vrow_off(base + tok_i), v);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float pv = s_lds[g][tok_i];
#pragma unroll
          for (int i = 0; i < 8; ++i) acc[g][i] += pv * v[i];
        }
      }
    }
    __syncthreads();
  }

  // Combine the C_PAR token parities via LDS, then write out.
  __shared__ float comb[G * DHEAD];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    if (cpar == 0) {
#pragma unroll
      for (int i = 0; i < 8; ++i) comb[g * DHEAD + d8 + i] = acc[g][i];
    }
  }
  __syncthreads();
  for (int p = 1; p < C_PAR; ++p) {
    if (cpar == p) {
#pragma unroll
      for (int g = 0; g < G; ++g)
#pragma unroll
        for (int i = 0; i < 8; ++i)
          comb[g * DHEAD + d8 + i] += acc[g][i];
    }
    __syncthreads();
  }
  if (cpar == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const int hq = hkv * G + g;
      if (nparts == 1) {
        const float inv_l = 1.f / fmaxf(head_l[g], 1e-20f);
        float o[8];
#pragma unroll
        for (int i = 0; i < 8; ++i)
          o[i] = comb[g * DHEAD + d8 + i] * inv_l;
        store_bf16x8(out + ((int64_t)seq * Hq + hq) * DHEAD + d8, o);
      } else {
        float* tp =
            tmp_out + (((int64_t)seq * Hq + hq) * max_parts + part) * DHEAD;
#pragma unroll
        for (int i = 0; i < 8; ++i) tp[d8 + i] = comb[g * DHEAD + d8 + i];
        if (d8 == 0) {
          float* ml =
              tmp_ml + (((int64_t)seq * Hq + hq) * max_parts + part) * 2;
          ml[0] = head_m[g];
          ml[1] = head_l[g];
        }
      }
    }
  }
}

// Reduce partials: out[seq, hq, :] = sum_p w_p * tmp_out[p] / L
__global__ __launch_bounds__(128) void paged_attn_reduce_kernel(
    uint16_t* __restrict__ out, const float* __restrict__ tmp_out,
    const float* __restrict__ tmp_ml, const int* __restrict__ seq_lens,
    int Hq, int D, int partition_size, int max_parts) {
  const int seq = blockIdx.x;
  const int hq = blockIdx.y;
  const int nparts =
      min(max_parts, (seq_lens[seq] + partition_size - 1) / partition_size);
  const float* ml = tmp_ml + ((int64_t)seq * Hq + hq) * max_parts * 2;

  __shared__ float w[512];
  __shared__ float l_sh;
  if (threadIdx.x == 0) {
    float M = -INFINITY;
    for (int p = 0; p < nparts; ++p) M = fmaxf(M, ml[p * 2]);
    float L = 0.f;
    for (int p = 0; p < nparts; ++p) {
      const float e = (ml[p * 2] == -INFINITY) ? 0.f : __expf(ml[p * 2] - M);
      w[p] = e;
      L += e * ml[p * 2 + 1];
    }
    l_sh = fmaxf(L, 1e-20f);
  }
  __syncthreads();
  const float* tp = tmp_out + ((int64_t)seq * Hq + hq) * max_parts * D;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float o = 0.f;
    for (int p = 0; p < nparts; ++p) {
      // skip dead partitions explicitly: their tmp_out is uninitialized
      // and 0 * NaN would poison the sum (windowed-out partitions)
      if (w[p] != 0.f) o += w[p] * tp[p * D + d];
    }
    out[((int64_t)seq * Hq + hq) * D + d] = f32_to_bf16(o / l_sh);
  }
}

static bool use_mfma_a() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("HELIX_DECODE_MFMA");
    v = (e != nullptr && e[0] == '1') ? 1 : 0;
  }
  return v == 1;
}

static int env_int(const char* name, int dflt) {
  const char* e = getenv(name);
  return e ? atoi(e) : dflt;
}
// A/B knobs for the round-2 decode pipeline (defaults = measured best)
static int decode_vps() {
  static int v = -1;
  if (v < 0) v = env_int("HELIX_DECODE_VPS", 4);
  return v;
}
static int decode_occ() {
  static int v = -1;
  if (v < 0) v = env_int("HELIX_DECODE_OCC", 0);
  return v;
}
static int decode_kpre() {
  static int v = -1;
  if (v < 0) v = env_int("HELIX_DECODE_KPRE", 1);
  return v;
}

template <int DHEAD, int G>
void launch_decode(uint16_t* out, float* tmp_out, float* tmp_ml,
                   const uint16_t* q, const uint16_t* kc, const uint16_t* vc,
                   const int* bt, const int* lens, float scale, int B,
                   int Hkv, int block_size, int max_blocks, int eff_part,
                   int nparts, int max_parts, int window, bool kv8,
                   hipStream_t stream) {
#define LAUNCH_PD(MF, K8, VPS, OCC, KP)                                      \
  hipLaunchKernelGGL(                                                        \
      (paged_attn_decode_kernel<DHEAD, G, MF, K8, VPS, OCC, KP>),            \
      dim3(B, Hkv, nparts), dim3(NTHREADS), 0, stream, out,                  \
      tmp_out, tmp_ml, q, kc, vc, bt, lens, scale, Hkv,                      \
      block_size, max_blocks, eff_part, max_parts, window)
  if (kv8) {
    if (decode_kpre()) LAUNCH_PD(false, true, 4, 2, true);
    else               LAUNCH_PD(false, true, 4, 0, false);
  } else if (use_mfma_a()) {
    LAUNCH_PD(true, false, 4, 0, false);
  } else {
    const int vps = decode_vps(), occ = decode_occ();
    if (decode_kpre()) {
      // measured defaults (profiles/r02_decode_kpre.md): K-prefetch +
      // waves/EU floor 2 everywhere (+5.4% e2e at B=512). When the
      // grid underfills the chip (< 2048 workgroups on 256 CUs x 4
      // SIMDs), per-wave pipelining has to replace occupancy-based
      // latency hiding: the 8-deep V staging wins those shapes
      // (4.25-4.29 vs 3.6-3.7 TB/s at B<=64 / L>=2048) but loses the
      // full-grid B=512 headline (27.5k vs 28.1k tok/s), so it is
      // grid-gated, not global. HELIX_DECODE_VPS / _OCC override.
      const bool vps_forced = getenv("HELIX_DECODE_VPS") != nullptr;
      const bool small_grid = B * Hkv * nparts < 2048;
      const bool deep = vps_forced ? (vps >= 8) : small_grid;
      if (deep)                      LAUNCH_PD(false, false, 8, 2, true);
      else if (occ == 0 || occ == 2) LAUNCH_PD(false, false, 4, 2, true);
      else                           LAUNCH_PD(false, false, 4, 0, true);
    }
    else if (occ == 2)        LAUNCH_PD(false, false, 4, 2, false);
    else if (vps >= 8 && occ >= 4) LAUNCH_PD(false, false, 8, 4, false);
    else if (vps >= 8)        LAUNCH_PD(false, false, 8, 0, false);
    else if (occ >= 4)        LAUNCH_PD(false, false, 4, 4, false);
    else                      LAUNCH_PD(false, false, 4, 0, false);
  }
#undef LAUNCH_PD
}

}  // namespace

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_tables, torch::Tensor seq_lens,
                       double scale, torch::Tensor tmp_out,
                       torch::Tensor tmp_ml, int64_t max_len_hint,
                       int64_t window) {
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k_cache.size(1);
  const int block_size = k_cache.size(2);
  const int max_blocks = block_tables.size(1);
  const int max_parts = tmp_out.size(2);
  const int G = Hq / Hkv;
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(G <= MAX_G && Hq % Hkv == 0);
  TORCH_CHECK(G == 1 || G == 2 || G == 4 || G == 8, "G must be 1/2/4/8");
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);

  const int max_len = (int)max_len_hint;
  const int max_useful = cdiv(max_len, PART_QUANT);
  // split-K sizing target: enough workgroups to fill 256 CUs. 1024 was
  // round-1's tuning; HELIX_DECODE_TARGET_WGS re-tunes it for the
  // round-2 pipeline on underfilled (small-B, long-L) shapes.
  static int target_wgs = [] {
    const char* e = getenv("HELIX_DECODE_TARGET_WGS");
    return e ? atoi(e) : 1024;
  }();
  int nparts = std::max(1, target_wgs / std::max(1, B * Hkv));
  nparts = std::min({nparts, max_useful, max_parts});
  int eff_part = cdiv(cdiv(max_len, nparts), PART_QUANT) * PART_QUANT;
  nparts = cdiv(max_len, eff_part);

  auto stream = at::hip::getCurrentHIPStream();
  auto* o = (uint16_t*)out.data_ptr();
  auto* to = tmp_out.data_ptr<float>();
  auto* tm = tmp_ml.data_ptr<float>();
  auto* qp = (const uint16_t*)q.data_ptr();
  auto* kp = (const uint16_t*)k_cache.data_ptr();
  auto* vp = (const uint16_t*)v_cache.data_ptr();
  auto* bp = block_tables.data_ptr<int>();
  auto* lp = seq_lens.data_ptr<int>();

  const bool kv8 = k_cache.scalar_type() == torch::kUInt8;
#define DISPATCH(DH, GG)                                                     \
  launch_decode<DH, GG>(o, to, tm, qp, kp, vp, bp, lp, (float)scale, B,     \
                        Hkv, block_size, max_blocks, eff_part, nparts,      \
                        max_parts, (int)window, kv8, stream)
  if (D == 128) {
    if (G == 1) DISPATCH(128, 1);
    else if (G == 2) DISPATCH(128, 2);
    else if (G == 4) DISPATCH(128, 4);
    else DISPATCH(128, 8);
  } else {
    if (G == 1) DISPATCH(64, 1);
    else if (G == 2) DISPATCH(64, 2);
    else if (G == 4) DISPATCH(64, 4);
    else DISPATCH(64, 8);
  }
#undef DISPATCH

  if (nparts > 1) {
    hipLaunchKernelGGL(paged_attn_reduce_kernel, dim3(B, Hq), dim3(128), 0,
                       stream, o, to, tm, lp, Hq, D, eff_part, max_parts);
    // note: dead (windowed-out) partitions wrote ml = -inf and are
    // skipped by the reduce weighting
  }
}
