// Paged-attention decode (single query token per sequence) for CDNA4.
//
// Replaces the decode attention the reference delegates to vLLM
// (SURVEY.md §2.8 "Decode attention (paged KV)").
//
// Design (MI355X-first):
//  - Decode attention is KV-bandwidth-bound: the kernel reads each KV byte
//    exactly once per kv-head and amortizes it across the G = Hq/Hkv query
//    heads of the GQA group (G-way traffic saving vs per-q-head kernels).
//  - grid = (num_seqs, Hkv, num_partitions): flash-decoding style split-K
//    over the sequence so small batches still fill 256 CUs.
//  - Each 256-thread block walks its partition in 256-token chunks:
//      A: thread t loads K row of token t (vectorized 16B), dots against the
//         G query vectors held in LDS (broadcast reads), scores -> LDS.
//      B: per-head online-softmax update (running m, l) by one wave per head.
//      C: V accumulation with dim-owned accumulators: thread owns dim d for
//         all G heads; V reads are lane-contiguous (coalesced 2B*64 = 128B).
//  - Single-partition grids write normalized bf16 straight to `out`;
//    multi-partition grids write fp32 partials + (m, l) for a reduce kernel.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int NTHREADS = 256;
constexpr int CHUNK = 256;
constexpr int MAX_G = 8;

__global__ __launch_bounds__(NTHREADS) void paged_attn_decode_kernel(
    uint16_t* __restrict__ out,          // [B, Hq, D] (used when nparts==1)
    float* __restrict__ tmp_out,         // [B, Hq, maxP, D]
    float* __restrict__ tmp_ml,          // [B, Hq, maxP, 2]
    const uint16_t* __restrict__ q,      // [B, Hq, D]
    const uint16_t* __restrict__ k_cache,// [nblocks, Hkv, bs, D]
    const uint16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,// [B, max_blocks]
    const int* __restrict__ seq_lens,    // [B]
    float scale, int Hq, int Hkv, int D, int block_size, int max_blocks,
    int partition_size, int max_parts) {
  const int seq = blockIdx.x;
  const int hkv = blockIdx.y;
  const int part = blockIdx.z;
  const int nparts = gridDim.z;
  const int len = seq_lens[seq];
  const int p_start = part * partition_size;
  if (p_start >= len) {
    // Dead partition: mark so the reduce kernel skips it.
    if (nparts > 1 && threadIdx.x == 0) {
      const int G = Hq / Hkv;
      for (int g = 0; g < G; ++g) {
        const int hq = hkv * G + g;
        float* ml = tmp_ml + (((int64_t)seq * Hq + hq) * max_parts + part) * 2;
        ml[0] = -INFINITY;
        ml[1] = 0.f;
      }
    }
    return;
  }
  const int p_end = min(len, p_start + partition_size);
  const int G = Hq / Hkv;

  __shared__ float q_lds[MAX_G][128];
  __shared__ float s_lds[MAX_G][CHUNK];
  __shared__ float head_m[MAX_G], head_l[MAX_G], head_corr[MAX_G];
  __shared__ float comb[MAX_G * NTHREADS];  // [par][g][d] flattened

  // Load the G query vectors (pre-scaled) into LDS.
  for (int idx = threadIdx.x; idx < G * D; idx += NTHREADS) {
    const int g = idx / D, d = idx % D;
    q_lds[g][d] =
        bf16_to_f32(q[((int64_t)seq * Hq + hkv * G + g) * D + d]) * scale;
  }
  if (threadIdx.x < MAX_G) {
    head_m[threadIdx.x] = -INFINITY;
    head_l[threadIdx.x] = 0.f;
  }
  __syncthreads();

  // Phase-C ownership: thread owns dim d for tokens of parity `par`.
  const int n_par = NTHREADS / D;          // D in {64, 128}
  const int d_own = threadIdx.x % D;
  const int par = threadIdx.x / D;
  float acc[MAX_G];
#pragma unroll
  for (int g = 0; g < MAX_G; ++g) acc[g] = 0.f;

  const int* btable = block_tables + (int64_t)seq * max_blocks;
  const int nwaves = NTHREADS / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  for (int base = p_start; base < p_end; base += CHUNK) {
    const int chunk_n = min(CHUNK, p_end - base);

    // --- Phase A: scores ---------------------------------------------------
    if ((int)threadIdx.x < chunk_n) {
      const int tok = base + threadIdx.x;
      const int64_t blk = btable[tok / block_size];
      const uint16_t* krow =
          k_cache + (((blk * Hkv + hkv) * (int64_t)block_size +
                      tok % block_size)) * D;
      float s[MAX_G];
#pragma unroll
      for (int g = 0; g < MAX_G; ++g) s[g] = 0.f;
      for (int j = 0; j < D; j += 8) {
        float kv[8];
        load_bf16x8(krow + j, kv);
        for (int g = 0; g < G; ++g) {
#pragma unroll
          for (int i = 0; i < 8; ++i) s[g] += q_lds[g][j + i] * kv[i];
        }
      }
      for (int g = 0; g < G; ++g) s_lds[g][threadIdx.x] = s[g];
    }
    __syncthreads();

    // --- Phase B: online softmax per head (one wave per head slot) --------
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

    // --- Phase C: V accumulation -------------------------------------------
    {
      for (int g = 0; g < G; ++g) acc[g] *= head_corr[g];
      for (int tok_i = par; tok_i < chunk_n; tok_i += n_par) {
        const int tok = base + tok_i;
        const int64_t blk = btable[tok / block_size];
        const float v = bf16_to_f32(
            v_cache[(((blk * Hkv + hkv) * (int64_t)block_size +
                      tok % block_size)) * D + d_own]);
        for (int g = 0; g < G; ++g) acc[g] += s_lds[g][tok_i] * v;
      }
    }
    __syncthreads();  // s_lds reused next chunk
  }

  // Combine parities via LDS: comb[par*G*D + g*D + d]
  for (int g = 0; g < G; ++g) comb[(par * G + g) * D + d_own] = acc[g];
  __syncthreads();
  if (par == 0) {
    for (int g = 0; g < G; ++g) {
      float o = comb[g * D + d_own];
      for (int p = 1; p < n_par; ++p) o += comb[(p * G + g) * D + d_own];
      const int hq = hkv * G + g;
      if (nparts == 1) {
        const float l = head_l[g];
        out[((int64_t)seq * Hq + hq) * D + d_own] =
            f32_to_bf16(o / fmaxf(l, 1e-20f));
      } else {
        tmp_out[(((int64_t)seq * Hq + hq) * max_parts + part) * D + d_own] = o;
        if (d_own == 0) {
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
__global__ void paged_attn_reduce_kernel(
    uint16_t* __restrict__ out, const float* __restrict__ tmp_out,
    const float* __restrict__ tmp_ml, const int* __restrict__ seq_lens,
    int Hq, int D, int partition_size, int max_parts) {
  const int seq = blockIdx.x;
  const int hq = blockIdx.y;
  const int nparts =
      min(max_parts, (seq_lens[seq] + partition_size - 1) / partition_size);
  const float* ml = tmp_ml + ((int64_t)seq * Hq + hq) * max_parts * 2;

  __shared__ float w[256];
  __shared__ float m_sh, l_sh;
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
    for (int p = 0; p < nparts; ++p) o += w[p] * tp[p * D + d];
    out[((int64_t)seq * Hq + hq) * D + d] = f32_to_bf16(o / l_sh);
  }
}

}  // namespace

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_tables, torch::Tensor seq_lens,
                       double scale, torch::Tensor tmp_out,
                       torch::Tensor tmp_ml, int64_t partition_size) {
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k_cache.size(1);
  const int block_size = k_cache.size(2);
  const int max_blocks = block_tables.size(1);
  const int max_parts = tmp_out.size(2);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(Hq / Hkv <= MAX_G && Hq % Hkv == 0);
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);

  const int max_len = seq_lens.max().item<int>();
  int nparts = cdiv(max_len, (int)partition_size);
  nparts = std::min(nparts, max_parts);
  // If a single partition covers everything, write out directly.
  if ((int64_t)B * Hkv >= 512 || max_len <= partition_size) nparts = 1;
  int eff_part = (int)partition_size;
  if (nparts == 1) eff_part = max_len;  // single pass over full length

  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(paged_attn_decode_kernel, dim3(B, Hkv, nparts),
                     dim3(NTHREADS), 0, stream, (uint16_t*)out.data_ptr(),
                     tmp_out.data_ptr<float>(), tmp_ml.data_ptr<float>(),
                     (const uint16_t*)q.data_ptr(),
                     (const uint16_t*)k_cache.data_ptr(),
                     (const uint16_t*)v_cache.data_ptr(),
                     block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
                     (float)scale, Hq, Hkv, D, block_size, max_blocks,
                     eff_part, max_parts);
  if (nparts > 1) {
    hipLaunchKernelGGL(paged_attn_reduce_kernel, dim3(B, Hq), dim3(256), 0,
                       stream, (uint16_t*)out.data_ptr(),
                       tmp_out.data_ptr<float>(), tmp_ml.data_ptr<float>(),
                       seq_lens.data_ptr<int>(), Hq, D, eff_part, max_parts);
  }
}
