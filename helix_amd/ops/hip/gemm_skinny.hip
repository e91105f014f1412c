// Skinny-M bf16 MFMA GEMM for the decode-path projections (gfx950).
//
// SURVEY.md §2.8 "Q/K/V + O projections, MLP GEMMs — hand-written":
// at decode, M = batch (<= 512) while N,K are model dims, so hipBLASLt's
// square-tile kernels run at 550-660 TF (NOTES.md item 2). This kernel
// keeps the verified m97 structure (128x128 tile, 2-barrier K-loop,
// global_load_lds width-16 staging — the plain-HIP sweet spot per the
// CDNA4 guide) and adds what skinny shapes need:
//   - BM=64 tile variant for small decode batches;
//   - split-K over blockIdx.z so shapes like M=512,N=4096 (128 tiles)
//     still put >= 3 workgroups on every CU (fp32 partials + a fused
//     reduce epilogue, bitwise-deterministic);
//   - bijective XCD-chunked workgroup swizzle (m-tiles of one n-tile
//     share an XCD's L2, so the 4x re-read of the B weight tile hits L2
//     instead of HBM).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int BN = 128, BK = 64;

// blockIdx.x -> linear tile id such that consecutive linear ids land on
// the SAME XCD (hardware round-robins workgroups across the 8 XCDs).
__device__ __forceinline__ int xcd_chunked(int orig, int nwg) {
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = orig % 8, pos = orig / 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
}

template <int BM>
__global__ __launch_bounds__(256) void gemm_skinny_kernel(
    uint16_t* __restrict__ out,       // [M,N] bf16 (gridDim.z == 1)
    float* __restrict__ scratch,      // [Z,M,N] fp32 partials (z > 1)
    const uint16_t* __restrict__ x,   // [M,K]
    const uint16_t* __restrict__ w,   // [N,K]
    const uint16_t* __restrict__ bias,
    int M, int N, int K, int mtiles, int kslice, int act, int swz) {
  __shared__ uint16_t a_lds[BM * BK];
  __shared__ uint16_t b_lds[BN * BK];

  int lin = blockIdx.x;
  if (swz) lin = xcd_chunked(lin, gridDim.x);
  const int tile_m = lin % mtiles;
  const int tile_n = lin / mtiles;
  const int m0 = tile_m * BM, n0 = tile_n * BN;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int lane_hi = lane >> 4, lane_lo = lane & 15;
  // wave grid: BM=128 -> 2x2 (64x64 each); BM=64 -> 1x4 (64x32 each)
  constexpr int WM = (BM == 128) ? 2 : 1;
  constexpr int WN = 4 / WM;
  constexpr int MI = (BM / WM) / 16;   // 4
  constexpr int NI = (BN / WN) / 16;   // 4 or 2
  const int wr = wid / WN, wc = wid % WN;

  floatx4 acc[MI][NI];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NI; ++j) acc[i][j] = floatx4{0, 0, 0, 0};

  const int kz0 = blockIdx.z * kslice;
  const int nk = kslice / BK;
  constexpr int A_ITS = BM * BK / (256 * 8);  // 16B/lane staging passes
  constexpr int B_ITS = BN * BK / (256 * 8);
  for (int kt = 0; kt < nk; ++kt) {
    const int k0 = kz0 + kt * BK;
    __syncthreads();
#pragma unroll
    for (int it = 0; it < A_ITS; ++it) {
      const int e = (it * 256 + tid) * 8;
      const int row = e / BK, col = e % BK;
      const int arow = min(m0 + row, M - 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(x +
              (int64_t)arow * K + k0 + col),
          (__attribute__((address_space(3))) uint32_t*)(a_lds + e), 16, 0, 0);
    }
#pragma unroll
    for (int it = 0; it < B_ITS; ++it) {
      const int e = (it * 256 + tid) * 8;
      const int row = e / BK, col = e % BK;
      const int brow = min(n0 + row, N - 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(w +
              (int64_t)brow * K + k0 + col),
          (__attribute__((address_space(3))) uint32_t*)(b_lds + e), 16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 a_frag[MI], b_frag[NI];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) {
        const int row = wr * (BM / WM) + mi * 16 + lane_lo;
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            a_lds + row * BK + kk * 32 + lane_hi * 8);
      }
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        const int row = wc * (BN / WN) + ni * 16 + lane_lo;
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            b_lds + row * BK + kk * 32 + lane_hi * 8);
      }
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
  }

  // Epilogue: direct bf16 store (+bias/act) when unsplit; fp32 partial
  // slab per z otherwise (reduce kernel applies bias/act).
  const bool split = gridDim.z > 1;
  float* part = scratch + (int64_t)blockIdx.z * M * N;
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      const int col = n0 + wc * (BN / WN) + ni * 16 + lane_lo;
      if (col >= N) continue;
      const float b = (!split && bias) ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * (BM / WM) + mi * 16 + lane_hi * 4 + r;
        if (row >= M) continue;
        float v = acc[mi][ni][r];
        if (split) {
          part[(int64_t)row * N + col] = v;
        } else {
          v += b;
          if (act == 1) {
            const float c = 0.7978845608028654f;
            v = 0.5f * v * (1.f + tanhf(c * (v + 0.044715f * v * v * v)));
          }
          out[(int64_t)row * N + col] = f32_to_bf16(v);
        }
      }
    }
  }
}

// Sum the Z fp32 partial slabs, apply bias/act, store bf16.
__global__ __launch_bounds__(256) void splitk_reduce_kernel(
    uint16_t* __restrict__ out, const float* __restrict__ scratch,
    const uint16_t* __restrict__ bias, int64_t MN, int N, int Z, int act) {
  const int64_t i0 = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 8;
  if (i0 + 8 > MN) {
    for (int64_t i = i0; i < MN; ++i) {
      float v = 0.f;
      for (int z = 0; z < Z; ++z) v += scratch[(int64_t)z * MN + i];
      if (bias) v += bf16_to_f32(bias[i % N]);
      if (act == 1) {
        const float c = 0.7978845608028654f;
        v = 0.5f * v * (1.f + tanhf(c * (v + 0.044715f * v * v * v)));
      }
      out[i] = f32_to_bf16(v);
    }
    return;
  }
  float v[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] = 0.f;
  for (int z = 0; z < Z; ++z) {
    const float* s = scratch + (int64_t)z * MN + i0;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] += s[j];
  }
  if (bias) {
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] += bf16_to_f32(bias[(i0 + j) % N]);
  }
  if (act == 1) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float c = 0.7978845608028654f;
      v[j] = 0.5f * v[j] *
             (1.f + tanhf(c * (v[j] + 0.044715f * v[j] * v[j] * v[j])));
    }
  }
  store_bf16x8(out + i0, v);
}

}  // namespace

void gemm_skinny_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> bias,
                      c10::optional<torch::Tensor> scratch, int64_t split_k,
                      int64_t act, int64_t swizzle) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  const int S = (int)split_k;
  TORCH_CHECK(K % (S * BK) == 0, "K must split into 64-multiples");
  const int BM = (M > 64) ? 128 : 64;
  const int mtiles = cdiv(M, BM), ntiles = cdiv(N, BN);
  const int kslice = K / S;
  auto stream = at::hip::getCurrentHIPStream();
  const uint16_t* bias_ptr =
      bias.has_value() ? (const uint16_t*)bias->data_ptr() : nullptr;
  float* scratch_ptr = nullptr;
  if (S > 1) {
    TORCH_CHECK(scratch.has_value() &&
                scratch->numel() >= (int64_t)S * M * N);
    scratch_ptr = scratch->data_ptr<float>();
  }
  dim3 grid(mtiles * ntiles, 1, S);
  if (BM == 128) {
    hipLaunchKernelGGL(gemm_skinny_kernel<128>, grid, dim3(256), 0, stream,
                       (uint16_t*)out.data_ptr(), scratch_ptr,
                       (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)w.data_ptr(), bias_ptr, M, N, K,
                       mtiles, kslice, (int)act, (int)swizzle);
  } else {
    hipLaunchKernelGGL(gemm_skinny_kernel<64>, grid, dim3(256), 0, stream,
                       (uint16_t*)out.data_ptr(), scratch_ptr,
                       (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)w.data_ptr(), bias_ptr, M, N, K,
                       mtiles, kslice, (int)act, (int)swizzle);
  }
  if (S > 1) {
    const int64_t MN = (int64_t)M * N;
    const unsigned nblk = (unsigned)((MN + 2047) / 2048);
    hipLaunchKernelGGL(splitk_reduce_kernel, dim3(nblk), dim3(256),
                       0, stream, (uint16_t*)out.data_ptr(), scratch_ptr,
                       bias_ptr, MN, N, S, (int)act);
  }
}
