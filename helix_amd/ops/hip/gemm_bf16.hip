// Hand-written bf16 MFMA GEMM for gfx950: out[M,N] = x[M,K] @ w[N,K]^T.
//
// This is the "embedding-model GEMM" of the north star (feeds the
// bge-class encoder -> PGVector RAG path) and the base for fused-epilogue
// linear layers. Plain library GEMMs elsewhere go through hipBLASLt.
//
// Structure: 128x128 output tile / 4 waves (each wave 64x64 via 4x4
// f32_16x16x32_bf16 fragments), BK=64 K-steps staged to LDS with
// global_load_lds width-16 (the guide's verified ~900 TF "m97 structure").
// Epilogue fuses optional bias + {none, silu-mul pairup?, gelu} later.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int BM = 128, BN = 128, BK = 64;

__global__ __launch_bounds__(256) void gemm_bf16_kernel(
    uint16_t* __restrict__ out, const uint16_t* __restrict__ x,
    const uint16_t* __restrict__ w, const uint16_t* __restrict__ bias,
    int M, int N, int K, int act /*0=none,1=gelu_tanh*/) {
  __shared__ uint16_t a_lds[BM * BK];
  __shared__ uint16_t b_lds[BN * BK];

  const int tile_n = blockIdx.x;
  const int tile_m = blockIdx.y;
  const int m0 = tile_m * BM, n0 = tile_n * BN;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int wr = wid / 2, wc = wid % 2;  // wave grid 2x2 -> 64x64 each
  const int lane_hi = lane >> 4, lane_lo = lane & 15;

  floatx4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = floatx4{0, 0, 0, 0};

  // Each wave stages 1/4 of each LDS tile per K-step: 128*64*2B / 4 waves
  // = 4 KiB/wave = 4 iterations of 64 lanes x 16 B (global_load_lds).
  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const int k0 = kt * BK;
    __syncthreads();
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      // element index within the [128][BK] tile, 8 bf16 per lane
      const int e = ((wid * 4 + it) * WAVE + lane) * 8;
      const int row = e / BK, col = e % BK;
      const int arow = min(m0 + row, M - 1);
      const int brow = min(n0 + row, N - 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(x +
              (int64_t)arow * K + k0 + col),
          (__attribute__((address_space(3))) uint32_t*)(a_lds +
              (wid * 4 + it) * WAVE * 8),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(w +
              (int64_t)brow * K + k0 + col),
          (__attribute__((address_space(3))) uint32_t*)(b_lds +
              (wid * 4 + it) * WAVE * 8),
          16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int row = wr * 64 + mi * 16 + lane_lo;
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            a_lds + row * BK + kk * 32 + lane_hi * 8);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int row = wc * 64 + ni * 16 + lane_lo;
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            b_lds + row * BK + kk * 32 + lane_hi * 8);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
  }

  // Epilogue
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wc * 64 + ni * 16 + lane_lo;
      if (col >= N) continue;
      const float b = bias ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + mi * 16 + lane_hi * 4 + r;
        if (row >= M) continue;
        float v = acc[mi][ni][r] + b;
        if (act == 1) {
          const float c = 0.7978845608028654f;
          v = 0.5f * v * (1.f + tanhf(c * (v + 0.044715f * v * v * v)));
        }
        out[(int64_t)row * N + col] = f32_to_bf16(v);
      }
    }
  }
}

}  // namespace

void gemm_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               c10::optional<torch::Tensor> bias, int64_t act) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K);
  TORCH_CHECK(K % BK == 0, "K must be a multiple of 64");
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  auto stream = at::hip::getCurrentHIPStream();
  const uint16_t* bias_ptr =
      bias.has_value() ? (const uint16_t*)bias->data_ptr() : nullptr;
  hipLaunchKernelGGL(gemm_bf16_kernel, dim3(cdiv(N, BN), cdiv(M, BM)),
                     dim3(256), 0, stream, (uint16_t*)out.data_ptr(),
                     (const uint16_t*)x.data_ptr(),
                     (const uint16_t*)w.data_ptr(), bias_ptr, M, N, K,
                     (int)act);
}
