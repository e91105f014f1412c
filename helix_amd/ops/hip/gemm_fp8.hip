// Hand-written MX-fp8 (OCP e4m3) MFMA GEMM for gfx950:
//   out[M,N] = dequant( x8[M,K] @ w8[N,K]^T ) with per-row activation
//   scales and per-output-channel weight scales applied in the EPILOGUE.
//
// Uses the gfx950-only block-scaled `mfma_scale_f32_16x16x128_f8f6f4`
// intrinsic (2x the bf16 MFMA rate; ~5 PF dense peak) with UNITY e8m0
// hardware scales (0x7F = 2^0 in every byte): the HW scale path is
// format-verified by tests/test_ops_gpu.py::test_mfma_probe_fp8_*, and
// keeping dequant in the epilogue sidesteps the per-32-block scale-lane
// mapping entirely while preserving full per-channel accuracy.
//
// Structure mirrors gemm_bf16.hip (m97 128x128 tile, 4 waves, LDS
// staging via global_load_lds width-16) with BK=128 to match the
// intrinsic's K and 1-byte elements (same 32 KiB LDS footprint).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

constexpr int BM = 128, BN = 128, BK = 128;
constexpr int UNITY = 0x7F7F7F7F;   // e8m0 2^0 in all four bytes

typedef int v8i __attribute__((ext_vector_type(8)));

__global__ __launch_bounds__(256) void gemm_fp8_kernel(
    uint16_t* __restrict__ out, const uint8_t* __restrict__ x,
    const uint8_t* __restrict__ w, const float* __restrict__ x_scale,
    const float* __restrict__ w_scale, const uint16_t* __restrict__ bias,
    int M, int N, int K, int act /*0=none,1=gelu_tanh*/) {
  __shared__ uint8_t a_lds[BM * BK];
  __shared__ uint8_t b_lds[BN * BK];

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

  // 128x128 B per tile / 4 waves = 4 KiB/wave = 4 x 64 lanes x 16 B.
  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const int k0 = kt * BK;
    __syncthreads();
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int e = ((wid * 4 + it) * WAVE + lane) * 16;  // byte index
      const int row = e / BK, col = e % BK;
      const int arow = min(m0 + row, M - 1);
      const int brow = min(n0 + row, N - 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(x +
              (int64_t)arow * K + k0 + col),
          (__attribute__((address_space(3))) uint32_t*)(a_lds +
              (wid * 4 + it) * WAVE * 16),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(w +
              (int64_t)brow * K + k0 + col),
          (__attribute__((address_space(3))) uint32_t*)(b_lds +
              (wid * 4 + it) * WAVE * 16),
          16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    v8i a_frag[4], b_frag[4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int row = wr * 64 + mi * 16 + lane_lo;
      a_frag[mi] = *reinterpret_cast<const v8i*>(
          a_lds + row * BK + lane_hi * 32);
    }
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int row = wc * 64 + ni * 16 + lane_lo;
      b_frag[ni] = *reinterpret_cast<const v8i*>(
          b_lds + row * BK + lane_hi * 32);
    }
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0, UNITY, 0, UNITY);
  }

  // Epilogue: per-row activation scale x per-column weight scale.
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wc * 64 + ni * 16 + lane_lo;
      if (col >= N) continue;
      const float ws = w_scale[col];
      const float b = bias ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + mi * 16 + lane_hi * 4 + r;
        if (row >= M) continue;
        float v = acc[mi][ni][r] * x_scale[row] * ws + b;
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

void gemm_fp8(torch::Tensor out, torch::Tensor x, torch::Tensor w,
              torch::Tensor x_scale, torch::Tensor w_scale,
              c10::optional<torch::Tensor> bias, int64_t act) {
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K);
  TORCH_CHECK(K % BK == 0, "K must be a multiple of 128");
  TORCH_CHECK(x.scalar_type() == torch::kUInt8 &&
              w.scalar_type() == torch::kUInt8,
              "x/w must be e4m3 bytes (view(torch.uint8))");
  TORCH_CHECK(x_scale.scalar_type() == torch::kFloat32 &&
              w_scale.scalar_type() == torch::kFloat32);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  auto stream = at::hip::getCurrentHIPStream();
  const uint16_t* bias_ptr =
      bias.has_value() ? (const uint16_t*)bias->data_ptr() : nullptr;
  hipLaunchKernelGGL(gemm_fp8_kernel, dim3(cdiv(N, BN), cdiv(M, BM)),
                     dim3(256), 0, stream, (uint16_t*)out.data_ptr(),
                     (const uint8_t*)x.data_ptr(),
                     (const uint8_t*)w.data_ptr(),
                     x_scale.data_ptr<float>(), w_scale.data_ptr<float>(),
                     bias_ptr, M, N, K, (int)act);
}
