// Fused activation kernels (CDNA4). Memory-bound → bf16x8 vectorized,
// grid-stride, grid capped (guide G11).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

// out[t, i] = silu(x[t, i]) * x[t, I + i]   — Llama/Mistral gated MLP.
__global__ void silu_and_mul_kernel(uint16_t* __restrict__ out,
                                    const uint16_t* __restrict__ x,
                                    int64_t T, int I) {
  const int nvec = I / 8;
  const int64_t total = T * nvec;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t t = idx / nvec;
    const int v = idx % nvec;
    float g[8], u[8];
    load_bf16x8(x + t * 2 * I + v * 8, g);
    load_bf16x8(x + t * 2 * I + I + v * 8, u);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float s = g[i] / (1.f + __expf(-g[i]));
      g[i] = s * u[i];
    }
    store_bf16x8(out + t * I + v * 8, g);
  }
}

// out = gelu(x) (tanh approx) — encoder (bge-class) MLP.
__global__ void gelu_tanh_kernel(uint16_t* __restrict__ out,
                                 const uint16_t* __restrict__ x,
                                 int64_t total_vec) {
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total_vec; idx += (int64_t)gridDim.x * blockDim.x) {
    float e[8];
    load_bf16x8(x + idx * 8, e);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float c = 0.7978845608028654f;  // sqrt(2/pi)
      const float t = tanhf(c * (e[i] + 0.044715f * e[i] * e[i] * e[i]));
      e[i] = 0.5f * e[i] * (1.f + t);
    }
    store_bf16x8(out + idx * 8, e);
  }
}

}  // namespace

void silu_and_mul(torch::Tensor out, torch::Tensor x) {
  const int I = out.size(-1);
  const int64_t T = out.numel() / I;
  TORCH_CHECK(x.size(-1) == 2 * I, "input must be [.., 2*I]");
  TORCH_CHECK(I % 8 == 0);
  auto stream = at::hip::getCurrentHIPStream();
  const int64_t total = T * (I / 8);
  const int grid = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(grid), dim3(256), 0, stream,
                     (uint16_t*)out.data_ptr(), (const uint16_t*)x.data_ptr(),
                     T, I);
}

void gelu_tanh(torch::Tensor out, torch::Tensor x) {
  TORCH_CHECK(x.numel() % 8 == 0);
  const int64_t total_vec = x.numel() / 8;
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = (int)std::min<int64_t>((total_vec + 255) / 256, 2048);
  hipLaunchKernelGGL(gelu_tanh_kernel, dim3(grid), dim3(256), 0, stream,
                     (uint16_t*)out.data_ptr(), (const uint16_t*)x.data_ptr(),
                     total_vec);
}
