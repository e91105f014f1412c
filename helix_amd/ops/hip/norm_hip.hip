#include "hip/hip_runtime.h"
// RMSNorm kernels (CDNA4 / gfx950).
//
// Memory-bound: fully vectorized bf16x8 loads/stores (guide G13), fp32
// accumulation, one workgroup per token row.
//
// Reference behavior: replaces the RMSNorm the reference delegates to vLLM
// containers (see SURVEY.md §2.8 "RMSNorm, residual add").
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

// out[t, :] = x[t, :] / rms(x[t, :]) * w
// H must be a multiple of 8. One block per token.
template <int BLOCK>
__global__ void rms_norm_kernel(uint16_t* __restrict__ out,
                                const uint16_t* __restrict__ x,
                                const uint16_t* __restrict__ w, float eps,
                                int H) {
  __shared__ float red[BLOCK / WAVE];
  const int t = blockIdx.x;
  const uint16_t* row = x + (int64_t)t * H;
  uint16_t* orow = out + (int64_t)t * H;

  float ss = 0.f;
  const int nvec = H / 8;
  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float e[8];
    load_bf16x8(row + v * 8, e);
#pragma unroll
    for (int i = 0; i < 8; ++i) ss += e[i] * e[i];
  }
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / H + eps);

  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float e[8], ww[8];
    load_bf16x8(row + v * 8, e);
    load_bf16x8(w + v * 8, ww);
#pragma unroll
    for (int i = 0; i < 8; ++i) e[i] = e[i] * inv * ww[i];
    store_bf16x8(orow + v * 8, e);
  }
}

// residual' = x + residual ; x' = rmsnorm(residual') * w    (both in-place)
template <int BLOCK>
__global__ void fused_add_rms_norm_kernel(uint16_t* __restrict__ x,
                                          uint16_t* __restrict__ residual,
                                          const uint16_t* __restrict__ w,
                                          float eps, int H) {
  __shared__ float red[BLOCK / WAVE];
  const int t = blockIdx.x;
  uint16_t* xrow = x + (int64_t)t * H;
  uint16_t* rrow = residual + (int64_t)t * H;

  float ss = 0.f;
  const int nvec = H / 8;
  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float a[8], b[8];
    load_bf16x8(xrow + v * 8, a);
    load_bf16x8(rrow + v * 8, b);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      a[i] += b[i];
      ss += a[i] * a[i];
    }
    store_bf16x8(rrow + v * 8, a);  // new residual
  }
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / H + eps);

  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float e[8], ww[8];
    load_bf16x8(rrow + v * 8, e);
    load_bf16x8(w + v * 8, ww);
#pragma unroll
    for (int i = 0; i < 8; ++i) e[i] = e[i] * inv * ww[i];
    store_bf16x8(xrow + v * 8, e);
  }
}

// LayerNorm (for encoder / bge-class models): out = (x - mu)/sigma * w + b
template <int BLOCK>
__global__ void layer_norm_kernel(uint16_t* __restrict__ out,
                                  const uint16_t* __restrict__ x,
                                  const uint16_t* __restrict__ w,
                                  const uint16_t* __restrict__ b, float eps,
                                  int H) {
  __shared__ float red[BLOCK / WAVE];
  const int t = blockIdx.x;
  const uint16_t* row = x + (int64_t)t * H;
  uint16_t* orow = out + (int64_t)t * H;

  float s = 0.f;
  const int nvec = H / 8;
  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float e[8];
    load_bf16x8(row + v * 8, e);
#pragma unroll
    for (int i = 0; i < 8; ++i) s += e[i];
  }
  const float mu = block_reduce_sum(s, red) / H;
  float ss = 0.f;
  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float e[8];
    load_bf16x8(row + v * 8, e);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float d = e[i] - mu;
      ss += d * d;
    }
  }
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / H + eps);
  for (int v = threadIdx.x; v < nvec; v += BLOCK) {
    float e[8], ww[8], bb[8];
    load_bf16x8(row + v * 8, e);
    load_bf16x8(w + v * 8, ww);
    load_bf16x8(b + v * 8, bb);
#pragma unroll
    for (int i = 0; i < 8; ++i) e[i] = (e[i] - mu) * inv * ww[i] + bb[i];
    store_bf16x8(orow + v * 8, e);
  }
}

}  // namespace

void rms_norm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
              double eps) {
  const int T = x.numel() / x.size(-1);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((rms_norm_kernel<256>), dim3(T), dim3(256), 0, stream,
                     (uint16_t*)out.data_ptr(), (const uint16_t*)x.data_ptr(),
                     (const uint16_t*)w.data_ptr(), (float)eps, H);
}

void fused_add_rms_norm(torch::Tensor x, torch::Tensor residual,
                        torch::Tensor w, double eps) {
  const int T = x.numel() / x.size(-1);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((fused_add_rms_norm_kernel<256>), dim3(T), dim3(256), 0,
                     stream, (uint16_t*)x.data_ptr(),
                     (uint16_t*)residual.data_ptr(),
                     (const uint16_t*)w.data_ptr(), (float)eps, H);
}

void layer_norm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                torch::Tensor b, double eps) {
  const int T = x.numel() / x.size(-1);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((layer_norm_kernel<256>), dim3(T), dim3(256), 0, stream,
                     (uint16_t*)out.data_ptr(), (const uint16_t*)x.data_ptr(),
                     (const uint16_t*)w.data_ptr(),
                     (const uint16_t*)b.data_ptr(), (float)eps, H);
}
