#include "hip/hip_runtime.h"
// Rotary position embedding (neox / Llama rotate-half style), in-place on
// Q and K. cos/sin tables are precomputed on HOST (guide Appendix B:
// on-device trig turns memory-bound into VALU-bound).
//
// q: [T, Hq*D], k: [T, Hk*D], positions: [T] int32/int64,
// cos_sin: [max_pos, D] fp32 — row = [cos(0..D/2), sin(0..D/2)].
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

__global__ void rope_kernel(const int64_t* __restrict__ positions,
                            uint16_t* __restrict__ q,
                            uint16_t* __restrict__ k,
                            const float* __restrict__ cos_sin, int num_q,
                            int num_k, int D) {
  const int t = blockIdx.x;
  const int64_t pos = positions[t];
  const float* cs = cos_sin + pos * D;
  const int half = D / 2;
  const int total = (num_q + num_k) * half;  // rotation pairs for this token

  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int h = idx / half;
    const int d = idx % half;
    uint16_t* base;
    if (h < num_q) {
      base = q + (int64_t)t * num_q * D + h * D;
    } else {
      base = k + (int64_t)t * num_k * D + (h - num_q) * D;
    }
    const float c = cs[d];
    const float s = cs[half + d];
    const float x1 = bf16_to_f32(base[d]);
    const float x2 = bf16_to_f32(base[d + half]);
    base[d] = f32_to_bf16(x1 * c - x2 * s);
    base[d + half] = f32_to_bf16(x2 * c + x1 * s);
  }
}

}  // namespace

void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, torch::Tensor cos_sin, int64_t head_dim) {
  const int T = positions.size(0);
  const int num_q = q.size(-1) / head_dim;
  const int num_k = k.size(-1) / head_dim;
  TORCH_CHECK(positions.scalar_type() == torch::kInt64);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  auto stream = at::hip::getCurrentHIPStream();
  int threads = std::min(256, (int)((num_q + num_k) * head_dim / 2));
  threads = std::max(threads, 64);
  hipLaunchKernelGGL(rope_kernel, dim3(T), dim3(threads), 0, stream,
                     positions.data_ptr<int64_t>(), (uint16_t*)q.data_ptr(),
                     (uint16_t*)k.data_ptr(), cos_sin.data_ptr<float>(),
                     num_q, num_k, (int)head_dim);
}
