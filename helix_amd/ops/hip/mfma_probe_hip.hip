#include "hip/hip_runtime.h"
// MFMA layout verification probe: computes D = A @ B for a single
// 16x16x32 bf16 tile using the fragment layouts assumed across the
// kernel library. tests/test_ops_gpu.py compares against torch with
// ASYMMETRIC random operands (guide §3 ERRATA: symmetric inputs can
// hide a transposed layout).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

__global__ void mfma_probe_kernel(float* __restrict__ d,
                                  const uint16_t* __restrict__ a,  // [16][32]
                                  const uint16_t* __restrict__ b)  // [32][16]
{
  const int lane = threadIdx.x & (WAVE - 1);
  const int lane_hi = lane >> 4, lane_lo = lane & 15;
  bf16x8 a_frag, b_frag;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    // A: lane holds A[row=lane_lo][k=lane_hi*8+i]
    reinterpret_cast<uint16_t*>(&a_frag)[i] = a[lane_lo * 32 + lane_hi * 8 + i];
    // B: lane holds B[k=lane_hi*8+i][col=lane_lo]
    reinterpret_cast<uint16_t*>(&b_frag)[i] = b[(lane_hi * 8 + i) * 16 + lane_lo];
  }
  floatx4 c = floatx4{0, 0, 0, 0};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, c, 0, 0, 0);
  // C/D: lane holds C[row=lane_hi*4+r][col=lane_lo]
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(lane_hi * 4 + r) * 16 + lane_lo] = c[r];
}

__global__ void mfma_probe_fp8_kernel(float* __restrict__ d,
                                      const uint8_t* __restrict__ a,  // [16][128]
                                      const uint8_t* __restrict__ b,  // [128][16]
                                      int scale_a, int scale_b)
{
  const int lane = threadIdx.x & (WAVE - 1);
  const int lane_hi = lane >> 4, lane_lo = lane & 15;
  typedef int v8i __attribute__((ext_vector_type(8)));
  v8i a_frag, b_frag;
#pragma unroll
  for (int i = 0; i < 32; ++i) {
    // assumed A: lane holds A[row=lane_lo][k=lane_hi*32+i]
    reinterpret_cast<uint8_t*>(&a_frag)[i] = a[lane_lo * 128 + lane_hi * 32 + i];
    // assumed B: lane holds B[k=lane_hi*32+i][col=lane_lo]
    reinterpret_cast<uint8_t*>(&b_frag)[i] = b[(lane_hi * 32 + i) * 16 + lane_lo];
  }
  floatx4 c = floatx4{0, 0, 0, 0};
  // fmt 0 = fp8(e4m3); unity scales via caller (0x7F7F7F7F = e8m0 2^0)
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a_frag, b_frag, c, 0, 0, 0, scale_a, 0, scale_b);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(lane_hi * 4 + r) * 16 + lane_lo] = c[r];
}

}  // namespace

void mfma_probe_fp8(torch::Tensor d, torch::Tensor a, torch::Tensor b,
                    int64_t scale_a, int64_t scale_b) {
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 128);
  TORCH_CHECK(b.size(0) == 128 && b.size(1) == 16);
  TORCH_CHECK(a.scalar_type() == torch::kUInt8);
  TORCH_CHECK(d.scalar_type() == torch::kFloat32);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_fp8_kernel, dim3(1), dim3(64), 0, stream,
                     d.data_ptr<float>(), (const uint8_t*)a.data_ptr(),
                     (const uint8_t*)b.data_ptr(), (int)scale_a,
                     (int)scale_b);
}

void mfma_probe(torch::Tensor d, torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32);
  TORCH_CHECK(b.size(0) == 32 && b.size(1) == 16);
  TORCH_CHECK(d.scalar_type() == torch::kFloat32);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     d.data_ptr<float>(), (const uint16_t*)a.data_ptr(),
                     (const uint16_t*)b.data_ptr());
}
