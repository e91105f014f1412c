#include "hip/hip_runtime.h"
// Paged KV cache write: scatter this step's K/V rows into their cache slots.
//
// Cache layout (chosen for the decode kernel's per-token contiguous reads):
//   k_cache, v_cache: [num_blocks, Hkv, block_size, D] bf16
// slot_mapping[t] = block_idx * block_size + offset   (int64, -1 = skip)
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

__global__ void reshape_and_cache_kernel(
    const uint16_t* __restrict__ k, const uint16_t* __restrict__ v,
    uint16_t* __restrict__ k_cache, uint16_t* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping, int Hkv, int D, int block_size) {
  const int t = blockIdx.x;
  const int64_t slot = slot_mapping[t];
  if (slot < 0) return;
  const int64_t block = slot / block_size;
  const int off = slot % block_size;
  const int nvec = Hkv * D / 8;

  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    const int h = (idx * 8) / D;
    const int d = (idx * 8) % D;
    const int64_t src = (int64_t)t * Hkv * D + idx * 8;
    const int64_t dst =
        ((block * Hkv + h) * (int64_t)block_size + off) * D + d;
    *reinterpret_cast<u16x8*>(k_cache + dst) =
        *reinterpret_cast<const u16x8*>(k + src);
    *reinterpret_cast<u16x8*>(v_cache + dst) =
        *reinterpret_cast<const u16x8*>(v + src);
  }
}

}  // namespace

void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping) {
  const int T = slot_mapping.size(0);
  const int Hkv = k_cache.size(1);
  const int block_size = k_cache.size(2);
  const int D = k_cache.size(3);
  TORCH_CHECK(D % 8 == 0);
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  auto stream = at::hip::getCurrentHIPStream();
  const int threads = std::min(256, Hkv * D / 8);
  hipLaunchKernelGGL(reshape_and_cache_kernel, dim3(T), dim3(threads), 0,
                     stream, (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(),
                     (uint16_t*)k_cache.data_ptr(),
                     (uint16_t*)v_cache.data_ptr(),
                     slot_mapping.data_ptr<int64_t>(), Hkv, D, block_size);
}
