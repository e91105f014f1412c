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

// EB = element bytes (2 = bf16 cache, 1 = fp8/e4m3 cache: inputs are
// already-quantized byte views with identical [T, Hkv, D] geometry).
template <int EB>
__global__ void reshape_and_cache_kernel(
    const uint8_t* __restrict__ k, const uint8_t* __restrict__ v,
    uint8_t* __restrict__ k_cache, uint8_t* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping, int Hkv, int D, int block_size) {
  const int t = blockIdx.x;
  const int64_t slot = slot_mapping[t];
  if (slot < 0) return;
  const int64_t block = slot / block_size;
  const int off = slot % block_size;
  const int row_bytes = Hkv * D * EB;
  const int nvec = row_bytes / 16;

  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    const int b16 = idx * 16;               // byte offset within the row
    const int h = b16 / (D * EB);
    const int d = b16 % (D * EB);
    const int64_t src = (int64_t)t * row_bytes + b16;
    const int64_t dst =
        (((block * Hkv + h) * (int64_t)block_size + off) * D) * EB + d;
    *reinterpret_cast<u8x16*>(k_cache + dst) =
        *reinterpret_cast<const u8x16*>(k + src);
    *reinterpret_cast<u8x16*>(v_cache + dst) =
        *reinterpret_cast<const u8x16*>(v + src);
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
  TORCH_CHECK(D % 16 == 0);
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  const bool kv8 = k_cache.scalar_type() == torch::kUInt8;
  TORCH_CHECK(!kv8 || k.scalar_type() == torch::kUInt8,
              "fp8 cache takes pre-quantized uint8 K/V views");
  auto stream = at::hip::getCurrentHIPStream();
  const int eb = kv8 ? 1 : 2;
  const int threads = std::min(256, Hkv * D * eb / 16);
  if (kv8) {
    hipLaunchKernelGGL(reshape_and_cache_kernel<1>, dim3(T), dim3(threads),
                       0, stream, (const uint8_t*)k.data_ptr(),
                       (const uint8_t*)v.data_ptr(),
                       (uint8_t*)k_cache.data_ptr(),
                       (uint8_t*)v_cache.data_ptr(),
                       slot_mapping.data_ptr<int64_t>(), Hkv, D, block_size);
  } else {
    hipLaunchKernelGGL(reshape_and_cache_kernel<2>, dim3(T), dim3(threads),
                       0, stream, (const uint8_t*)k.data_ptr(),
                       (const uint8_t*)v.data_ptr(),
                       (uint8_t*)k_cache.data_ptr(),
                       (uint8_t*)v_cache.data_ptr(),
                       slot_mapping.data_ptr<int64_t>(), Hkv, D, block_size);
  }
}
