// Fused QKV epilogue: one pass over the fused qkv projection output.
//
// Reads qkv [T, (Hq+2*Hkv)*D] (row stride = the projection's output
// width, so no .contiguous() splits), applies rotate-half rope to Q and
// K, writes compact roped Q for the attention kernels, scatters roped K
// and V straight into the paged cache, and optionally emits compact K/V
// for the fresh-prefill attention path.
//
// Replaces 5 kernels (3 split copies + rope + reshape_and_cache) with 1:
// the split copies alone were ~0.5 ms/step at B=512 (see
// profiles/r01_decode_profile_v1.md batch-512 window breakdown).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

__global__ void rope_qkv_cache_kernel(
    const int64_t* __restrict__ positions,
    const uint16_t* __restrict__ qkv, int64_t qkv_stride,
    uint16_t* __restrict__ q_out,
    uint16_t* __restrict__ k_cache, uint16_t* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,
    uint16_t* __restrict__ k_out, uint16_t* __restrict__ v_out,
    const float* __restrict__ cos_sin,
    int Hq, int Hkv, int D, int block_size, int kv8) {
  const int t = blockIdx.x;
  const int64_t pos = positions[t];
  const float* cs = cos_sin + pos * D;
  const int half = D / 2;
  const uint16_t* src = qkv + (int64_t)t * qkv_stride;
  const int64_t slot = slot_mapping ? slot_mapping[t] : -1;
  int64_t cblock = 0;
  int coff = 0;
  if (slot >= 0) {
    cblock = slot / block_size;
    coff = (int)(slot % block_size);
  }

  // --- Q and K rotation pairs (q heads first, then k heads) ---
  const int qk_pairs = (Hq + Hkv) * half;
  for (int idx = threadIdx.x; idx < qk_pairs; idx += blockDim.x) {
    const int h = idx / half;
    const int d = idx % half;
    const float c = cs[d];
    const float sn = cs[half + d];
    const uint16_t* b = src + h * D;
    const float x1 = bf16_to_f32(b[d]);
    const float x2 = bf16_to_f32(b[d + half]);
    const uint16_t y1 = f32_to_bf16(x1 * c - x2 * sn);
    const uint16_t y2 = f32_to_bf16(x2 * c + x1 * sn);
    if (h < Hq) {
      uint16_t* o = q_out + ((int64_t)t * Hq + h) * D;
      o[d] = y1;
      o[d + half] = y2;
    } else {
      const int hk = h - Hq;
      if (k_out != nullptr) {
        uint16_t* o = k_out + ((int64_t)t * Hkv + hk) * D;
        o[d] = y1;
        o[d + half] = y2;
      }
      if (slot >= 0) {
        const int64_t ko =
            ((cblock * Hkv + hk) * (int64_t)block_size + coff) * D;
        if (kv8) {           // OCP e4m3 cache (kv_cache_dtype="fp8")
          uint8_t* o = (uint8_t*)k_cache + ko;
          o[d] = f32_to_fp8(x1 * c - x2 * sn);
          o[d + half] = f32_to_fp8(x2 * c + x1 * sn);
        } else {
          uint16_t* o = k_cache + ko;
          o[d] = y1;
          o[d + half] = y2;
        }
      }
    }
  }

  // --- V pass-through (vectorized) ---
  const uint16_t* vsrc = src + (Hq + Hkv) * D;
  const int nvec = Hkv * D / 8;
  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    const int h = (idx * 8) / D;
    const int d = (idx * 8) % D;
    const u16x8 val = *reinterpret_cast<const u16x8*>(vsrc + idx * 8);
    if (v_out != nullptr)
      *reinterpret_cast<u16x8*>(v_out + (int64_t)t * Hkv * D + idx * 8) =
          val;
    if (slot >= 0) {
      const int64_t vo =
          ((cblock * Hkv + h) * (int64_t)block_size + coff) * D + d;
      if (kv8) {
        u8x8 q8;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          q8[e] = f32_to_fp8(bf16_to_f32(val[e]));
        *reinterpret_cast<u8x8*>((uint8_t*)v_cache + vo) = q8;
      } else {
        *reinterpret_cast<u16x8*>(v_cache + vo) = val;
      }
    }
  }
}

}  // namespace

void rope_qkv_cache(torch::Tensor positions, torch::Tensor qkv,
                    torch::Tensor q_out,
                    c10::optional<torch::Tensor> k_cache,
                    c10::optional<torch::Tensor> v_cache,
                    c10::optional<torch::Tensor> slot_mapping,
                    c10::optional<torch::Tensor> k_out,
                    c10::optional<torch::Tensor> v_out,
                    torch::Tensor cos_sin, int64_t num_q_heads,
                    int64_t num_kv_heads, int64_t head_dim) {
  const int T = positions.size(0);
  if (T == 0) return;
  TORCH_CHECK(positions.scalar_type() == torch::kInt64);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  TORCH_CHECK(qkv.stride(-1) == 1, "qkv innermost must be contiguous");
  TORCH_CHECK(head_dim % 8 == 0);
  int block_size = 1;
  int kv8 = 0;
  uint16_t* kc = nullptr;
  uint16_t* vc = nullptr;
  const int64_t* slots = nullptr;
  if (k_cache.has_value()) {
    kv8 = (k_cache->scalar_type() == torch::kUInt8) ? 1 : 0;
    TORCH_CHECK(v_cache.has_value() && slot_mapping.has_value());
    TORCH_CHECK(slot_mapping->scalar_type() == torch::kInt64);
    block_size = k_cache->size(2);
    kc = (uint16_t*)k_cache->data_ptr();
    vc = (uint16_t*)v_cache->data_ptr();
    slots = slot_mapping->data_ptr<int64_t>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  const int threads = 256;
  hipLaunchKernelGGL(
      rope_qkv_cache_kernel, dim3(T), dim3(threads), 0, stream,
      positions.data_ptr<int64_t>(), (const uint16_t*)qkv.data_ptr(),
      qkv.stride(0), (uint16_t*)q_out.data_ptr(), kc, vc, slots,
      k_out.has_value() ? (uint16_t*)k_out->data_ptr() : nullptr,
      v_out.has_value() ? (uint16_t*)v_out->data_ptr() : nullptr,
      cos_sin.data_ptr<float>(), (int)num_q_heads, (int)num_kv_heads,
      (int)head_dim, block_size, kv8);
}
