// Common helpers for helix_amd CDNA4 (gfx950) kernels.
//
// MI355X-native: wave width is 64, LDS is 160 KiB/CU with 32 x 4B banks,
// MFMA bf16 tile is 16x16x32 (gfx950 doubled-K family). No CUDA compat.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64

#define HIP_CHECK(expr)                                                      \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                      \
      abort();                                                               \
    }                                                                        \
  } while (0)

namespace helix {

// ---------------------------------------------------------------------------
// Vector types. bf16 handled as raw ushort bits; fp32 accumulate everywhere.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) float f32x8;
typedef __attribute__((ext_vector_type(4))) uint16_t u16x4;
typedef __attribute__((ext_vector_type(8))) uint16_t u16x8;   // 8 bf16 = 16B
typedef __attribute__((ext_vector_type(2))) uint32_t u32x2;
typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;
// MFMA fragment types for f32_16x16x32_bf16
typedef __attribute__((ext_vector_type(8))) uint8_t u8x8;     // 8 fp8 = 8B
typedef __attribute__((ext_vector_type(16))) uint8_t u8x16;   // 16 fp8 = 16B
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;    // A/B operand (4 VGPRs)
typedef __attribute__((ext_vector_type(4))) float floatx4;    // C/D accumulator

__device__ __forceinline__ float bf16_to_f32(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = uint32_t(h) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  union { float f; uint32_t u; } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t lsb = (v.u >> 16) & 1u;
  v.u += 0x7fffu + lsb;
  return uint16_t(v.u >> 16);
}

// Load 8 bf16 (16 bytes) and widen to 8 fp32.
__device__ __forceinline__ void load_bf16x8(const uint16_t* p, float* out) {
  u16x8 v = *reinterpret_cast<const u16x8*>(p);
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf16_to_f32(v[i]);
}

__device__ __forceinline__ void store_bf16x8(uint16_t* p, const float* in) {
  u16x8 v;
#pragma unroll
  for (int i = 0; i < 8; ++i) v[i] = f32_to_bf16(in[i]);
  *reinterpret_cast<u16x8*>(p) = v;
}

// ---------------------------------------------------------------------------
// Wave (64-lane) reductions via shuffles.
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-level reduction helpers. `scratch` must hold >= blockDim.x/WAVE floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = blockDim.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) r += __shfl_xor(r, off, WAVE);
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  float out = scratch[0];
  __syncthreads();
  return out;
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = blockDim.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) r = fmaxf(r, __shfl_xor(r, off, WAVE));
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  float out = scratch[0];
  __syncthreads();
  return out;
}

constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }

// splitmix64 — cheap per-element RNG for sampling kernels.
__device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

// uniform in (0,1]
__device__ __forceinline__ float u64_to_uniform(uint64_t r) {
  return (float)((r >> 40) + 1) * (1.0f / 16777216.0f);
}

// OCP e4m3 <-> f32 via the gfx950 cvt units (NOT the MI300X fnuz
// encoding). Byte-select variant for scalar use; pack for stores.
__device__ __forceinline__ float fp8_to_f32(uint8_t b) {
  return __builtin_amdgcn_cvt_f32_fp8((int)b, 0);
}
__device__ __forceinline__ uint8_t f32_to_fp8(float f) {
  return (uint8_t)(__builtin_amdgcn_cvt_pk_fp8_f32(f, f, 0, false) & 0xFF);
}

}  // namespace helix
