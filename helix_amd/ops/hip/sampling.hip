// Token sampling kernels (CDNA4).
//
// Replaces the "logits -> next token" sampling the reference delegates to
// vLLM (SURVEY.md §2.8 "Logits GEMM + sampling").
//
// Exact temperature sampling without a vocab sort via the Gumbel-max trick:
//   next = argmax(logits / T + G_i),  G_i = -log(-log(U_i))
// Greedy (T == 0) is a plain argmax. One 256-thread block per row;
// vectorized bf16x8 logit reads (memory-bound over V ~ 128k).
// top-k / top-p restriction is applied upstream (engine masks logits).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace helix;

namespace {

template <typename T>
__device__ __forceinline__ float as_f32(T v);
template <>
__device__ __forceinline__ float as_f32<uint16_t>(uint16_t v) {
  return bf16_to_f32(v);
}
template <>
__device__ __forceinline__ float as_f32<float>(float v) {
  return v;
}

template <typename T>
__global__ void sample_kernel(int64_t* __restrict__ out,
                              const T* __restrict__ logits,
                              const float* __restrict__ temperatures,
                              const uint64_t* __restrict__ seeds, int V) {
  const int row = blockIdx.x;
  const T* lrow = logits + (int64_t)row * V;
  const float temp = temperatures[row];
  const uint64_t seed = seeds[row];
  const bool greedy = temp <= 0.f;
  const float inv_t = greedy ? 1.f : 1.f / temp;

  float best = -INFINITY;
  int best_i = 0;
  // Vectorized main loop: 8 logits per iteration (memory-bound over V).
  const int v8 = (V / 8) * 8;
  for (int base = threadIdx.x * 8; base < v8; base += blockDim.x * 8) {
    float vals[8];
    if constexpr (sizeof(T) == 2) {
      load_bf16x8(lrow + base, vals);
    } else {
      const f32x4* p = reinterpret_cast<const f32x4*>(lrow + base);
      f32x4 a = p[0], b = p[1];
#pragma unroll
      for (int i = 0; i < 4; ++i) { vals[i] = a[i]; vals[4 + i] = b[i]; }
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float val = vals[i] * inv_t;
      const int idx = base + i;
      if (!greedy) {
        const float u = u64_to_uniform(splitmix64(seed ^ (uint64_t)idx));
        val += -__logf(-__logf(u));
      }
      if (val > best || (val == best && idx < best_i)) {
        best = val;
        best_i = idx;
      }
    }
  }
  for (int i = v8 + threadIdx.x; i < V; i += blockDim.x) {
    float val = as_f32(lrow[i]) * inv_t;
    if (!greedy) {
      const float u = u64_to_uniform(splitmix64(seed ^ (uint64_t)i));
      val += -__logf(-__logf(u));
    }
    if (val > best || (val == best && i < best_i)) {
      best = val;
      best_i = i;
    }
  }
  // block argmax reduce
  __shared__ float smax[4];
  __shared__ int sidx[4];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(best_i, off, WAVE);
    if (ov > best || (ov == best && oi < best_i)) {
      best = ov;
      best_i = oi;
    }
  }
  if (lane == 0) {
    smax[wid] = best;
    sidx[wid] = best_i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w) {
      if (smax[w] > best || (smax[w] == best && sidx[w] < best_i)) {
        best = smax[w];
        best_i = sidx[w];
      }
    }
    out[row] = best_i;
  }
}

// Two-stage variant: V split across NSPLIT blocks per row so small
// batches still fill the chip; partial (max, idx) pairs reduced by a
// second tiny kernel.
template <typename T>
__global__ void sample_partial_kernel(float* __restrict__ pmax,
                                      int* __restrict__ pidx,
                                      const T* __restrict__ logits,
                                      const float* __restrict__ temperatures,
                                      const uint64_t* __restrict__ seeds,
                                      int V, int nsplit) {
  const int row = blockIdx.x;
  const int split = blockIdx.y;
  const int seg = (V + nsplit - 1) / nsplit;
  const int lo = split * seg;
  const int hi = min(V, lo + seg);
  const T* lrow = logits + (int64_t)row * V;
  const float temp = temperatures[row];
  const uint64_t seed = seeds[row];
  const bool greedy = temp <= 0.f;
  const float inv_t = greedy ? 1.f : 1.f / temp;

  float best = -INFINITY;
  int best_i = lo;
  const int lo8 = lo + ((hi - lo) / 8) * 8;
  for (int base = lo + threadIdx.x * 8; base + 8 <= hi;
       base += blockDim.x * 8) {
    float vals[8];
    if constexpr (sizeof(T) == 2) {
      load_bf16x8((const uint16_t*)lrow + base, vals);
    } else {
      const f32x4* p = reinterpret_cast<const f32x4*>(lrow + base);
      f32x4 a = p[0], b = p[1];
#pragma unroll
      for (int i = 0; i < 4; ++i) { vals[i] = a[i]; vals[4 + i] = b[i]; }
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = base + i;
      float val = vals[i] * inv_t;
      if (!greedy) {
        const float u = u64_to_uniform(splitmix64(seed ^ (uint64_t)idx));
        val += -__logf(-__logf(u));
      }
      if (val > best || (val == best && idx < best_i)) {
        best = val;
        best_i = idx;
      }
    }
  }
  for (int i = lo8 + threadIdx.x; i < hi; i += blockDim.x) {
    float val = as_f32(lrow[i]) * inv_t;
    if (!greedy) {
      const float u = u64_to_uniform(splitmix64(seed ^ (uint64_t)i));
      val += -__logf(-__logf(u));
    }
    if (val > best || (val == best && i < best_i)) {
      best = val;
      best_i = i;
    }
  }
  // block argmax
  __shared__ float smax[4];
  __shared__ int sidx[4];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(best_i, off, WAVE);
    if (ov > best || (ov == best && oi < best_i)) {
      best = ov;
      best_i = oi;
    }
  }
  if (lane == 0) { smax[wid] = best; sidx[wid] = best_i; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w)
      if (smax[w] > best || (smax[w] == best && sidx[w] < best_i)) {
        best = smax[w];
        best_i = sidx[w];
      }
    pmax[row * nsplit + split] = best;
    pidx[row * nsplit + split] = best_i;
  }
}

__global__ void sample_reduce_kernel(int64_t* __restrict__ out,
                                     const float* __restrict__ pmax,
                                     const int* __restrict__ pidx,
                                     int nsplit) {
  const int row = blockIdx.x;  // one thread per row
  float best = -INFINITY;
  int best_i = 0;
  for (int s = 0; s < nsplit; ++s) {
    const float v = pmax[row * nsplit + s];
    const int i = pidx[row * nsplit + s];
    if (v > best || (v == best && i < best_i)) {
      best = v;
      best_i = i;
    }
  }
  out[row] = best_i;
}

}  // namespace

void sample_tokens(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temperatures, torch::Tensor seeds) {
  const int B = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(out.scalar_type() == torch::kInt64);
  TORCH_CHECK(temperatures.scalar_type() == torch::kFloat32);
  TORCH_CHECK(seeds.scalar_type() == torch::kUInt64 ||
              seeds.scalar_type() == torch::kInt64);
  auto stream = at::hip::getCurrentHIPStream();
  // fill the chip: ~1024 blocks
  int nsplit = std::max(1, std::min(64, 1024 / std::max(1, B)));
  auto opts = torch::TensorOptions().device(logits.device());
  auto pmax = torch::empty({B, nsplit}, opts.dtype(torch::kFloat32));
  auto pidx = torch::empty({B, nsplit}, opts.dtype(torch::kInt32));
  if (logits.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((sample_partial_kernel<uint16_t>),
                       dim3(B, nsplit), dim3(256), 0, stream,
                       pmax.data_ptr<float>(), pidx.data_ptr<int>(),
                       (const uint16_t*)logits.data_ptr(),
                       temperatures.data_ptr<float>(),
                       (const uint64_t*)seeds.data_ptr(), V, nsplit);
  } else {
    TORCH_CHECK(logits.scalar_type() == torch::kFloat32);
    hipLaunchKernelGGL((sample_partial_kernel<float>),
                       dim3(B, nsplit), dim3(256), 0, stream,
                       pmax.data_ptr<float>(), pidx.data_ptr<int>(),
                       logits.data_ptr<float>(),
                       temperatures.data_ptr<float>(),
                       (const uint64_t*)seeds.data_ptr(), V, nsplit);
  }
  hipLaunchKernelGGL(sample_reduce_kernel, dim3(B), dim3(1), 0, stream,
                     out.data_ptr<int64_t>(), pmax.data_ptr<float>(),
                     pidx.data_ptr<int>(), nsplit);
}
