// Standalone ablation probe for the v3 paged-decode kernel structure.
// Compile: hipcc --offload-arch=gfx950 -O3 -std=c++17 -o abl_decode \
//            scripts/abl_decode_probe.hip
// Variants (template ABL bitmask), timed back-to-back on identical data:
//   0 FULL        — the shipped structure
//   1 A_MATH_OFF  — K loads kept live (asm), score math removed
//   2 B_OFF       — softmax removed (p = raw score)
//   4 C_LOADS_OFF — V loads removed (constant v), accumulate kept
//   8 A_OFF       — phase A fully removed (scores constant)
// Rule-17 guard: skipped values are kept live with asm volatile.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define WAVE 64
typedef __attribute__((ext_vector_type(8))) uint16_t u16x8;

__device__ __forceinline__ float bf2f(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = uint32_t(h) << 16;
  return v.f;
}
__device__ __forceinline__ float wrmax(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
__device__ __forceinline__ float wrsum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

constexpr int NT = 128, CHUNK = 128, DH = 128, G = 4;

template <int ABL>
__global__ __launch_bounds__(NT, 3) void decode_abl(
    uint16_t* __restrict__ out, const uint16_t* __restrict__ q,
    const uint16_t* __restrict__ k_cache, const uint16_t* __restrict__ v_cache,
    const int* __restrict__ btab, const int* __restrict__ lens,
    float scale, int Hkv, int bs, int max_blocks, int part_sz) {
  const int seq = blockIdx.x, hkv = blockIdx.y, part = blockIdx.z;
  const int Hq = Hkv * G;
  const int len = lens[seq];
  const int p0 = part * part_sz;
  if (p0 >= len) return;
  const int p1 = min(len, p0 + part_sz);

  __shared__ float q_lds[G][DH];
  __shared__ float s_lds[G][CHUNK];
  __shared__ float hm[G], hl[G], hc[G];
  for (int i = threadIdx.x; i < G * DH; i += NT)
    q_lds[i / DH][i % DH] =
        bf2f(q[((long)seq * Hq + hkv * G + i / DH) * DH + i % DH]) * scale;
  if (threadIdx.x < G) { hm[threadIdx.x] = -INFINITY; hl[threadIdx.x] = 0.f; }
  __syncthreads();

  constexpr int NGRP = DH / 8;
  constexpr int C_PAR = NT / NGRP;
  const int d8 = (threadIdx.x % NGRP) * 8;
  const int cpar = threadIdx.x / NGRP;
  float acc[G][8];
#pragma unroll
  for (int g = 0; g < G; ++g)
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[g][i] = 0.f;
  const int* bt = btab + (long)seq * max_blocks;
  const int wid = threadIdx.x / WAVE, lane = threadIdx.x & 63;
  const int nw = NT / WAVE;

  for (int base = p0; base < p1; base += CHUNK) {
    const int cn = min(CHUNK, p1 - base);
    if ((int)threadIdx.x < cn) {
      const int tok = base + threadIdx.x;
      const long blk = bt[tok / bs];
      const uint16_t* kr =
          k_cache + ((blk * Hkv + hkv) * (long)bs + tok % bs) * DH;
      u16x8 kraw[DH / 8];
      if (!(ABL & 8)) {
#pragma unroll
        for (int j = 0; j < DH / 8; ++j)
          kraw[j] = *reinterpret_cast<const u16x8*>(kr + j * 8);
      }
      float s[G];
#pragma unroll
      for (int g = 0; g < G; ++g) s[g] = 1.0f;
      if (!(ABL & 9)) {          // full score math
#pragma unroll
        for (int g = 0; g < G; ++g) s[g] = 0.f;
#pragma unroll
        for (int j = 0; j < DH / 8; ++j) {
          float kv[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) kv[i] = bf2f(kraw[j][i]);
#pragma unroll
          for (int g = 0; g < G; ++g)
#pragma unroll
            for (int i = 0; i < 8; ++i)
              s[g] += q_lds[g][j * 8 + i] * kv[i];
        }
      } else if (!(ABL & 8)) {   // keep loads live, no math
#pragma unroll
        for (int j = 0; j < DH / 8; ++j)
          asm volatile("" ::"v"(kraw[j][0]), "v"(kraw[j][7]));
      }
#pragma unroll
      for (int g = 0; g < G; ++g) s_lds[g][threadIdx.x] = s[g];
    }
    __syncthreads();

    for (int g = wid; g < G; g += nw) {
      if (ABL & 2) {
        if (lane == 0) { hm[g] = 0.f; hl[g] = 1.f; hc[g] = 1.f; }
      } else {
        float mc = -INFINITY;
        for (int i = lane; i < cn; i += WAVE)
          mc = fmaxf(mc, s_lds[g][i]);
        mc = wrmax(mc);
        const float mo = hm[g];
        const float mn = fmaxf(mo, mc);
        const float corr = (mo == -INFINITY) ? 0.f : __expf(mo - mn);
        float la = 0.f;
        for (int i = lane; i < cn; i += WAVE) {
          const float p = __expf(s_lds[g][i] - mn);
          s_lds[g][i] = p;
          la += p;
        }
        la = wrsum(la);
        if (lane == 0) { hl[g] = hl[g] * corr + la; hm[g] = mn; hc[g] = corr; }
      }
    }
    __syncthreads();

    {
#pragma unroll
      for (int g = 0; g < G; ++g)
#pragma unroll
        for (int i = 0; i < 8; ++i) acc[g][i] *= hc[g];
      const int npass = (cn - cpar + C_PAR - 1) / C_PAR;
      int ps = 0;
      for (; ps + 4 <= npass; ps += 4) {
        u16x8 vv[4];
        if (ABL & 4) {
#pragma unroll
          for (int u = 0; u < 4; ++u)
#pragma unroll
            for (int i = 0; i < 8; ++i) vv[u][i] = 0x3f80;
        } else {
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            const int tok = base + (ps + u) * C_PAR + cpar;
            const long blk = bt[tok / bs];
            vv[u] = *reinterpret_cast<const u16x8*>(
                v_cache + ((blk * Hkv + hkv) * (long)bs + tok % bs) * DH +
                d8);
          }
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int ti = (ps + u) * C_PAR + cpar;
          float v[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = bf2f(vv[u][i]);
#pragma unroll
          for (int g = 0; g < G; ++g) {
            const float pv = s_lds[g][ti];
#pragma unroll
            for (int i = 0; i < 8; ++i) acc[g][i] += pv * v[i];
          }
        }
      }
      for (; ps < npass; ++ps) {
        const int ti = ps * C_PAR + cpar;
        const int tok = base + ti;
        const long blk = bt[tok / bs];
        u16x8 vv;
        if (ABL & 4) {
#pragma unroll
          for (int i = 0; i < 8; ++i) vv[i] = 0x3f80;
        } else {
          vv = *reinterpret_cast<const u16x8*>(
              v_cache + ((blk * Hkv + hkv) * (long)bs + tok % bs) * DH + d8);
        }
        float v[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] = bf2f(vv[i]);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float pv = s_lds[g][ti];
#pragma unroll
          for (int i = 0; i < 8; ++i) acc[g][i] += pv * v[i];
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int g = 0; g < G; ++g) {
    const int hq = hkv * G + g;
    const float inv = 1.f / fmaxf(hl[g], 1e-20f);
    if (cpar == 0)
#pragma unroll
      for (int i = 0; i < 8; ++i)
        out[((long)seq * Hq + hq) * DH + d8 + i] =
            (uint16_t)(__float_as_uint(acc[g][i] * inv) >> 16);
  }
}

int main() {
  const int B = 256, Hkv = 8, bs = 16, L = 560;
  const int nb = (L + bs - 1) / bs;
  const long nblocks = (long)B * nb + 1;
  uint16_t *q, *kc, *vc, *out;
  int *btab, *lens;
  hipMalloc(&q, (long)B * Hkv * G * DH * 2);
  hipMalloc(&kc, nblocks * Hkv * bs * DH * 2);
  hipMalloc(&vc, nblocks * Hkv * bs * DH * 2);
  hipMalloc(&out, (long)B * Hkv * G * DH * 2);
  hipMalloc(&btab, (long)B * nb * 4);
  hipMalloc(&lens, B * 4);
  std::vector<int> hb(B * nb), hl(B, L);
  for (int i = 0; i < B * nb; ++i) hb[i] = 1 + i;
  hipMemcpy(btab, hb.data(), hb.size() * 4, hipMemcpyHostToDevice);
  hipMemcpy(lens, hl.data(), B * 4, hipMemcpyHostToDevice);
  hipMemset(kc, 0x3c, nblocks * Hkv * bs * DH * 2);
  hipMemset(vc, 0x3c, nblocks * Hkv * bs * DH * 2);
  hipMemset(q, 0x3c, (long)B * Hkv * G * DH * 2);

  const float scale = 0.0884f;
  const int part_sz = ((L + 127) / 128) * 128;   // nparts = 1 (B*Hkv=2048)
  dim3 grid(B, Hkv, 1), blk(NT);

  auto bench = [&](auto kern, const char* name) {
    for (int i = 0; i < 10; ++i)
      hipLaunchKernelGGL(kern, grid, blk, 0, 0, out, q, kc, vc, btab, lens,
                         scale, Hkv, bs, nb, part_sz);
    hipDeviceSynchronize();
    hipEvent_t a, b;
    hipEventCreate(&a);
    hipEventCreate(&b);
    hipEventRecord(a);
    for (int i = 0; i < 50; ++i)
      hipLaunchKernelGGL(kern, grid, blk, 0, 0, out, q, kc, vc, btab, lens,
                         scale, Hkv, bs, nb, part_sz);
    hipEventRecord(b);
    hipEventSynchronize(b);
    float ms;
    hipEventElapsedTime(&ms, a, b);
    const double bytes = (double)B * Hkv * L * DH * 2 * 2;
    printf("%-14s %8.1f us  %5.2f TB/s\n", name, ms * 1000 / 50,
           bytes / (ms / 50 / 1000) / 1e12);
  };
  // interleaved repetitions for within-probe stability
  for (int rep = 0; rep < 2; ++rep) {
    bench(decode_abl<0>, "FULL");
    bench(decode_abl<1>, "A_MATH_OFF");
    bench(decode_abl<2>, "B_OFF");
    bench(decode_abl<4>, "C_LOADS_OFF");
    bench(decode_abl<8>, "A_OFF");
    bench(decode_abl<12>, "A+C_LOADS_OFF");
  }
  return 0;
}
