// One-shot all-reduce over xGMI-mapped peer mailboxes (gfx950).
//
// The TP decode path's all-reduces are hidden-size-small (B*hidden bf16,
// 16 KB - 4 MB). RCCL's ring all-reduce is per-link bound and pays
// 2*(N-1) latency hops; on one MI355X node every GPU pair has its own
// xGMI link (7 links/GPU, ~153 GB/s each), so an all-reduce that PULLS
// every peer's buffer in parallel over the point-to-point links is both
// lower-latency and higher-aggregate-bandwidth for these sizes
// (SURVEY.md §2.6 "one-shot/two-shot allreduce"; replaces the NCCL
// collective vLLM uses inside the reference's model containers,
// reference design/2026-04-28-cloud-gpu-smoke-results.md:28).
//
// Protocol per rank: a fine-grained HBM "mailbox" shared with all peers
// via hipIpc (dmabuf mode). Layout:
//   [ start_flags[MAX_WORLD][MAX_BLOCKS] | done_flags[...] | data ]
// Flags are PUSHED (remote store into each peer's mailbox) so the spin
// loops poll LOCAL memory; data is PULLED (remote reads) for the
// reduction. A per-block device-local sequence counter makes the kernel
// hipGraph-replay-safe: every launch bumps the counter, so no flag
// resets are ever needed.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstring>

#include "common.h"

namespace helix {

constexpr int AR_MAX_WORLD = 8;
constexpr int AR_MAX_BLOCKS = 256;
// flags: 2 regions x MAX_WORLD x MAX_BLOCKS u32, padded to 64 KiB so the
// data region starts on a large aligned boundary.
constexpr size_t AR_FLAGS_BYTES = 64 * 1024;

struct ARMailboxes {
  // peer mailbox base pointers (index == rank); own rank's entry is the
  // local buffer. Unused ranks null.
  uint32_t* box[AR_MAX_WORLD];
};

__device__ __forceinline__ uint32_t* start_flag(uint32_t* box, int rank,
                                                int blk) {
  return box + rank * AR_MAX_BLOCKS + blk;
}
__device__ __forceinline__ uint32_t* done_flag(uint32_t* box, int rank,
                                               int blk) {
  return box + AR_MAX_WORLD * AR_MAX_BLOCKS + rank * AR_MAX_BLOCKS + blk;
}
__device__ __forceinline__ uint16_t* data_region(uint32_t* box) {
  return reinterpret_cast<uint16_t*>(reinterpret_cast<char*>(box) +
                                     AR_FLAGS_BYTES);
}

__device__ __forceinline__ void st_system_release(uint32_t* p, uint32_t v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}
__device__ __forceinline__ uint32_t ld_system_acquire(uint32_t* p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
}

// One kernel does the whole one-shot all-reduce for its chunk:
//  1. wait until every peer consumed the previous call's data (done >= seq-1)
//  2. stage my chunk into my mailbox (local HBM write)
//  3. push start=seq into every peer's mailbox (remote release store)
//  4. spin on LOCAL start flags until all ranks arrived
//  5. pull all ranks' chunks over xGMI, sum in fp32, write out
//  6. push done=seq to every peer
__global__ void __launch_bounds__(256)
one_shot_allreduce_bf16(const uint16_t* __restrict__ inp,
                        uint16_t* __restrict__ out, int n, int world,
                        int rank, uint32_t* counters, ARMailboxes boxes) {
  const int blk = blockIdx.x;
  const int tid = threadIdx.x;
  uint32_t* mybox = boxes.box[rank];

  // per-block monotonically increasing sequence number (graph-safe)
  __shared__ uint32_t s_seq;
  if (tid == 0) s_seq = ++counters[blk];
  __syncthreads();
  const uint32_t seq = s_seq;

  // 1. previous-call consumption barrier (local poll; peers push done)
  if (tid < world && tid != rank) {
    while (ld_system_acquire(done_flag(mybox, tid, blk)) < seq - 1) {
    }
  }
  __syncthreads();

  // chunking: contiguous [start, end) per block, 8-elem vector units
  const int chunk = ((n / 8 + gridDim.x - 1) / gridDim.x) * 8;
  const int start = blk * chunk;
  const int end = min(n, start + chunk);

  // 2. stage into my mailbox
  uint16_t* mydata = data_region(mybox);
  for (int i = start + tid * 8; i < end; i += blockDim.x * 8) {
    if (i + 8 <= end) {
      *reinterpret_cast<u16x8*>(mydata + i) =
          *reinterpret_cast<const u16x8*>(inp + i);
    } else {
      for (int j = i; j < end; ++j) mydata[j] = inp[j];
    }
  }
  __syncthreads();
  // 3. publish: data must be system-visible before the flag
  __threadfence_system();
  if (tid < world && tid != rank) {
    st_system_release(start_flag(boxes.box[tid], rank, blk), seq);
  }
  // 4. arrival barrier (local poll)
  if (tid < world && tid != rank) {
    while (ld_system_acquire(start_flag(mybox, tid, blk)) < seq) {
    }
  }
  __syncthreads();

  // 5. pull + reduce. Each rank's chunk arrives over its own xGMI link.
  for (int i = start + tid * 8; i < end; i += blockDim.x * 8) {
    if (i + 8 <= end) {
      float acc[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] = 0.f;
      for (int r = 0; r < world; ++r) {
        const uint16_t* d = data_region(boxes.box[r]);
        u16x8 v = *reinterpret_cast<const u16x8*>(d + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32(v[j]);
      }
      u16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = f32_to_bf16(acc[j]);
      *reinterpret_cast<u16x8*>(out + i) = o;
    } else {
      for (int j = i; j < end; ++j) {
        float acc = 0.f;
        for (int r = 0; r < world; ++r)
          acc += bf16_to_f32(data_region(boxes.box[r])[j]);
        out[j] = f32_to_bf16(acc);
      }
    }
  }
  __syncthreads();
  // 6. release: my reads of everyone's data for `seq` are done
  __threadfence_system();
  if (tid < world && tid != rank) {
    st_system_release(done_flag(boxes.box[tid], rank, blk), seq);
  }
}

// ---------------------------------------------------------------------------
// Host-side context (one per process).
// ---------------------------------------------------------------------------
struct ARContext {
  int world = 0;
  int rank = -1;
  size_t capacity = 0;  // data bytes
  uint32_t* mybox = nullptr;
  uint32_t* counters = nullptr;
  ARMailboxes boxes{};
  bool open = false;
};

static ARContext g_ar;

}  // namespace helix

using namespace helix;

// Allocate the local mailbox (fine-grained so peer loads/atomics are
// system-coherent over xGMI) and return its IPC handle for exchange.
torch::Tensor ar_create(int64_t world, int64_t rank, int64_t capacity) {
  TORCH_CHECK(world >= 2 && world <= AR_MAX_WORLD, "world must be 2..8");
  TORCH_CHECK(g_ar.mybox == nullptr, "allreduce context already created");
  size_t bytes = AR_FLAGS_BYTES + (size_t)capacity;
  void* p = nullptr;
  hipError_t e = hipExtMallocWithFlags(&p, bytes, hipDeviceMallocFinegrained);
  TORCH_CHECK(e == hipSuccess, "hipExtMallocWithFlags: ",
              hipGetErrorString(e));
  TORCH_CHECK(hipMemset(p, 0, AR_FLAGS_BYTES) == hipSuccess);
  void* c = nullptr;
  TORCH_CHECK(hipMalloc(&c, AR_MAX_BLOCKS * sizeof(uint32_t)) == hipSuccess);
  TORCH_CHECK(hipMemset(c, 0, AR_MAX_BLOCKS * sizeof(uint32_t)) ==
              hipSuccess);
  TORCH_CHECK(hipDeviceSynchronize() == hipSuccess);
  g_ar.world = (int)world;
  g_ar.rank = (int)rank;
  g_ar.capacity = (size_t)capacity;
  g_ar.mybox = (uint32_t*)p;
  g_ar.counters = (uint32_t*)c;

  hipIpcMemHandle_t h;
  e = hipIpcGetMemHandle(&h, p);
  TORCH_CHECK(e == hipSuccess, "hipIpcGetMemHandle: ", hipGetErrorString(e));
  auto t = torch::empty({(int64_t)sizeof(h)}, torch::kUInt8);
  memcpy(t.data_ptr(), &h, sizeof(h));
  return t;
}

// Map every peer's mailbox. `handles` is the all-gathered list (rank order).
void ar_open(std::vector<torch::Tensor> handles) {
  TORCH_CHECK(g_ar.mybox != nullptr, "call ar_create first");
  TORCH_CHECK((int)handles.size() == g_ar.world, "need one handle per rank");
  for (int r = 0; r < g_ar.world; ++r) {
    if (r == g_ar.rank) {
      g_ar.boxes.box[r] = g_ar.mybox;
      continue;
    }
    hipIpcMemHandle_t h;
    TORCH_CHECK(handles[r].numel() == (int64_t)sizeof(h), "bad handle size");
    memcpy(&h, handles[r].data_ptr(), sizeof(h));
    void* p = nullptr;
    hipError_t e =
        hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess);
    TORCH_CHECK(e == hipSuccess, "hipIpcOpenMemHandle(rank ", r,
                "): ", hipGetErrorString(e));
    g_ar.boxes.box[r] = (uint32_t*)p;
  }
  g_ar.open = true;
}

int64_t ar_capacity() { return g_ar.open ? (int64_t)g_ar.capacity : 0; }

void ar_allreduce(torch::Tensor inp, torch::Tensor out) {
  TORCH_CHECK(g_ar.open, "allreduce context not opened");
  TORCH_CHECK(inp.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(inp.scalar_type() == torch::kBFloat16 &&
              out.scalar_type() == torch::kBFloat16);
  int64_t n = inp.numel();
  TORCH_CHECK(out.numel() == n);
  TORCH_CHECK((size_t)n * 2 <= g_ar.capacity, "message exceeds AR capacity");
  // ~16 KB of staged data per block, capped at AR_MAX_BLOCKS
  int blocks = (int)std::min<int64_t>(AR_MAX_BLOCKS, (n * 2 + 16383) / 16384);
  blocks = std::max(blocks, 1);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(one_shot_allreduce_bf16, dim3(blocks), dim3(256), 0,
                     stream, (const uint16_t*)inp.data_ptr(),
                     (uint16_t*)out.data_ptr(), (int)n, g_ar.world, g_ar.rank,
                     g_ar.counters, g_ar.boxes);
}

void ar_destroy() {
  if (!g_ar.mybox) return;
  for (int r = 0; r < g_ar.world; ++r) {
    if (r != g_ar.rank && g_ar.boxes.box[r]) {
      (void)hipIpcCloseMemHandle(g_ar.boxes.box[r]);
    }
    g_ar.boxes.box[r] = nullptr;
  }
  (void)hipFree(g_ar.mybox);
  (void)hipFree(g_ar.counters);
  g_ar = ARContext{};
}
