// One-sided v2: owner-side apply-queue RINGS over xGMI (K12b).
//
// Reference semantics being reproduced: the ET op queue serializes ALL
// writes to a block on its owner (CommManager.java:36-155 — per-block
// comm-thread serialization is the PS consistency mechanism), which is
// what lets arbitrary UpdateFunctions (NMF's clamp(old - step*delta))
// run asynchronously. v1 one-sided tables (onesided.hip) only supported
// add-algebra fns (atomic adds ARE the apply). v2: per (owner, writer)
// a fixed-capacity ring in the owner's uncached HBM; writers enqueue
// (key, delta-row) items with a system-scope tail fetch-add + payload
// write + release-ordered ready flag; the OWNER alone drains and applies
// its update function between batches — single applier = the reference's
// per-block write serialization.
//
// Ring layout inside one uncached allocation (per owner table):
//   hdr     : u64[2*W]            (tail_w = hdr[2w], head_w = hdr[2w+1])
//   ready   : u32[W*cap]          (holds low 32 bits of the item's seq;
//                                  slot reuse never false-positives: the
//                                  next seq mapping to the slot differs
//                                  by cap)
//   keys    : u32[W*cap]
//   payload : f32[W*cap*vd]
//
// Lessons from round 1 baked in (docs/ROADMAP.md appendix): payload is
// plain stores to UNCACHED memory, ordered before the ready flag by a
// per-thread system fence + block barrier + post-barrier store (the
// split-K publish recipe); the tail counter is an INTEGER system-scope
// atomic (fp32 atomics can be silently dropped on some memory configs);
// backpressure is host-side (bounded), never an unbounded in-kernel spin.

#include "hip_common.h"

namespace {

struct RingGeom {
  unsigned long long* hdr;
  unsigned int* ready;
  unsigned int* keys;
  float* payload;
};

__host__ __device__ inline RingGeom geom(void* base, int W, long cap,
                                         int vd) {
  RingGeom g;
  char* p = (char*)base;
  g.hdr = (unsigned long long*)p;
  p += (size_t)2 * W * 8;
  g.ready = (unsigned int*)p;
  p += (size_t)W * cap * 4;
  g.keys = (unsigned int*)p;
  p += (size_t)W * cap * 4;
  g.payload = (float*)p;
  return g;
}

__global__ void ring_reserve_kernel(void* base, int W, long cap, int vd,
                                    int writer, long n,
                                    unsigned long long* out_base) {
  RingGeom g = geom(base, W, cap, vd);
  // returns the OLD tail; items get seqs old+1 .. old+n
  *out_base = __hip_atomic_fetch_add(&g.hdr[2 * writer], (unsigned long long)n,
                                     __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_SYSTEM);
}

__global__ void ring_read_head_kernel(void* base, int W, long cap, int vd,
                                      int writer,
                                      unsigned long long* out_head) {
  RingGeom g = geom(base, W, cap, vd);
  *out_head = __hip_atomic_load(&g.hdr[2 * writer + 1], __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_SYSTEM);
}

// one BLOCK per item: threads write the payload row, fence, barrier,
// thread 0 publishes the ready flag
__global__ void ring_push_kernel(void* base, int W, long cap, int vd,
                                 int writer,
                                 const unsigned long long* seq_base,
                                 const int64_t* __restrict__ keys,
                                 const float* __restrict__ deltas, long n) {
  RingGeom g = geom(base, W, cap, vd);
  const long item = blockIdx.x;
  if (item >= n) return;
  const unsigned long long seq = *seq_base + item + 1;
  const long slot = (long)((seq - 1) % (unsigned long long)cap);
  const long soff = (long)writer * cap + slot;
  for (int j = threadIdx.x; j < vd; j += blockDim.x)
    g.payload[soff * vd + j] = deltas[item * vd + j];
  if (threadIdx.x == 0) g.keys[soff] = (unsigned int)keys[item];
  __threadfence_system();
  __syncthreads();
  if (threadIdx.x == 0)
    __hip_atomic_store(&g.ready[soff], (unsigned int)seq, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
}

// OWNER-side drain: one block per writer, in-order scan from head while
// items are ready, copy into local out buffers, publish the new head.
__global__ void ring_drain_kernel(void* base, int W, long cap, int vd,
                                  long max_per,
                                  int64_t* __restrict__ out_keys,  // [W*max_per]
                                  float* __restrict__ out_deltas,  // [W*max_per*vd]
                                  int* __restrict__ out_counts) {  // [W]
  RingGeom g = geom(base, W, cap, vd);
  const int w = blockIdx.x;
  __shared__ int ok;
  const unsigned long long head = g.hdr[2 * w + 1];
  const unsigned long long tail = __hip_atomic_load(
      &g.hdr[2 * w], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
  long consumed = 0;
  unsigned long long seq = head + 1;
  while (seq <= tail && consumed < max_per) {
    const long slot = (long)((seq - 1) % (unsigned long long)cap);
    const long soff = (long)w * cap + slot;
    if (threadIdx.x == 0) {
      ok = (__hip_atomic_load(&g.ready[soff], __ATOMIC_ACQUIRE,
                              __HIP_MEMORY_SCOPE_SYSTEM)
            == (unsigned int)seq);
    }
    __syncthreads();
    if (!ok) break;
    const long o = (long)w * max_per + consumed;
    for (int j = threadIdx.x; j < vd; j += blockDim.x)
      out_deltas[o * vd + j] = g.payload[soff * vd + j];
    if (threadIdx.x == 0) out_keys[o] = (int64_t)g.keys[soff];
    __syncthreads();
    ++consumed;
    ++seq;
  }
  if (threadIdx.x == 0) {
    out_counts[w] = (int)consumed;
    // the head store also releases the slot for reuse by the writer
    __hip_atomic_store(&g.hdr[2 * w + 1], head + (unsigned long long)consumed,
                       __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

}  // namespace

int64_t os_ring_bytes(int64_t W, int64_t cap, int64_t vd) {
  return 2 * W * 8 + W * cap * 4 + W * cap * 4 + W * cap * vd * 4;
}

void os_ring_reserve(int64_t base, int64_t W, int64_t cap, int64_t vd,
                     int64_t writer, int64_t n, torch::Tensor scratch) {
  hipLaunchKernelGGL(ring_reserve_kernel, dim3(1), dim3(1), 0,
                     current_stream(), (void*)(uintptr_t)base, (int)W, cap,
                     (int)vd, (int)writer, n,
                     (unsigned long long*)scratch.data_ptr());
}

void os_ring_read_head(int64_t base, int64_t W, int64_t cap, int64_t vd,
                       int64_t writer, torch::Tensor scratch) {
  hipLaunchKernelGGL(ring_read_head_kernel, dim3(1), dim3(1), 0,
                     current_stream(), (void*)(uintptr_t)base, (int)W, cap,
                     (int)vd, (int)writer,
                     (unsigned long long*)scratch.data_ptr());
}

void os_ring_push(int64_t base, int64_t W, int64_t cap, int64_t vd,
                  int64_t writer, torch::Tensor seq_base,
                  torch::Tensor keys, torch::Tensor deltas) {
  CHECK_IN(keys); CHECK_IN(deltas);
  const long n = keys.numel();
  if (n == 0) return;
  TORCH_CHECK(deltas.size(0) == n && deltas.size(1) == vd);
  hipLaunchKernelGGL(ring_push_kernel, dim3((unsigned)n), dim3(64), 0,
                     current_stream(), (void*)(uintptr_t)base, (int)W, cap,
                     (int)vd, (int)writer,
                     (const unsigned long long*)seq_base.data_ptr(),
                     keys.data_ptr<int64_t>(), deltas.data_ptr<float>(), n);
}

std::vector<torch::Tensor> os_ring_drain(int64_t base, int64_t W,
                                         int64_t cap, int64_t vd,
                                         int64_t max_per) {
  auto dev = torch::TensorOptions().device(torch::kCUDA);
  auto keys = torch::empty({W * max_per}, dev.dtype(torch::kInt64));
  auto deltas = torch::empty({W * max_per, vd}, dev.dtype(torch::kFloat32));
  auto counts = torch::zeros({W}, dev.dtype(torch::kInt32));
  hipLaunchKernelGGL(ring_drain_kernel, dim3((unsigned)W), dim3(64), 0,
                     current_stream(), (void*)(uintptr_t)base, (int)W, cap,
                     (int)vd, max_per, keys.data_ptr<int64_t>(),
                     deltas.data_ptr<float>(), counts.data_ptr<int>());
  return {keys, deltas, counts};
}
