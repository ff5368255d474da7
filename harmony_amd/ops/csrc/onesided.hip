// One-sided xGMI data plane primitives (K12).
//
// Reference: the ET remote-access path (RemoteAccessOpSender/Handler,
// CommManager — SURVEY §2.1) gives workers ASYNC per-key access to remote
// shards over sockets. The collective data plane (et/comm.py) replaces it
// with bulk-synchronous RCCL ops; THIS file is the async analogue the
// MI355X way: each rank's shard is exported with hipIpcGetMemHandle
// (dmabuf mode — HSA_ENABLE_IPC_MODE_LEGACY=0), peers map it once and
// dereference it directly from gather/scatter kernels. On one node the
// mapping is peer HBM over xGMI (p2p loads/stores, ≈153 GB/s per link);
// no message, no rendezvous, no collective — a pull is a kernel.
//
// Shards for one-sided tables are allocated FINE-GRAINED
// (hipExtMallocWithFlags(hipDeviceMallocFinegrained)) and NOT through the
// torch caching allocator: IPC handles need the allocation base, and
// cross-process visibility needs cache-coherent (fine-grained) memory —
// MI355X L2s are per-XCD, so coarse-grained atomics from one process are
// not reliably visible to another process's loads until a flush. The
// scatter atomics are system-scope for the same reason.

#include "hip_common.h"

namespace {

constexpr int GATHER_THREADS = 256;

// rows from a (possibly remote) shard: out[i] = shard[idx[i]]
__global__ void gather_rows_kernel(const float* __restrict__ shard,
                                   const int64_t* __restrict__ idx,
                                   float* __restrict__ out,
                                   int n, int k) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = (int64_t)n * k;
  for (int64_t i = t; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = i / k, c = i - r * k;
    // system-scope load: another process's completed atomics must be
    // visible even if this process cached the line on an earlier pull
    out[i] = __hip_atomic_load(&shard[idx[r] * k + c], __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

// fp32 add via integer CAS at system scope: CDNA hardware silently DROPS
// unsupported fp atomics on some memory types (the gfx90a-era
// unsafe-fp-atomics behavior; box-dependent via XNACK/page setup) —
// integer compare-exchange is architecturally guaranteed everywhere.
__device__ __forceinline__ void sys_atomic_add_f32(float* addr, float v) {
  unsigned int* a = reinterpret_cast<unsigned int*>(addr);
  unsigned int old = __hip_atomic_load(a, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_SYSTEM);
  while (true) {
    unsigned int assumed = old;
    const float f = __uint_as_float(assumed) + v;
    const unsigned int want = __float_as_uint(f);
    if (__hip_atomic_compare_exchange_strong(
            a, &old, want, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_SYSTEM))
      break;
  }
}

// atomic add into a (possibly remote) shard: shard[idx[i]] += delta[i]
__global__ void scatter_add_rows_kernel(float* __restrict__ shard,
                                        const int64_t* __restrict__ idx,
                                        const float* __restrict__ delta,
                                        int n, int k) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = (int64_t)n * k;
  for (int64_t i = t; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = i / k, c = i - r * k;
    sys_atomic_add_f32(&shard[idx[r] * k + c], delta[i]);
  }
}

// int32 variants (LDA count tables): integer atomics are natively
// supported at system scope — no CAS loop needed
__global__ void gather_rows_i32_kernel(const int* __restrict__ shard,
                                       const int64_t* __restrict__ idx,
                                       int* __restrict__ out,
                                       int n, int k) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = (int64_t)n * k;
  for (int64_t i = t; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = i / k, c = i - r * k;
    out[i] = __hip_atomic_load(&shard[idx[r] * k + c], __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

__global__ void scatter_add_rows_i32_kernel(int* __restrict__ shard,
                                            const int64_t* __restrict__ idx,
                                            const int* __restrict__ delta,
                                            int n, int k) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = (int64_t)n * k;
  for (int64_t i = t; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = i / k, c = i - r * k;
    const int d = delta[i];
    if (d != 0)
      __hip_atomic_fetch_add(&shard[idx[r] * k + c], d, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

dim3 grid_for(int64_t total) {
  int64_t blocks = (total + GATHER_THREADS - 1) / GATHER_THREADS;
  if (blocks > 4096) blocks = 4096;   // grid-stride; >> 256 CUs
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

}  // namespace

torch::Tensor os_shard_alloc(int64_t rows, int64_t k, int64_t dtype_i32) {
  // uncached-memory tensor usable as an IPC export base (f32 or i32)
  void* p = nullptr;
  size_t bytes = (size_t)rows * k * 4;
  // uncached (MTYPE_UC): no XCD-L2 lines exist for shard memory, so a
  // plain write from one process can never linger dirty and later stomp
  // a peer's committed atomic — fine-grained alone still flaked (~1/5)
  TORCH_CHECK(hipExtMallocWithFlags(&p, bytes, hipDeviceMallocUncached)
                  == hipSuccess, "uncached hipMalloc failed");
  TORCH_CHECK(hipMemset(p, 0, bytes) == hipSuccess);
  int dev = 0;
  TORCH_CHECK(hipGetDevice(&dev) == hipSuccess);
  auto opts = torch::TensorOptions()
                  .dtype(dtype_i32 ? torch::kInt32 : torch::kFloat32)
                  .device(torch::kCUDA, dev);
  return torch::from_blob(
      p, {rows, k}, [](void* q) { hipFree(q); }, opts);
}

torch::Tensor os_ipc_handle(torch::Tensor shard) {
  CHECK_IN(shard);
  hipIpcMemHandle_t h;
  TORCH_CHECK(hipIpcGetMemHandle(&h, shard.data_ptr()) == hipSuccess,
              "hipIpcGetMemHandle failed (needs dmabuf IPC: "
              "HSA_ENABLE_IPC_MODE_LEGACY=0)");
  auto out = torch::empty({(int64_t)sizeof(h)},
                          torch::TensorOptions().dtype(torch::kUInt8));
  memcpy(out.data_ptr(), &h, sizeof(h));
  return out;
}

int64_t os_ipc_open(torch::Tensor handle_bytes) {
  TORCH_CHECK(handle_bytes.numel() == (int64_t)sizeof(hipIpcMemHandle_t));
  hipIpcMemHandle_t h;
  memcpy(&h, handle_bytes.contiguous().data_ptr(), sizeof(h));
  void* p = nullptr;
  TORCH_CHECK(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess)
                  == hipSuccess, "hipIpcOpenMemHandle failed");
  return (int64_t)(uintptr_t)p;
}

void os_ipc_close(int64_t ptr) {
  TORCH_CHECK(hipIpcCloseMemHandle((void*)(uintptr_t)ptr) == hipSuccess);
}

torch::Tensor os_gather(int64_t ptr, torch::Tensor idx, int64_t k,
                        int64_t dtype_i32) {
  CHECK_IN(idx);
  const int n = idx.numel();
  auto out = torch::empty(
      {(int64_t)n, k},
      idx.options().dtype(dtype_i32 ? torch::kInt32 : torch::kFloat32));
  if (n == 0) return out;
  if (dtype_i32)
    hipLaunchKernelGGL(gather_rows_i32_kernel, grid_for((int64_t)n * k),
                       dim3(GATHER_THREADS), 0, current_stream(),
                       (const int*)(uintptr_t)ptr,
                       idx.data_ptr<int64_t>(), out.data_ptr<int>(),
                       n, (int)k);
  else
    hipLaunchKernelGGL(gather_rows_kernel, grid_for((int64_t)n * k),
                       dim3(GATHER_THREADS), 0, current_stream(),
                       (const float*)(uintptr_t)ptr,
                       idx.data_ptr<int64_t>(), out.data_ptr<float>(),
                       n, (int)k);
  return out;
}

void os_scatter_add(int64_t ptr, torch::Tensor idx, torch::Tensor delta) {
  CHECK_IN(idx); CHECK_IN(delta);
  const int n = idx.numel();
  if (n == 0) return;
  const int k = delta.size(1);
  if (delta.scalar_type() == torch::kInt32)
    hipLaunchKernelGGL(scatter_add_rows_i32_kernel, grid_for((int64_t)n * k),
                       dim3(GATHER_THREADS), 0, current_stream(),
                       (int*)(uintptr_t)ptr, idx.data_ptr<int64_t>(),
                       delta.data_ptr<int>(), n, k);
  else
    hipLaunchKernelGGL(scatter_add_rows_kernel, grid_for((int64_t)n * k),
                       dim3(GATHER_THREADS), 0, current_stream(),
                       (float*)(uintptr_t)ptr, idx.data_ptr<int64_t>(),
                       delta.data_ptr<float>(), n, k);
}

// ---- CU-masked streams (multi-tenant GPU partitioning) ----------------
// MI355X has no MPS-style preemption; co-located jobs' full-chip kernels
// timeshare CUs and thrash each other's per-XCD L2. hipExtStreamCreate-
// WithCUMask pins a stream's kernels to a CU subset — the MI355X-native
// analogue of the reference's per-executor resource arbitration
// (LocalTaskUnitScheduler's CPU semaphores). Masks are bit-per-CU words.

int64_t os_cu_masked_stream(torch::Tensor mask_words) {
  TORCH_CHECK(mask_words.dtype() == torch::kInt32);
  auto m = mask_words.contiguous();
  hipStream_t s = nullptr;
  TORCH_CHECK(hipExtStreamCreateWithCUMask(
                  &s, (uint32_t)m.numel(),
                  (const uint32_t*)m.data_ptr<int>()) == hipSuccess,
              "hipExtStreamCreateWithCUMask failed");
  return (int64_t)(uintptr_t)s;
}

void os_stream_destroy(int64_t stream) {
  TORCH_CHECK(hipStreamDestroy((hipStream_t)(uintptr_t)stream)
              == hipSuccess);
}
