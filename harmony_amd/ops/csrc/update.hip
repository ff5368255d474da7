// K3/K9: fused owner-side update application — the "server" compute.
// Reference semantics: BlockImpl.update via per-app UpdateFunction
// (NMFETModelUpdateFunction.java:48-52 axpy+clamp,
// MLRETModelUpdateFunction.java:60-62 add,
// LDAETModelUpdateFunction.java:43-64 count merge with clamp>=0).
// Here the aggregated deltas arriving from the push all-to-all are applied
// in ONE gather-modify-scatter kernel per table (vs 3+ torch launches).

#include "hip_common.h"

namespace {

enum Mode { ADD = 0, ASSIGN = 1, NMF_SGD = 2, LDA_COUNTS = 3 };

template <typename T>
__device__ __forceinline__ T apply_one(T v, T d, int mode, float step,
                                       float maxval) {
  switch (mode) {
    case ADD: return v + d;
    case ASSIGN: return d;
    case NMF_SGD: {
      float x = (float)v - step * (float)d;
      x = fminf(fmaxf(x, 0.f), maxval);
      return (T)x;
    }
    case LDA_COUNTS: {
      T x = v + d;
      return x < (T)0 ? (T)0 : x;
    }
  }
  return v;
}

template <typename T>
__global__ void scatter_apply_kernel(T* __restrict__ shard,
                                     const int64_t* __restrict__ rows,
                                     const T* __restrict__ deltas,
                                     int64_t n, int vd, int mode, float step,
                                     float maxval) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n * vd) return;
  int64_t r = rows[i / vd];
  int64_t col = i % vd;
  T* dst = shard + r * vd + col;
  *dst = apply_one(*dst, deltas[i], mode, step, maxval);
}

template <typename T>
__global__ void dense_apply_kernel(T* __restrict__ shard,
                                   const T* __restrict__ delta, int64_t n,
                                   int mode, float step, float maxval) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  shard[i] = apply_one(shard[i], delta[i], mode, step, maxval);
}

}  // namespace

void scatter_apply(torch::Tensor shard, torch::Tensor rows,
                   torch::Tensor deltas, int64_t mode, double step,
                   double maxval) {
  CHECK_IN(shard); CHECK_IN(rows); CHECK_IN(deltas);
  const int64_t n = rows.size(0);
  const int vd = shard.size(1);
  if (n == 0) return;
  const int64_t total = n * vd;
  dim3 blk(256), grid((unsigned)((total + 255) / 256));
  if (shard.dtype() == torch::kFloat32) {
    hipLaunchKernelGGL(scatter_apply_kernel<float>, grid, blk, 0,
                       current_stream(), shard.data_ptr<float>(),
                       rows.data_ptr<int64_t>(), deltas.data_ptr<float>(),
                       n, vd, (int)mode, (float)step, (float)maxval);
  } else if (shard.dtype() == torch::kInt32) {
    hipLaunchKernelGGL(scatter_apply_kernel<int>, grid, blk, 0,
                       current_stream(), shard.data_ptr<int>(),
                       rows.data_ptr<int64_t>(), deltas.data_ptr<int>(),
                       n, vd, (int)mode, (float)step, (float)maxval);
  } else {
    TORCH_CHECK(false, "scatter_apply: unsupported dtype");
  }
}

void dense_apply(torch::Tensor shard, torch::Tensor delta, int64_t mode,
                 double step, double maxval) {
  CHECK_IN(shard); CHECK_IN(delta);
  const int64_t total = shard.numel();
  TORCH_CHECK(delta.numel() == total, "size mismatch");
  if (total == 0) return;
  dim3 blk(256), grid((unsigned)((total + 255) / 256));
  if (shard.dtype() == torch::kFloat32) {
    hipLaunchKernelGGL(dense_apply_kernel<float>, grid, blk, 0,
                       current_stream(), shard.data_ptr<float>(),
                       delta.data_ptr<float>(), total, (int)mode, (float)step,
                       (float)maxval);
  } else if (shard.dtype() == torch::kInt32) {
    hipLaunchKernelGGL(dense_apply_kernel<int>, grid, blk, 0,
                       current_stream(), shard.data_ptr<int>(),
                       delta.data_ptr<int>(), total, (int)mode, (float)step,
                       (float)maxval);
  } else {
    TORCH_CHECK(false, "dense_apply: unsupported dtype");
  }
}
