// K4: MLR fused softmax + label-subtract + CE/accuracy.
// Replaces reference MLRTrainer.java:374-398 (gradient) + :475-489
// (predict/softmax with log-sum-exp guard) + the loss/accuracy pass — one
// kernel, one read of the logits, instead of five torch launches.
//
// Shape regime: B up to ~64k rows, C small (10..1024). One thread per row:
// each thread streams its C logits three times (max, sum-exp, write) —
// L2-resident for the typical 40 B..4 KB rows; lanes of a wave touch
// consecutive rows so the per-pass footprint is contiguous.

#include "hip_common.h"

constexpr int MLR_MAXC = 16;
constexpr int ROWS_PER_WG = 4;
constexpr int FWD_THREADS = 256;

namespace {

__global__ void mlr_softmax_grad_kernel(const float* __restrict__ logits,
                                        const int64_t* __restrict__ labels,
                                        float* __restrict__ grad,
                                        float* __restrict__ loss,
                                        int* __restrict__ correct,
                                        int B, int C) {
  int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= B) return;
  const float* z = logits + (int64_t)row * C;
  float* g = grad + (int64_t)row * C;
  float m = -1e30f;
  int argmax = 0;
  for (int j = 0; j < C; ++j) {
    float v = z[j];
    if (v > m) { m = v; argmax = j; }
  }
  float s = 0.f;
  for (int j = 0; j < C; ++j) s += __expf(z[j] - m);
  float inv = 1.0f / s;
  int64_t lab = labels[row];
  for (int j = 0; j < C; ++j)
    g[j] = __expf(z[j] - m) * inv - (j == (int)lab ? 1.0f : 0.0f);
  // CE = log-sum-exp - z[label]
  float ce = m + __logf(s) - z[lab];
  atomicAdd(loss, ce);
  if (argmax == (int)lab) atomicAdd(correct, 1);
}

// Fused forward: logits = X @ W^T + row softmax + label-subtract + CE/acc,
// reading X exactly once. rocBLAS/Tensile handles the skinny-N (C ~ 10)
// GEMM poorly (measured 600 us vs the ~160 us HBM floor for X = 1 GB);
// a row-block reduction kernel is the right shape: each workgroup owns
// ROWS_PER_WG rows, threads stride the F dimension, per-thread partial dots
// for all C classes live in registers, then a wave+LDS tree reduce.
__global__ __launch_bounds__(FWD_THREADS)
void mlr_fwd_kernel(const float* __restrict__ X,
                    const float* __restrict__ W,
                    const int64_t* __restrict__ labels,
                    float* __restrict__ grad,
                    float* __restrict__ loss,
                    int* __restrict__ correct,
                    int B, int F, int C) {
  // Each thread owns 4 consecutive f (float4 loads: the scalar-load version
  // was VMEM-ISSUE-bound at 3x the HBM floor); threads stride F in steps of
  // 4*FWD_THREADS; ROWS_PER_WG rows share every W load.
  __shared__ float part[ROWS_PER_WG][FWD_THREADS / WAVE][MLR_MAXC];
  const int row0 = blockIdx.x * ROWS_PER_WG;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);

  float acc[ROWS_PER_WG][MLR_MAXC];
#pragma unroll
  for (int r = 0; r < ROWS_PER_WG; ++r)
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c) acc[r][c] = 0.f;

  const int f4 = F >> 2;                  // F % 4 handled by the tail loop
  for (int i = tid; i < f4; i += FWD_THREADS) {
    const int f = i * 4;
    float4 w4[MLR_MAXC];
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c)
      w4[c] = (c < C) ? *(const float4*)&W[(int64_t)c * F + f]
                      : make_float4(0.f, 0.f, 0.f, 0.f);
#pragma unroll
    for (int r = 0; r < ROWS_PER_WG; ++r) {
      const float4 x = (row0 + r < B)
          ? *(const float4*)&X[(int64_t)(row0 + r) * F + f]
          : make_float4(0.f, 0.f, 0.f, 0.f);
#pragma unroll
      for (int c = 0; c < MLR_MAXC; ++c)
        if (c < C)
          acc[r][c] += x.x * w4[c].x + x.y * w4[c].y + x.z * w4[c].z +
                       x.w * w4[c].w;
    }
  }
  for (int f = f4 * 4 + tid; f < F; f += FWD_THREADS) {   // tail (F % 4)
#pragma unroll
    for (int r = 0; r < ROWS_PER_WG; ++r) {
      const float x = (row0 + r < B) ? X[(int64_t)(row0 + r) * F + f] : 0.f;
#pragma unroll
      for (int c = 0; c < MLR_MAXC; ++c)
        if (c < C) acc[r][c] += x * W[(int64_t)c * F + f];
    }
  }
#pragma unroll
  for (int r = 0; r < ROWS_PER_WG; ++r)
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c) {
      float v = wave_reduce_sum(acc[r][c]);
      if (lane == 0 && c < C) part[r][wave][c] = v;
    }
  __syncthreads();
  // one thread per row finishes: sum waves, softmax w/ LSE guard, write grad
  if (tid < ROWS_PER_WG && row0 + tid < B) {
    const int row = row0 + tid;
    float z[MLR_MAXC];
    float m = -1e30f;
    int argmax = 0;
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c) {
      float v = 0.f;
#pragma unroll
      for (int wv = 0; wv < FWD_THREADS / WAVE; ++wv) v += part[tid][wv][c];
      z[c] = (c < C) ? v : -1e30f;
      if (c < C && v > m) { m = v; argmax = c; }
    }
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c)
      if (c < C) s += __expf(z[c] - m);
    const float inv = 1.0f / s;
    const int64_t lab = labels[row];
    float zlab = 0.f;
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c)
      if (c < C) {
        grad[(int64_t)row * C + c] =
            __expf(z[c] - m) * inv - (c == (int)lab ? 1.0f : 0.0f);
        if (c == (int)lab) zlab = z[c];
      }
    atomicAdd(loss, m + __logf(s) - zlab);
    if (argmax == (int)lab) atomicAdd(correct, 1);
  }
}

// Gradient GEMM grad[C,F] = P^T @ X for skinny C: B-tile per workgroup, the
// P tile staged in LDS (broadcast reads), X streamed coalesced once, C
// partials per thread in registers, one atomicAdd per (c, f) per tile into
// the L2-resident 640 KB grad buffer. Tensile's generic GEMM measured ~3x
// the HBM floor on this shape.
constexpr int GRAD_BTILE = 512;
constexpr int GRAD_FTILE = 4 * FWD_THREADS;   // 4 consecutive f per thread

__global__ __launch_bounds__(FWD_THREADS)
void mlr_grad_kernel(const float* __restrict__ P,
                     const float* __restrict__ X,
                     float* __restrict__ grad,
                     int B, int F, int C) {
  // 2D grid (f-tile, b-tile): the single-axis version launched only B/256
  // workgroups (1 wave/CU on a 16k batch - measured 12.7 ms). P tile in
  // LDS (broadcast), X float4-streamed once, one atomicAdd per (c, f) per
  // b-tile into the L2-resident grad buffer.
  __shared__ float pl[GRAD_BTILE * MLR_MAXC];
  const int f0 = blockIdx.x * GRAD_FTILE;
  const int b0 = blockIdx.y * GRAD_BTILE;
  const int tid = threadIdx.x;
  const int bmax = min(GRAD_BTILE, B - b0);
  for (int idx = tid; idx < bmax * C; idx += FWD_THREADS)
    pl[(idx / C) * MLR_MAXC + (idx % C)] =
        P[(int64_t)(b0 + idx / C) * C + idx % C];
  __syncthreads();
  const int f = f0 + tid * 4;
  if (f + 3 < F) {
    float4 facc[MLR_MAXC];
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c)
      facc[c] = make_float4(0.f, 0.f, 0.f, 0.f);
    for (int b = 0; b < bmax; ++b) {
      const float4 x = *(const float4*)&X[(int64_t)(b0 + b) * F + f];
#pragma unroll
      for (int c = 0; c < MLR_MAXC; ++c)
        if (c < C) {
          const float p = pl[b * MLR_MAXC + c];
          facc[c].x += p * x.x;
          facc[c].y += p * x.y;
          facc[c].z += p * x.z;
          facc[c].w += p * x.w;
        }
    }
#pragma unroll
    for (int c = 0; c < MLR_MAXC; ++c)
      if (c < C) {
        atomicAdd(&grad[(int64_t)c * F + f + 0], facc[c].x);
        atomicAdd(&grad[(int64_t)c * F + f + 1], facc[c].y);
        atomicAdd(&grad[(int64_t)c * F + f + 2], facc[c].z);
        atomicAdd(&grad[(int64_t)c * F + f + 3], facc[c].w);
      }
  } else {
    for (int ff = f; ff < min(f + 4, F); ++ff) {
      float facc[MLR_MAXC];
#pragma unroll
      for (int c = 0; c < MLR_MAXC; ++c) facc[c] = 0.f;
      for (int b = 0; b < bmax; ++b) {
        const float x = X[(int64_t)(b0 + b) * F + ff];
#pragma unroll
        for (int c = 0; c < MLR_MAXC; ++c)
          if (c < C) facc[c] += pl[b * MLR_MAXC + c] * x;
      }
#pragma unroll
      for (int c = 0; c < MLR_MAXC; ++c)
        if (c < C) atomicAdd(&grad[(int64_t)c * F + ff], facc[c]);
    }
  }
}

}  // namespace

torch::Tensor mlr_grad(torch::Tensor P, torch::Tensor X) {
  CHECK_IN(P); CHECK_IN(X);
  const int B = X.size(0), F = X.size(1), C = P.size(1);
  TORCH_CHECK(P.size(0) == B && C <= MLR_MAXC);
  auto grad = torch::zeros({C, F}, X.options());
  if (B > 0) {
    dim3 blk(FWD_THREADS);
    dim3 grid((F + GRAD_FTILE - 1) / GRAD_FTILE,
              (B + GRAD_BTILE - 1) / GRAD_BTILE);
    hipLaunchKernelGGL(mlr_grad_kernel, grid, blk, 0, current_stream(),
                       P.data_ptr<float>(), X.data_ptr<float>(),
                       grad.data_ptr<float>(), B, F, C);
  }
  return grad;
}

std::vector<torch::Tensor> mlr_fwd(torch::Tensor X, torch::Tensor W,
                                   torch::Tensor labels) {
  CHECK_IN(X); CHECK_IN(W); CHECK_IN(labels);
  TORCH_CHECK(X.dtype() == torch::kFloat32 && W.dtype() == torch::kFloat32);
  const int B = X.size(0), F = X.size(1), C = W.size(0);
  TORCH_CHECK(W.size(1) == F, "W must be [C, F]");
  TORCH_CHECK(C <= MLR_MAXC, "mlr_fwd supports C <= ", MLR_MAXC);
  auto grad = torch::empty({B, C}, X.options());
  auto loss = torch::zeros({}, X.options());
  auto correct = torch::zeros({}, X.options().dtype(torch::kInt32));
  if (B > 0) {
    dim3 blk(FWD_THREADS), grid((B + ROWS_PER_WG - 1) / ROWS_PER_WG);
    hipLaunchKernelGGL(mlr_fwd_kernel, grid, blk, 0, current_stream(),
                       X.data_ptr<float>(), W.data_ptr<float>(),
                       labels.data_ptr<int64_t>(), grad.data_ptr<float>(),
                       loss.data_ptr<float>(), correct.data_ptr<int>(),
                       B, F, C);
  }
  return {grad, loss, correct.to(torch::kInt64)};
}

std::vector<torch::Tensor> mlr_softmax_grad(torch::Tensor logits,
                                            torch::Tensor labels) {
  CHECK_IN(logits);
  CHECK_IN(labels);
  TORCH_CHECK(logits.dtype() == torch::kFloat32, "logits must be f32");
  int B = logits.size(0), C = logits.size(1);
  auto grad = torch::empty_like(logits);
  auto loss = torch::zeros({}, logits.options());
  auto correct = torch::zeros({}, logits.options().dtype(torch::kInt32));
  if (B > 0) {
    dim3 blk(256), grid((B + 255) / 256);
    hipLaunchKernelGGL(mlr_softmax_grad_kernel, grid, blk, 0,
                       current_stream(),
                       logits.data_ptr<float>(), labels.data_ptr<int64_t>(),
                       grad.data_ptr<float>(), loss.data_ptr<float>(),
                       correct.data_ptr<int>(), B, C);
  }
  return {grad, loss, correct.to(torch::kInt64)};
}
