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

}  // namespace

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
