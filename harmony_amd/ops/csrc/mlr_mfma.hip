// K4-MFMA: MLR fused forward + gradient on gfx950 matrix cores.
//
// Replaces the rocBLAS GEMM pair of the MLR step (reference
// MLRTrainer.java:374-398 fwd+grad, :475-489 softmax) with two hand-written
// f32-input MFMA kernels (v_mfma_f32_16x16x4_f32 — exact f32, the CDNA4
// f32 rate; there is no xf32 on gfx950):
//
//   fwd : logits[b][c] = sum_f X[b][f] * Wt[f][c], then an in-register
//         quarter-wave softmax + label-subtract + CE/accuracy epilogue —
//         P lands in HBM padded to 16 classes, X is read exactly once.
//   grad: gradT[c][f] = sum_b P[b][c] * X[b][f]   (computed as P^T·X with
//         P as the MFMA A operand so X, the 1 GiB streaming operand, is
//         read coalesced row-major; split-B partials combine with fp32
//         global atomics).
//
// The host side (ops.mlr_step_mfma) can run the pair row-blocked so the
// grad pass re-reads its X block from the 256 MiB Infinity Cache instead
// of HBM — the whole step is X-bandwidth-bound (2 passes over B*F fp32).
//
// Shapes: C <= 16 (bench C=10), F % 64 == 0, rows % 64 == 0. A-fragments
// of X are staged through LDS [64][65] (the +1 pad keeps the 16-lane
// column read conflict-free, §2 of the CDNA4 guide); Wt ([F][16], padded)
// is read straight from L2 (1 MiB, re-read per row-block); P ([B][16]) is
// read straight from L2 in the grad kernel.

#include "hip_common.h"

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int CPAD = 16;     // class dim padded to one MFMA tile
constexpr int MT = 64;       // rows per workgroup (4 waves x 16)
constexpr int KT = 64;       // f-tile per LDS stage

// ---------------------------------------------------------------- forward
// grid.x = rows/64, grid.y = SPLITF; block = 256 threads (4 waves).
// Each wave owns 16 rows; accumulates logits over its f-range.
// atomic_out=1: atomicAdd partial logits (split-F); else direct store.
__global__ __launch_bounds__(256)
void mlr_fwd_mfma_kernel(const float* __restrict__ X,
                         const float* __restrict__ Wt,   // [F][16]
                         float* __restrict__ logits,     // [B][16]
                         int rows, long F, long row_off, int atomic_out) {
  __shared__ float xs[MT][KT + 1];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int r = lane & 15;           // fragment row / col index
  const int k = lane >> 4;           // fragment k index (0..3)
  const long row0 = row_off + (long)blockIdx.x * MT;
  const long f_per = F / gridDim.y;
  const long f_lo = (long)blockIdx.y * f_per;
  const long f_hi = f_lo + f_per;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  // cooperative X stage: thread t loads 4 float4 of row t/4
  const int srow = threadIdx.x >> 2;               // 0..63
  const int scol0 = (threadIdx.x & 3) * 16;        // 0,16,32,48
  const float* xsrc = X + (row0 + srow) * F;

  for (long f0 = f_lo; f0 < f_hi; f0 += KT) {
    __syncthreads();
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const float4 v = *reinterpret_cast<const float4*>(
          xsrc + f0 + scol0 + u * 4);
      xs[srow][scol0 + u * 4 + 0] = v.x;
      xs[srow][scol0 + u * 4 + 1] = v.y;
      xs[srow][scol0 + u * 4 + 2] = v.z;
      xs[srow][scol0 + u * 4 + 3] = v.w;
    }
    __syncthreads();
    const float* wt = Wt + f0 * CPAD;
#pragma unroll
    for (int kk = 0; kk < KT; kk += 4) {
      const float a = xs[wave * 16 + r][kk + k];
      const float b = wt[(kk + k) * CPAD + r];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
  }
  // acc: lane holds cols c=r of rows (lane>>4)*4 + reg
  float* dst = logits + (row0 + wave * 16) * CPAD;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int row = (lane >> 4) * 4 + q;
    if (atomic_out)
      atomicAdd(&dst[row * CPAD + r], acc[q]);
    else
      dst[row * CPAD + r] = acc[q];
  }
}

// ------------------------------------------------- softmax epilogue on P
// One thread per (row, col) quarter: operates on logits [B][16] in place ->
// P = softmax(z) - onehot(label); z cols >= C treated as -inf.
// 64-lane waves: lane = row*16 split... simple: one WAVE per 4 rows like the
// fwd fragment map is unnecessary here — use one thread per row (C<=16 fits
// a scalar loop; B threads, trivially bandwidth-bound on 1 MiB).
__global__ void mlr_softmax_pad_kernel(float* __restrict__ P,
                                       const int64_t* __restrict__ labels,
                                       float* __restrict__ loss,
                                       int* __restrict__ correct,
                                       int B, int C) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= B) return;
  float* z = P + (long)row * CPAD;
  float m = -1e30f;
  int argmax = 0;
  for (int j = 0; j < C; ++j) {
    if (z[j] > m) { m = z[j]; argmax = j; }
  }
  float s = 0.f;
  for (int j = 0; j < C; ++j) s += __expf(z[j] - m);
  const float inv = 1.f / s;
  const int lab = (int)labels[row];
  const float ce = m + __logf(s) - z[lab];
  for (int j = 0; j < C; ++j)
    z[j] = __expf(z[j] - m) * inv - (j == lab ? 1.f : 0.f);
  for (int j = C; j < CPAD; ++j) z[j] = 0.f;
  atomicAdd(loss, ce);
  if (argmax == lab) atomicAdd(correct, 1);
}

// ---------------------------------------------------------------- gradient
// gradT[c][f] += sum_b P[b][c] * X[b][f]. grid.x = F/64, grid.y = SPLITB;
// 4 waves per block, each owning a 16-wide f-tile. A (P) and B (X) are read
// straight from global: both fragment reads are 64 B-contiguous per
// 16-lane group (P row-major padded, X row-major).
__global__ __launch_bounds__(256)
void mlr_grad_mfma_kernel(const float* __restrict__ P,   // [B][16]
                          const float* __restrict__ X,   // [B][F]
                          float* __restrict__ gradT,     // [16][F]
                          int rows, long F, long row_off) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int r = lane & 15;
  const int k = lane >> 4;
  const long f0 = (long)blockIdx.x * MT + wave * 16;
  const long b_per = rows / gridDim.y;
  const long b_lo = row_off + (long)blockIdx.y * b_per;
  const long b_hi = b_lo + b_per;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (long b = b_lo; b < b_hi; b += 4) {
    const float a = P[(b + k) * CPAD + r];       // A[c=r][k]
    const float bb = X[(b + k) * F + f0 + r];    // B[k][f=r]
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bb, acc, 0, 0, 0);
  }
  // D: col f = f0 + r, row c = (lane>>4)*4 + reg
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int c = (lane >> 4) * 4 + q;
    atomicAdd(&gradT[(long)c * F + f0 + r], acc[q]);
  }
}

}  // namespace

// Host entry: one MLR step's compute = fused fwd+softmax+grad.
// Returns {gradT [16,F] (rows 0..C-1 are P^T X), loss_sum, n_correct}.
// row_block > 0: process rows in blocks of that size, fwd+grad back to back
// per block, so the grad pass re-reads X from L2/Infinity Cache.
std::vector<torch::Tensor> mlr_step_mfma(torch::Tensor X, torch::Tensor Wt,
                                         torch::Tensor labels,
                                         int64_t row_block, int64_t C,
                                         int64_t splitf, int64_t splitb) {
  CHECK_IN(X); CHECK_IN(Wt); CHECK_IN(labels);
  const long B = X.size(0), F = X.size(1);
  TORCH_CHECK(Wt.size(0) == F && Wt.size(1) == CPAD, "Wt must be [F,16]");
  TORCH_CHECK(C <= CPAD && F % KT == 0 && B % MT == 0,
              "mlr_step_mfma shape: C<=16, F%64==0, B%64==0");
  auto opts = X.options();
  // zeros: the split-F forward accumulates partial logits with atomicAdd
  auto P = torch::zeros({B, (long)CPAD}, opts);
  auto gradT = torch::zeros({(long)CPAD, F}, opts);
  auto loss = torch::zeros({1}, opts);
  auto correct = torch::zeros({1}, opts.dtype(torch::kInt32));
  hipStream_t s = current_stream();

  long rb = row_block > 0 ? row_block : B;
  TORCH_CHECK(rb % MT == 0, "row_block % 64 != 0");
  for (long off = 0; off < B; off += rb) {
    const long n = std::min(rb, B - off);
    const int sf = (int)std::max<long>(1, splitf);
    TORCH_CHECK(F % ((long)sf * KT) == 0, "F % (splitf*64) != 0");
    hipLaunchKernelGGL(mlr_fwd_mfma_kernel, dim3(n / MT, sf), dim3(256), 0,
                       s, X.data_ptr<float>(), Wt.data_ptr<float>(),
                       P.data_ptr<float>(), (int)n, F, off, sf > 1 ? 1 : 0);
    hipLaunchKernelGGL(mlr_softmax_pad_kernel, dim3((n + 255) / 256),
                       dim3(256), 0, s, P.data_ptr<float>() + off * CPAD,
                       labels.data_ptr<int64_t>() + off,
                       loss.data_ptr<float>(), correct.data_ptr<int>(),
                       (int)n, (int)C);
    const int sb = (int)std::max<long>(1, splitb);
    TORCH_CHECK(n % ((long)sb * 4) == 0, "rows % (splitb*4) != 0");
    hipLaunchKernelGGL(mlr_grad_mfma_kernel, dim3(F / MT, sb), dim3(256), 0,
                       s, P.data_ptr<float>(), X.data_ptr<float>(),
                       gradT.data_ptr<float>(), (int)n, F, off);
  }
  return {gradT, loss, correct};
}
