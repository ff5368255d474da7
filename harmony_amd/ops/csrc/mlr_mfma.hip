// K4-MFMA: MLR fused forward + gradient on gfx950 matrix cores.
//
// Replaces the rocBLAS GEMM pair of the MLR step (reference
// MLRTrainer.java:374-398 fwd+grad, :475-489 softmax) with two hand-written
// f32-input MFMA kernels (v_mfma_f32_16x16x4_f32 — exact f32; there is no
// xf32 on gfx950, and f32-in MFMA runs at the f32 vector rate while leaving
// the VALU free for the softmax epilogue):
//
//   fwd : logits[b][c] = sum_f X[b][f] * Wt[f][c]; X tiles and the padded
//         W^T tile are staged in LDS; the softmax/label-subtract/CE
//         epilogue is a separate trivial kernel over P (1 MiB).
//   grad: gradT[c][f] = sum_b P[b][c] * X[b][f]  (P^T·X with P as the MFMA
//         A operand so X — the 1 GiB streaming operand — is staged through
//         LDS in full coalesced rows; split-B partials combine with fp32
//         global atomics).
//
// Both kernels keep TWO independent accumulator chains per output tile:
// v_mfma_f32_16x16x4_f32 has a 40-cycle dependent-accumulator latency vs a
// 32-cycle issue interval, so a single chain leaves the MFMA pipe 20% idle
// even at full occupancy (CDNA4 guide §3).
//
// Shapes: C <= 16 (bench C=10), F % 64 == 0, rows % 64 == 0. LDS X tiles
// are [64][65] (the +1 pad keeps 16-lane column reads conflict-free).
// Measured A/B vs the rocBLAS pair: scripts/mlr_mfma_ab.py,
// profiles/r02_mlr_mfma.md.

#include "hip_common.h"

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int CPAD = 16;     // class dim padded to one MFMA tile
constexpr int MT = 64;       // rows per workgroup (4 waves x 16)
constexpr int KT = 64;       // f-tile (fwd) / b-tile (grad) per LDS stage

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
  __shared__ float ws[KT * CPAD];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int r = lane & 15;           // fragment row / col index
  const int k = lane >> 4;           // fragment k index (0..3)
  const long row0 = row_off + (long)blockIdx.x * MT;
  const long f_per = F / gridDim.y;
  const long f_lo = (long)blockIdx.y * f_per;
  const long f_hi = f_lo + f_per;

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  // cooperative X stage: thread t loads 4 float4 of row t/4
  const int srow = threadIdx.x >> 2;               // 0..63
  const int scol0 = (threadIdx.x & 3) * 16;        // 0,16,32,48
  const float* xsrc = X + (row0 + srow) * F;
  const int xr = wave * 16 + r;

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
    // W^T tile: KT*16 floats, 4 per thread, coalesced
    *reinterpret_cast<float4*>(&ws[threadIdx.x * 4]) =
        *reinterpret_cast<const float4*>(Wt + f0 * CPAD + threadIdx.x * 4);
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < KT; kk += 8) {
      const float a0 = xs[xr][kk + k];
      const float b0 = ws[(kk + k) * CPAD + r];
      acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc0, 0, 0, 0);
      const float a1 = xs[xr][kk + 4 + k];
      const float b1 = ws[(kk + 4 + k) * CPAD + r];
      acc1 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc1, 0, 0, 0);
    }
  }
  // acc: lane holds col c=r of rows (lane>>4)*4 + reg
  float* dst = logits + (row0 + wave * 16) * CPAD;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int row = (lane >> 4) * 4 + q;
    const float v = acc0[q] + acc1[q];
    if (atomic_out)
      atomicAdd(&dst[row * CPAD + r], v);
    else
      dst[row * CPAD + r] = v;
  }
}

// ------------------------------------------------- softmax epilogue on P
// One thread per row: P = softmax(z) - onehot(label) in place on [B][16];
// cols >= C are zeroed. Trivially bandwidth-bound on 1 MiB.
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
// 4 waves per block, each owning a 16-wide f-tile; X (streaming) and P
// (L2-resident) are staged through LDS in coalesced full rows.
__global__ __launch_bounds__(256)
void mlr_grad_mfma_kernel(const float* __restrict__ P,   // [B][16]
                          const float* __restrict__ X,   // [B][F]
                          float* __restrict__ gradT,     // [16][F]
                          int rows, long F, long row_off) {
  __shared__ float xs[KT][MT + 1];   // [b_local][f_local]
  __shared__ float ps[KT * CPAD];    // [b_local][c]
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int r = lane & 15;
  const int k = lane >> 4;
  const long fblk = (long)blockIdx.x * MT;
  const long b_per = rows / gridDim.y;
  const long b_lo = row_off + (long)blockIdx.y * b_per;
  const long b_hi = b_lo + b_per;

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  const int srow = threadIdx.x >> 2;               // 0..63 (b_local)
  const int scol0 = (threadIdx.x & 3) * 16;
  const int fcol = wave * 16 + r;

  for (long b0 = b_lo; b0 < b_hi; b0 += KT) {
    __syncthreads();
    {
      const float* xsrc = X + (b0 + srow) * F + fblk;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const float4 v = *reinterpret_cast<const float4*>(
            xsrc + scol0 + u * 4);
        xs[srow][scol0 + u * 4 + 0] = v.x;
        xs[srow][scol0 + u * 4 + 1] = v.y;
        xs[srow][scol0 + u * 4 + 2] = v.z;
        xs[srow][scol0 + u * 4 + 3] = v.w;
      }
      *reinterpret_cast<float4*>(&ps[threadIdx.x * 4]) =
          *reinterpret_cast<const float4*>(P + b0 * CPAD + threadIdx.x * 4);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < KT; kk += 8) {
      const float a0 = ps[(kk + k) * CPAD + r];
      const float b0v = xs[kk + k][fcol];
      acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0v, acc0, 0, 0, 0);
      const float a1 = ps[(kk + 4 + k) * CPAD + r];
      const float b1v = xs[kk + 4 + k][fcol];
      acc1 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1v, acc1, 0, 0, 0);
    }
  }
  // D: col f = fblk + wave*16 + r, row c = (lane>>4)*4 + reg
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int c = (lane >> 4) * 4 + q;
    atomicAdd(&gradT[(long)c * F + fblk + wave * 16 + r],
              acc0[q] + acc1[q]);
  }
}

}  // namespace

// Piecewise entries (used by the python two-stream pipeline: fwd of row
// block i+1 overlaps grad of block i, so grad re-reads X from the
// Infinity Cache while fwd streams the next block from HBM).

void mlr_fwd_mfma_part(torch::Tensor X, torch::Tensor Wt, torch::Tensor P,
                       int64_t off, int64_t n, int64_t splitf) {
  const long F = X.size(1);
  const int sf = (int)std::max<int64_t>(1, splitf);
  TORCH_CHECK(F % ((long)sf * KT) == 0 && n % MT == 0, "fwd part shape");
  hipLaunchKernelGGL(mlr_fwd_mfma_kernel, dim3(n / MT, sf), dim3(256), 0,
                     current_stream(), X.data_ptr<float>(),
                     Wt.data_ptr<float>(), P.data_ptr<float>(), (int)n, F,
                     off, sf > 1 ? 1 : 0);
}

void mlr_softmax_part(torch::Tensor P, torch::Tensor labels,
                      torch::Tensor loss, torch::Tensor correct,
                      int64_t off, int64_t n, int64_t C) {
  hipLaunchKernelGGL(mlr_softmax_pad_kernel, dim3((n + 255) / 256),
                     dim3(256), 0, current_stream(),
                     P.data_ptr<float>() + off * CPAD,
                     labels.data_ptr<int64_t>() + off,
                     loss.data_ptr<float>(), correct.data_ptr<int>(),
                     (int)n, (int)C);
}

void mlr_grad_mfma_part(torch::Tensor P, torch::Tensor X,
                        torch::Tensor gradT, int64_t off, int64_t n,
                        int64_t splitb) {
  const long F = X.size(1);
  const int sb = (int)std::max<int64_t>(1, splitb);
  TORCH_CHECK(n % ((long)sb * KT) == 0, "grad part shape");
  hipLaunchKernelGGL(mlr_grad_mfma_kernel, dim3(F / MT, sb), dim3(256), 0,
                     current_stream(), P.data_ptr<float>(),
                     X.data_ptr<float>(), gradT.data_ptr<float>(), (int)n,
                     F, off);
}

// Host entry: one MLR step's compute = fused fwd+softmax+grad.
// Returns {gradT [16,F] (rows 0..C-1 are P^T X), loss_sum, n_correct}.
// row_block > 0: process rows in blocks of that size, fwd+grad back to back
// per block (the grad pass then re-reads X from L2/Infinity Cache).
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
    TORCH_CHECK(n % ((long)sb * KT) == 0, "rows % (splitb*64) != 0");
    hipLaunchKernelGGL(mlr_grad_mfma_kernel, dim3(F / MT, sb), dim3(256), 0,
                       s, P.data_ptr<float>(), X.data_ptr<float>(),
                       gradT.data_ptr<float>(), (int)n, F, off);
  }
  return {gradT, loss, correct};
}
