// K1+K2: NMF gradient over a sparse batch.
// Reference hot loop NMFTrainer.java:328-367 (per nonzero (i,j,v):
// e = L_i.R_j - v; lGrad += 2e R_j + L2; rGrad_j += 2e L_i + L2) and the
// per-thread gradient-map merge :375-406.
//
// CDNA4 design: one WAVE per L-row (CSR). The row's L vector lives in
// registers (rank <= 256 -> <=4 f32 per lane), its lGrad accumulates in
// registers across all of the row's nonzeros and is written once —
// no atomics on the L side (this replaces the reference's per-thread
// hashmap merge). rGrad contributions scatter with atomicAdd: distinct rows
// rarely collide on a column within a cycle, and f32 atomicAdd on HBM/L2 is
// cheap on gfx950. The rank-dot is a 64-lane butterfly reduction.

#include "hip_common.h"
#include <cstdlib>

namespace {

constexpr int MAXC = 4;  // supports rank <= 4*64 = 256

__global__ void nmf_grad_kernel(const float* __restrict__ L,
                                const float* __restrict__ R,
                                const int64_t* __restrict__ row_ptr,
                                const int64_t* __restrict__ col_idx,
                                const float* __restrict__ vals,
                                float* __restrict__ lgrad,
                                float* __restrict__ rgrad,
                                float* __restrict__ sqerr,
                                int n_rows, int k, float lam2) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int row = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  if (row >= n_rows) return;
  const int nchunk = (k + WAVE - 1) / WAVE;

  float l[MAXC], lg[MAXC];
#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * WAVE + lane;
    l[c] = (c < nchunk && idx < k) ? L[(int64_t)row * k + idx] : 0.f;
    lg[c] = 0.f;
  }

  float sq = 0.f;
  const int64_t p0 = row_ptr[row], p1 = row_ptr[row + 1];
  for (int64_t p = p0; p < p1; ++p) {
    const int64_t j = col_idx[p];
    float r[MAXC];
    float part = 0.f;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) {
      int idx = c * WAVE + lane;
      r[c] = (c < nchunk && idx < k) ? R[j * k + idx] : 0.f;
      part += l[c] * r[c];
    }
    const float e = wave_reduce_sum(part) - vals[p];
    const float ge = 2.f * e;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) {
      int idx = c * WAVE + lane;
      if (c < nchunk && idx < k) {
        lg[c] += ge * r[c] + lam2 * l[c];
        atomicAdd(&rgrad[j * k + idx], ge * l[c] + lam2 * r[c]);
      }
    }
    sq += (lane == 0) ? e * e : 0.f;
  }

#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * WAVE + lane;
    if (c < nchunk && idx < k) lgrad[(int64_t)row * k + idx] = lg[c];
  }
  if (lane == 0 && sq != 0.f) atomicAdd(sqerr, sq);
}

// Two-pass variant (no atomics): the batch's nonzeros are ALSO indexed by a
// static column-sorted permutation (precomputed once per data block, like
// the reference's per-feature pre-sort in GBT). Pass A (row-major) computes
// e[p] + lgrad with L/lgrad in registers; pass B walks each column's
// segment, accumulates rgrad_j in registers and writes it once. This
// replaces the reference's per-thread gradient hashmap + merge
// (NMFTrainer.aggregateGradient:375-406) with a segmented reduction and
// removes 200M+ HBM/L2 atomics per batch.

template <int G>
__device__ __forceinline__ float group_sum(float v) {
#pragma unroll
  for (int m = G / 2; m > 0; m >>= 1) v += __shfl_xor(v, m, G);
  return v;
}

template <int G>
__global__ void nmf_grad_e_kernel(const float* __restrict__ L,
                                  const float* __restrict__ R,
                                  const int64_t* __restrict__ row_ptr,
                                  const int64_t* __restrict__ col_idx,
                                  const float* __restrict__ vals,
                                  float* __restrict__ lgrad,
                                  float* __restrict__ e_out,
                                  float* __restrict__ sqerr,
                                  int n_rows, int k, float lam2) {
  const int lane = threadIdx.x & (G - 1);
  const int row = (blockIdx.x * blockDim.x + threadIdx.x) / G;
  if (row >= n_rows) return;
  const int nchunk = (k + G - 1) / G;
  float l[MAXC], lg[MAXC];
#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * G + lane;
    l[c] = (c < nchunk && idx < k) ? L[(int64_t)row * k + idx] : 0.f;
    lg[c] = 0.f;
  }
  float sq = 0.f;
  const int64_t p0 = row_ptr[row], p1 = row_ptr[row + 1];
  // software-pipelined 2-deep: issue the NEXT nonzero's R-row loads before
  // the current dot's butterfly reduce — the serial per-nonzero chain
  // (load R_j -> dot -> reduce) is latency-bound otherwise
  float r[MAXC], rn[MAXC];
  if (p0 < p1) {
    const int64_t j0 = col_idx[p0];
#pragma unroll
    for (int c = 0; c < MAXC; ++c) {
      int idx = c * G + lane;
      r[c] = (c < nchunk && idx < k) ? R[j0 * k + idx] : 0.f;
    }
  }
  for (int64_t p = p0; p < p1; ++p) {
    if (p + 1 < p1) {
      const int64_t jn = col_idx[p + 1];
#pragma unroll
      for (int c = 0; c < MAXC; ++c) {
        int idx = c * G + lane;
        rn[c] = (c < nchunk && idx < k) ? R[jn * k + idx] : 0.f;
      }
    }
    float part = 0.f;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) part += l[c] * r[c];
    const float e = group_sum<G>(part) - vals[p];
    const float ge = 2.f * e;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) lg[c] += ge * r[c] + lam2 * l[c];
    if (lane == 0) e_out[p] = e;
    sq += (lane == 0) ? e * e : 0.f;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) r[c] = rn[c];
  }
#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * G + lane;
    if (c < nchunk && idx < k) lgrad[(int64_t)row * k + idx] = lg[c];
  }
  if (lane == 0 && sq != 0.f) atomicAdd(sqerr, sq);
}

template <int G>
__global__ void nmf_rgrad_kernel(const float* __restrict__ L,
                                 const float* __restrict__ R,
                                 const float* __restrict__ e_in,
                                 const int64_t* __restrict__ perm,
                                 const int64_t* __restrict__ seg_ptr,
                                 const int64_t* __restrict__ row_sorted,
                                 float* __restrict__ rgrad,
                                 int n_cols, int k, float lam2) {
  const int lane = threadIdx.x & (G - 1);
  const int col = (blockIdx.x * blockDim.x + threadIdx.x) / G;
  if (col >= n_cols) return;
  const int nchunk = (k + G - 1) / G;
  float acc[MAXC];
#pragma unroll
  for (int c = 0; c < MAXC; ++c) acc[c] = 0.f;
  const int64_t p0 = seg_ptr[col], p1 = seg_ptr[col + 1];
  // 2-deep pipeline over the segment's (e, L-row) loads
  float lv[MAXC], lvn[MAXC];
  float ge = 0.f, gen = 0.f;
  if (p0 < p1) {
    ge = 2.f * e_in[perm[p0]];
    const int64_t i0 = row_sorted[p0];
#pragma unroll
    for (int c = 0; c < MAXC; ++c) {
      int idx = c * G + lane;
      lv[c] = (c < nchunk && idx < k) ? L[i0 * k + idx] : 0.f;
    }
  }
  for (int64_t p = p0; p < p1; ++p) {
    if (p + 1 < p1) {
      gen = 2.f * e_in[perm[p + 1]];
      const int64_t in_ = row_sorted[p + 1];
#pragma unroll
      for (int c = 0; c < MAXC; ++c) {
        int idx = c * G + lane;
        lvn[c] = (c < nchunk && idx < k) ? L[in_ * k + idx] : 0.f;
      }
    }
#pragma unroll
    for (int c = 0; c < MAXC; ++c) acc[c] += ge * lv[c];
#pragma unroll
    for (int c = 0; c < MAXC; ++c) lv[c] = lvn[c];
    ge = gen;
  }
  // L2 term: lam2 * (#nonzeros in this column) * R_j
  const float nl = lam2 * (float)(p1 - p0);
#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * G + lane;
    if (c < nchunk && idx < k)
      rgrad[(int64_t)col * k + idx] = acc[c] + nl * R[(int64_t)col * k + idx];
  }
}

}  // namespace

std::vector<torch::Tensor> nmf_grad_twopass(
    torch::Tensor L, torch::Tensor R, torch::Tensor row_ptr,
    torch::Tensor col_idx, torch::Tensor vals, torch::Tensor perm,
    torch::Tensor seg_ptr, torch::Tensor row_sorted, double lam) {
  CHECK_IN(L); CHECK_IN(R); CHECK_IN(row_ptr); CHECK_IN(col_idx);
  CHECK_IN(vals); CHECK_IN(perm); CHECK_IN(seg_ptr); CHECK_IN(row_sorted);
  const int n = L.size(0), k = L.size(1), m = R.size(0);
  TORCH_CHECK(k <= 64 * MAXC);
  TORCH_CHECK(seg_ptr.numel() == m + 1, "seg_ptr must cover all R rows");
  auto lgrad = torch::empty_like(L);
  auto rgrad = torch::empty_like(R);
  auto e = torch::empty_like(vals);
  auto sqerr = torch::zeros({}, L.options());
  // lane-group width: 32 (2 rows/wave) when the rank fits 32*MAXC —
  // same co-scheduling reasoning as the LDA sampler; HARMONY_NMF_G overrides
  int G = (k <= 32 * MAXC) ? 32 : 64;
  const char* env = getenv("HARMONY_NMF_G");
  if (env && atoi(env) == 64) G = 64;
  if (env && atoi(env) == 32 && k <= 32 * MAXC) G = 32;
  int threads = 256;   // HARMONY_NMF_THREADS: occupancy A/B knob
  const char* te = getenv("HARMONY_NMF_THREADS");
  if (te) { int t = atoi(te); if (t == 128 || t == 256 || t == 512 || t == 1024) threads = t; }
  if (n > 0) {
    dim3 grid((n + threads / G - 1) / (threads / G));
    if (G == 32)
      hipLaunchKernelGGL(nmf_grad_e_kernel<32>, grid, dim3(threads), 0,
                         current_stream(),
                         L.data_ptr<float>(), R.data_ptr<float>(),
                         row_ptr.data_ptr<int64_t>(),
                         col_idx.data_ptr<int64_t>(),
                         vals.data_ptr<float>(), lgrad.data_ptr<float>(),
                         e.data_ptr<float>(), sqerr.data_ptr<float>(),
                         n, k, 2.f * (float)lam);
    else
      hipLaunchKernelGGL(nmf_grad_e_kernel<64>, grid, dim3(threads), 0,
                         current_stream(),
                         L.data_ptr<float>(), R.data_ptr<float>(),
                         row_ptr.data_ptr<int64_t>(),
                         col_idx.data_ptr<int64_t>(),
                         vals.data_ptr<float>(), lgrad.data_ptr<float>(),
                         e.data_ptr<float>(), sqerr.data_ptr<float>(),
                         n, k, 2.f * (float)lam);
  }
  if (m > 0) {
    dim3 grid((m + threads / G - 1) / (threads / G));
    if (G == 32)
      hipLaunchKernelGGL(nmf_rgrad_kernel<32>, grid, dim3(threads), 0,
                         current_stream(),
                         L.data_ptr<float>(), R.data_ptr<float>(),
                         e.data_ptr<float>(), perm.data_ptr<int64_t>(),
                         seg_ptr.data_ptr<int64_t>(),
                         row_sorted.data_ptr<int64_t>(),
                         rgrad.data_ptr<float>(), m, k, 2.f * (float)lam);
    else
      hipLaunchKernelGGL(nmf_rgrad_kernel<64>, grid, dim3(threads), 0,
                         current_stream(),
                         L.data_ptr<float>(), R.data_ptr<float>(),
                         e.data_ptr<float>(), perm.data_ptr<int64_t>(),
                         seg_ptr.data_ptr<int64_t>(),
                         row_sorted.data_ptr<int64_t>(),
                         rgrad.data_ptr<float>(), m, k, 2.f * (float)lam);
  }
  return {lgrad, rgrad, sqerr};
}

std::vector<torch::Tensor> nmf_grad(torch::Tensor L, torch::Tensor R,
                                    torch::Tensor row_ptr,
                                    torch::Tensor col_idx, torch::Tensor vals,
                                    double lam) {
  CHECK_IN(L); CHECK_IN(R); CHECK_IN(row_ptr); CHECK_IN(col_idx); CHECK_IN(vals);
  TORCH_CHECK(L.dtype() == torch::kFloat32 && R.dtype() == torch::kFloat32);
  TORCH_CHECK(L.size(1) == R.size(1), "rank mismatch");
  TORCH_CHECK(L.size(1) <= 64 * MAXC, "rank > ", 64 * MAXC, " unsupported");
  const int n = L.size(0), k = L.size(1);
  TORCH_CHECK(row_ptr.numel() == n + 1, "row_ptr size");
  auto lgrad = torch::empty_like(L);
  auto rgrad = torch::zeros_like(R);
  auto sqerr = torch::zeros({}, L.options());
  if (n > 0) {
    const int waves_per_block = 4;                 // 256 threads
    dim3 blk(WAVE * waves_per_block);
    dim3 grid((n + waves_per_block - 1) / waves_per_block);
    hipLaunchKernelGGL(nmf_grad_kernel, grid, blk, 0, current_stream(),
                       L.data_ptr<float>(), R.data_ptr<float>(),
                       row_ptr.data_ptr<int64_t>(), col_idx.data_ptr<int64_t>(),
                       vals.data_ptr<float>(), lgrad.data_ptr<float>(),
                       rgrad.data_ptr<float>(), sqerr.data_ptr<float>(),
                       n, k, 2.f * (float)lam);
  }
  return {lgrad, rgrad, sqerr};
}
