// K7: LDA collapsed Gibbs sweep — one wave per document.
// Reference: SparseLDASampler.java:141-274 (s/r/q bucket sampling). The
// bucket decomposition is a CPU sparsity trick; on CDNA4 the dense K-way
// distribution p(k) = (n_dk+a)(n_wk+b)/(n_k+Vb) is computed by 64 lanes in
// K/64 register chunks, the draw is a wave butterfly total + inclusive-scan
// select, and the doc-topic row lives in LDS for the whole document.
//
// Within-sweep semantics: word-topic and topic-sum counts are the pulled
// snapshot (batch-stale, standard for distributed GPU LDA); doc-topic counts
// update token-by-token (the sequential dependency that matters for mixing).
// RNG: counter-based hash (hip_common.h) keyed by (seed, token index) —
// bit-reproduced by the torch reference for cross-checking.

#include "hip_common.h"
#include <cstdlib>

namespace {

constexpr int MAXC = 16;           // K <= 64*G_chunks
constexpr int BLOCK_THREADS = 256;

// Templated on the lane-group width G: one document per G-lane group.
// Measured (1M tokens, K=256, MI355X): G=64 is faster IN ISOLATION
// (0.74 vs 0.93 ms step) but G=32 wins the 3-concurrent-job bench
// (1.52 vs 1.72 ms/step) — the halved grid and lane width leave more CU
// issue slots to the co-scheduled jobs' kernels. The default follows the
// multi-tenant headline (G=32 when K <= 512); HARMONY_LDA_G overrides.
template <int G>
__device__ __forceinline__ float group_reduce_sum(float v) {
#pragma unroll
  for (int m = G / 2; m > 0; m >>= 1) v += __shfl_xor(v, m, G);
  return v;
}

template <int G>
__device__ __forceinline__ float group_inclusive_scan(float v, int lane) {
#pragma unroll
  for (int d = 1; d < G; d <<= 1) {
    float o = __shfl_up(v, d, G);
    if (lane >= d) v += o;
  }
  return v;
}

template <int G>
__global__ void lda_gibbs_kernel(int* __restrict__ doc_topic,
                                 const int* __restrict__ word_topic,
                                 const int* __restrict__ topic_sum,
                                 const int64_t* __restrict__ doc_offsets,
                                 const int64_t* __restrict__ word_ids,
                                 int* __restrict__ z,
                                 float alpha, float beta, float vbeta,
                                 int n_docs, int K, unsigned int seed) {
  constexpr int GROUPS = BLOCK_THREADS / G;
  extern __shared__ int smem[];                  // [GROUPS][K] nd + [K] invden
  const int group = threadIdx.x / G;
  const int lane = threadIdx.x % G;
  const int sub = (threadIdx.x % WAVE) / G;      // group index within wave
  int* nd = smem + group * K;
  float* invden = (float*)(smem + GROUPS * K);
  const int nchunk = (K + G - 1) / G;

  for (int idx = threadIdx.x; idx < K; idx += blockDim.x)
    invden[idx] = 1.0f / ((float)topic_sum[idx] + vbeta);
  __syncthreads();

  const int doc = blockIdx.x * GROUPS + group;
  if (doc >= n_docs) return;

#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * G + lane;
    if (c < nchunk && idx < K) nd[idx] = doc_topic[(int64_t)doc * K + idx];
  }

  const int64_t p0 = doc_offsets[doc], p1 = doc_offsets[doc + 1];
  // NOTE: a 2-deep prefetch of the next token's word-topic row was tried
  // and measured SLOWER (0.51 -> 0.71 ms at 1M tokens, K=256): the extra
  // registers cost occupancy, which was already hiding the row-load
  // latency across the resident waves. Keep the simple form.
  for (int64_t p = p0; p < p1; ++p) {
    const int64_t w = word_ids[p];
    const int old = z[p];
    if (lane == 0) nd[old] -= 1;

    float pr[MAXC];
    float part = 0.f;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) {
      int idx = c * G + lane;
      if (c < nchunk && idx < K) {
        pr[c] = ((float)nd[idx] + alpha) *
                ((float)word_topic[w * K + idx] + beta) * invden[idx];
      } else {
        pr[c] = 0.f;
      }
      part += pr[c];
    }
    const float tot = group_reduce_sum<G>(part);
    const float u = rng_uniform(seed, (unsigned int)p) * tot;

    float run = 0.f;
    int knew = -1;
#pragma unroll
    for (int c = 0; c < MAXC; ++c) {
      if (c >= nchunk) break;
      const float csum = group_reduce_sum<G>(pr[c]);
      if (knew < 0 && run + csum > u) {
        const float pref = group_inclusive_scan<G>(pr[c], lane);
        unsigned long long b = __ballot(run + pref > u);
        unsigned long long gm = (G == WAVE)
            ? b : ((b >> (sub * G)) & ((1ull << G) - 1));
        int sel = (gm != 0) ? (__ffsll((long long)gm) - 1) : (G - 1);
        knew = c * G + sel;
      }
      run += csum;
    }
    if (knew < 0 || knew >= K) knew = old;   // numeric edge: keep old topic
    if (lane == 0) {
      nd[knew] += 1;
      z[p] = knew;
    }
  }

#pragma unroll
  for (int c = 0; c < MAXC; ++c) {
    int idx = c * G + lane;
    if (c < nchunk && idx < K) doc_topic[(int64_t)doc * K + idx] = nd[idx];
  }
}

}  // namespace

namespace {

// K9 (sparse form): apply (row, old_topic, new_topic) +/-1 pairs to the
// owner's word-topic shard — the wire format is the reference's TopicChanges
// delta pairs (lda/TopicChanges, LDAETModelUpdateFunction.java:43-64), 12 B
// per changed token instead of a dense K-int row per touched word.
__global__ void lda_apply_pairs_kernel(int* __restrict__ shard,
                                       const int64_t* __restrict__ rows,
                                       const int* __restrict__ old_t,
                                       const int* __restrict__ new_t,
                                       int64_t n, int K) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int64_t base = rows[i] * K;
  atomicSub(&shard[base + old_t[i]], 1);
  atomicAdd(&shard[base + new_t[i]], 1);
}

// Fully-fused local update: for every token, if the topic changed, apply
// the +/-1 to the word-topic shard row AND the topic-summary row in one
// pass. Replaces (nonzero sync + 3 gathers + 2 bincounts + scatter +
// summary push) on the single-owner path. The summary deltas stage through
// an LDS histogram per block (grid-stride loop, bounded grid): per-token
// GLOBAL atomics on the K summary counters were measured catastrophic
// (~1M conflicting updates on 256 addresses: LDA step 1.19 -> 1.84 ms).
__global__ void lda_apply_all_kernel(int* __restrict__ shard,
                                     const int64_t* __restrict__ word_rows,
                                     const int* __restrict__ old_t,
                                     const int* __restrict__ new_t,
                                     int64_t summary_row,
                                     int64_t n, int K) {
  extern __shared__ int ssum[];                  // [K]
  for (int k = threadIdx.x; k < K; k += blockDim.x) ssum[k] = 0;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const int o = old_t[i], nw = new_t[i];
    if (o == nw) continue;
    const int64_t base = word_rows[i] * K;
    atomicSub(&shard[base + o], 1);
    atomicAdd(&shard[base + nw], 1);
    atomicSub(&ssum[o], 1);
    atomicAdd(&ssum[nw], 1);
  }
  __syncthreads();
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    const int v = ssum[k];
    if (v != 0) atomicAdd(&shard[summary_row * K + k], v);
  }
}

}  // namespace

void lda_apply_all(torch::Tensor shard, torch::Tensor word_rows,
                   torch::Tensor old_t, torch::Tensor new_t,
                   int64_t summary_row) {
  CHECK_IN(shard); CHECK_IN(word_rows); CHECK_IN(old_t); CHECK_IN(new_t);
  TORCH_CHECK(shard.dtype() == torch::kInt32);
  const int64_t n = word_rows.size(0);
  if (n == 0) return;
  const int K = shard.size(1);
  const unsigned nblk = (unsigned)std::min<int64_t>((n + 255) / 256, 1024);
  dim3 blk(256), grid(nblk);
  hipLaunchKernelGGL(lda_apply_all_kernel, grid, blk, (size_t)K * 4,
                     current_stream(),
                     shard.data_ptr<int>(), word_rows.data_ptr<int64_t>(),
                     old_t.data_ptr<int>(), new_t.data_ptr<int>(),
                     summary_row, n, K);
}

void lda_apply_pairs(torch::Tensor shard, torch::Tensor rows,
                     torch::Tensor old_t, torch::Tensor new_t) {
  CHECK_IN(shard); CHECK_IN(rows); CHECK_IN(old_t); CHECK_IN(new_t);
  TORCH_CHECK(shard.dtype() == torch::kInt32);
  const int64_t n = rows.size(0);
  if (n == 0) return;
  const int K = shard.size(1);
  dim3 blk(256), grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(lda_apply_pairs_kernel, grid, blk, 0, current_stream(),
                     shard.data_ptr<int>(), rows.data_ptr<int64_t>(),
                     old_t.data_ptr<int>(), new_t.data_ptr<int>(), n, K);
}

torch::Tensor lda_gibbs(torch::Tensor doc_topic, torch::Tensor word_topic,
                        torch::Tensor topic_sum, torch::Tensor doc_offsets,
                        torch::Tensor word_ids, torch::Tensor assignments,
                        double alpha, double beta, int64_t num_vocabs,
                        int64_t seed) {
  CHECK_IN(doc_topic); CHECK_IN(word_topic); CHECK_IN(topic_sum);
  CHECK_IN(doc_offsets); CHECK_IN(word_ids); CHECK_IN(assignments);
  TORCH_CHECK(doc_topic.dtype() == torch::kInt32);
  TORCH_CHECK(word_topic.dtype() == torch::kInt32);
  const int D = doc_topic.size(0), K = doc_topic.size(1);
  TORCH_CHECK(K <= WAVE * MAXC, "num_topics > ", WAVE * MAXC, " unsupported");
  if (D == 0) return assignments;
  int G = (K <= 32 * MAXC) ? 32 : 64;             // 2 docs/wave when K fits
  const char* env = getenv("HARMONY_LDA_G");      // A/B override
  if (env && atoi(env) == 64) G = 64;
  if (env && atoi(env) == 32 && K <= 32 * MAXC) G = 32;
  if (env && atoi(env) == 16 && K <= 16 * MAXC) G = 16;
  const int groups = BLOCK_THREADS / G;
  dim3 blk(BLOCK_THREADS);
  dim3 grid((D + groups - 1) / groups);
  const size_t shmem = (size_t)(groups * K + K) * 4;
  if (G == 16) {
    hipLaunchKernelGGL(lda_gibbs_kernel<16>, grid, blk, shmem,
                       current_stream(),
                       doc_topic.data_ptr<int>(), word_topic.data_ptr<int>(),
                       topic_sum.data_ptr<int>(),
                       doc_offsets.data_ptr<int64_t>(),
                       word_ids.data_ptr<int64_t>(),
                       assignments.data_ptr<int>(),
                       (float)alpha, (float)beta,
                       (float)(num_vocabs * beta), D, K,
                       (unsigned int)(seed & 0xffffffff));
  } else if (G == 32) {
    hipLaunchKernelGGL(lda_gibbs_kernel<32>, grid, blk, shmem,
                       current_stream(),
                       doc_topic.data_ptr<int>(), word_topic.data_ptr<int>(),
                       topic_sum.data_ptr<int>(),
                       doc_offsets.data_ptr<int64_t>(),
                       word_ids.data_ptr<int64_t>(),
                       assignments.data_ptr<int>(),
                       (float)alpha, (float)beta,
                       (float)(num_vocabs * beta), D, K,
                       (unsigned int)(seed & 0xffffffff));
  } else {
    hipLaunchKernelGGL(lda_gibbs_kernel<64>, grid, blk, shmem,
                       current_stream(),
                       doc_topic.data_ptr<int>(), word_topic.data_ptr<int>(),
                       topic_sum.data_ptr<int>(),
                       doc_offsets.data_ptr<int64_t>(),
                       word_ids.data_ptr<int64_t>(),
                       assignments.data_ptr<int>(),
                       (float)alpha, (float)beta,
                       (float)(num_vocabs * beta), D, K,
                       (unsigned int)(seed & 0xffffffff));
  }
  return assignments;
}
