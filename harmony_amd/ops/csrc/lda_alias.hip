// K7b: Metropolis-Hastings alias LDA sampler (LightLDA-style), opt-in.
//
// The exact dense sampler (lda.hip) costs O(K) per token. With the
// word-topic snapshot batch-stale (which it already is — see mlapps/lda.py),
// a 2-proposal MH step per token has the SAME stationary distribution at
// O(1) per token:
//   word proposal  t1 ~ q_w(k) ∝ (n_wk + b)/(n_k + Vb)  via a per-word
//                  alias table built once per pull; because q_w equals the
//                  word factor of the (stale) posterior exactly, the
//                  acceptance ratio collapses to the doc factor
//                  (n_d,t1 + a)/(n_d,s + a).
//   doc proposal   t2 ~ q_d(k) ∝ n~_dk + a  (n~ includes the current token)
//                  drawn in O(1): with prob aK/(aK+L_d) a uniform topic,
//                  else the topic of a uniformly chosen token of the doc;
//                  acceptance carries the full ratio with the doc-proposal
//                  correction.
//
// Execution shape: ONE THREAD PER DOCUMENT (the per-token work is scalar),
// the thread's doc-topic counts live as a u8 LDS row (counts <= doc length
// <= 255), tokens walk serially per doc, thousands of docs in flight.
// RNG: counter hash (hip_common.h) at ctr = token*8 + draw — mirrored by
// the torch reference for sample-exact CPU/GPU tests.

#include "hip_common.h"

namespace {

constexpr int ALIAS_THREADS = 128;

// Vose alias construction, one thread per word row, deterministic order
// (ascending k for the small/large queues) so the torch reference builds
// bit-identical tables. scratch: [rows][K] int32 workspace.
__global__ void alias_build_kernel(const int* __restrict__ word_topic,
                                   const float* __restrict__ invden,
                                   float beta,
                                   float* __restrict__ prob,     // [rows][K]
                                   int* __restrict__ alias,      // [rows][K]
                                   float* __restrict__ qsum,     // [rows]
                                   int* __restrict__ scratch,    // [rows][K]
                                   int rows, int K) {
  const int w = blockIdx.x * blockDim.x + threadIdx.x;
  if (w >= rows) return;
  const int64_t base = (int64_t)w * K;
  float s = 0.f;
  for (int k = 0; k < K; ++k) {
    const float p = ((float)word_topic[base + k] + beta) * invden[k];
    prob[base + k] = p;
    s += p;
  }
  qsum[w] = s;
  const float scale = (float)K / s;
  // normalized to mean 1: prob[k]*scale; two-queue pairing
  int* idx = scratch + base;
  int small_top = -1, large_top = -1;   // intrusive stacks via idx[]
  for (int k = K - 1; k >= 0; --k) {    // descending so pop order ascends
    const float pk = prob[base + k] * scale;
    prob[base + k] = pk;
    if (pk < 1.f) { idx[k] = small_top; small_top = k; }
    else          { idx[k] = large_top; large_top = k; }
  }
  while (small_top >= 0 && large_top >= 0) {
    const int sm = small_top; small_top = idx[sm];
    const int lg = large_top;
    alias[base + sm] = lg;
    const float rem = (prob[base + lg] + prob[base + sm]) - 1.f;
    prob[base + lg] = rem;
    large_top = idx[lg];
    if (rem < 1.f) { idx[lg] = small_top; small_top = lg; }
    else           { idx[lg] = large_top; large_top = lg; }
  }
  while (large_top >= 0) { const int lg = large_top; large_top = idx[lg];
                           prob[base + lg] = 1.f; alias[base + lg] = lg; }
  while (small_top >= 0) { const int sm = small_top; small_top = idx[sm];
                           prob[base + sm] = 1.f; alias[base + sm] = sm; }
}

__global__ __launch_bounds__(ALIAS_THREADS)
void lda_mh_kernel(int* __restrict__ doc_topic,        // [D][K] int32
                   const int* __restrict__ word_topic, // [rows][K] (stale)
                   const float* __restrict__ invden,   // [K]
                   const float* __restrict__ prob,     // alias prob
                   const int* __restrict__ alias,      // alias index
                   const int64_t* __restrict__ doc_offsets,
                   const int64_t* __restrict__ word_ids,
                   int* __restrict__ z,
                   float alpha, float beta,
                   int n_docs, int K, unsigned int seed) {
  extern __shared__ unsigned char nd8[];               // [THREADS][K] u8
  const int tid = threadIdx.x;
  const int doc = blockIdx.x * blockDim.x + tid;
  unsigned char* nd = nd8 + (size_t)tid * K;
  if (doc >= n_docs) return;
  for (int k = 0; k < K; ++k)
    nd[k] = (unsigned char)doc_topic[(int64_t)doc * K + k];
  const int64_t p0 = doc_offsets[doc], p1 = doc_offsets[doc + 1];
  const float Ld = (float)(p1 - p0);
  const float aK = alpha * (float)K;
  const float p_uniform = aK / (aK + Ld);
  for (int64_t p = p0; p < p1; ++p) {
    const int64_t w = word_ids[p];
    const int64_t wbase = w * K;
    int s = z[p];
    nd[s] -= 1;                                       // exclude the token
    const unsigned int c0 = (unsigned int)(p * 8);
    // ---- word proposal (alias): acceptance = doc factor only -------
    {
      const float u = rng_uniform(seed, c0 + 0) * (float)K;
      int bin = (int)u;
      if (bin >= K) bin = K - 1;
      const float frac = u - (float)bin;
      const int t1 = (frac < prob[wbase + bin]) ? bin : alias[wbase + bin];
      const float a1 = ((float)nd[t1] + alpha) / ((float)nd[s] + alpha);
      if (rng_uniform(seed, c0 + 1) < a1) s = t1;
    }
    // ---- doc proposal: q_d ∝ n~_dk + a (n~ includes current token) --
    {
      int t2;
      if (rng_uniform(seed, c0 + 2) < p_uniform) {
        t2 = (int)(rng_uniform(seed, c0 + 3) * (float)K);
        if (t2 >= K) t2 = K - 1;
      } else {
        int64_t j = p0 + (int64_t)(rng_uniform(seed, c0 + 4) * Ld);
        if (j >= p1) j = p1 - 1;
        t2 = (j == p) ? s : z[j];
      }
      const float nds = (float)nd[s], ndt = (float)nd[t2];
      // n~ (proposal counts) include the current assignment s
      const float qs = nds + 1.f + alpha, qt = ndt + (t2 == s ? 1.f : 0.f) + alpha;
      const float pis = (nds + alpha) *
          ((float)word_topic[wbase + s] + beta) * invden[s];
      const float pit = (ndt + alpha) *
          ((float)word_topic[wbase + t2] + beta) * invden[t2];
      const float a2 = (pit * qs) / (pis * qt);
      if (rng_uniform(seed, c0 + 5) < a2) s = t2;
    }
    nd[s] += 1;
    z[p] = s;
  }
  for (int k = 0; k < K; ++k)
    doc_topic[(int64_t)doc * K + k] = (int)nd[k];
}

}  // namespace

std::vector<torch::Tensor> lda_alias_build(torch::Tensor word_topic,
                                           torch::Tensor topic_sum,
                                           double beta, int64_t num_vocabs) {
  CHECK_IN(word_topic); CHECK_IN(topic_sum);
  TORCH_CHECK(word_topic.dtype() == torch::kInt32);
  const int rows = word_topic.size(0), K = word_topic.size(1);
  auto invden = 1.0 / (topic_sum.to(torch::kFloat32)
                       + (double)num_vocabs * beta);
  invden = invden.contiguous();
  auto prob = torch::empty({rows, K}, word_topic.options()
                                          .dtype(torch::kFloat32));
  auto alias = torch::empty({rows, K}, word_topic.options());
  auto qsum = torch::empty({rows}, prob.options());
  auto scratch = torch::empty({rows, K}, word_topic.options());
  if (rows > 0) {
    dim3 blk(256), grid((rows + 255) / 256);
    hipLaunchKernelGGL(alias_build_kernel, grid, blk, 0, current_stream(),
                       word_topic.data_ptr<int>(), invden.data_ptr<float>(),
                       (float)beta, prob.data_ptr<float>(),
                       alias.data_ptr<int>(), qsum.data_ptr<float>(),
                       scratch.data_ptr<int>(), rows, K);
  }
  return {prob, alias, qsum, invden};
}

torch::Tensor lda_mh(torch::Tensor doc_topic, torch::Tensor word_topic,
                     torch::Tensor invden, torch::Tensor prob,
                     torch::Tensor alias, torch::Tensor doc_offsets,
                     torch::Tensor word_ids, torch::Tensor assignments,
                     double alpha, double beta, int64_t seed) {
  CHECK_IN(doc_topic); CHECK_IN(word_topic); CHECK_IN(invden);
  CHECK_IN(prob); CHECK_IN(alias); CHECK_IN(doc_offsets);
  CHECK_IN(word_ids); CHECK_IN(assignments);
  const int D = doc_topic.size(0), K = doc_topic.size(1);
  if (D == 0) return assignments;
  dim3 blk(ALIAS_THREADS), grid((D + ALIAS_THREADS - 1) / ALIAS_THREADS);
  const size_t shmem = (size_t)ALIAS_THREADS * K;     // u8 rows
  TORCH_CHECK(shmem <= 160 * 1024, "K too large for u8 LDS rows");
  hipLaunchKernelGGL(lda_mh_kernel, grid, blk, shmem, current_stream(),
                     doc_topic.data_ptr<int>(), word_topic.data_ptr<int>(),
                     invden.data_ptr<float>(), prob.data_ptr<float>(),
                     alias.data_ptr<int>(), doc_offsets.data_ptr<int64_t>(),
                     word_ids.data_ptr<int64_t>(),
                     assignments.data_ptr<int>(),
                     (float)alpha, (float)beta, D, K,
                     (unsigned int)(seed & 0xffffffff));
  return assignments;
}
