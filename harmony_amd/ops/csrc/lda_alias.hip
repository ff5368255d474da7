// K7b: Metropolis-Hastings alias LDA sampler (LightLDA-style), opt-in.
//
// The exact dense sampler (lda.hip) costs O(K) per token. With the
// word-topic snapshot batch-stale (which it already is — see mlapps/lda.py),
// a 2-proposal MH step per token has the SAME stationary distribution at
// O(1) per token:
//   word proposal  t1 ~ q_w(k) ∝ (n_wk + b)/(n_k + Vb)  via a per-word
//                  TWO-LEVEL alias table built once per pull; because q_w
//                  equals the word factor of the (stale) posterior exactly,
//                  the acceptance ratio collapses to the doc factor
//                  (n_d,t1 + a)/(n_d,s + a).
//   doc proposal   t2 ~ q_d(k) ∝ n~_dk + a  (n~ includes the current token)
//                  drawn in O(1): with prob aK/(aK+L_d) a uniform topic,
//                  else the topic of a uniformly chosen token of the doc;
//                  acceptance carries the full ratio with the doc-proposal
//                  correction.
//
// Two-level alias (exact): the row's K probabilities split into 64 lane
// segments; a 64-entry TOP alias selects the segment (mass-proportional),
// a per-segment alias selects the entry. Sampling is exact:
// P(k) = mass(g)/sum * p(k)/mass(g) = p(k)/sum. Construction is the win:
// all 64 segment tables build CONCURRENTLY (one lane each, K/64 serial
// steps) and only the 64-entry top table is a serial chain — the flat
// one-wave Vose was a K-long dependent-LDS chain (measured 1.03 ms per
// 46k x 256 build; thread-per-row before that: 2.26 ms).
//
// Sampler shape: ONE THREAD PER DOCUMENT (per-token work is scalar), the
// doc-topic counts as a u8 LDS row (counts <= doc length <= 255), tokens
// serial per doc, thousands of docs in flight. RNG: counter hash
// (hip_common.h) at ctr = token*8 + draw — mirrored by the torch reference
// for sample-exact CPU/GPU tests.

#include "hip_common.h"
#include <cstdlib>
#include <algorithm>

namespace {

constexpr int ALIAS_THREADS = 64;   // same-box interleaved A/B: 64 beats 128
                                    // (isolated 0.59 vs 0.655 ms; 3-job bench
                                    // 1.59/1.62 vs 1.63/1.66 ms/step) — the
                                    // earlier '64 = 1.74 ms' was cross-box noise
constexpr int BUILD_WAVES = 4;

// In-LDS serial Vose over n entries at pr[0..n), links lk, output al.
// Deterministic order (descending init so pops ascend) — the torch
// reference replicates it exactly.
__device__ void vose_serial(float* pr, int* al, int* lk, int n) {
  int small_top = -1, large_top = -1;
  for (int k = n - 1; k >= 0; --k) {
    if (pr[k] < 1.f) { lk[k] = small_top; small_top = k; }
    else             { lk[k] = large_top; large_top = k; }
  }
  while (small_top >= 0 && large_top >= 0) {
    const int sm = small_top; small_top = lk[sm];
    const int lg = large_top;
    al[sm] = lg;
    const float rem = (pr[lg] + pr[sm]) - 1.f;
    pr[lg] = rem;
    large_top = lk[lg];
    if (rem < 1.f) { lk[lg] = small_top; small_top = lg; }
    else           { lk[lg] = large_top; large_top = lg; }
  }
  while (large_top >= 0) { const int lg = large_top; large_top = lk[lg];
                           pr[lg] = 1.f; al[lg] = lg; }
  while (small_top >= 0) { const int sm = small_top; small_top = lk[sm];
                           pr[sm] = 1.f; al[sm] = sm; }
}

__global__ void alias_build_kernel(const int* __restrict__ word_topic,
                                   const float* __restrict__ invden,
                                   float beta,
                                   float* __restrict__ prob,     // [rows][K]
                                   int* __restrict__ alias,      // [rows][K]
                                   float* __restrict__ top_prob, // [rows][64]
                                   int* __restrict__ top_alias,  // [rows][64]
                                   float* __restrict__ qv,       // [rows][K]
                                   float* __restrict__ qsum,     // [rows]
                                   int rows, int K) {
  extern __shared__ float smem_f[];
  // per wave: pr[K] f32 | al[K] i32 | lk[K] i32 | tp[64] f32 | ta[64] i32 | tl[64] i32
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const size_t per_wave = 3 * (size_t)K + 3 * WAVE;
  float* pr = smem_f + (size_t)wave * per_wave;
  int* al = (int*)(pr + K);
  int* lk = al + K;
  float* tp = (float*)(lk + K);
  int* ta = (int*)(tp + WAVE);
  int* tl = ta + WAVE;
  const int S = K / WAVE;                  // entries per lane segment
  const int waves_wg = blockDim.x / WAVE;
  for (int row = blockIdx.x * waves_wg + wave; row < rows;
       row += gridDim.x * waves_wg) {
    const int64_t base = (int64_t)row * K;
    // lane's contiguous segment [lane*S, (lane+1)*S)
    float seg_mass = 0.f;
    for (int i = 0; i < S; ++i) {
      const int k = lane * S + i;
      const float p = ((float)word_topic[base + k] + beta) * invden[k];
      pr[k] = p;
      seg_mass += p;
    }
    const float total = wave_reduce_sum(seg_mass);
    // proposal density actually encoded by the tables (for stale-table
    // acceptance correction): qv = p / total
    const float inv_total = 1.f / total;
    for (int i = 0; i < S; ++i)
      qv[base + lane * S + i] = pr[lane * S + i] * inv_total;
    // per-segment alias: normalize within segment to mean 1 (concurrent
    // across all 64 lanes; each runs a tiny S-entry serial Vose)
    const float sscale = (seg_mass > 0.f) ? (float)S / seg_mass : 0.f;
    for (int i = 0; i < S; ++i) pr[lane * S + i] *= sscale;
    vose_serial(pr + lane * S, al + lane * S, lk + lane * S, S);
    // top alias over segment masses (normalized to mean 1 across 64)
    tp[lane] = seg_mass * (float)WAVE / total;
    if (lane == 0) {
      vose_serial(tp, ta, tl, WAVE);
      qsum[row] = total;
    }
    // coalesced write-back
    const int nchunk = K / WAVE;
    for (int c = 0; c < nchunk; ++c) {
      const int k = c * WAVE + lane;
      prob[base + k] = pr[k];
      alias[base + k] = al[k];
    }
    top_prob[(int64_t)row * WAVE + lane] = tp[lane];
    top_alias[(int64_t)row * WAVE + lane] = ta[lane];
  }
}

template <bool PF>
__global__ __launch_bounds__(256)
void lda_mh_kernel(int* __restrict__ doc_topic,        // [D][K] int32
                   const int* __restrict__ word_topic, // [rows][K] (fresh)
                   const float* __restrict__ invden,   // [K]
                   const float* __restrict__ prob,     // entry alias prob
                   const int* __restrict__ alias,      // entry alias index
                   const float* __restrict__ top_prob, // [rows][64]
                   const int* __restrict__ top_alias,  // [rows][64]
                   const float* __restrict__ qv,       // proposal density
                   const int64_t* __restrict__ doc_offsets,
                   const int64_t* __restrict__ word_ids,
                   int* __restrict__ z,
                   float alpha, float beta,
                   int n_docs, int K, unsigned int seed) {
  extern __shared__ unsigned char nd8[];               // [THREADS][K] u8
  const int tid = threadIdx.x;
  const int doc = blockIdx.x * blockDim.x + tid;
  unsigned char* nd = nd8 + (size_t)tid * K;
  // coalesced block-wide row staging: the WG's docs are contiguous, so the
  // [docs x K] slice loads as one flat stream (thread-private loops would
  // put adjacent lanes 1 KB apart — every 4 B load its own cacheline)
  const int64_t gbase = (int64_t)blockIdx.x * blockDim.x * K;
  const int ndocs_wg =
      min((int)blockDim.x, n_docs - (int)(blockIdx.x * blockDim.x));
  for (int i = tid; i < ndocs_wg * K; i += blockDim.x)
    nd8[i] = (unsigned char)doc_topic[gbase + i];
  __syncthreads();
  // inactive tail threads skip the token loop but MUST reach the final
  // barrier (an early return would strand the block-wide __syncthreads)
  const bool active = doc < n_docs;
  const int64_t p0 = active ? doc_offsets[doc] : 0;
  const int64_t p1 = active ? doc_offsets[doc + 1] : 0;
  const float Ld = (float)(p1 - p0);
  const float aK = alpha * (float)K;
  const float p_uniform = aK / (aK + Ld);
  const int S = K / WAVE;
  // software-prefetched word-proposal top row: token p+1's first-level
  // alias lookup depends only on (seed, p+1, word_ids[p+1]) — issue its
  // two loads before token p's dependent acceptance chain
  float u1_pf = 0.f; float tpv_pf = 0.f; int tav_pf = 0; int gb_pf = 0;
  if (PF && p0 < p1) {
    const float u1 = rng_uniform(seed, (unsigned int)(p0 * 8)) * (float)WAVE;
    int gb = (int)u1;
    if (gb >= WAVE) gb = WAVE - 1;
    const int64_t tbase = word_ids[p0] * WAVE;
    u1_pf = u1; gb_pf = gb;
    tpv_pf = top_prob[tbase + gb]; tav_pf = top_alias[tbase + gb];
  }
  for (int64_t p = p0; p < p1; ++p) {
    const int64_t w = word_ids[p];
    const int64_t wbase = w * K;
    const float u1_cur = u1_pf; const float tpv = tpv_pf;
    const int tav = tav_pf; const int gb_cur = gb_pf;
    if (PF && p + 1 < p1) {
      const float u1n =
          rng_uniform(seed, (unsigned int)((p + 1) * 8)) * (float)WAVE;
      int gbn = (int)u1n;
      if (gbn >= WAVE) gbn = WAVE - 1;
      const int64_t tbn = word_ids[p + 1] * WAVE;
      u1_pf = u1n; gb_pf = gbn;
      tpv_pf = top_prob[tbn + gbn]; tav_pf = top_alias[tbn + gbn];
    }
    int s = z[p];
    nd[s] -= 1;                                       // exclude the token
    const unsigned int c0 = (unsigned int)(p * 8);
    // ---- word proposal (two-level alias): acceptance = doc factor ---
    {
      int g;
      if (PF) {
        g = (u1_cur - (float)gb_cur < tpv) ? gb_cur : tav;
      } else {
        const float u1 = rng_uniform(seed, c0 + 0) * (float)WAVE;
        int gb = (int)u1;
        if (gb >= WAVE) gb = WAVE - 1;
        const int64_t tb = w * WAVE;
        g = (u1 - (float)gb < top_prob[tb + gb]) ? gb : top_alias[tb + gb];
      }
      const float u2 = rng_uniform(seed, c0 + 6) * (float)S;
      int eb = (int)u2;
      if (eb >= S) eb = S - 1;
      const int64_t ebase = wbase + g * S;
      const int t1 = g * S + ((u2 - (float)eb < prob[ebase + eb])
                                  ? eb : alias[ebase + eb]);
      // acceptance: pi uses the CURRENT snapshot, q the (possibly older)
      // table's encoded density — with a fresh table the word factors
      // cancel mathematically; the explicit form stays correct when the
      // tables are reused across several pulls (alias_refresh)
      const float pi_s = ((float)nd[s] + alpha) *
          ((float)word_topic[wbase + s] + beta) * invden[s];
      const float pi_t = ((float)nd[t1] + alpha) *
          ((float)word_topic[wbase + t1] + beta) * invden[t1];
      const float a1 = (pi_t * qv[wbase + s]) / (pi_s * qv[wbase + t1]);
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
      const float qs = nds + 1.f + alpha;
      const float qt = ndt + (t2 == s ? 1.f : 0.f) + alpha;
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
  __syncthreads();
  for (int i = tid; i < ndocs_wg * K; i += blockDim.x)
    doc_topic[gbase + i] = (int)nd8[i];
}


// K7c: wave-per-doc MH sweep. The serial kernel (above) is exact vs the
// CPU oracle but parallelism-starved: one thread per doc = 16k threads on
// a 512k-thread chip, and each token's acceptance chain is a dependent
// global-load sequence (~4.6 us/token measured) with too few waves to
// hide it. Here a FULL WAVE cooperates on one doc: lanes take tokens
// round-robin and share the doc-topic row in LDS via atomics — within-doc
// token updates become approximately parallel (standard GPU-LDA
// relaxation; acceptance reads slightly stale counts). Convergence is
// validated against the serial sampler by scripts/lda_convergence.py.
__global__ __launch_bounds__(256)
void lda_mh_wave_kernel(int* __restrict__ doc_topic,
                        const int* __restrict__ word_topic,
                        const float* __restrict__ invden,
                        const float* __restrict__ prob,
                        const int* __restrict__ alias,
                        const float* __restrict__ top_prob,
                        const int* __restrict__ top_alias,
                        const float* __restrict__ qv,
                        const int64_t* __restrict__ doc_offsets,
                        const int64_t* __restrict__ word_ids,
                        int* __restrict__ z,
                        float alpha, float beta,
                        int n_docs, int K, unsigned int seed) {
  extern __shared__ int ndw[];                 // [waves_wg][K] int32
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int waves_wg = blockDim.x >> 6;
  const int doc = blockIdx.x * waves_wg + wave;
  int* nd = ndw + (size_t)wave * K;
  const int64_t gbase = (int64_t)blockIdx.x * waves_wg * K;
  const int ndocs_wg = min(waves_wg, n_docs - blockIdx.x * waves_wg);
  for (int i = tid; i < ndocs_wg * K; i += blockDim.x)
    ndw[i] = doc_topic[gbase + i];
  __syncthreads();
  const bool active = doc < n_docs;
  const int64_t p0 = active ? doc_offsets[doc] : 0;
  const int64_t p1 = active ? doc_offsets[doc + 1] : 0;
  const float Ld = (float)(p1 - p0);
  const float aK = alpha * (float)K;
  const float p_uniform = aK / (aK + Ld);
  const int S = K / WAVE;
  for (int64_t p = p0 + lane; p < p1; p += WAVE) {
    const int64_t w = word_ids[p];
    const int64_t wbase = w * K;
    int s = z[p];
    // exclude this token (value after MY decrement; other lanes race —
    // the acceptance below reads approximately-current counts)
    const int nds0 = atomicAdd(&nd[s], -1) - 1;
    const unsigned int c0 = (unsigned int)(p * 8);
    // ---- word proposal (two-level alias) ----
    {
      const float u1 = rng_uniform(seed, c0 + 0) * (float)WAVE;
      int gb = (int)u1;
      if (gb >= WAVE) gb = WAVE - 1;
      const int64_t tb = w * WAVE;
      const int g =
          (u1 - (float)gb < top_prob[tb + gb]) ? gb : top_alias[tb + gb];
      const float u2 = rng_uniform(seed, c0 + 6) * (float)S;
      int eb = (int)u2;
      if (eb >= S) eb = S - 1;
      const int64_t ebase = wbase + g * S;
      const int t1 = g * S + ((u2 - (float)eb < prob[ebase + eb])
                                  ? eb : alias[ebase + eb]);
      const float pi_s = ((float)nds0 + alpha) *
          ((float)word_topic[wbase + s] + beta) * invden[s];
      const float pi_t = ((float)nd[t1] + alpha) *
          ((float)word_topic[wbase + t1] + beta) * invden[t1];
      const float a1 = (pi_t * qv[wbase + s]) / (pi_s * qv[wbase + t1]);
      if (rng_uniform(seed, c0 + 1) < a1) s = t1;
    }
    // ---- doc proposal ----
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
      const float qs = nds + 1.f + alpha;
      const float qt = ndt + (t2 == s ? 1.f : 0.f) + alpha;
      const float pis = (nds + alpha) *
          ((float)word_topic[wbase + s] + beta) * invden[s];
      const float pit = (ndt + alpha) *
          ((float)word_topic[wbase + t2] + beta) * invden[t2];
      const float a2 = (pit * qs) / (pis * qt);
      if (rng_uniform(seed, c0 + 5) < a2) s = t2;
    }
    atomicAdd(&nd[s], 1);
    z[p] = s;
  }
  __syncthreads();
  for (int i = tid; i < ndocs_wg * K; i += blockDim.x)
    doc_topic[gbase + i] = ndw[i];
}

}  // namespace

std::vector<torch::Tensor> lda_alias_build(torch::Tensor word_topic,
                                           torch::Tensor topic_sum,
                                           double beta, int64_t num_vocabs) {
  CHECK_IN(word_topic); CHECK_IN(topic_sum);
  TORCH_CHECK(word_topic.dtype() == torch::kInt32);
  const int rows = word_topic.size(0), K = word_topic.size(1);
  TORCH_CHECK(K % WAVE == 0, "alias sampler requires K % 64 == 0");
  auto invden = 1.0 / (topic_sum.to(torch::kFloat32)
                       + (double)num_vocabs * beta);
  invden = invden.contiguous();
  auto prob = torch::empty({rows, K},
                           word_topic.options().dtype(torch::kFloat32));
  auto alias = torch::empty({rows, K}, word_topic.options());
  auto top_prob = torch::empty({rows, WAVE},
                               word_topic.options().dtype(torch::kFloat32));
  auto top_alias = torch::empty({rows, WAVE}, word_topic.options());
  auto qv = torch::empty({rows, K},
                         word_topic.options().dtype(torch::kFloat32));
  auto qsum = torch::empty({rows}, prob.options());
  if (rows > 0) {
    int waves = BUILD_WAVES;       // HARMONY_LDA_BUILD_WAVES: occupancy A/B
    const char* bw = getenv("HARMONY_LDA_BUILD_WAVES");
    if (bw) { int w = atoi(bw); if (w >= 1 && w <= 8) waves = w; }
    dim3 blk(WAVE * waves);
    dim3 grid(std::min((rows + waves - 1) / waves, 8192));
    const size_t shmem = (size_t)waves * (3 * K + 3 * WAVE) * 4;
    hipLaunchKernelGGL(alias_build_kernel, grid, blk, shmem, current_stream(),
                       word_topic.data_ptr<int>(), invden.data_ptr<float>(),
                       (float)beta, prob.data_ptr<float>(),
                       alias.data_ptr<int>(), top_prob.data_ptr<float>(),
                       top_alias.data_ptr<int>(), qv.data_ptr<float>(),
                       qsum.data_ptr<float>(), rows, K);
  }
  return {prob, alias, top_prob, top_alias, qv, qsum, invden};
}

torch::Tensor lda_mh(torch::Tensor doc_topic, torch::Tensor word_topic,
                     torch::Tensor invden, torch::Tensor prob,
                     torch::Tensor alias, torch::Tensor top_prob,
                     torch::Tensor top_alias, torch::Tensor qv,
                     torch::Tensor doc_offsets,
                     torch::Tensor word_ids, torch::Tensor assignments,
                     double alpha, double beta, int64_t seed) {
  CHECK_IN(doc_topic); CHECK_IN(word_topic); CHECK_IN(invden);
  CHECK_IN(prob); CHECK_IN(alias); CHECK_IN(top_prob); CHECK_IN(top_alias);
  CHECK_IN(qv); CHECK_IN(doc_offsets); CHECK_IN(word_ids);
  CHECK_IN(assignments);
  const int D = doc_topic.size(0), K = doc_topic.size(1);
  if (D == 0) return assignments;
  int threads = ALIAS_THREADS;    // HARMONY_LDA_MH_THREADS: fill/LDS A/B
  const char* te = getenv("HARMONY_LDA_MH_THREADS");
  if (te) { int t = atoi(te); if (t==64 || t==128 || t==256) threads = t; }
  dim3 blk(threads), grid((D + threads - 1) / threads);
  const size_t shmem = (size_t)threads * K;           // u8 rows
  TORCH_CHECK(shmem <= 160 * 1024, "K too large for u8 LDS rows");
  const char* pe = getenv("HARMONY_LDA_MH_PREFETCH");
  const bool pf = !(pe && atoi(pe) == 0);   // default: prefetch on
  auto kern = pf ? lda_mh_kernel<true> : lda_mh_kernel<false>;
  hipLaunchKernelGGL(kern, grid, blk, shmem, current_stream(),
                     doc_topic.data_ptr<int>(), word_topic.data_ptr<int>(),
                     invden.data_ptr<float>(), prob.data_ptr<float>(),
                     alias.data_ptr<int>(), top_prob.data_ptr<float>(),
                     top_alias.data_ptr<int>(), qv.data_ptr<float>(),
                     doc_offsets.data_ptr<int64_t>(),
                     word_ids.data_ptr<int64_t>(),
                     assignments.data_ptr<int>(),
                     (float)alpha, (float)beta, D, K,
                     (unsigned int)(seed & 0xffffffff));
  return assignments;
}

torch::Tensor lda_mh_wave(torch::Tensor doc_topic, torch::Tensor word_topic,
                          torch::Tensor invden, torch::Tensor prob,
                          torch::Tensor alias, torch::Tensor top_prob,
                          torch::Tensor top_alias, torch::Tensor qv,
                          torch::Tensor doc_offsets,
                          torch::Tensor word_ids,
                          torch::Tensor assignments,
                          double alpha, double beta, int64_t seed) {
  CHECK_IN(doc_topic); CHECK_IN(word_topic); CHECK_IN(invden);
  CHECK_IN(prob); CHECK_IN(alias); CHECK_IN(top_prob); CHECK_IN(top_alias);
  CHECK_IN(qv); CHECK_IN(doc_offsets); CHECK_IN(word_ids);
  CHECK_IN(assignments);
  const int D = doc_topic.size(0), K = doc_topic.size(1);
  if (D == 0) return assignments;
  int waves_wg = 4;                    // 256 threads, 4 docs per block
  const char* we = getenv("HARMONY_LDA_WAVE_WG");   // occupancy A/B
  if (we) { int v = atoi(we); if (v >= 1 && v <= 4) waves_wg = v; }  // <=4: 256-thread launch bound
  dim3 blk(WAVE * waves_wg), grid((D + waves_wg - 1) / waves_wg);
  const size_t shmem = (size_t)waves_wg * K * 4;
  hipLaunchKernelGGL(lda_mh_wave_kernel, grid, blk, shmem, current_stream(),
                     doc_topic.data_ptr<int>(), word_topic.data_ptr<int>(),
                     invden.data_ptr<float>(), prob.data_ptr<float>(),
                     alias.data_ptr<int>(), top_prob.data_ptr<float>(),
                     top_alias.data_ptr<int>(), qv.data_ptr<float>(),
                     doc_offsets.data_ptr<int64_t>(),
                     word_ids.data_ptr<int64_t>(),
                     assignments.data_ptr<int>(),
                     (float)alpha, (float)beta, D, K,
                     (unsigned int)(seed & 0xffffffff));
  return assignments;
}
