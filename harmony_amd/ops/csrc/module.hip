// Python bindings for the CDNA4 kernel set (built in-tree as
// harmony_amd/ops/_hip_ops by setup_ops.py; PYTORCH_ROCM_ARCH=gfx950).

#include <torch/extension.h>

std::vector<torch::Tensor> mlr_softmax_grad(torch::Tensor logits,
                                            torch::Tensor labels);
std::vector<torch::Tensor> nmf_grad(torch::Tensor L, torch::Tensor R,
                                    torch::Tensor row_ptr,
                                    torch::Tensor col_idx, torch::Tensor vals,
                                    double lam);
std::vector<torch::Tensor> nmf_grad_twopass(
    torch::Tensor L, torch::Tensor R, torch::Tensor row_ptr,
    torch::Tensor col_idx, torch::Tensor vals, torch::Tensor perm,
    torch::Tensor seg_ptr, torch::Tensor row_sorted, double lam);
std::vector<torch::Tensor> mlr_fwd(torch::Tensor X, torch::Tensor W,
                                   torch::Tensor labels);
torch::Tensor mlr_grad(torch::Tensor P, torch::Tensor X);
std::vector<torch::Tensor> mlr_step_mfma(torch::Tensor X, torch::Tensor Wt,
                                         torch::Tensor labels,
                                         int64_t row_block, int64_t C,
                                         int64_t splitf, int64_t splitb);
void mlr_fwd_mfma_part(torch::Tensor X, torch::Tensor Wt, torch::Tensor P,
                       int64_t off, int64_t n, int64_t splitf);
void mlr_softmax_part(torch::Tensor P, torch::Tensor labels,
                      torch::Tensor loss, torch::Tensor correct,
                      int64_t off, int64_t n, int64_t C);
void mlr_grad_mfma_part(torch::Tensor P, torch::Tensor X,
                        torch::Tensor gradT, int64_t off, int64_t n,
                        int64_t splitb);
torch::Tensor lda_gibbs(torch::Tensor doc_topic, torch::Tensor word_topic,
                        torch::Tensor topic_sum, torch::Tensor doc_offsets,
                        torch::Tensor word_ids, torch::Tensor assignments,
                        double alpha, double beta, int64_t num_vocabs,
                        int64_t seed);
void lda_apply_pairs(torch::Tensor shard, torch::Tensor rows,
                     torch::Tensor old_t, torch::Tensor new_t);
void lda_apply_all(torch::Tensor shard, torch::Tensor word_rows,
                   torch::Tensor old_t, torch::Tensor new_t,
                   int64_t summary_row);
std::vector<torch::Tensor> lda_alias_build(torch::Tensor word_topic,
                                           torch::Tensor topic_sum,
                                           double beta, int64_t num_vocabs);
torch::Tensor lda_mh_wave(torch::Tensor doc_topic, torch::Tensor word_topic,
                          torch::Tensor invden, torch::Tensor prob,
                          torch::Tensor alias, torch::Tensor top_prob,
                          torch::Tensor top_alias, torch::Tensor qv,
                          torch::Tensor doc_offsets,
                          torch::Tensor word_ids, torch::Tensor assignments,
                          double alpha, double beta, int64_t seed);
torch::Tensor lda_mh(torch::Tensor doc_topic, torch::Tensor word_topic,
                     torch::Tensor invden, torch::Tensor prob,
                     torch::Tensor alias, torch::Tensor top_prob,
                     torch::Tensor top_alias, torch::Tensor qv,
                     torch::Tensor doc_offsets,
                     torch::Tensor word_ids, torch::Tensor assignments,
                     double alpha, double beta, int64_t seed);
void lasso_cd(torch::Tensor Xt, torch::Tensor r, torch::Tensor w,
              torch::Tensor col_sq, double lam_n);
torch::Tensor os_shard_alloc(int64_t rows, int64_t k, int64_t dtype_i32);
torch::Tensor os_ipc_handle(torch::Tensor shard);
int64_t os_ipc_open(torch::Tensor handle_bytes);
void os_ipc_close(int64_t ptr);
torch::Tensor os_gather(int64_t ptr, torch::Tensor idx, int64_t k,
                        int64_t dtype_i32);
void os_scatter_add(int64_t ptr, torch::Tensor idx, torch::Tensor delta);
int64_t os_cu_masked_stream(torch::Tensor mask_words);
void os_stream_destroy(int64_t stream);
int64_t os_ring_bytes(int64_t W, int64_t cap, int64_t vd);
void os_ring_reserve(int64_t base, int64_t W, int64_t cap, int64_t vd,
                     int64_t writer, int64_t n, torch::Tensor scratch);
void os_ring_read_head(int64_t base, int64_t W, int64_t cap, int64_t vd,
                       int64_t writer, torch::Tensor scratch);
void os_ring_push(int64_t base, int64_t W, int64_t cap, int64_t vd,
                  int64_t writer, torch::Tensor seq_base,
                  torch::Tensor keys, torch::Tensor deltas);
std::vector<torch::Tensor> os_ring_drain(int64_t base, int64_t W,
                                         int64_t cap, int64_t vd,
                                         int64_t max_per);
void gbt_hist(torch::Tensor bins, torch::Tensor resid, torch::Tensor node,
              torch::Tensor cnt, torch::Tensor sum);
void scatter_apply(torch::Tensor shard, torch::Tensor rows,
                   torch::Tensor deltas, int64_t mode, double step,
                   double maxval);
void dense_apply(torch::Tensor shard, torch::Tensor delta, int64_t mode,
                 double step, double maxval);
std::vector<torch::Tensor> parse_nmf_bytes(const std::string& buf);
std::vector<torch::Tensor> parse_libsvm_bytes(const std::string& buf,
                                              int64_t num_features);
std::vector<torch::Tensor> parse_lda_bytes(const std::string& buf);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("mlr_softmax_grad", &mlr_softmax_grad,
        "fused softmax + label-subtract + CE/accuracy (K4)");
  m.def("nmf_grad", &nmf_grad, "NMF sparse-batch gradient (K1+K2)");
  m.def("nmf_grad_twopass", &nmf_grad_twopass,
        "NMF gradient, segmented-reduce rgrad, no atomics (K1+K2)");
  m.def("mlr_fwd", &mlr_fwd,
        "fused MLR forward: X@W^T + softmax + grad + CE/acc (K4)");
  m.def("mlr_grad", &mlr_grad, "skinny-C gradient GEMM P^T @ X (K5)");
  m.def("mlr_step_mfma", &mlr_step_mfma,
        "fused MLR step on f32 MFMA: fwd+softmax+grad, optional L3 "
        "row-blocking (K4-MFMA)");
  m.def("mlr_fwd_mfma_part", &mlr_fwd_mfma_part, "K4-MFMA fwd piece");
  m.def("mlr_softmax_part", &mlr_softmax_part, "K4-MFMA softmax piece");
  m.def("mlr_grad_mfma_part", &mlr_grad_mfma_part, "K4-MFMA grad piece");
  m.def("lda_gibbs", &lda_gibbs, "LDA collapsed Gibbs sweep (K7)");
  m.def("lda_apply_pairs", &lda_apply_pairs,
        "apply TopicChanges +/-1 pairs to the word-topic shard (K9 sparse)");
  m.def("lda_apply_all", &lda_apply_all,
        "fused local token-delta + summary update (K9, single-owner path)");
  m.def("lda_alias_build", &lda_alias_build,
        "per-word Vose alias tables over the stale word factor (K7b)");
  m.def("lda_mh", &lda_mh,
        "Metropolis-Hastings alias LDA sweep, thread-per-doc (K7b)");
  m.def("lda_mh_wave", &lda_mh_wave,
        "wave-per-doc MH sweep (K7c, approximate within-doc parallelism)");
  m.def("gbt_hist", &gbt_hist, "GBT level histogram build (K10)");
  m.def("lasso_cd", &lasso_cd, "Lasso persistent CD sweep (K11)");
  m.def("os_shard_alloc", &os_shard_alloc, "one-sided shard (hipMalloc)");
  m.def("os_ipc_handle", &os_ipc_handle, "export shard via hipIpc");
  m.def("os_ipc_open", &os_ipc_open, "map a peer shard (xGMI)");
  m.def("os_ipc_close", &os_ipc_close, "unmap a peer shard");
  m.def("os_gather", &os_gather, "gather rows from a mapped shard (K12)");
  m.def("os_scatter_add", &os_scatter_add,
        "atomic scatter-add into a mapped shard (K12)");
  m.def("os_cu_masked_stream", &os_cu_masked_stream,
        "create a HIP stream pinned to a CU-mask (multi-tenant partitioning)");
  m.def("os_stream_destroy", &os_stream_destroy, "destroy a raw HIP stream");
  m.def("os_ring_bytes", &os_ring_bytes, "ring buffer size (K12b)");
  m.def("os_ring_reserve", &os_ring_reserve, "reserve seq range in a peer ring");
  m.def("os_ring_read_head", &os_ring_read_head, "read peer ring head");
  m.def("os_ring_push", &os_ring_push, "enqueue items into a peer ring");
  m.def("os_ring_drain", &os_ring_drain, "owner-side in-order ring drain");
  m.def("scatter_apply", &scatter_apply, "owner-side sparse update (K9)");
  m.def("dense_apply", &dense_apply, "owner-side dense update (K3)");
  m.def("parse_nmf_bytes", &parse_nmf_bytes, "native NMF text parser");
  m.def("parse_libsvm_bytes", &parse_libsvm_bytes,
        "native libsvm-style parser (mlr/gbt/lasso)");
  m.def("parse_lda_bytes", &parse_lda_bytes, "native LDA doc parser");
}
