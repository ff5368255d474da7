"""Compute ops: CDNA4 HIP kernels with fp32 torch reference implementations.

Dispatch policy:
  * CPU tensors -> torch reference implementation (also the numerics oracle).
  * CUDA (ROCm) tensors -> the in-tree HIP extension `_hip_ops`. If the
    extension is missing on a GPU machine this raises immediately — a silent
    eager fallback would invalidate every GPU benchmark (and the round-end
    "native code not loaded" check).
Set HARMONY_FORCE_TORCH_OPS=1 to force the torch path on GPU (debug only).

Kernel inventory (reference hot loops):
  K1  nmf_grad            NMFTrainer.java:328-367
  K3  (fused in push)     NMFETModelUpdateFunction.java:48-52
  K4  mlr softmax+grad    MLRTrainer.java:374-398,475-489
  K7  lda_gibbs           SparseLDASampler.java:141-274
  K9  scatter-apply       LDAETModelUpdateFunction.java:43-64
  K12 loss reductions     NMFTrainer.java:414-456 etc.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

_hip = None
_hip_err: Optional[str] = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    try:
        from harmony_amd.ops import _hip_ops  # built in-tree by setup_ops.py

        _hip = _hip_ops
    except ImportError:
        try:
            import importlib.util
            from pathlib import Path

            cands = list(Path(__file__).parent.glob("_hip_ops*.so"))
            if cands:
                spec = importlib.util.spec_from_file_location("_hip_ops", cands[0])
                mod = importlib.util.module_from_spec(spec)
                spec.loader.exec_module(mod)
                _hip = mod
            else:
                _hip_err = "harmony_amd/ops/_hip_ops*.so not built " \
                           "(run: python setup_ops.py build)"
        except Exception as e:  # noqa: BLE001
            _hip_err = str(e)
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _use_hip(t: torch.Tensor) -> bool:
    if t.device.type != "cuda":
        return False
    if os.environ.get("HARMONY_FORCE_TORCH_OPS") == "1":
        return False
    if _load_hip() is None:
        raise RuntimeError(
            f"GPU tensor but HIP extension not available: {_hip_err}. "
            "Refusing silent eager fallback on a GPU machine.")
    return True


# ---------------------------------------------------------------------------
# K4: MLR fused softmax + label-subtract + CE/accuracy
# ---------------------------------------------------------------------------

def mlr_forward(X: torch.Tensor, W: torch.Tensor, labels: torch.Tensor
                ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """MLR forward: logits = X @ W^T (rocBLAS), then the fused
    softmax + label-subtract + CE/accuracy kernel (K4).
    Returns (p - onehot [B,C], loss_sum, n_correct)."""
    # Measured (scripts/mlr_ab.py, isolated A/B on MI355X): rocBLAS X@W^T is
    # near the HBM floor (0.204 ms at 16k x 16k x 10) — the custom fused
    # kernel (0.685 ms) only looked competitive against concurrency-inflated
    # profile times. GEMM + the small fused softmax kernel is the fast path;
    # mlr_fwd stays available for C<=16 regression testing.
    return softmax_grad_ce(X @ W.t(), labels)


def mlr_grad_gemm(P: torch.Tensor, X: torch.Tensor) -> torch.Tensor:
    """grad = P^T @ X. rocBLAS measured 0.212 ms vs the custom B-tile kernel's
    0.520 ms (scripts/mlr_ab.py) — Tensile's split-K wins this shape."""
    return P.t() @ X


def mlr_step_ok(X: torch.Tensor, C: int) -> bool:
    """Shapes the fused MFMA step kernel supports (see mlr_mfma.hip)."""
    B, F = X.shape
    return (C <= 16 and F % 64 == 0 and B % 64 == 0
            and X.dtype == torch.float32)


def _pow2_div(limit: int, n: int) -> int:
    """Largest power of two <= limit that divides n."""
    p = 1
    while p * 2 <= limit and n % (p * 2) == 0:
        p *= 2
    return p


def mlr_step_mfma(X: torch.Tensor, W: torch.Tensor, labels: torch.Tensor,
                  row_block: int = 0, splitf: Optional[int] = None,
                  splitb: Optional[int] = None,
                  Wt_buf: Optional[torch.Tensor] = None
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """One fused MLR step on the gfx950 matrix cores (K4-MFMA,
    ops/csrc/mlr_mfma.hip): logits = X @ W^T via v_mfma_f32_16x16x4_f32,
    in-kernel softmax - onehot, grad = P^T @ X via MFMA with split-B
    atomics. With row_block > 0 the fwd/grad pair runs per row block so the
    grad pass re-reads X from the Infinity Cache (the step is X-bandwidth
    bound). Returns (grad [C,F], loss_sum, n_correct) — exact-f32 MFMA
    numerics (a k-ordered fmaf chain).

    Replaces reference MLRTrainer.java:374-398 + :475-489 + the rocBLAS
    GEMM pair (measured A/B: scripts/mlr_mfma_ab.py)."""
    C, F = W.shape
    if not _use_hip(X):                # CPU oracle path
        p, loss, correct = mlr_forward(X, W, labels)
        return mlr_grad_gemm(p, X), loss, correct
    B = X.shape[0]
    rows = row_block if row_block > 0 else B
    if splitf is None:
        # target ~512 workgroups per launch (2 blocks/CU on 256 CUs) —
        # the sweep's optimum on the bench shape (scripts/mlr_mfma_ab.py)
        splitf = _pow2_div(max(1, 512 // (rows // 64)), F // 64)
    if splitb is None:
        splitb = _pow2_div(max(1, 512 // (F // 64)), rows // 64)
    if Wt_buf is None:
        Wt_buf = torch.zeros((F, 16), dtype=X.dtype, device=X.device)
    Wt_buf[:, :C] = W.t()
    gradT, loss, correct = _hip.mlr_step_mfma(
        X.contiguous(), Wt_buf, labels.contiguous(), row_block, C,
        splitf, splitb)
    return gradT[:C], loss[0], correct[0].to(torch.int64)


_mlr_aux_streams = {}


def mlr_step_mfma_pipelined(X: torch.Tensor, W: torch.Tensor,
                            labels: torch.Tensor, row_block: int = 2048,
                            splitf: Optional[int] = None,
                            splitb: Optional[int] = None,
                            Wt_buf: Optional[torch.Tensor] = None
                            ) -> Tuple[torch.Tensor, torch.Tensor,
                                       torch.Tensor]:
    """Row-blocked K4-MFMA step with a two-stream software pipeline:
    fwd+softmax of block i+1 (stream A) overlaps grad of block i (stream
    B). The grad pass then re-reads its X block from the 256 MiB Infinity
    Cache while fwd streams the next block from HBM — the step's HBM
    traffic drops from 2 passes over X toward 1 (the step is X-bandwidth
    bound). Falls back to the single-stream path when shapes don't block
    evenly."""
    C, F = W.shape
    if not _use_hip(X):
        return mlr_step_mfma(X, W, labels)
    B = X.shape[0]
    if row_block <= 0 or B % row_block or row_block % 64 or B == row_block:
        return mlr_step_mfma(X, W, labels, row_block=0, splitf=splitf,
                             splitb=splitb, Wt_buf=Wt_buf)
    if splitf is None:
        splitf = _pow2_div(max(1, 512 // (row_block // 64)), F // 64)
    if splitb is None:
        splitb = _pow2_div(max(2, 1024 // (F // 64)), row_block // 64)
    if Wt_buf is None:
        Wt_buf = torch.zeros((F, 16), dtype=X.dtype, device=X.device)
    Wt_buf[:, :C] = W.t()
    dev = X.device
    key = dev.index
    aux = _mlr_aux_streams.get(key)
    if aux is None:
        aux = (torch.cuda.Stream(device=dev), torch.cuda.Stream(device=dev))
        _mlr_aux_streams[key] = aux
    sA, sB = aux
    X = X.contiguous()
    labels = labels.contiguous()
    P = torch.zeros((B, 16), dtype=X.dtype, device=dev)
    gradT = torch.zeros((16, F), dtype=X.dtype, device=dev)
    loss = torch.zeros((1,), dtype=X.dtype, device=dev)
    correct = torch.zeros((1,), dtype=torch.int32, device=dev)
    cur = torch.cuda.current_stream(dev)
    start = torch.cuda.Event()
    start.record(cur)
    sA.wait_event(start)
    sB.wait_event(start)
    for off in range(0, B, row_block):
        ev = torch.cuda.Event()
        with torch.cuda.stream(sA):
            _hip.mlr_fwd_mfma_part(X, Wt_buf, P, off, row_block, splitf)
            _hip.mlr_softmax_part(P, labels, loss, correct, off, row_block,
                                  C)
            ev.record(sA)
        sB.wait_event(ev)
        with torch.cuda.stream(sB):
            _hip.mlr_grad_mfma_part(P, X, gradT, off, row_block, splitb)
    endA, endB = torch.cuda.Event(), torch.cuda.Event()
    endA.record(sA)
    endB.record(sB)
    cur.wait_event(endA)
    cur.wait_event(endB)
    for t in (X, Wt_buf, P, gradT, loss, correct, labels):
        t.record_stream(sA)
        t.record_stream(sB)
    return gradT[:C], loss[0], correct[0].to(torch.int64)


def softmax_grad_ce(logits: torch.Tensor, labels: torch.Tensor
                    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Row softmax with log-sum-exp guard; returns (p - onehot(label),
    sum of CE loss, #correct). Fuses reference MLRTrainer predict/softmax
    (:475-489) + label subtract (:374-398) + CE/accuracy metrics."""
    if _use_hip(logits):
        return _hip.mlr_softmax_grad(logits.contiguous(), labels.contiguous())
    z = logits.float()
    m = z.max(dim=1, keepdim=True).values
    e = torch.exp(z - m)
    s = e.sum(dim=1, keepdim=True)
    p = e / s
    lse = m.squeeze(1) + torch.log(s.squeeze(1))
    loss = (lse - z.gather(1, labels.view(-1, 1)).squeeze(1)).sum()
    correct = (z.argmax(dim=1) == labels).sum()
    grad = p
    grad.scatter_add_(1, labels.view(-1, 1),
                      torch.full((z.shape[0], 1), -1.0, device=z.device))
    return grad.to(logits.dtype), loss, correct


# ---------------------------------------------------------------------------
# K1: NMF gradient over a sparse batch (CSR by row)
# ---------------------------------------------------------------------------

def nmf_grad(L: torch.Tensor, R: torch.Tensor, row_ptr: torch.Tensor,
             col_idx: torch.Tensor, vals: torch.Tensor, lam: float,
             col_sorted=None) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """For each nonzero (i, j, v): e = L_i . R_j - v;
    lgrad_i += 2 e R_j, rgrad_j += 2 e L_i; plus L2 terms 2*lam*{L_i,R_j}
    per nonzero (reference NMFTrainer.updateGradient:328-367).
    L: [n_rows, k], CSR over rows: row_ptr [n_rows+1], col_idx/vals [nnz]
    (col_idx indexes R's rows). Returns (lgrad, rgrad, sq_err_sum).

    col_sorted: optional static precompute (perm, seg_ptr, row_sorted) — the
    column-sorted view of the nonzeros (seg_ptr covers ALL R rows) — enables
    the atomic-free two-pass kernel on GPU."""
    if _use_hip(L):
        if col_sorted is not None:
            perm, seg_ptr, row_sorted = col_sorted
            return tuple(_hip.nmf_grad_twopass(
                L.contiguous(), R.contiguous(), row_ptr.contiguous(),
                col_idx.contiguous(), vals.contiguous(), perm.contiguous(),
                seg_ptr.contiguous(), row_sorted.contiguous(), float(lam)))
        return tuple(_hip.nmf_grad(L.contiguous(), R.contiguous(),
                                   row_ptr.contiguous(), col_idx.contiguous(),
                                   vals.contiguous(), float(lam)))
    row_idx = torch.repeat_interleave(
        torch.arange(L.shape[0], device=L.device),
        row_ptr[1:] - row_ptr[:-1])
    Lr = L[row_idx]                       # [nnz, k]
    Rr = R[col_idx]
    e = (Lr * Rr).sum(dim=1) - vals       # [nnz]
    ge = (2.0 * e).unsqueeze(1)
    lcontrib = ge * Rr + 2.0 * lam * Lr
    rcontrib = ge * Lr + 2.0 * lam * Rr
    lgrad = torch.zeros_like(L)
    rgrad = torch.zeros_like(R)
    lgrad.index_add_(0, row_idx, lcontrib)
    rgrad.index_add_(0, col_idx, rcontrib)
    return lgrad, rgrad, (e * e).sum()


# ---------------------------------------------------------------------------
# K7: LDA collapsed Gibbs sampling over a doc block
# ---------------------------------------------------------------------------

def lda_gibbs(doc_topic: torch.Tensor, word_topic: torch.Tensor,
              topic_sum: torch.Tensor, doc_offsets: torch.Tensor,
              word_ids: torch.Tensor, assignments: torch.Tensor,
              alpha: float, beta: float, num_vocabs: int, seed: int
              ) -> torch.Tensor:
    """One Gibbs sweep over the batch's tokens; returns new assignments.

    doc_topic: [n_docs, K] int32 (updated in place),
    word_topic: [n_words_in_batch, K] (pulled rows, LOCAL indices; treated as
    fixed within the sweep — batch-stale counts, see mlapps/lda.py),
    topic_sum: [K], doc_offsets: [n_docs+1] CSR over tokens, word_ids: local
    word index per token, assignments: [n_tokens] int32 (updated in place).

    p(k) ∝ (n_dk + α) (n_wk + β) / (n_k + V β)
    (reference SparseLDASampler.java:141-274's s/r/q bucket decomposition is a
    CPU sparsity optimization; on CDNA4 the dense K-way distribution is
    computed by a wave per document — see ops/csrc/lda.hip.)

    The torch path mirrors the device kernel: same counter-based RNG keyed by
    (seed, token index), same float32 probability terms, same "first k with
    cumsum > u" draw — CPU and GPU agree except at float summation-order
    tie-breaks (tests require >=99% identical samples + exact invariants)."""
    if _use_hip(word_topic):
        return _hip.lda_gibbs(doc_topic, word_topic, topic_sum, doc_offsets,
                              word_ids, assignments, float(alpha), float(beta),
                              int(num_vocabs), int(seed))
    from harmony_amd.ops.rng import rng_uniform

    n_docs = doc_topic.shape[0]
    inv_den = (1.0 / (topic_sum.float() + num_vocabs * beta)).float()   # [K]
    # Lockstep over token positions: docs advance one token per step so the
    # per-doc sequential dependency (n_dk) is honored while steps stay
    # vectorized over docs.
    lengths = doc_offsets[1:] - doc_offsets[:-1]
    max_len = int(lengths.max()) if n_docs else 0
    for pos in range(max_len):
        active = (lengths > pos).nonzero(as_tuple=True)[0]
        tok = doc_offsets[active] + pos
        w = word_ids[tok].long()
        old = assignments[tok].long()
        ar = torch.arange(active.shape[0])
        dt = doc_topic[active]
        dt[ar, old] -= 1
        probs = ((dt.float() + alpha) * (word_topic[w].float() + beta)
                 * inv_den)                                  # [n_active, K]
        tot = probs.sum(dim=1)
        u = rng_uniform(seed & 0xFFFFFFFF, tok.long()) * tot
        cdf = probs.cumsum(dim=1)
        new = torch.searchsorted(cdf, u.unsqueeze(1).to(cdf.dtype),
                                 right=True).squeeze(1)
        bad = new >= probs.shape[1]
        new = torch.where(bad, old, new)                     # numeric edge
        dt[ar, new] += 1
        doc_topic[active] = dt
        assignments[tok] = new.to(assignments.dtype)
    return assignments


def lda_apply_pairs(shard: torch.Tensor, rows: torch.Tensor,
                    old_t: torch.Tensor, new_t: torch.Tensor) -> None:
    """Apply (row, old_topic, new_topic) ±1 count pairs to the word-topic
    shard (the reference's TopicChanges wire format — 12 B per changed token
    instead of a dense K-int row per touched word)."""
    if _use_hip(shard):
        _hip.lda_apply_pairs(shard, rows.contiguous(),
                             old_t.contiguous(), new_t.contiguous())
        return
    K = shard.shape[1]
    flat = shard.view(-1)
    ones = torch.ones(rows.shape[0], dtype=shard.dtype, device=shard.device)
    flat.scatter_add_(0, rows * K + old_t.long(), -ones)
    flat.scatter_add_(0, rows * K + new_t.long(), ones)


def _butterfly_sum64(x: torch.Tensor) -> torch.Tensor:
    """Bitwise replica of the wave64 shfl_xor butterfly sum over dim -1."""
    idx = torch.arange(64)
    for m in (32, 16, 8, 4, 2, 1):
        x = x + x[..., idx ^ m]
    return x[..., 0]


def _vose_serial(pr, al, lk, n):
    """Python replica of the device vose_serial (same order)."""
    small_top, large_top = -1, -1
    for k in range(n - 1, -1, -1):
        if float(pr[k]) < 1.0:
            lk[k] = small_top
            small_top = k
        else:
            lk[k] = large_top
            large_top = k
    while small_top >= 0 and large_top >= 0:
        sm = small_top
        small_top = lk[sm]
        lg = large_top
        al[sm] = lg
        rem = float(pr[lg]) + float(pr[sm]) - 1.0
        pr[lg] = rem
        large_top = lk[lg]
        if rem < 1.0:
            lk[lg] = small_top
            small_top = lg
        else:
            lk[lg] = large_top
            large_top = lg
    while large_top >= 0:
        lg = large_top
        large_top = lk[lg]
        pr[lg] = 1.0
        al[lg] = lg
    while small_top >= 0:
        sm = small_top
        small_top = lk[sm]
        pr[sm] = 1.0
        al[sm] = sm


def lda_alias_build(word_topic: torch.Tensor, topic_sum: torch.Tensor,
                    beta: float, num_vocabs: int):
    """Two-level per-word alias tables over the (batch-stale) word factor
    q_w(k) = (n_wk + b)/(n_k + V b): a 64-entry top alias over lane-segment
    masses + per-segment aliases (exact — see ops/csrc/lda_alias.hip).
    Returns (prob, alias, top_prob, top_alias, qsum, invden). Construction
    is deterministic and float-order-matched to the device kernel
    (butterfly total, ascending serial segment sums) so CPU and GPU build
    bit-identical tables."""
    if _use_hip(word_topic):
        return tuple(_hip.lda_alias_build(word_topic.contiguous(),
                                          topic_sum.contiguous(),
                                          float(beta), int(num_vocabs)))
    rows, K = word_topic.shape
    W = 64
    assert K % W == 0, "alias sampler requires K % 64 == 0"
    S = K // W
    invden = (1.0 / (topic_sum.float() + num_vocabs * beta)).contiguous()
    p = (word_topic.float() + beta) * invden            # [rows, K]
    pseg = p.view(rows, W, S)
    seg_mass = torch.zeros(rows, W)
    for i in range(S):                                   # ascending = kernel
        seg_mass = seg_mass + pseg[:, :, i]
    total = _butterfly_sum64(seg_mass)
    qv = (p * (1.0 / total).unsqueeze(1)).contiguous()   # encoded density
    sscale = torch.where(seg_mass > 0, S / seg_mass,
                         torch.zeros_like(seg_mass))
    prob = (pseg * sscale.unsqueeze(2)).reshape(rows, K).contiguous()
    alias = torch.empty((rows, K), dtype=torch.int32)
    top_prob = (seg_mass * W / total.unsqueeze(1)).contiguous()
    top_alias = torch.empty((rows, W), dtype=torch.int32)
    lk = [0] * max(K, W)
    for r in range(rows):
        for g in range(W):
            _vose_serial(prob[r, g * S:(g + 1) * S],
                         alias[r, g * S:(g + 1) * S], lk, S)
        _vose_serial(top_prob[r], top_alias[r], lk, W)
    return prob, alias, top_prob, top_alias, qv, total, invden


def lda_mh_wave(doc_topic: torch.Tensor, word_topic: torch.Tensor,
                invden: torch.Tensor, prob: torch.Tensor,
                alias: torch.Tensor, top_prob: torch.Tensor,
                top_alias: torch.Tensor, qv: torch.Tensor,
                doc_offsets: torch.Tensor, word_ids: torch.Tensor,
                assignments: torch.Tensor, alpha: float, beta: float,
                seed: int) -> torch.Tensor:
    """Wave-per-doc MH sweep (K7c): 64 lanes cooperate on each doc with
    the doc-topic row shared in LDS via atomics — within-doc token
    updates are approximately parallel (acceptance may read slightly
    stale counts; standard GPU-LDA relaxation, convergence validated by
    scripts/lda_convergence.py). On CPU this falls back to the SERIAL
    sampler: the wave kernel's interleaving is hardware-scheduling
    dependent, so there is no bit-exact oracle — the two are compared
    statistically, not elementwise."""
    if _use_hip(word_topic):
        return _hip.lda_mh_wave(doc_topic, word_topic, invden, prob, alias,
                                top_prob, top_alias, qv, doc_offsets,
                                word_ids, assignments, float(alpha),
                                float(beta), int(seed))
    return lda_mh(doc_topic, word_topic, invden, prob, alias, top_prob,
                  top_alias, qv, doc_offsets, word_ids, assignments,
                  alpha, beta, seed)


def lda_mh(doc_topic: torch.Tensor, word_topic: torch.Tensor,
           invden: torch.Tensor, prob: torch.Tensor, alias: torch.Tensor,
           top_prob: torch.Tensor, top_alias: torch.Tensor,
           qv: torch.Tensor, doc_offsets: torch.Tensor,
           word_ids: torch.Tensor, assignments: torch.Tensor, alpha: float,
           beta: float, seed: int) -> torch.Tensor:
    """One Metropolis-Hastings alias sweep (K7b; proposal/acceptance
    derivation in ops/csrc/lda_alias.hip). Same stationary distribution as
    the exact sampler under the batch-stale word-topic snapshot; O(1) per
    token. Torch path mirrors the kernel's RNG and float math."""
    if _use_hip(word_topic):
        return _hip.lda_mh(doc_topic, word_topic, invden, prob, alias,
                           top_prob, top_alias, qv, doc_offsets, word_ids,
                           assignments, float(alpha), float(beta), int(seed))
    from harmony_amd.ops.rng import rng_uniform

    K = word_topic.shape[1]
    W = 64
    S = K // W
    lengths = doc_offsets[1:] - doc_offsets[:-1]
    n_docs = doc_topic.shape[0]
    max_len = int(lengths.max()) if n_docs else 0
    aK = alpha * K
    for pos in range(max_len):
        active = (lengths > pos).nonzero(as_tuple=True)[0]
        tok = doc_offsets[active] + pos
        w = word_ids[tok].long()
        s = assignments[tok].long()
        ar = torch.arange(active.shape[0])
        dt = doc_topic[active]
        dt[ar, s] -= 1
        c0 = (tok * 8).long()
        sd = seed & 0xFFFFFFFF
        # word proposal: two-level alias draw
        u1 = rng_uniform(sd, c0) * W
        gb = u1.long().clamp_(max=W - 1)
        g = torch.where(u1 - gb.float() < top_prob[w, gb], gb,
                        top_alias[w, gb].long())
        u2 = rng_uniform(sd, c0 + 6) * S
        eb = u2.long().clamp_(max=S - 1)
        ecol = g * S + eb
        t1 = g * S + torch.where(u2 - eb.float() < prob[w, ecol], eb,
                                 alias[w, ecol].long())
        # stale-table-safe acceptance: pi from the CURRENT snapshot, q from
        # the table's encoded density (cancels when the table is fresh)
        pi_s = (dt[ar, s].float() + alpha) * \
            (word_topic[w, s].float() + beta) * invden[s]
        pi_t = (dt[ar, t1].float() + alpha) * \
            (word_topic[w, t1].float() + beta) * invden[t1]
        a1 = (pi_t * qv[w, s]) / (pi_s * qv[w, t1])
        s = torch.where(rng_uniform(sd, c0 + 1) < a1, t1, s)
        # doc proposal
        Ld = lengths[active].float()
        uni = rng_uniform(sd, c0 + 2) < (aK / (aK + Ld))
        t_uni = (rng_uniform(sd, c0 + 3) * K).long().clamp_(max=K - 1)
        j = doc_offsets[active] + (rng_uniform(sd, c0 + 4) * Ld).long()
        j = torch.minimum(j, doc_offsets[active] + lengths[active] - 1)
        t_copy = torch.where(j == tok, s, assignments[j].long())
        t2 = torch.where(uni, t_uni, t_copy)
        nds = dt[ar, s].float()
        ndt = dt[ar, t2].float()
        qs = nds + 1.0 + alpha
        qt = ndt + (t2 == s).float() + alpha
        pis = (nds + alpha) * (word_topic[w, s].float() + beta) * invden[s]
        pit = (ndt + alpha) * (word_topic[w, t2].float() + beta) * invden[t2]
        a2 = (pit * qs) / (pis * qt)
        s = torch.where(rng_uniform(sd, c0 + 5) < a2, t2, s)
        dt[ar, s] += 1
        doc_topic[active] = dt
        assignments[tok] = s.to(assignments.dtype)
    return assignments


def lda_apply_all(shard, word_rows, old_t, new_t, summary_row: int) -> None:
    """Fused single-owner update: per token, if the topic changed, apply
    ±1 to the word row and the summary row (K9; replaces the whole
    nonzero/gather/bincount/scatter pipeline on the local path)."""
    if _use_hip(shard):
        _hip.lda_apply_all(shard, word_rows.contiguous(), old_t.contiguous(),
                           new_t.contiguous(), int(summary_row))
        return
    changed = (old_t != new_t)
    rows = word_rows[changed]
    o = old_t[changed].long()
    nw = new_t[changed].long()
    K = shard.shape[1]
    flat = shard.view(-1)
    ones = torch.ones(rows.shape[0], dtype=shard.dtype, device=shard.device)
    flat.scatter_add_(0, rows * K + o, -ones)
    flat.scatter_add_(0, rows * K + nw, ones)
    srow = summary_row * K
    flat.scatter_add_(0, srow + o, -ones)
    flat.scatter_add_(0, srow + nw, ones)


# ---------------------------------------------------------------------------
# K3/K9: fused owner-side update application ("server" compute)
# ---------------------------------------------------------------------------

# update-function name (et.update_functions) -> device kernel mode
_APPLY_MODES = {"add": 0, "assign": 1, "nmf_sgd": 2, "lda_counts": 3}


def gbt_hist(bins: torch.Tensor, resid: torch.Tensor, node: torch.Tensor,
             n_nodes: int, num_bins: int
             ) -> Tuple[torch.Tensor, torch.Tensor]:
    """GBT level histogram (K10): per-(node, feature, bin) sample counts and
    residual sums for one tree level. Reference GBTTrainer.java:244+ scans
    sorted raw values per node/feature on the CPU; the GPU histogram method
    (pre-quantized bins) is the standard redesign. Returns (cnt, sum),
    each float32 [n_nodes, F, num_bins]."""
    B, F = bins.shape
    if _use_hip(bins) and n_nodes * num_bins <= 8192:
        cnt = torch.zeros(n_nodes, F, num_bins, device=bins.device)
        s = torch.zeros_like(cnt)
        _hip.gbt_hist(bins.int().contiguous(), resid.float().contiguous(),
                      node.int().contiguous(), cnt, s)
        return cnt, s
    idx = (node.long().unsqueeze(1) * F * num_bins
           + torch.arange(F, device=bins.device) * num_bins + bins)
    cnt = torch.zeros(n_nodes * F * num_bins, device=bins.device)
    s = torch.zeros_like(cnt)
    cnt.scatter_add_(0, idx.reshape(-1),
                     torch.ones(B, 1, device=bins.device).expand(B, F)
                     .reshape(-1))
    s.scatter_add_(0, idx.reshape(-1),
                   resid.float().unsqueeze(1).expand(B, F).reshape(-1))
    return (cnt.view(n_nodes, F, num_bins), s.view(n_nodes, F, num_bins))


def lasso_cd(X: torch.Tensor, r: torch.Tensor, w: torch.Tensor,
             col_sq: torch.Tensor, lam_n: float
             ) -> Tuple[torch.Tensor, torch.Tensor]:
    """One full cyclic coordinate-descent sweep (K11): for each coordinate
    the closed-form soft-threshold update (reference LassoTrainer.java:
    164-190), residual maintained incrementally. Returns (w_new, r_new);
    inputs are not mutated. GPU: single persistent workgroup, residual in
    LDS (ops/csrc/lasso.hip); CPU: the tensor-op loop."""
    F = X.shape[1]
    w = w.clone()
    r = r.clone()
    if _use_hip(X) and X.shape[0] + 17 <= 16384:
        Xt = getattr(X, "_harmony_xt", None)
        if Xt is None:
            Xt = X.t().contiguous()
            X._harmony_xt = Xt          # batches are static per block
        _hip.lasso_cd(Xt, r, w, col_sq.contiguous(), float(lam_n))
        return w, r
    for i in range(F):
        xi = X[:, i]
        c = xi @ r + w[i] * col_sq[i]
        wn = (torch.clamp(c.abs() - lam_n, min=0.0) * torch.sign(c)
              / col_sq[i])
        r = r + xi * (w[i] - wn)
        w[i] = wn
    return w, r


def fused_apply_supported(update_fn_name: str) -> bool:
    return update_fn_name in _APPLY_MODES


def scatter_apply(shard: torch.Tensor, rows: torch.Tensor,
                  deltas: torch.Tensor, update_fn_name: str,
                  step_size: float = 0.0, max_val: float = 0.0) -> None:
    """shard[rows] = f(shard[rows], deltas) in one kernel (GPU only)."""
    assert _use_hip(shard), "scatter_apply is the GPU fused path"
    _hip.scatter_apply(shard, rows.contiguous(), deltas.contiguous(),
                       _APPLY_MODES[update_fn_name], float(step_size),
                       float(max_val))


def dense_apply(shard: torch.Tensor, delta: torch.Tensor,
                update_fn_name: str, step_size: float = 0.0,
                max_val: float = 0.0) -> None:
    assert _use_hip(shard), "dense_apply is the GPU fused path"
    _hip.dense_apply(shard, delta.contiguous(), _APPLY_MODES[update_fn_name],
                     float(step_size), float(max_val))


# ---------------------------------------------------------------------------
# segment / scatter helpers
# ---------------------------------------------------------------------------

def segment_sum(keys: torch.Tensor, deltas: torch.Tensor
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Aggregate duplicate keys -> (unique_keys, summed deltas)."""
    uniq, inv = torch.unique(keys, return_inverse=True)
    out = torch.zeros((uniq.shape[0], deltas.shape[1]), dtype=deltas.dtype,
                      device=deltas.device)
    out.index_add_(0, inv, deltas)
    return uniq, out
