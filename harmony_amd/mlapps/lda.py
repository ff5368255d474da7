"""LDA — collapsed Gibbs sampling on the PS.

Reference: dolphin/mlapps/lda/ — model table wordId -> topic-count row with a
summary row at key numVocabs (LDATrainer.java:210-213), local table docId ->
doc-topic counts + per-token assignments; per token the SparseLDA s/r/q
bucket sampler (SparseLDASampler.java:141-274); deltas pushed as
(topicIdx, +/-count) pairs merged server-side with clamp>=0
(LDAETModelUpdateFunction.java:29,43-64).

MI355X redesign (documented divergences):
  * Rows are DENSE int32 [num_topics] vectors, not the reference's sparse
    pair encoding — 100k vocab x 1k topics x 4B = 400 MB, trivially resident
    in 288 GB HBM3E, and dense rows make the sampler a coalesced wave-per-
    document kernel (K7, ops/csrc/lda.hip) instead of branchy bucket walks.
  * Word-topic counts are batch-stale: a sweep samples against the pulled
    snapshot and pushes the net delta afterwards (standard GPU/distributed
    LDA; the reference's within-batch incremental cache update is a
    single-thread CPU optimization).

App args: num_docs (global), num_vocabs, num_topics, tokens_per_doc, alpha,
beta, docs_per_batch.
"""

from __future__ import annotations

import torch

from harmony_amd.config import JobConfig, TableConfig
from harmony_amd.dolphin.data_provider import TrainingDataProvider
from harmony_amd.dolphin.model_accessor import ETModelAccessor
from harmony_amd.dolphin.trainer import Trainer, TrainerContext
from harmony_amd.et.table import Table
from harmony_amd.utils import stable_seed
from harmony_amd import ops

MODEL_TABLE = "lda_model"


def defaults(job: JobConfig) -> dict:
    a = dict(num_docs=32768, num_vocabs=30000, num_topics=256,
             tokens_per_doc=64, alpha=0.1, beta=0.01, docs_per_batch=4096,
             sampler="exact",   # "exact" (dense Gibbs) | "alias" (MH, K7b)
                                # | "alias_wave" (K7c wave-per-doc MH)
             alias_refresh=4)   # rebuild alias tables every N pulls
    a.update(job.app_args)
    return a


def model_table_cfg(job: JobConfig, world_size: int) -> TableConfig:
    a = defaults(job)
    # key space: word ids [0, V) + summary row at key V (topic totals)
    return TableConfig(
        table_id=f"{job.job_id}/{MODEL_TABLE}",
        num_keys=a["num_vocabs"] + 1,
        value_dim=a["num_topics"],
        dtype="int32",
        num_blocks=max(world_size, min(512, a["num_vocabs"] + 1)),
        update_fn="lda_counts",
        init_fn="zeros",
    )


class LDABatch:
    def __init__(self, doc_ids: torch.Tensor, doc_offsets: torch.Tensor,
                 word_ids: torch.Tensor, num_vocabs: int):
        self.doc_ids = doc_ids          # [n_docs] local doc indices
        self.doc_offsets = doc_offsets  # [n_docs+1] CSR over tokens
        self.word_ids = word_ids        # [n_tokens] global word ids
        self.uniq_words, self.word_local = torch.unique(word_ids,
                                                        return_inverse=True)
        # pull keys = batch's words + the summary row
        self.pull_keys = torch.cat([
            self.uniq_words,
            torch.tensor([num_vocabs], device=word_ids.device)])
        # static per block -> routing is cached (et/comm.py _route)
        self.pull_keys._harmony_static = True
        self.num_examples = doc_ids.shape[0]
        # contiguous doc ranges let the sampler mutate a doc_topic SLICE
        # in place (no gather/scatter copies — ~130 us/batch on GPU)
        n = doc_ids.shape[0]
        self.doc_lo = int(doc_ids[0]) if n else 0
        self.docs_contiguous = bool(n == 0 or (
            int(doc_ids[-1]) - self.doc_lo == n - 1
            and torch.equal(doc_ids, torch.arange(
                self.doc_lo, self.doc_lo + n, device=doc_ids.device))))


def make_batches(job: JobConfig, rank: int, device: torch.device,
                 world_size: int = 1):
    a = defaults(job)
    n_blocks = job.num_worker_blocks or job.num_mini_batches
    if job.app_args.get("input"):
        # sample_lda format: one doc per line of word ids (reference
        # LDA data; each rank takes its file split, docs chunked to blocks)
        from harmony_amd import dataloader as dl

        offsets, words = dl.parse_lda_split(job.app_args["input"], rank,
                                            world_size)
        n_docs = offsets.shape[0] - 1
        blocks = []
        per = max(1, n_docs // n_blocks)
        for b in range(n_blocks):
            lo, hi = b * per, min((b + 1) * per, n_docs)
            if lo >= hi:
                break
            tok_lo, tok_hi = int(offsets[lo]), int(offsets[hi])
            w = words[tok_lo:tok_hi]
            # sort tokens within each doc (see synthetic path note)
            off = (offsets[lo:hi + 1] - offsets[lo]).to(device)
            batch = LDABatch(torch.arange(lo, hi).to(device), off,
                             w.to(device), a["num_vocabs"])
            batch.block_idx = len(blocks)
            blocks.append(batch)
        return blocks, n_docs
    D, T = a["docs_per_batch"], a["tokens_per_doc"]
    g = torch.Generator().manual_seed(stable_seed(job.job_id, "data", rank))
    blocks = []
    for b in range(n_blocks):
        doc_ids = torch.arange(b * D, (b + 1) * D)
        offsets = torch.arange(0, (D + 1) * T, T)
        # Zipf-ish word draw: square a uniform to skew mass to low ids
        u = torch.rand(D * T, generator=g)
        word_ids = (u * u * a["num_vocabs"]).long().clamp_(0, a["num_vocabs"] - 1)
        # sort tokens by word WITHIN each doc: Gibbs is valid under any
        # within-doc token order, and consecutive same-word tokens let the
        # sampler reuse the word-topic row from cache (big win on Zipf data)
        word_ids = word_ids.view(D, T).sort(dim=1).values.reshape(-1)
        batch = LDABatch(doc_ids.to(device), offsets.to(device),
                         word_ids.to(device), a["num_vocabs"])
        batch.block_idx = b
        blocks.append(batch)
    local_docs = D * n_blocks
    return blocks, local_docs


class LDATrainer(Trainer):
    def __init__(self, ctx: TrainerContext, num_local_docs: int,
                 blocks=None):
        super().__init__(ctx)
        self.a = defaults(JobConfig(job_id=ctx.job_id, app="lda",
                                    app_args=ctx.app_args))
        table = ctx.table(MODEL_TABLE)
        from harmony_amd.et.onesided import OneSidedTable

        if isinstance(table, OneSidedTable):
            # async mode: pulls/pushes are direct xGMI kernels (atomic int
            # adds); LDA's count algebra is pure add (clamp is a no-op)
            from harmony_amd.dolphin.model_accessor import OneSidedAccessor

            self.accessor = OneSidedAccessor(table)
        else:
            self.accessor = ETModelAccessor(table)
        K = self.a["num_topics"]
        self.doc_topic = torch.zeros(num_local_docs, K, dtype=torch.int32,
                                     device=ctx.device)
        self._blocks = blocks or []
        self._assignments = {}          # block index -> [n_tokens] int32
        self._epoch_seed = stable_seed(ctx.job_id, "gibbs", ctx.rank)
        self._step = 0

    def initialize(self) -> None:
        """Random initial topic assignments; push the implied word-topic
        counts so the global table is consistent with local assignments
        (collective: every rank participates)."""
        K, V = self.a["num_topics"], self.a["num_vocabs"]
        g = torch.Generator().manual_seed(self._epoch_seed)
        all_keys, all_deltas = [], []
        for i, b in enumerate(self._blocks):
            z = torch.randint(0, K, (b.word_ids.shape[0],), generator=g,
                              dtype=torch.int32).to(self.ctx.device)
            self._assignments[i] = z
            # doc-topic counts
            tok_doc = torch.repeat_interleave(
                torch.arange(b.doc_ids.shape[0], device=z.device),
                b.doc_offsets[1:] - b.doc_offsets[:-1])
            idx = b.doc_ids[tok_doc] * K + z.long()
            self.doc_topic.view(-1).scatter_add_(
                0, idx, torch.ones_like(idx, dtype=torch.int32))
            # word-topic deltas (one-hot rows summed per word)
            wt = torch.zeros(b.uniq_words.shape[0], K, dtype=torch.int32,
                             device=z.device)
            flat = b.word_local * K + z.long()
            wt.view(-1).scatter_add_(0, flat,
                                     torch.ones_like(flat, dtype=torch.int32))
            all_keys.append(b.uniq_words)
            all_deltas.append(wt)
            # summary row delta
            summ = torch.bincount(z.long(), minlength=K).to(torch.int32)
            all_keys.append(torch.tensor([V], device=z.device))
            all_deltas.append(summ.unsqueeze(0))
        self.accessor.push(torch.cat(all_keys), torch.cat(all_deltas))

    def set_batch_data(self, batch) -> None:
        self.batch = batch
        self._block_idx = batch.block_idx

    def pull_model(self) -> None:
        pulled = self.accessor.pull(self.batch.pull_keys)
        from harmony_amd.dolphin.model_accessor import OneSidedAccessor

        if isinstance(self.accessor, OneSidedAccessor):
            # async pushes are per-CELL atomic, not per-row: a concurrent
            # pull can see a row where the -1 landed but the +1 hasn't,
            # i.e. transiently negative counts -> negative pmf mass in the
            # samplers. Clamp at the consumer (reference's defensive clamp,
            # LDAETModelUpdateFunction.java:43-64; advisor r01).
            pulled = pulled.clamp_min_(0)
        self.word_topic = pulled[:-1]          # [n_uniq_words, K]
        self.topic_sum = pulled[-1]            # [K]

    def _is_alias(self) -> bool:
        return self.a["sampler"] in ("alias", "alias_wave")

    def _ensure_alias(self) -> None:
        # two-level alias tables over the word factor (K7b), rebuilt every
        # alias_refresh uses OF THIS BLOCK (tables index the block's local
        # word ids) — the MH acceptance corrects for table staleness with
        # the stored proposal density qv. Built in the COMP phase: the
        # build is pure local compute, and keeping it inside the NET
        # ticket (round 1) serialized co-located jobs behind ~300 us of
        # kernel per refresh.
        if not hasattr(self, "_alias_cache"):
            self._alias_cache = {}
            self._alias_age = {}
        bid = self._block_idx
        refresh = max(1, int(self.a["alias_refresh"]))
        age = self._alias_age.get(bid, refresh)
        if age >= refresh:
            self._alias_cache[bid] = ops.lda_alias_build(
                self.word_topic, self.topic_sum, self.a["beta"],
                self.a["num_vocabs"])
            age = 0
        self._alias_age[bid] = age + 1
        self._alias = self._alias_cache[bid]

    def local_compute(self) -> None:
        b = self.batch
        z = self._assignments[self._block_idx]
        old = z.clone()
        self._step += 1
        if b.docs_contiguous:
            # in-place slice view: the sampler kernels mutate doc_topic
            # directly (no gather/scatter round trip)
            dt = self.doc_topic[b.doc_lo:b.doc_lo + b.num_examples]
        else:
            dt = self.doc_topic[b.doc_ids]      # gather copy
        if self._is_alias():
            self._ensure_alias()
            prob, alias, tprob, talias, qv, _, invden = self._alias
            fn = (ops.lda_mh_wave if self.a["sampler"] == "alias_wave"
                  else ops.lda_mh)
            # NOTE: invden/qv index LOCAL word ids of the batch on which the
            # tables were built; with one static key set per block this is
            # consistent across refreshes of the same block
            fn(dt, self.word_topic, invden, prob, alias, tprob,
               talias, qv, b.doc_offsets, b.word_local, z,
               self.a["alpha"], self.a["beta"],
               self._epoch_seed + self._step)
        else:
            ops.lda_gibbs(dt, self.word_topic,
                          self.topic_sum, b.doc_offsets, b.word_local, z,
                          self.a["alpha"], self.a["beta"],
                          self.a["num_vocabs"],
                          self._epoch_seed + self._step)
        if not b.docs_contiguous:
            self.doc_topic[b.doc_ids] = dt      # write back
        self._old_z = old
        self._new_z = z

    def push_update(self) -> None:
        import time

        t0 = time.perf_counter()
        table = self.accessor.table
        b = self.batch
        old, z = self._old_z, self._new_z
        from harmony_amd.et.onesided import OneSidedTable

        if isinstance(table, OneSidedTable):
            # async path: dense ±1 delta rows for CHANGED words, one atomic
            # scatter into whichever rank owns each word (denser on the
            # wire than the 12 B pair format, but requires no rendezvous)
            K = self.a["num_topics"]
            changed = (z != old).nonzero(as_tuple=True)[0]
            if changed.numel():
                pw = b.word_ids[changed]
                uniqw, inv = torch.unique(pw, return_inverse=True)
                delta = torch.zeros(uniqw.shape[0], K, dtype=torch.int32,
                                    device=z.device)
                ones = torch.ones_like(changed, dtype=torch.int32)
                delta.view(-1).scatter_add_(
                    0, inv * K + z[changed].long(), ones)
                delta.view(-1).scatter_add_(
                    0, inv * K + old[changed].long(), -ones)
                summ = (torch.bincount(z[changed].long(), minlength=K)
                        - torch.bincount(old[changed].long(), minlength=K)
                        ).to(torch.int32)
                table.push(torch.cat([uniqw, b.pull_keys[-1:]]),
                           torch.cat([delta, summ.unsqueeze(0)]))
        elif table.comm is None or table.world_size == 1:
            # single-owner path: ONE fused kernel applies every changed
            # token's ±1 to its word row AND the summary row — no host sync
            # (the distributed path's nonzero() is only needed to size the
            # all-to-all)
            if not hasattr(b, "word_rows"):
                b.word_rows = table.local_rows_of(b.word_ids)
                b.summary_row = int(table.local_rows_of(b.pull_keys[-1:]))
            ops.lda_apply_all(table.shard, b.word_rows, old, z, b.summary_row)
        else:
            # compressed delta: (word, old_topic, new_topic) for CHANGED
            # tokens only — the reference's TopicChanges pair format, 12 B
            # per change on the xGMI wire instead of a dense K-int row per
            # touched word
            K = self.a["num_topics"]
            changed = (z != old).nonzero(as_tuple=True)[0]
            pw = b.word_ids[changed]
            po = old[changed]
            pn = z[changed]

            def apply_pairs(tbl, keys, payload):
                rows = tbl.local_rows_of(keys)
                ops.lda_apply_pairs(tbl.shard, rows,
                                    payload[:, 0].contiguous(),
                                    payload[:, 1].contiguous())

            table.comm.push_pairs(table, pw, torch.stack([po, pn], dim=1),
                                  apply_pairs)
            summ = (torch.bincount(pn.long(), minlength=K)
                    - torch.bincount(po.long(), minlength=K)).to(torch.int32)
            self.accessor.push(b.pull_keys[-1:], summ.unsqueeze(0),
                               assume_unique=True)
        self.accessor.metrics["total_push_time_sec"] += time.perf_counter() - t0

    def evaluate_model(self):
        """Collapsed LDA log-likelihood (reference LDAStatCalculator:
        log p(w|z) from the word-topic table + log p(z) from the local
        doc-topic counts; doc part is per-rank, word part global)."""
        import torch as T

        a = self.a
        K, V = a["num_topics"], a["num_vocabs"]
        alpha, beta = float(a["alpha"]), float(a["beta"])
        full = self.accessor.table.pull_all()[:V + 1].float()
        wt = full[:V]                              # [V, K]
        nk = full[V]                               # [K] topic totals
        lpw = (T.lgamma(wt + beta).sum()
               - T.lgamma(nk + V * beta).sum()
               + K * (T.lgamma(T.tensor(V * beta)) -
                      V * T.lgamma(T.tensor(beta))))
        dt = self.doc_topic.float()
        nd = dt.sum(dim=1)
        lpz = (T.lgamma(dt + alpha).sum()
               - T.lgamma(nd + K * alpha).sum()
               + dt.shape[0] * (T.lgamma(T.tensor(K * alpha)) -
                                K * T.lgamma(T.tensor(alpha))))
        return {"log_likelihood": float(lpw + lpz),
                "log_pw_given_z": float(lpw),
                "log_pz_local": float(lpz)}

    def num_batch_examples(self) -> int:
        return self.batch.num_examples



def build(job: JobConfig, ctx, cp):
    a = defaults(job)
    cfg = model_table_cfg(job, ctx.world_size)
    if str(a.get("one_sided", "")).lower() in ("true", "1"):
        from harmony_amd.et.onesided import OneSidedTable

        table = OneSidedTable(cfg, ctx.rank, ctx.world_size, ctx.device,
                              store=ctx.store)
        cp.barrier(f"{job.job_id}/os_alloc", ctx.world_size)
        table.connect()
        cp.barrier(f"{job.job_id}/os_conn", ctx.world_size)
    else:
        comm = ctx.new_data_plane()
        table = Table(cfg, ctx.rank, ctx.world_size, ctx.device, comm=comm)
    blocks, local_docs = make_batches(job, ctx.rank, ctx.device,
                                      ctx.world_size)
    tctx = TrainerContext(job_id=job.job_id, rank=ctx.rank,
                          world_size=ctx.world_size, device=ctx.device,
                          tables={MODEL_TABLE: table}, app_args=job.app_args)
    trainer = LDATrainer(tctx, local_docs, blocks)
    provider = TrainingDataProvider(blocks)
    return {MODEL_TABLE: table}, trainer, provider
