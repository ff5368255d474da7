"""NMF — async-SGD non-negative matrix factorization on the PS.

Reference: dolphin/mlapps/nmf/NMFTrainer.java — model table holds R columns
(colIdx -> Vector(rank)), a worker-local table holds L rows; per nonzero
(i,j,v): e = L_i.R_j - v, lGrad += 2e R_j (+L2), rGrad_j += 2e L_i (+L2)
(updateGradient:328-367); per-thread gradient maps merged then pushed
(aggregateGradient:375-406); server applies new = clamp(old - step*delta)
(NMFETModelUpdateFunction.java:48-52); loss = squared error (:414-456).

MI355X shape: the batch's nonzeros are device-resident CSR; K1 (nmf_grad,
ops/csrc/nmf.hip) computes e/lgrad/rgrad in one kernel with rank-wide lanes;
the per-thread hashmap merge of the reference (K2) becomes an on-device
segment-sum over sorted column keys; pull = all-to-all-v key gather; push =
all-to-all-v of column deltas with the clamp update fused on the owner.

App args: num_rows (global), num_cols, rank, nnz_per_row, batch_size (rows),
step_size, lam, max_val, decay_rate, decay_period.
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

MODEL_TABLE = "nmf_model"


def defaults(job: JobConfig) -> dict:
    a = dict(num_rows=16384, num_cols=16384, rank=100, nnz_per_row=64,
             rows_per_batch=2048, step_size=0.01, lam=0.0, max_val=1e6,
             decay_rate=0.9, decay_period=5, dtype="float32")
    a.update(job.app_args)
    return a


def model_table_cfg(job: JobConfig, world_size: int) -> TableConfig:
    a = defaults(job)
    return TableConfig(
        table_id=f"{job.job_id}/{MODEL_TABLE}",
        num_keys=a["num_cols"],
        value_dim=a["rank"],
        dtype=a["dtype"],
        num_blocks=max(world_size, min(256, a["num_cols"])),
        update_fn="nmf_sgd",
        update_args={"step_size": a["step_size"], "max_val": a["max_val"]},
        init_fn="uniform_clamped",
        init_args={"max_val": 1.0},
    )


class NMFBatch:
    """One mini-batch: CSR nonzeros of a contiguous range of local L rows,
    with the batch's unique columns precomputed (static per block)."""

    def __init__(self, l_rows: torch.Tensor, row_ptr: torch.Tensor,
                 col_idx: torch.Tensor, vals: torch.Tensor):
        self.l_rows = l_rows            # [n_rows] indices into local L
        self.row_ptr = row_ptr          # [n_rows+1] CSR offsets over nonzeros
        self.col_idx = col_idx          # [nnz] global column key
        self.vals = vals                # [nnz]
        self.uniq_cols, self.col_local = torch.unique(col_idx, return_inverse=True)
        self.uniq_cols._harmony_static = True   # routing cached (et/comm.py)
        self.num_examples = l_rows.shape[0]
        # static column-sorted view for the atomic-free two-pass gradient
        # (ops.nmf_grad col_sorted): perm sorts nonzeros by LOCAL column,
        # seg_ptr covers every pulled R row, row_sorted = L row per nonzero
        perm = torch.argsort(self.col_local, stable=True)
        counts = torch.bincount(self.col_local,
                                minlength=self.uniq_cols.shape[0])
        seg_ptr = torch.zeros(self.uniq_cols.shape[0] + 1, dtype=torch.int64,
                              device=col_idx.device)
        seg_ptr[1:] = counts.cumsum(0)
        row_of = torch.repeat_interleave(
            torch.arange(row_ptr.shape[0] - 1, device=col_idx.device),
            row_ptr[1:] - row_ptr[:-1])
        self.col_sorted = (perm, seg_ptr, row_of[perm])


def make_batches(job: JobConfig, rank: int, device: torch.device,
                 world_size: int = 1):
    a = defaults(job)
    n_blocks = job.num_worker_blocks or job.num_mini_batches
    if job.app_args.get("input"):
        # sample_nmf format "rowId: col,val ..." (reference NMFETDataParser):
        # this rank's rows become local L rows, CSR per row, rows chunked
        # into blocks
        from harmony_amd import dataloader as dl

        rows, cols, vals = dl.parse_nmf_split(job.app_args["input"], rank,
                                              world_size)
        uniq_rows, local_row = torch.unique(rows, return_inverse=True)
        n_rows = uniq_rows.shape[0]
        order = torch.argsort(local_row, stable=True)
        local_row, cols, vals = local_row[order], cols[order], vals[order]
        counts = torch.bincount(local_row, minlength=n_rows)
        row_ptr_all = torch.zeros(n_rows + 1, dtype=torch.int64)
        row_ptr_all[1:] = counts.cumsum(0)
        blocks = []
        per = max(1, n_rows // n_blocks)
        for b in range(n_blocks):
            lo, hi = b * per, min((b + 1) * per, n_rows)
            if lo >= hi:
                break
            p0, p1 = int(row_ptr_all[lo]), int(row_ptr_all[hi])
            blocks.append(NMFBatch(
                torch.arange(lo, hi).to(device),
                (row_ptr_all[lo:hi + 1] - row_ptr_all[lo]).to(device),
                cols[p0:p1].to(device), vals[p0:p1].to(device)))
        return blocks, n_rows
    rows_local = a["rows_per_batch"] * n_blocks
    g = torch.Generator().manual_seed(stable_seed(job.job_id, "data", rank))
    blocks = []
    for b in range(n_blocks):
        lo = b * a["rows_per_batch"]
        l_rows = torch.arange(lo, lo + a["rows_per_batch"])
        nnz = a["rows_per_batch"] * a["nnz_per_row"]
        row_ptr = torch.arange(0, nnz + 1, a["nnz_per_row"])
        col_idx = torch.randint(0, a["num_cols"], (nnz,), generator=g)
        vals = torch.rand(nnz, generator=g)
        blocks.append(NMFBatch(l_rows.to(device), row_ptr.to(device),
                               col_idx.to(device), vals.to(device)))
    return blocks, rows_local


class NMFTrainer(Trainer):
    def __init__(self, ctx: TrainerContext, num_local_rows: int):
        super().__init__(ctx)
        self.a = defaults(JobConfig(job_id=ctx.job_id, app="nmf",
                                    app_args=ctx.app_args))
        table = ctx.table(MODEL_TABLE)
        from harmony_amd.et.onesided import OneSidedTable

        if isinstance(table, OneSidedTable):
            # async mode (one_sided=true): pulls are xGMI gather kernels;
            # pushes go through the owner-side apply-queue RINGS (v2) since
            # nmf_sgd is not add-algebra — reference per-block op-queue
            # semantics (CommManager.java:36-155)
            from harmony_amd.dolphin.model_accessor import OneSidedAccessor

            self.accessor = OneSidedAccessor(table)
        else:
            self.accessor = ETModelAccessor(table)
        # Worker-local model table (reference: local table rowKey -> L row,
        # DolphinJobEntity.java:100-110): device-resident, never leaves HBM.
        g = torch.Generator().manual_seed(stable_seed(ctx.job_id, "L", ctx.rank))
        self.L = (torch.rand(num_local_rows, self.a["rank"], generator=g)
                  .to(ctx.device))
        self.step_size = self.a["step_size"]
        self.use_pull_all = False   # set job-wide by build() (see pull_model)
        # device-resident loss accumulator: a per-batch float() would force
        # a host sync and serialize the async step pipeline
        self._sq_err = torch.zeros((), device=ctx.device)
        self.R_batch = None
        # hipGraph capture of the compute phase: the pulled R rows land in a
        # stable per-block buffer; step size lives in a device scalar so
        # epoch decay does not invalidate the graph
        import os

        from harmony_amd.utils.graphs import GraphRunner

        # measured on MI355X: graph replay was neutral in the 3-job bench
        # and slightly SLOWER per-app (kernels are large; launches were not
        # the bottleneck) — opt-in via HARMONY_GRAPHS=1
        self._graphs = GraphRunner(
            enabled=ctx.device.type == "cuda"
            and os.environ.get("HARMONY_GRAPHS") == "1")
        self._R_bufs = {}
        self._bodies = {}
        self._step_t = torch.tensor(float(self.step_size), device=ctx.device)

    def pull_model(self) -> None:
        b = self.batch
        # when batches touch most of the column space (dense-ish data), one
        # all-gather beats the two-sided all-to-all key exchange. The choice
        # MUST be identical on every rank (both branches are collectives; a
        # per-rank data-dependent branch would issue mismatched RCCL
        # collectives inside one NET phase and deadlock) — build() agrees on
        # it once via the control store and sets self.use_pull_all.
        if self.use_pull_all:
            pulled = self.accessor.pull_all()[b.uniq_cols]
        else:
            pulled = self.accessor.pull(b.uniq_cols)
        if self._graphs.enabled:
            # graphs need a stable pointer -> stage into a per-block buffer
            buf = self._R_bufs.get(id(b))
            if buf is None:
                buf = self._R_bufs.setdefault(id(b), pulled.clone())
            else:
                buf.copy_(pulled)
            self.R_batch = buf
        else:
            self.R_batch = pulled

    def local_compute(self) -> None:
        b = self.batch
        if b.num_examples == 0:   # stopped worker: empty push, no kernel
            self.rgrad = torch.zeros((0, self.a["rank"]),
                                     device=self.ctx.device)
            return
        key = id(b)
        body = self._bodies.get(key)
        if body is None:
            R_buf = self.R_batch

            def body(b=b, R_buf=R_buf):
                L_batch = self.L[b.l_rows]
                lgrad, rgrad, sq = ops.nmf_grad(
                    L_batch, R_buf, b.row_ptr, b.col_local, b.vals,
                    self.a["lam"], col_sorted=b.col_sorted)
                # local L update (worker-side SGD apply, same rule as the
                # server's; step from a device scalar so decay stays live)
                self.L[b.l_rows] = (L_batch - self._step_t * lgrad).clamp_(
                    0.0, self.a["max_val"])
                self.rgrad = rgrad
                self._sq_err += sq

            self._bodies[key] = body
        self._graphs.run(key, body, state=(self.L, self._sq_err))

    def push_update(self) -> None:
        # uniq_cols are unique and rgrad is already per-key aggregated by K1
        from harmony_amd.dolphin.model_accessor import OneSidedAccessor

        if isinstance(self.accessor, OneSidedAccessor):
            self.accessor.push(self.batch.uniq_cols, self.rgrad)
            self.accessor.drain()      # apply peers' queued pushes (owner)
        else:
            self.accessor.push(self.batch.uniq_cols, self.rgrad,
                               assume_unique=True)

    def on_epoch_finished(self, epoch: int) -> None:
        if (epoch + 1) % self.a["decay_period"] == 0:
            self.step_size *= self.a["decay_rate"]
            self._step_t.fill_(float(self.step_size))

    def evaluate_model(self):
        out = {"sq_err": float(self._sq_err)}
        self._sq_err = torch.zeros((), device=self.ctx.device)
        return out

    def num_batch_examples(self) -> int:
        return self.batch.num_examples


def build(job: JobConfig, ctx, cp):
    cfg = model_table_cfg(job, ctx.world_size)
    a = defaults(job)
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
    blocks, rows_local = make_batches(job, ctx.rank, ctx.device,
                                      ctx.world_size)
    tctx = TrainerContext(job_id=job.job_id, rank=ctx.rank,
                          world_size=ctx.world_size, device=ctx.device,
                          tables={MODEL_TABLE: table}, app_args=job.app_args)
    trainer = NMFTrainer(tctx, rows_local)
    # Agree job-wide (store-based, not a collective) whether batches are
    # dense enough for the all-gather pull: max uniq-column count over every
    # rank's blocks, decided ONCE — a per-rank/per-batch decision could put
    # ranks of one job on different collectives (advisor r01, high).
    local_max = max((b.uniq_cols.shape[0] for b in blocks), default=0)
    agreed = cp.agree_max(f"{job.job_id}/nmf_uniqmax", int(local_max),
                          n=ctx.world_size) if cp is not None else local_max
    trainer.use_pull_all = agreed * 2 > defaults(job)["num_cols"]
    def _reslice(b, frac):
        # row-prefix re-slice for SetBatchShareOp: rebuild the static
        # precomputes (uniq_cols, col_sorted) for the smaller batch once
        n = 0 if frac <= 0 else max(1, int(b.l_rows.shape[0] * frac))
        nnz = int(b.row_ptr[n])
        nb = NMFBatch(b.l_rows[:n], b.row_ptr[:n + 1].contiguous(),
                      b.col_idx[:nnz], b.vals[:nnz])
        nb.block_idx = getattr(b, "block_idx", None)
        return nb

    provider = TrainingDataProvider(blocks, reslice=_reslice)
    return {MODEL_TABLE: table}, trainer, provider
