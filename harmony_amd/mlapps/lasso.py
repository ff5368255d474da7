"""Lasso — cyclic coordinate descent on the PS.

Reference: dolphin/mlapps/lasso/LassoTrainer.java:164-190 — model is
partitioned partIdx -> weight Vector(featuresPerPartition), workers pull all
partitions each batch (:292); per coordinate i the closed-form update
optimal_i = dot(x_i, y - sum_{j!=i} x_j w_j) / dot(x_i, x_i) followed by a
soft-threshold by lambda (ZERO_THRESHOLD 1e-9 at :61); server applies vector
add (LassoETModelUpdateFunction.java:33).

MI355X shape: the residual r = y - Xw is maintained incrementally so each
coordinate step is two rocBLAS dot/axpy column ops on the device-resident
batch; the full sweep stays on the GPU (F small relative to batch).

App args: num_features, num_parts, batch_size, lam, step_size(unused),
noise.
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

MODEL_TABLE = "lasso_model"
ZERO_THRESHOLD = 1e-9


def defaults(job: JobConfig) -> dict:
    a = dict(num_features=256, num_parts=16, batch_size=2048, lam=0.05,
             noise=0.1, density=0.25)
    a.update(job.app_args)
    return a


def model_table_cfg(job: JobConfig, world_size: int) -> TableConfig:
    a = defaults(job)
    P = a["num_parts"]
    assert a["num_features"] % P == 0
    return TableConfig(
        table_id=f"{job.job_id}/{MODEL_TABLE}",
        num_keys=P,
        value_dim=a["num_features"] // P,
        dtype="float32",
        num_blocks=P,
        update_fn="add",
        init_fn="zeros",
    )


def make_batches(job: JobConfig, rank: int, device: torch.device,
                 world_size: int = 1):
    a = defaults(job)
    F = a["num_features"]
    if job.app_args.get("input"):
        # sample_lasso rows "label idx:val ..." (reference LassoParser)
        from harmony_amd import dataloader as dl

        X, y = dl.parse_libsvm_split(job.app_args["input"], rank,
                                     world_size, F)
        n_blocks = max(1, job.num_worker_blocks or job.num_mini_batches)
        return [(xb.to(device), yb.to(device)) for xb, yb in
                zip(torch.chunk(X, n_blocks), torch.chunk(y, n_blocks))], None
    g = torch.Generator().manual_seed(stable_seed(job.job_id, "data", rank))
    # sparse ground-truth weights
    w_true = torch.randn(F, generator=g)
    mask = torch.rand(F, generator=g) < a["density"]
    w_true = w_true * mask
    blocks = []
    n_blocks = job.num_worker_blocks or job.num_mini_batches
    for _ in range(n_blocks):
        X = torch.randn(a["batch_size"], F, generator=g)
        y = X @ w_true + a["noise"] * torch.randn(a["batch_size"], generator=g)
        blocks.append((X.to(device), y.to(device)))
    return blocks, w_true


class LassoTrainer(Trainer):
    def __init__(self, ctx: TrainerContext):
        super().__init__(ctx)
        self.a = defaults(JobConfig(job_id=ctx.job_id, app="lasso",
                                    app_args=ctx.app_args))
        table = ctx.table(MODEL_TABLE)
        from harmony_amd.et.onesided import OneSidedTable

        if isinstance(table, OneSidedTable):
            from harmony_amd.dolphin.model_accessor import OneSidedAccessor

            self.accessor = OneSidedAccessor(table)
        else:
            self.accessor = ETModelAccessor(table)
        self._loss = torch.zeros(())

    def pull_model(self) -> None:
        self.w = self.accessor.pull_all()[:self.a["num_parts"]].reshape(-1)

    def local_compute(self) -> None:
        X, y = self.batch
        if X.shape[0] == 0:      # stopped worker: no work, zero delta
            self.delta = torch.zeros_like(self.w)
            self._loss = torch.zeros(())
            return
        F = X.shape[1]
        lam_n = self.a["lam"] * X.shape[0]
        r = y - X @ self.w                     # residual
        col_sq = getattr(X, "_harmony_colsq", None)
        if col_sq is None:
            col_sq = (X * X).sum(dim=0).clamp_min(ZERO_THRESHOLD)
            X._harmony_colsq = col_sq          # batches are static per block
        # full cyclic sweep in one persistent kernel on GPU (K11); the torch
        # path queues F small device ops asynchronously (no host syncs)
        w, r = ops.lasso_cd(X, r, self.w, col_sq, lam_n)
        _ = F
        self.delta = w - self.w
        self._loss = (r * r).mean()          # device scalar; no batch sync

    def push_update(self) -> None:
        P = self.a["num_parts"]
        self.accessor.push_dense(self.delta.reshape(P, -1))

    def evaluate_model(self):
        return {"mse": float(self._loss)}

    def num_batch_examples(self) -> int:
        return self.batch[0].shape[0]


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
    blocks, _ = make_batches(job, ctx.rank, ctx.device, ctx.world_size)
    tctx = TrainerContext(job_id=job.job_id, rank=ctx.rank,
                          world_size=ctx.world_size, device=ctx.device,
                          tables={MODEL_TABLE: table}, app_args=job.app_args)
    trainer = LassoTrainer(tctx)
    def _reslice(b, frac):
        # frac<=0 -> EMPTY batch: a stopped worker (StopWorkerOp) does
        # zero work and its sparse pulls/pushes carry zero keys
        n = 0 if frac <= 0 else max(1, int(b[0].shape[0] * frac))
        return (b[0][:n], b[1][:n])

    provider = TrainingDataProvider(reslice=_reslice, local_blocks=blocks)
    return {MODEL_TABLE: table}, trainer, provider
