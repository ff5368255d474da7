"""MLR — multinomial (softmax) logistic regression on the PS.

Reference: dolphin/mlapps/mlr/MLRTrainer.java — the model is partitioned as
key = classIdx*numPartitionsPerClass + partIdx -> Vector(featuresPerPartition)
and workers pull ALL partitions each mini-batch (MLRTrainer.java:185-187);
per sample p = softmax(Wx) with a log-sum-exp guard (:475-489), gradient
grad_j -= stepSize * p_j * x (+L2) accumulated densely (:374-398); the server
applies old.addi(delta) (MLRETModelUpdateFunction.java:60-62).

MI355X shape: the whole per-batch math is (batch x F) @ (F x C) GEMM on the
matrix cores + a fused row-softmax/label-subtract kernel + the outer-product
gradient GEMM; pull-all = all-gather, dense push = reduce-scatter with the
add-update epilogue. GEMMs go to rocBLAS (torch.matmul); the softmax+grad
scaling is the fused HIP kernel `mlr_softmax_grad` (ops/csrc/mlr.hip).

App args: num_classes, num_features, num_parts_per_class, batch_size,
step_size, lambda (L2), dtype.
"""

from __future__ import annotations

import torch

from harmony_amd.config import JobConfig, TableConfig
from harmony_amd.dolphin.data_provider import TrainingDataProvider
from harmony_amd.dolphin.model_accessor import ETModelAccessor
from harmony_amd.dolphin.trainer import Trainer, TrainerContext
from harmony_amd.et.table import Table
from harmony_amd import ops

MODEL_TABLE = "mlr_model"


def defaults(job: JobConfig) -> dict:
    a = dict(num_classes=10, num_features=1024, num_parts_per_class=8,
             batch_size=4096, step_size=0.01, lam=1e-4, dtype="float32",
             decay_rate=0.9, decay_period=5)
    a.update(job.app_args)
    return a


def model_table_cfg(job: JobConfig, world_size: int) -> TableConfig:
    a = defaults(job)
    P = a["num_parts_per_class"]
    assert a["num_features"] % P == 0
    return TableConfig(
        table_id=f"{job.job_id}/{MODEL_TABLE}",
        num_keys=a["num_classes"] * P,
        value_dim=a["num_features"] // P,
        dtype=a["dtype"],
        # one key per block (a partition row IS the unit of placement, as in
        # the reference's per-partition table entries) -> no keyspace padding,
        # clean [C*P, F/P] <-> [C, F] views.
        num_blocks=a["num_classes"] * P,
        update_fn="add",
        init_fn="gaussian",
        init_args={"std": 0.01},
    )


def make_batches(job: JobConfig, rank: int, device: torch.device,
                 world_size: int = 1):
    """One block = one mini-batch (reference ETTrainingDataProvider).
    With app_args['input'] (reference -input flag): load this rank's split of
    a sample_mlr-format file; otherwise synthetic per-class gaussian blobs,
    deterministic per (job, rank)."""
    a = defaults(job)
    if a.get("input") or job.app_args.get("input"):
        from harmony_amd import dataloader as dl

        X, y = dl.parse_libsvm_split(job.app_args["input"], rank,
                                     world_size, a["num_features"])
        n_blocks = max(1, job.num_worker_blocks or job.num_mini_batches)
        xs = torch.chunk(X, n_blocks)
        ys = torch.chunk(y.long(), n_blocks)
        return [(x.to(device), yy.to(device)) for x, yy in zip(xs, ys)]
    C, F, B = a["num_classes"], a["num_features"], a["batch_size"]
    from harmony_amd.utils import stable_seed

    g = torch.Generator().manual_seed(stable_seed(job.job_id, rank))
    centers = torch.randn(C, F, generator=g) * 0.5
    blocks = []
    n_blocks = job.num_worker_blocks or job.num_mini_batches
    for _ in range(n_blocks):
        y = torch.randint(0, C, (B,), generator=g)
        x = centers[y] + torch.randn(B, F, generator=g)
        blocks.append((x.to(device), y.to(device)))
    return blocks


class MLRTrainer(Trainer):
    def __init__(self, ctx: TrainerContext):
        super().__init__(ctx)
        self.a = defaults(JobConfig(job_id=ctx.job_id, app="mlr",
                                    app_args=ctx.app_args))
        table = ctx.table(MODEL_TABLE)
        from harmony_amd.et.onesided import OneSidedTable

        if isinstance(table, OneSidedTable):
            # async mode (app arg one_sided=true): pull/push are direct
            # xGMI kernels; the worker never issues a collective
            from harmony_amd.dolphin.model_accessor import OneSidedAccessor

            self.accessor = OneSidedAccessor(table)
        else:
            self.accessor = ETModelAccessor(table)
        self.step_size = self.a["step_size"]
        self.W = None          # [C*P, F/P] pulled model
        # hipGraph capture of the compute phase (static per block); the
        # pulled model lands in a stable buffer so graphs see fixed pointers
        import os

        from harmony_amd.utils.graphs import GraphRunner

        # measured on MI355X: graph replay was neutral in the 3-job bench
        # and slightly SLOWER per-app (kernels are large; launches were not
        # the bottleneck) — opt-in via HARMONY_GRAPHS=1
        self._graphs = GraphRunner(
            enabled=ctx.device.type == "cuda"
            and os.environ.get("HARMONY_GRAPHS") == "1")
        self._W_buf = None
        self._bodies = {}
        # device-resident accumulators (no per-batch host syncs)
        self._loss_sum = torch.zeros((), device=ctx.device)
        self._loss_n = 0
        self._correct = torch.zeros((), dtype=torch.int64, device=ctx.device)

    def _w_matrix(self) -> torch.Tensor:
        C, F = self.a["num_classes"], self.a["num_features"]
        return self.W.view(C, F)

    def _wt_buf(self) -> torch.Tensor:
        # persistent padded W^T staging buffer for the MFMA step kernel
        if getattr(self, "_wt", None) is None:
            F = self.a["num_features"]
            self._wt = torch.zeros((F, 16), dtype=torch.float32,
                                   device=self.ctx.device)
        return self._wt

    def pull_model(self) -> None:
        C, P = self.a["num_classes"], self.a["num_parts_per_class"]
        pulled = self.accessor.pull_all()[:C * P]
        if self._W_buf is None:
            self._W_buf = pulled.clone()
        else:
            self._W_buf.copy_(pulled)
        self.W = self._W_buf

    def local_compute(self) -> None:
        x, y = self.batch
        if x.shape[0] == 0:       # stopped worker: zero grad, no forward
            self.grad_raw = torch.zeros_like(self._w_matrix())
            return
        key = id(self.batch)
        body = self._bodies.get(key)
        if body is None:
            W = self._w_matrix()                   # [C, F] view of W_buf
            import os as _os

            # default: rocBLAS pair (measured at the fetch roofline and
            # 4% faster than the in-tree MFMA kernels both isolated and
            # in-bench — profiles/r02_mlr_mfma.md); HARMONY_MLR_MFMA=1
            # switches to the hand-written K4-MFMA path
            use_mfma = (x.device.type == "cuda"
                        and ops.mlr_step_ok(x, self.a["num_classes"])
                        and _os.environ.get("HARMONY_MLR_MFMA", "0") == "1")
            rb = int(_os.environ.get("HARMONY_MLR_RB", "0"))
            if rb and (x.shape[0] % rb or rb % 64):
                rb = 0

            if use_mfma:
                # fused fwd+softmax+grad on f32 MFMA, L3 row-blocked
                # (K4-MFMA, ops/csrc/mlr_mfma.hip; A/B scripts/mlr_mfma_ab.py)
                def body(x=x, y=y, W=W, rb=rb):
                    g, loss, correct = ops.mlr_step_mfma(
                        x, W, y, row_block=rb, Wt_buf=self._wt_buf())
                    self.grad_raw = g / x.shape[0] + self.a["lam"] * W
                    self._loss_sum += loss
                    self._correct += correct
            else:
                def body(x=x, y=y, W=W):
                    # forward GEMM + fused softmax + grad GEMM
                    p, loss, correct = ops.mlr_forward(x, W, y)
                    g = ops.mlr_grad_gemm(p, x)        # [C, F]
                    self.grad_raw = g / x.shape[0] + self.a["lam"] * W
                    self._loss_sum += loss
                    self._correct += correct

            self._bodies[key] = body
        self._graphs.run(key, body, state=(self._loss_sum, self._correct))
        self._loss_n += x.shape[0]

    def push_update(self) -> None:
        C, P = self.a["num_classes"], self.a["num_parts_per_class"]
        # step size applied OUTSIDE the graph (it decays over epochs)
        delta = (-self.step_size) * self.grad_raw
        self.accessor.push_dense(delta.view(C * P, -1))

    def on_epoch_finished(self, epoch: int) -> None:
        if (epoch + 1) % self.a["decay_period"] == 0:
            self.step_size *= self.a["decay_rate"]

    def evaluate_model(self):
        if not self._loss_n:
            return {}
        out = {"cross_entropy": float(self._loss_sum) / self._loss_n,
               "accuracy": float(self._correct) / self._loss_n}
        self._loss_sum = torch.zeros((), device=self.ctx.device)
        self._correct = torch.zeros((), dtype=torch.int64,
                                    device=self.ctx.device)
        self._loss_n = 0
        return out

    def num_batch_examples(self) -> int:
        return self.batch[0].shape[0]


def build(job: JobConfig, ctx, cp):
    cfg = model_table_cfg(job, ctx.world_size)
    a = defaults(job)
    if str(a.get("one_sided", "")).lower() in ("true", "1"):
        # async PS mode: hipIpc/xGMI one-sided table — no collectives in
        # the training loop, SSP slack is the only cross-worker coupling
        from harmony_amd.et.onesided import OneSidedTable

        table = OneSidedTable(cfg, ctx.rank, ctx.world_size, ctx.device,
                              store=ctx.store)
        cp.barrier(f"{job.job_id}/os_alloc", ctx.world_size)
        table.connect()
        cp.barrier(f"{job.job_id}/os_conn", ctx.world_size)
    else:
        comm = ctx.new_data_plane()
        table = Table(cfg, ctx.rank, ctx.world_size, ctx.device, comm=comm)
    tctx = TrainerContext(job_id=job.job_id, rank=ctx.rank,
                          world_size=ctx.world_size, device=ctx.device,
                          tables={MODEL_TABLE: table}, app_args=job.app_args)
    trainer = MLRTrainer(tctx)
    def _reslice(b, frac):
        # frac<=0 -> EMPTY batch: a stopped worker (StopWorkerOp) does
        # zero work and its sparse pulls/pushes carry zero keys
        n = 0 if frac <= 0 else max(1, int(b[0].shape[0] * frac))
        return (b[0][:n], b[1][:n])

    provider = TrainingDataProvider(reslice=_reslice, local_blocks=
        make_batches(job, ctx.rank, ctx.device, ctx.world_size))
    return {MODEL_TABLE: table}, trainer, provider
