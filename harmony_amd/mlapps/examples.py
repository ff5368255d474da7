"""Verification apps: addinteger / addvector.

Reference: dolphin/mlapps/examples/{addinteger,addvector} — every worker
pushes a known delta to every key each batch; a validator then checks the
final server values equal the expected sum. These are the correctness probes
the reference's integration and ownership-first-migration tests drive
(OwnershipFirstMigrationTest.java:23-96): if a live migration loses or
double-applies an update, validation fails.
"""

from __future__ import annotations

import torch

from harmony_amd.config import JobConfig, TableConfig
from harmony_amd.dolphin.data_provider import TrainingDataProvider
from harmony_amd.dolphin.model_accessor import ETModelAccessor
from harmony_amd.dolphin.trainer import Trainer, TrainerContext
from harmony_amd.et.table import Table

MODEL_TABLE = "add_model"


def defaults(job: JobConfig) -> dict:
    a = dict(num_keys=32, vector_dim=16, delta=1.0, batch_size=64)
    if job.app == "addinteger":
        a["vector_dim"] = 1
    a.update(job.app_args)
    return a


def model_table_cfg(job: JobConfig, world_size: int) -> TableConfig:
    a = defaults(job)
    return TableConfig(
        table_id=f"{job.job_id}/{MODEL_TABLE}",
        num_keys=a["num_keys"],
        value_dim=a["vector_dim"],
        dtype="float32",
        num_blocks=min(a["num_keys"], 32),
        update_fn="add",
        init_fn="zeros",
    )


class AddTrainer(Trainer):
    """Each batch: pull all keys, then push `delta` to every key."""

    def __init__(self, ctx: TrainerContext):
        super().__init__(ctx)
        self.a = defaults(JobConfig(job_id=ctx.job_id, app="addvector",
                                    app_args=ctx.app_args))
        self.accessor = ETModelAccessor(ctx.table(MODEL_TABLE))
        self.keys = torch.arange(self.a["num_keys"], device=ctx.device)
        self.num_pushes = 0

    def pull_model(self) -> None:
        self.pulled = self.accessor.pull(self.keys)

    def local_compute(self) -> None:
        self.delta = torch.full((self.a["num_keys"], self.a["vector_dim"]),
                                float(self.a["delta"]),
                                device=self.ctx.device)

    def push_update(self) -> None:
        self.accessor.push(self.keys, self.delta)
        self.num_pushes += 1

    def num_batch_examples(self) -> int:
        return self.a["batch_size"]


def expected_value(job: JobConfig, world_size: int, total_batches_per_rank: int) -> float:
    """Every worker pushed delta to every key once per batch."""
    a = defaults(job)
    return a["delta"] * world_size * total_batches_per_rank


def validate(table: Table, job: JobConfig, world_size: int,
             total_batches_per_rank: int) -> bool:
    """Reference ValidatorTask: all values equal the expected sum."""
    a = defaults(job)
    exp = expected_value(job, world_size, total_batches_per_rank)
    full = table.pull_all()[:a["num_keys"]]
    return bool(torch.allclose(full, torch.full_like(full, exp)))


def build(job: JobConfig, ctx, cp):
    cfg = model_table_cfg(job, ctx.world_size)
    comm = ctx.new_data_plane()
    table = Table(cfg, ctx.rank, ctx.world_size, ctx.device, comm=comm)
    tctx = TrainerContext(job_id=job.job_id, rank=ctx.rank,
                          world_size=ctx.world_size, device=ctx.device,
                          tables={MODEL_TABLE: table}, app_args=job.app_args)
    trainer = AddTrainer(tctx)
    n_blocks = job.num_worker_blocks or job.num_mini_batches
    provider = TrainingDataProvider([i for i in range(n_blocks)], shuffle=False)
    return {MODEL_TABLE: table}, trainer, provider
