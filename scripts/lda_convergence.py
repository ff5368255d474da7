"""Convergence evidence: LDA log-likelihood per epoch, exact vs alias
sampler, at bench scale on one GPU."""

import sys
import time

sys.path.insert(0, ".")

from harmony_amd import mlapps
from harmony_amd.config import JobConfig, RuntimeConfig
from harmony_amd.dolphin.worker import WorkerTasklet
from harmony_amd.runtime.bootstrap import init_executor
from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler


def run(sampler, epochs=8, K=256):
    import torch

    ctx = init_executor(RuntimeConfig(device="auto"))
    job = JobConfig(job_id=f"cv_{sampler}_{K}", app="lda",
                    max_num_epochs=epochs, num_mini_batches=4,
                    app_args={"num_vocabs": 100000, "num_topics": K,
                              "tokens_per_doc": 128, "docs_per_batch": 8192,
                              "sampler": sampler})
    cp = ControlPlane(ctx.store, 0, 1)
    app = mlapps.get_app("lda")
    tables, trainer, provider = app.build(job, ctx, cp)
    tus = TaskUnitScheduler(cp, {job.job_id})
    out = []
    t0 = time.perf_counter()

    def _hook(epoch):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t = time.perf_counter() - t0
        ll = trainer.evaluate_model()["log_likelihood"]
        out.append((epoch, round(t, 3), round(ll / 1e6, 3)))

    trainer.on_epoch_finished = _hook
    WorkerTasklet(job, trainer, provider, cp, tus, 0, 1).run()
    return out


if __name__ == "__main__":
    for sampler in ("exact", "alias", "alias_wave"):
        print(f"== {sampler}")
        for ep, t, ll in run(sampler):
            print(f"epoch {ep}  cum_time {t:8.3f}s  ll {ll:10.3f}M nats")
