"""Chaos: kill an executor mid-run -> fail-fast within the heartbeat
budget; restart the server and resume the job from its epoch checkpoint
(reference recovery story = fail-fast + restart +
checkpoint restore, JobServerDriver.java:271-299 TODO #677)."""

import json
import multiprocessing as mp
import os
import shutil
import threading
import time

from tests.dist_helper import free_port


CHKP = "/tmp/harmony_chaos_chkp"
COMMIT = "/tmp/harmony_chaos_commit"


def _rank_main(rank, world, dist_port, js_port, q):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(dist_port))
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.jobserver import client
    from harmony_amd.jobserver.server import JobServerDriver
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    driver = JobServerDriver(ctx, scheduler="default", port=js_port,
                             hb_period=0.2)   # fail-fast budget = 2 s

    if rank == 1:
        # the victim: die abruptly — but only once at least one epoch
        # checkpoint EXISTS on disk (a wall-clock delay flaked on loaded
        # machines: imports + first epochs could exceed it, leaving no
        # checkpoint to restore from). epoch1's dir appearing means
        # epoch0's snapshot completed.
        def die():
            deadline = time.monotonic() + 60
            marker = os.path.join(CHKP, "chaos1", "epoch1")
            while time.monotonic() < deadline and not os.path.isdir(marker):
                time.sleep(0.1)
            time.sleep(0.5)       # let a couple more epochs through
            os._exit(1)

        threading.Thread(target=die, daemon=True).start()

    if rank == 0:
        def submit():
            time.sleep(0.3)
            job = JobConfig(job_id="chaos1", app="mlr", max_num_epochs=400,
                            num_mini_batches=2, model_chkp_per_epoch=True,
                            chkp_path=CHKP, chkp_commit_path=COMMIT,
                            app_args={"num_classes": 3, "num_features": 64,
                                      "num_parts_per_class": 2,
                                      "batch_size": 128})
            client.submit(job, port=js_port)

        threading.Thread(target=submit, daemon=True).start()

    t0 = time.monotonic()
    driver.run()
    elapsed = time.monotonic() - t0
    if rank == 0:
        q.put({"failed_flag": driver.cp.flag_set("js/failed"),
               "elapsed": elapsed})


def test_chaos_kill_rank_then_restore(tmp_path):
    for d in (CHKP, COMMIT):
        shutil.rmtree(d, ignore_errors=True)
    mp.set_start_method("spawn", force=True)
    dist_port, js_port = free_port(), free_port()
    q = mp.Queue()
    ps = [mp.Process(target=_rank_main, args=(r, 2, dist_port, js_port, q))
          for r in range(2)]
    t0 = time.monotonic()
    for p in ps:
        p.start()
    res = q.get(timeout=120)
    for p in ps:
        p.join(timeout=60)
    total = time.monotonic() - t0
    # rank 1 died at ~3 s; heartbeat budget is 10 * 0.2 = 2 s; the server
    # must have failed fast and shut down well within the test window
    assert res["failed_flag"], res
    assert total < 110, total      # bounded well below the queue timeout
    #                                (fail-fast, not the 600 s clean join)
    # checkpoints from completed epochs exist (both ranks' block files)
    epochs = sorted(os.listdir(os.path.join(CHKP, "chaos1")))
    assert epochs, "no epoch checkpoints written before the kill"

    # ---- restart (fresh world-1 server) + restore from the first epoch
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR",
              "MASTER_PORT"):
        os.environ.pop(k, None)
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="chaos1", app="mlr", max_num_epochs=2,
                    num_mini_batches=2, chkp_path=CHKP,
                    chkp_commit_path=COMMIT, restore_chkp="epoch0",
                    app_args={"num_classes": 3, "num_features": 64,
                              "num_parts_per_class": 2, "batch_size": 128})
    m = run_job(job, ctx)
    s = m.summary()
    assert s["num_batches"] == 4          # resumed and completed
