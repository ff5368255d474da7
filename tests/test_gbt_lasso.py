import torch

from harmony_amd.config import JobConfig, RuntimeConfig
from harmony_amd.dolphin.master import run_job
from harmony_amd.runtime.bootstrap import init_executor
from tests.dist_helper import run_dist


def test_lasso_recovers_sparse_signal():
    job = JobConfig(job_id="t_lasso", app="lasso", max_num_epochs=4,
                    num_mini_batches=2,
                    app_args={"num_features": 64, "num_parts": 8,
                              "batch_size": 512, "lam": 0.02,
                              "density": 0.2, "noise": 0.05})
    ctx = init_executor(RuntimeConfig(device="cpu"))
    m = run_job(job, ctx)
    s = m.summary()
    assert s["num_batches"] == 8
    # converged to small residual: mse near noise level
    assert s["mse"] < 0.2, s


def test_gbt_fits_nonlinear_signal():
    job = JobConfig(job_id="t_gbt", app="gbt", max_num_epochs=8,
                    num_mini_batches=2,
                    app_args={"num_features": 8, "batch_size": 1024,
                              "num_bins": 32, "max_depth": 5,
                              "step_size": 0.3, "noise": 0.05})
    ctx = init_executor(RuntimeConfig(device="cpu"))
    m = run_job(job, ctx)
    s = m.summary()
    assert s["num_batches"] == 16
    # replicated forest includes the last push (r02 tensorized sync
    # appends at push time; the r01 pull-lag reported 15)
    assert s["num_trees"] == 16
    # boosting reduced mse well below the signal variance (~9)
    assert s["mse"] < 1.5, s


def test_gbt_tree_predict_routing():
    from harmony_amd.mlapps.gbt import GBTree

    t = GBTree(depth=2, feature=[0, 1, 1], threshold=[2, 1, -1],
               leaf_value=[10.0, 20.0, 30.0, 40.0])
    bins = torch.tensor([[0, 0], [0, 5], [5, 0], [5, 9]])
    out = t.predict_bins(bins)
    # f0<=2 -> left subtree: f1<=1 ? leaf0 : leaf1
    # f0>2  -> right subtree: threshold -1 -> always left -> leaf2
    assert out.tolist() == [10.0, 20.0, 30.0, 30.0]


def _gbt_2rank_worker(rank, world):
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="t_gbt2", app="gbt", max_num_epochs=2,
                    num_mini_batches=2,
                    app_args={"num_features": 8, "batch_size": 256,
                              "num_bins": 16, "max_depth": 3})
    m = run_job(job, ctx)
    s = m.summary()
    # each batch, BOTH ranks push one tree -> forest grows by 2 per batch
    return (s["num_batches"], s["num_trees"])


def test_gbt_two_ranks_forest_shared():
    res = run_dist(_gbt_2rank_worker, world=2, timeout=120)
    for nb, nt in res:
        assert nb == 4
        # r02 replicated forest includes the final collective push:
        # 4 batches * 2 ranks (r01's pull-lag view saw 6)
        assert nt == 8


def test_gbt_incremental_pred_matches_full():
    # the append-only prediction cache must give the same residuals as a
    # full forest re-predict, and must reset when the forest is replaced
    import torch

    from harmony_amd.mlapps.gbt import GBTree

    torch.manual_seed(0)
    bins = torch.randint(0, 16, (256, 8))
    y = torch.randn(256)
    trees = []
    for i in range(5):
        t = GBTree(2, [i % 8, (i + 1) % 8, (i + 2) % 8], [3, 7, 11],
                   [0.1 * i, -0.2, 0.3, 0.05])
        trees.append(t)
    full = torch.zeros_like(y)
    for t in trees:
        full = full + 0.1 * t.predict_bins(bins)

    class _Tr:
        pass

    from harmony_amd.mlapps.gbt import GBTTrainer

    tr = _Tr()
    tr._pred_cache = {}
    tr.forest = []
    tr.batch = (bins, y)
    tr.a = {"step_size": 0.1, "num_bins": 16, "max_depth": 2, "lam": 1.0,
            "objective": "regression"}
    inc = None
    for k in (2, 4, 5):            # grow the forest incrementally
        tr.forest = trees[:k]
        GBTTrainer.local_compute(tr)
        inc = tr._mse
    want = float(((y - full) ** 2).mean())
    assert abs(inc - want) < 1e-5
    # replace the forest with a *different* object list of the same length:
    # cache must reset, not reuse
    clones = [GBTree(t.depth, t.feature, t.threshold, t.leaf_value)
              for t in trees]
    tr.forest = clones
    GBTTrainer.local_compute(tr)
    assert abs(tr._mse - want) < 1e-5


def test_gbt_multiclass_end_to_end():
    # one forest per label (reference multi-label GBT): error rate falls
    # well below chance on separable synthetic classes
    import torch

    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="gbt_mc", app="gbt", max_num_epochs=6,
                    num_mini_batches=2,
                    app_args={"num_features": 8, "batch_size": 1024,
                              "num_bins": 32, "max_depth": 3,
                              "step_size": 0.5, "objective": "multiclass",
                              "num_classes": 3, "noise": 0.1})
    s = run_job(job, ctx).summary()
    assert s["num_batches"] == 12
    assert s["error_rate"] < 0.25, s       # chance = 0.67 for 3 classes
    # C forests, one tree per batch each; replica includes the last push
    assert s["num_trees"] == 3 * 12
