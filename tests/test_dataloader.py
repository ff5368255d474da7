"""Data loader: exactly-N splits + reference sample-format parsers + file
input through the apps (reference DataLoadTest / ExistKeyBulkDataLoader)."""

import torch

from harmony_amd import dataloader as dl


def _write(tmp_path, name, text):
    p = tmp_path / name
    p.write_text(text)
    return str(p)


def test_splits_cover_all_lines_exactly_once(tmp_path):
    lines = [f"line {i} {'y' * (i % 17 + 1)}" for i in range(101)]
    p = _write(tmp_path, "f.txt", "\n".join(lines) + "\n")
    for n in (1, 2, 3, 7):
        got = []
        splits = dl.compute_splits(p, n)
        assert len(splits) == n
        for s in splits:
            got.extend(dl.read_split(p, s))
        assert got == lines, f"n={n}"


def test_parse_nmf_format(tmp_path):
    p = _write(tmp_path, "nmf.txt",
               "# comment\n1: 1,3 2,1 5,4\n2: 3,2\n")
    rows, cols, vals = dl.parse_nmf(dl.read_split(p, (0, 10 ** 9)))
    assert rows.tolist() == [1, 1, 1, 2]
    assert cols.tolist() == [1, 2, 5, 3]
    assert vals.tolist() == [3.0, 1.0, 4.0, 2.0]


def test_parse_libsvm_format(tmp_path):
    p = _write(tmp_path, "mlr.txt", "5 1:0.5 3:0.25\n0 0:1.0\n")
    X, y = dl.parse_libsvm(dl.read_split(p, (0, 10 ** 9)), num_features=4)
    assert y.tolist() == [5.0, 0.0]
    assert X[0].tolist() == [0.0, 0.5, 0.0, 0.25]
    assert X[1].tolist() == [1.0, 0.0, 0.0, 0.0]


def test_parse_lda_format(tmp_path):
    p = _write(tmp_path, "lda.txt", "95 163 172\n271 367\n")
    off, words = dl.parse_lda(dl.read_split(p, (0, 10 ** 9)))
    assert off.tolist() == [0, 3, 5]
    assert words.tolist() == [95, 163, 172, 271, 367]


def test_parse_gbt_meta(tmp_path):
    p = _write(tmp_path, "m.txt", "0:0 1:0 2:3\n")
    meta = dl.parse_gbt_meta(dl.read_split(p, (0, 10 ** 9)))
    assert meta == {0: 0, 1: 0, 2: 3}


def test_mlr_job_from_file(tmp_path):
    import random

    random.seed(0)
    lines = []
    for i in range(256):
        lab = i % 3
        # separable: feature `lab` is hot, plus a shared noise-ish feature
        lines.append(f"{lab} {lab}:1.0 3:{0.1 * (i % 7)}")
    p = _write(tmp_path, "data.txt", "\n".join(lines) + "\n")

    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    job = JobConfig(job_id="file_mlr", app="mlr", max_num_epochs=3,
                    num_mini_batches=4,
                    app_args={"input": p, "num_classes": 3,
                              "num_features": 4, "num_parts_per_class": 2,
                              "step_size": 1.0})
    ctx = init_executor(RuntimeConfig(device="cpu"))
    s = run_job(job, ctx).summary()
    assert s["num_batches"] == 12
    assert s["total_examples"] == 3 * 256
    assert s["accuracy"] > 0.6


def test_nmf_job_from_file(tmp_path):
    lines = [f"{r}: " + " ".join(f"{c},{(r + c) % 5}" for c in range(8))
             for r in range(64)]
    p = _write(tmp_path, "nmf.txt", "\n".join(lines) + "\n")

    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    job = JobConfig(job_id="file_nmf", app="nmf", max_num_epochs=2,
                    num_mini_batches=2,
                    app_args={"input": p, "num_cols": 8, "rank": 4})
    ctx = init_executor(RuntimeConfig(device="cpu"))
    s = run_job(job, ctx).summary()
    assert s["num_batches"] == 4
    assert s["total_examples"] == 2 * 64


def test_keyless_bulk_loader_global_keys(tmp_path):
    # NoneKeyBulkDataLoader parity: keyless records get globally unique
    # sequential keys; ranks' ranges tile with no gaps or overlaps
    p = tmp_path / "raw.txt"
    p.write_text("".join(f"row{i}\n" for i in range(23)))
    import torch

    from harmony_amd.dataloader import load_keyless_split

    allk, allv = [], []
    for r in range(3):
        k, v = load_keyless_split(str(p), r, 3)
        assert k.numel() == len(v)
        allk.append(k)
        allv.extend(v)
    cat = torch.cat(allk)
    assert torch.equal(cat, torch.arange(23))
    assert allv == [f"row{i}" for i in range(23)]
