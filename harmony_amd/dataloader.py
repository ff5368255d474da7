"""Data loading: exactly-N file splits + reference-format parsers.

Reference: common/dataloader (HdfsSplitManager computes exactly-N splits,
ExactNumSplitFileInputFormat.java:44; HdfsDataSet iterates records) and the
evaluator-side bulk loaders (ExistKeyBulkDataLoader.java:40 parses
Pair<K,V> lines, NoneKeyBulkDataLoader assigns local keys). There is no
HDFS here — input is a local path (file:// accepted) on the node; splits
are byte ranges rounded to line boundaries, one per executor, exactly N.

Parsers accept the reference's sample file formats
(jobserver/bin/sample_{nmf,mlr,lda,gbt,lasso}):
  nmf:   "rowId: col,val col,val ..."
  mlr:   "label featIdx:val ..."         (sparse libsvm-like)
  lda:   "word word word ..."            (one doc per line)
  lasso: "label featIdx:val ..."
  gbt:   "label featIdx:val ..." + .meta "featIdx:type ..."
Comment lines (#) and blank lines are skipped.
"""

from __future__ import annotations

import os
from typing import Iterator, List, Tuple

import torch


def _strip_scheme(path: str) -> str:
    return path[len("file://"):] if path.startswith("file://") else path


def compute_splits(path: str, num_splits: int) -> List[Tuple[int, int]]:
    """Exactly num_splits byte ranges aligned to line starts (a line belongs
    to the split its first byte falls in — same rule as text input splits)."""
    path = _strip_scheme(path)
    size = os.path.getsize(path)
    raw = [(size * i) // num_splits for i in range(num_splits + 1)]
    return [(raw[i], raw[i + 1]) for i in range(num_splits)]


def read_split(path: str, split: Tuple[int, int]) -> Iterator[str]:
    """Yield the lines whose first byte lies in [start, end)."""
    path = _strip_scheme(path)
    start, end = split
    with open(path, "rb") as f:
        if start > 0:
            f.seek(start - 1)
            # if the previous byte isn't a newline, the line belongs to the
            # previous split — skip to the next line start
            if f.read(1) != b"\n":
                f.readline()
        while f.tell() < end:
            line = f.readline()
            if not line:
                break
            s = line.decode("utf-8", "replace").strip()
            if s and not s.startswith("#"):
                yield s


# ------------------------------------------------------------------ parsers

def parse_nmf(lines) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """-> (row_ids, col_ids, vals) flat COO (reference NMFETDataParser.java:39)."""
    rows, cols, vals = [], [], []
    for ln in lines:
        head, _, rest = ln.partition(":")
        r = int(head.strip())
        for tok in rest.split():
            c, v = tok.split(",")
            rows.append(r)
            cols.append(int(c))
            vals.append(float(v))
    return (torch.tensor(rows, dtype=torch.int64),
            torch.tensor(cols, dtype=torch.int64),
            torch.tensor(vals, dtype=torch.float32))


def parse_libsvm(lines, num_features: int
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    """-> (X dense [n, F], y [n]) for mlr/gbt/lasso sample formats."""
    xs, ys = [], []
    for ln in lines:
        toks = ln.split()
        ys.append(float(toks[0]))
        row = torch.zeros(num_features)
        for tok in toks[1:]:
            i, v = tok.split(":")
            if int(i) < num_features:
                row[int(i)] = float(v)
        xs.append(row)
    X = torch.stack(xs) if xs else torch.zeros(0, num_features)
    return X, torch.tensor(ys)


def parse_lda(lines) -> Tuple[torch.Tensor, torch.Tensor]:
    """-> (doc_offsets [n+1], word_ids flat) — one doc per line."""
    offsets = [0]
    words: List[int] = []
    for ln in lines:
        ws = [int(t) for t in ln.split()]
        words.extend(ws)
        offsets.append(len(words))
    return (torch.tensor(offsets, dtype=torch.int64),
            torch.tensor(words, dtype=torch.int64))


def parse_gbt_meta(lines) -> dict:
    """featIdx -> type (0 numeric, >0 = #categories)
    (reference GBTMetadataParser.java)."""
    out = {}
    for ln in lines:
        for tok in ln.split():
            i, t = tok.split(":")
            out[int(i)] = int(t)
    return out


def load_rank_lines(path: str, rank: int, world_size: int) -> List[str]:
    """This rank's split of the input file."""
    splits = compute_splits(path, world_size)
    return list(read_split(path, splits[rank]))


# ------------------------------------------------- native fast-path parsing

def _read_split_bytes(path: str, split: Tuple[int, int]) -> bytes:
    """Raw bytes of a split, trimmed to line boundaries (same rule as
    read_split)."""
    path = _strip_scheme(path)
    start, end = split
    with open(path, "rb") as f:
        if start > 0:
            f.seek(start - 1)
            if f.read(1) != b"\n":
                f.readline()
        start = f.tell()
        f.seek(end)
        if end > 0:
            f.seek(end - 1)
            if f.read(1) != b"\n":
                f.readline()
        end = f.tell()
        f.seek(start)
        return f.read(max(0, end - start))


def _native(world_size: int = 1):
    from harmony_amd import ops

    nat = ops._load_hip()
    if nat is not None and "HARMONY_PARSE_THREADS" not in os.environ:
        # one loader process per GPU: cap scan threads to a fair core share
        # (oversubscription measured 5x slower at 8 ranks — ingest_bench.py)
        os.environ["HARMONY_PARSE_THREADS"] = str(
            max(1, (os.cpu_count() or 1) // max(1, world_size)))
    return nat


def parse_nmf_split(path: str, rank: int, world_size: int):
    """Parse this rank's split of a sample_nmf file (native C++ parser when
    the extension is built — multi-threaded byte scan, measured 5-10x the
    Python line parser — else Python)."""
    split = compute_splits(path, world_size)[rank]
    nat = _native(world_size)
    if nat is not None:
        buf = _read_split_bytes(path, split)
        return tuple(nat.parse_nmf_bytes(buf))
    return parse_nmf(read_split(path, split))


def parse_libsvm_split(path: str, rank: int, world_size: int,
                       num_features: int):
    split = compute_splits(path, world_size)[rank]
    nat = _native(world_size)
    if nat is not None:
        buf = _read_split_bytes(path, split)
        return tuple(nat.parse_libsvm_bytes(buf,
                                            num_features))
    return parse_libsvm(read_split(path, split), num_features)


def parse_lda_split(path: str, rank: int, world_size: int):
    split = compute_splits(path, world_size)[rank]
    nat = _native(world_size)
    if nat is not None:
        buf = _read_split_bytes(path, split)
        return tuple(nat.parse_lda_bytes(buf))
    return parse_lda(read_split(path, split))


def load_keyless_split(path: str, rank: int, world_size: int,
                       parse_line=None):
    """Keyless bulk load (reference NoneKeyBulkDataLoader + its local
    key generator, et/evaluator/impl/NoneKeyBulkDataLoader): records with
    no key column get GLOBALLY UNIQUE sequential keys — each rank counts
    every split's records (cheap line scan) so its keys start after all
    earlier splits' records, with no communication.

    Returns (keys int64 tensor, values list) for this rank's split;
    parse_line maps a stripped line to a value (default: the line itself).
    """
    parse_line = parse_line or (lambda s: s)
    splits = compute_splits(path, world_size)
    start = 0
    for r in range(rank):
        start += sum(1 for ln in read_split(path, splits[r]) if ln.strip())
    values = [parse_line(ln.strip()) for ln in read_split(path, splits[rank])
              if ln.strip()]
    keys = torch.arange(start, start + len(values), dtype=torch.int64)
    return keys, values
