"""Property-based fuzz: the dense table vs a plain dict reference model
through random op sequences including migrations (hypothesis)."""

import torch
from hypothesis import given, settings, strategies as st

from harmony_amd.config import TableConfig
from harmony_amd.et.table import Table

NUM_KEYS = 48
VDIM = 3


def mk_table():
    cfg = TableConfig(table_id="fuzz", num_keys=NUM_KEYS, value_dim=VDIM,
                      num_blocks=6, init_fn="zeros", update_fn="add")
    return Table(cfg, 0, 1, torch.device("cpu"))


op_st = st.one_of(
    st.tuples(st.just("put"), st.lists(st.integers(0, NUM_KEYS - 1),
                                       min_size=1, max_size=8),
              st.integers(-5, 5)),
    st.tuples(st.just("update"), st.lists(st.integers(0, NUM_KEYS - 1),
                                          min_size=1, max_size=8),
              st.integers(-3, 3)),
    st.tuples(st.just("remove"), st.lists(st.integers(0, NUM_KEYS - 1),
                                          min_size=1, max_size=4),
              st.just(0)),
    st.tuples(st.just("migrate"), st.lists(st.integers(0, 5), min_size=1,
                                           max_size=3), st.just(0)),
)


@settings(max_examples=60, deadline=None)
@given(st.lists(op_st, min_size=1, max_size=24))
def test_table_matches_dict_model(ops_list):
    t = mk_table()
    ref = {k: [0.0] * VDIM for k in range(NUM_KEYS)}
    for op, keys, val in ops_list:
        if op == "put":
            kt = torch.tensor(sorted(set(keys)))
            t.put(kt, torch.full((kt.numel(), VDIM), float(val)))
            for k in kt.tolist():
                ref[k] = [float(val)] * VDIM
        elif op == "update":
            kt = torch.tensor(keys)  # duplicates allowed: they aggregate
            t.update(kt, torch.full((kt.numel(), VDIM), float(val)))
            for k in keys:
                ref[k] = [x + val for x in ref[k]]
        elif op == "remove":
            kt = torch.tensor(sorted(set(keys)))
            t.remove(kt)
            for k in kt.tolist():
                ref[k] = [0.0] * VDIM   # zeros init
        elif op == "migrate":
            # single-rank "migration": drop + adopt must preserve data
            blocks = sorted(set(keys))
            data = t.drop_blocks(blocks)
            t.adopt_blocks(data)
    got = t.get(torch.arange(NUM_KEYS))
    want = torch.tensor([ref[k] for k in range(NUM_KEYS)])
    assert torch.allclose(got, want), (got, want)


def mk_min_table():
    cfg = TableConfig(table_id="fuzzmin", num_keys=NUM_KEYS, value_dim=VDIM,
                      num_blocks=6, init_fn="zeros", update_fn="min",
                      init_args={})
    return Table(cfg, 0, 1, torch.device("cpu"))


@settings(max_examples=40, deadline=None)
@given(st.lists(st.tuples(st.lists(st.integers(0, NUM_KEYS - 1),
                                   min_size=1, max_size=8),
                          st.integers(-9, 9)),
                min_size=1, max_size=16))
def test_min_update_fn_matches_model(updates):
    # the "min" merge algebra (SSSP-style): duplicates within one push and
    # across pushes must both take elementwise minimums
    t = mk_min_table()
    ref = {k: [0.0] * VDIM for k in range(NUM_KEYS)}
    for keys, val in updates:
        kt = torch.tensor(keys)
        t.update(kt, torch.full((kt.numel(), VDIM), float(val)))
        for k in keys:
            ref[k] = [min(x, float(val)) for x in ref[k]]
    got = t.get(torch.arange(NUM_KEYS))
    want = torch.tensor([ref[k] for k in range(NUM_KEYS)])
    assert torch.allclose(got, want), (got, want)
