"""Server-side update/init functions (the PS "apply" semantics).

Reference: et/evaluator/api/UpdateFunction.java:25 and the per-app
implementations ({Nmf,Mlr,Lda,Gbt,Lasso}ETModelUpdateFunction). In the
reference every update runs one key at a time inside BlockImpl.update
(BlockImpl.java:71). Here updates are applied as *batched tensor ops* on the
owner GPU: deltas arriving from all workers for one superstep are aggregated
(sum) first, then the update function runs once per key — a fused epilogue of
the reduce-scatter / all-to-all push. Hot paths dispatch to HIP kernels via
harmony_amd.ops; the torch implementations below are the fp32 reference and
the CPU path.

An update function must be well-defined for an *aggregated* delta: all
reference update functions (axpy+clamp, add, sparse count merge) satisfy
delta-merge associativity.
"""

from __future__ import annotations

from typing import Any, Callable, Dict

import torch

# ---------------------------------------------------------------------------
# init functions: (shape, dtype, device, **args) -> tensor
# ---------------------------------------------------------------------------

_INIT_FNS: Dict[str, Callable[..., torch.Tensor]] = {}


def register_init(name: str):
    def deco(fn):
        _INIT_FNS[name] = fn
        return fn

    return deco


def init_fn(name: str) -> Callable[..., torch.Tensor]:
    return _INIT_FNS[name]


@register_init("zeros")
def _init_zeros(shape, dtype, device, **_):
    return torch.zeros(shape, dtype=dtype, device=device)


@register_init("gaussian")
def _init_gaussian(shape, dtype, device, std: float = 0.01, seed: int = 0, **_):
    # Per-table deterministic init (reference MLR init ~ N(0, gaussian),
    # MLRETModelUpdateFunction.java:50-62).
    g = torch.Generator(device="cpu").manual_seed(seed)
    t = torch.randn(shape, generator=g, dtype=torch.float32) * std
    return t.to(dtype=dtype, device=device)


@register_init("uniform_clamped")
def _init_uniform_clamped(shape, dtype, device, max_val: float = 1.0, seed: int = 0, **_):
    # Reference NMFModelGenerator: random init clamped to valid (>=0) range.
    g = torch.Generator(device="cpu").manual_seed(seed)
    t = torch.rand(shape, generator=g, dtype=torch.float32) * max_val
    return t.to(dtype=dtype, device=device)


# ---------------------------------------------------------------------------
# update functions: (values, agg_delta, **args) -> new values (may be in-place)
# values: [n, value_dim] owner rows; agg_delta: [n, value_dim] summed deltas
# ---------------------------------------------------------------------------

_UPDATE_FNS: Dict[str, Callable[..., torch.Tensor]] = {}


def register_update(name: str):
    def deco(fn):
        _UPDATE_FNS[name] = fn
        return fn

    return deco


def update_fn(name: str) -> Callable[..., torch.Tensor]:
    return _UPDATE_FNS[name]


def has_update_fn(name: str) -> bool:
    return name in _UPDATE_FNS


@register_update("add")
def _update_add(values: torch.Tensor, delta: torch.Tensor, **_) -> torch.Tensor:
    # Reference MLR/Lasso servers: old.addi(delta)
    # (MLRETModelUpdateFunction.java:60-62, LassoETModelUpdateFunction.java:33).
    values.add_(delta)
    return values


@register_update("assign")
def _update_assign(values: torch.Tensor, delta: torch.Tensor, **_) -> torch.Tensor:
    values.copy_(delta)
    return values


@register_update("nmf_sgd")
def _update_nmf(values: torch.Tensor, delta: torch.Tensor, step_size: float = 0.01,
                max_val: float = 1e6, **_) -> torch.Tensor:
    # new = clamp(old - step*delta, 0, max) — reference
    # NMFETModelUpdateFunction.java:48-52 (step applied server-side, values
    # clamped to the valid non-negative range).
    values.add_(delta, alpha=-step_size).clamp_(0.0, max_val)
    return values


@register_update("lda_counts")
def _update_lda(values: torch.Tensor, delta: torch.Tensor, **_) -> torch.Tensor:
    # Topic-count merge with clamp >= 0 — reference
    # LDAETModelUpdateFunction.java:43-64 merges (topicIdx, +/-count) pairs and
    # clamps at zero. Rows here are dense int32 topic-count vectors.
    values.add_(delta).clamp_(min=0)
    return values


@register_update("min")
def _update_min(values: torch.Tensor, delta: torch.Tensor, **_) -> torch.Tensor:
    # Min-combiner (Pregel shortest-path message semantics).
    torch.minimum(values, delta.to(values.dtype), out=values)
    return values


# How duplicate-key deltas are merged BEFORE the update function applies
# (push aggregation in et/comm.py). Must match the update function's algebra:
# f(f(v,a),b) == f(v, merge(a,b)).
MERGE_MODE = {
    "add": "sum",
    "assign": "last",
    "nmf_sgd": "sum",
    "lda_counts": "sum",
    "min": "min",
}


def merge_key_deltas(keys: torch.Tensor, deltas: torch.Tensor,
                     update_fn_name: str):
    """Aggregate duplicate keys per the update function's merge algebra ->
    (unique_keys, merged_deltas)."""
    mode = MERGE_MODE.get(update_fn_name, "sum")
    uniq, inv = torch.unique(keys, return_inverse=True)
    if mode == "last":
        out = torch.empty((uniq.shape[0], deltas.shape[1]), dtype=deltas.dtype,
                          device=deltas.device)
        out[inv] = deltas          # later duplicates win (any is valid)
        return uniq, out
    if mode == "min":
        out = torch.full((uniq.shape[0], deltas.shape[1]),
                         torch.finfo(deltas.dtype).max
                         if deltas.is_floating_point() else
                         torch.iinfo(deltas.dtype).max,
                         dtype=deltas.dtype, device=deltas.device)
        out.scatter_reduce_(0, inv.unsqueeze(1).expand_as(deltas), deltas,
                            reduce="amin")
        return uniq, out
    out = torch.zeros((uniq.shape[0], deltas.shape[1]), dtype=deltas.dtype,
                      device=deltas.device)
    out.index_add_(0, inv, deltas)
    return uniq, out


ALL_INIT = _INIT_FNS
ALL_UPDATE = _UPDATE_FNS
