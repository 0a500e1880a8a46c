"""Engine-side reducer implementations (reference src/engine/reduce.rs:27-46).

Two families, mirroring the reference:
  * additive (semigroup) reducers — count / int_sum / float_sum / avg:
    per-key accumulators merge by addition, retraction is subtraction;
    state is an AdditiveState (segmented-sum per batch, O(changed keys)).
  * multiset reducers — min/max/argmin/argmax/unique/any/tuple/sorted_tuple/
    count_distinct/earliest/latest: state is the full weighted multiset of
    (group, value) rows; aggregates of changed groups are recomputed by
    segmented scans over the group's slice (reference ReducerImpl::combine,
    reduce.rs:126-158).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Callable

import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.engine.column import (
    Column,
    ObjectColumn,
    TensorColumn,
    column_from_pylist,
    infer_and_build_column,
)


@dataclass
class ReducerSpec:
    """Descriptor of one reducer call inside a reduce()."""

    name: str  # count, sum, min, ...
    family: str  # 'additive' | 'multiset' | 'host'
    n_args: int = 1
    # additive: delta accumulators from (value tensors, diffs)
    make_acc: Callable | None = None
    finalize: Callable | None = None
    # multiset: aggregate over a sorted per-group slice
    segment_agg: Callable | None = None
    host_agg: Callable | None = None
    out_dtype: Callable | None = None


def _count_acc(args: list[torch.Tensor], diffs: torch.Tensor) -> torch.Tensor:
    return diffs.to(torch.float64)


def _sum_acc(args: list[torch.Tensor], diffs: torch.Tensor) -> torch.Tensor:
    return args[0].to(torch.float64) * diffs.to(torch.float64)


REDUCERS: dict[str, ReducerSpec] = {}


def _register(spec: ReducerSpec) -> ReducerSpec:
    REDUCERS[spec.name] = spec
    return spec


_register(
    ReducerSpec(
        name="count",
        family="additive",
        n_args=0,
        make_acc=_count_acc,
        finalize=lambda acc, args_dt: TensorColumn(acc.round().to(torch.int64), dt.INT),
        out_dtype=lambda args_dt: dt.INT,
    )
)

_register(
    ReducerSpec(
        name="sum",
        family="additive",
        n_args=1,
        make_acc=_sum_acc,
        finalize=lambda acc, args_dt: (
            TensorColumn(acc.round().to(torch.int64), dt.INT)
            if dt.unoptionalize(args_dt[0]) == dt.INT
            else TensorColumn(acc, dt.FLOAT)
        ),
        out_dtype=lambda args_dt: dt.unoptionalize(args_dt[0]),
    )
)

_register(
    ReducerSpec(
        name="avg",
        family="additive",
        n_args=1,
        make_acc=None,  # handled specially: needs sum and count
        finalize=None,
        out_dtype=lambda args_dt: dt.FLOAT,
    )
)


def _seg_min(values: torch.Tensor, seg: torch.Tensor, nseg: int, argcol=None):
    out = torch.full((nseg,), float("inf"), dtype=torch.float64, device=values.device)
    out.scatter_reduce_(0, seg, values.to(torch.float64), reduce="amin", include_self=True)
    return out


def _seg_max(values: torch.Tensor, seg: torch.Tensor, nseg: int, argcol=None):
    out = torch.full((nseg,), float("-inf"), dtype=torch.float64, device=values.device)
    out.scatter_reduce_(0, seg, values.to(torch.float64), reduce="amax", include_self=True)
    return out


def _multiset(name: str, seg_agg, host_agg, out_dtype, n_args: int = 1):
    _register(
        ReducerSpec(
            name=name,
            family="multiset",
            n_args=n_args,
            segment_agg=seg_agg,
            host_agg=host_agg,
            out_dtype=out_dtype,
        )
    )


# host_agg receives list of (value_tuple, weight) for one group, sorted by value
_multiset(
    "min",
    _seg_min,
    lambda rows: min(r[0][0] for r in rows),
    lambda args_dt: dt.unoptionalize(args_dt[0]),
)
_multiset(
    "max",
    _seg_max,
    lambda rows: max(r[0][0] for r in rows),
    lambda args_dt: dt.unoptionalize(args_dt[0]),
)
_multiset(
    "argmin",
    None,
    lambda rows: min(rows, key=lambda r: r[0][0])[0][1],
    lambda args_dt: dt.POINTER,
    n_args=2,
)
_multiset(
    "argmax",
    None,
    lambda rows: max(rows, key=lambda r: r[0][0])[0][1],
    lambda args_dt: dt.POINTER,
    n_args=2,
)
_multiset(
    "unique",
    None,
    lambda rows: _unique_host(rows),
    lambda args_dt: dt.unoptionalize(args_dt[0]),
)
_multiset(
    "any",
    None,
    lambda rows: rows[0][0][0],
    lambda args_dt: dt.unoptionalize(args_dt[0]),
)
_multiset(
    "sorted_tuple",
    None,
    lambda rows: tuple(
        v
        for r in sorted(rows, key=lambda r: (r[0][0] is not None, r[0][0]))
        for v in [r[0][0]] * r[1]
    ),
    lambda args_dt: dt.List(dt.unoptionalize(args_dt[0])),
)
_multiset(
    "tuple",
    None,
    lambda rows: tuple(
        v
        for r in sorted(rows, key=lambda r: (r[0][1] is None, r[0][1]))
        for v in [r[0][0]] * r[1]
    ),
    lambda args_dt: dt.List(dt.unoptionalize(args_dt[0])),
    n_args=2,  # (value, order_key)
)
_multiset(
    "count_distinct",
    None,
    lambda rows: len({r[0][0] for r in rows}),
    lambda args_dt: dt.INT,
)
_multiset(
    "count_distinct_approximate",
    None,
    lambda rows: len({r[0][0] for r in rows}),
    lambda args_dt: dt.INT,
)
_multiset(
    "earliest",
    None,
    lambda rows: min(rows, key=lambda r: r[0][1])[0][0],
    lambda args_dt: dt.unoptionalize(args_dt[0]),
    n_args=2,  # (value, arrival_seq)
)
_multiset(
    "latest",
    None,
    lambda rows: max(rows, key=lambda r: r[0][1])[0][0],
    lambda args_dt: dt.unoptionalize(args_dt[0]),
    n_args=2,
)
_multiset(
    "ndarray",
    None,
    lambda rows: _ndarray_host(sorted(rows, key=lambda r: (r[0][1] is None, r[0][1]))),
    lambda args_dt: dt.Array(),
    n_args=2,
)
_multiset(
    # sum over non-numeric-column values (ndarrays etc. — reference
    # Reducer::ArraySum, reduce.rs); weight-scaled so retractions cancel
    "array_sum",
    None,
    lambda rows: _array_sum_host(rows),
    lambda args_dt: dt.unoptionalize(args_dt[0]),
)


def _array_sum_host(rows):
    tot = None
    for r in rows:
        v, w = r[0][0], r[1]
        if v is None:
            continue
        contrib = v * w
        tot = contrib if tot is None else tot + contrib
    return tot


def _unique_host(rows):
    vals = {r[0][0] for r in rows}
    if len(vals) != 1:
        raise ValueError(f"unique() on non-unique values: {sorted(map(repr, vals))}")
    return next(iter(vals))


def _ndarray_host(rows):
    import numpy as np

    ordered = sorted(rows, key=lambda r: r[0][1])
    return np.array([v for r in ordered for v in [r[0][0]] * r[1]])


@dataclass
class StatefulReducerSpec(ReducerSpec):
    """Custom python reducer (pw.reducers.stateful_many / udf_reducer)."""

    combine_many: Callable | None = None
