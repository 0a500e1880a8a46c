"""GroupedTable: groupby().reduce() lowering (reference internals/groupbys.py:158)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.config import get_device
from pathway_amd.internals.type_inference import infer_dtype
from pathway_amd.internals.universe import Universe


def _raw_ref(name: str) -> ex.ColumnReference:
    return ex.ColumnReference(None, name)


class GroupedTable:
    def __init__(
        self,
        table,
        gb: list[ex.ColumnReference],
        instance: Any = None,
        sort_by: Any = None,
        by_id: Any = None,
    ):
        self._table = table
        self._gb = gb
        self._instance = instance
        self._sort_by = sort_by
        self._by_id = by_id
        self._gb_names = {r.name for r in gb}

    def reduce(self, *args: Any, **kwargs: Any):
        from pathway_amd.engine.nodes import ExprMapNode, GroupReduceNode
        from pathway_amd.internals.table import Table

        table = self._table
        out_exprs: dict[str, ex.ColumnExpression] = {}
        for a in args:
            a = thisclass.substitute_this(ex.wrap_expr(a), {thisclass.this: table})
            if not isinstance(a, ex.ColumnReference):
                raise ValueError("positional reduce arguments must be column references")
            out_exprs[a.name] = a
        for name, e in kwargs.items():
            out_exprs[name] = thisclass.substitute_this(
                ex.wrap_expr(e), {thisclass.this: table}
            )

        group_exprs: dict[str, ex.ColumnExpression] = {}
        for r in self._gb:
            group_exprs[r.name] = r
        if self._instance is not None:
            inst = thisclass.substitute_this(
                ex.wrap_expr(self._instance), {thisclass.this: table}
            )
            group_exprs["_pw_instance"] = inst
            self._gb_names.add("_pw_instance")

        reducer_calls: dict[str, tuple[str, list, dict]] = {}
        counter = [0]

        def rewrite(e: ex.ColumnExpression) -> ex.ColumnExpression:
            if isinstance(e, ex.ReducerExpression):
                rname = e._reducer
                rargs = [
                    thisclass.substitute_this(a, {thisclass.this: table})
                    for a in e._args
                ]
                # engine arg augmentation
                if rname == "sum" and len(rargs) == 1:
                    # non-numeric sums (ndarray columns, reference
                    # Reducer::ArraySum) take the host multiset family
                    d = dt.unoptionalize(infer_dtype(rargs[0], table._dtypes))
                    if d not in (
                        dt.INT, dt.FLOAT, dt.BOOL, dt.DURATION,
                    ):
                        rname = "array_sum"
                if rname in ("argmin", "argmax") and len(rargs) == 1:
                    rargs = [rargs[0], ex.ColumnReference(table, "id")]
                elif rname in ("earliest", "latest", "tuple", "ndarray") and len(rargs) == 1:
                    if self._sort_by is not None:
                        # sort_by overrides arrival order as the ordering
                        # key of order-sensitive reducers (reference
                        # groupby(sort_by=...) semantics)
                        order = thisclass.substitute_this(
                            ex.wrap_expr(self._sort_by), {thisclass.this: table}
                        )
                    else:
                        order = _raw_ref("__seq__")
                    rargs = [rargs[0], order]
                out_name = f"_pw_r{counter[0]}"
                counter[0] += 1
                rkw = dict(e._kwargs)
                if hasattr(e, "_combine_many"):
                    rkw["_combine_many"] = e._combine_many
                if hasattr(e, "_accumulator_cls"):
                    rkw["_accumulator_cls"] = e._accumulator_cls
                reducer_calls[out_name] = (rname, rargs, rkw)
                return _raw_ref(out_name)
            if isinstance(e, ex.ColumnReference):
                if e.table is table or (
                    hasattr(e.table, "_node")
                    and getattr(e.table, "_node", None) is table._node
                ):
                    if e.name == "id":
                        return _raw_ref("id")
                    if e.name in self._gb_names:
                        return _raw_ref(e.name)
                    raise ValueError(
                        f"column {e.name!r} is neither a grouping column nor "
                        "wrapped in a reducer"
                    )
                return e
            new = object.__new__(type(e))
            new.__dict__.update(e.__dict__)
            for attr, val in list(e.__dict__.items()):
                if isinstance(val, ex.ColumnExpression):
                    new.__dict__[attr] = rewrite(val)
                elif isinstance(val, tuple) and any(
                    isinstance(v, ex.ColumnExpression) for v in val
                ):
                    new.__dict__[attr] = tuple(
                        rewrite(v) if isinstance(v, ex.ColumnExpression) else v
                        for v in val
                    )
                elif isinstance(val, dict) and any(
                    isinstance(v, ex.ColumnExpression) for v in val.values()
                ):
                    new.__dict__[attr] = {
                        k: rewrite(v) if isinstance(v, ex.ColumnExpression) else v
                        for k, v in val.items()
                    }
            return new

        final_exprs = {name: rewrite(e) for name, e in out_exprs.items()}

        key_expr = None
        if self._by_id is not None:
            key_expr = self._by_id

        node = GroupReduceNode(
            table._node,
            group_exprs,
            reducer_calls,
            get_device(),
            sort_by=self._sort_by,
        )
        if self._instance is not None:
            node.instance_name = "_pw_instance"
        if key_expr is not None:
            node.key_expr = key_expr

        # dtypes of the intermediate reduce output
        mid_dtypes: dict[str, dt.DType] = {}
        for n, e in group_exprs.items():
            mid_dtypes[n] = infer_dtype(e, table._dtypes)
        for out_name, (rname, rargs, rkw) in reducer_calls.items():
            rexpr = ex.ReducerExpression(rname, *rargs)
            mid_dtypes[out_name] = infer_dtype(rexpr, table._dtypes)

        proj = ExprMapNode(node, final_exprs, get_device())
        dtypes = {n: infer_dtype(e, mid_dtypes) for n, e in final_exprs.items()}
        return Table(proj, dtypes, Universe())
