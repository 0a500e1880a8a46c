"""JoinResult and join lowering (reference internals/joins.py behavior;
engine semantics: join_tables, graph.rs:482 JoinType)."""

from __future__ import annotations

import enum
from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.config import get_device
from pathway_amd.internals.type_inference import infer_dtype
from pathway_amd.internals.universe import Universe


class JoinMode(enum.Enum):
    INNER = "inner"
    LEFT = "left"
    RIGHT = "right"
    OUTER = "outer"


def _raw_ref(name: str) -> ex.ColumnReference:
    return ex.ColumnReference(None, name)


class Joinable:
    pass


class JoinResult(Joinable):
    def __init__(
        self,
        left,
        right,
        on: list[Any],
        mode: JoinMode,
        assign_id: Any = None,
        left_instance: Any = None,
        right_instance: Any = None,
        probe_only_left: bool = False,
    ):
        self._left = left
        self._right = right
        self._mode = mode if isinstance(mode, JoinMode) else JoinMode(mode)
        self._assign_id = assign_id
        self._probe_only_left = probe_only_left
        self._left_on: list[ex.ColumnExpression] = []
        self._right_on: list[ex.ColumnExpression] = []
        for cond in on:
            cond = thisclass.substitute_this(
                ex.wrap_expr(cond),
                {thisclass.left: left, thisclass.right: right, thisclass.this: left},
            )
            if (
                not isinstance(cond, ex.ColumnBinaryOpExpression)
                or cond._symbol != "=="
            ):
                raise ValueError("join conditions must be equality expressions")
            self._left_on.append(cond._left)
            self._right_on.append(cond._right)
        # columns equated by name in the on-conditions are unified (pw.this.x
        # is unambiguous for them, reference joins.py behavior)
        self._unified: set[str] = set()
        for l, r in zip(self._left_on, self._right_on):
            if (
                isinstance(l, ex.ColumnReference)
                and isinstance(r, ex.ColumnReference)
                and l.name == r.name
            ):
                self._unified.add(l.name)
        if left_instance is not None and right_instance is not None:
            self._left_on.append(
                thisclass.substitute_this(ex.wrap_expr(left_instance), {thisclass.this: left})
            )
            self._right_on.append(
                thisclass.substitute_this(ex.wrap_expr(right_instance), {thisclass.this: right})
            )

    def _key_mode(self) -> str:
        aid = self._assign_id
        if aid is None:
            return "pair"
        if isinstance(aid, ex.ColumnReference):
            if aid.name == "id" and aid.table is self._left:
                return "left"
            if aid.name == "id" and aid.table is self._right:
                return "right"
        if isinstance(aid, type):
            if aid is thisclass.left:
                return "left"
            if aid is thisclass.right:
                return "right"
        return "pair"

    def select(self, *args: Any, **kwargs: Any):
        from pathway_amd.engine.nodes import ExprMapNode
        from pathway_amd.engine.nodes_join import JoinNode
        from pathway_amd.internals.table import Table

        left, right = self._left, self._right
        lmap = {f"l.{n}": n for n in left._dtypes}
        rmap = {f"r.{n}": n for n in right._dtypes}
        node = JoinNode(
            left._node,
            right._node,
            self._left_on,
            self._right_on,
            lmap,
            rmap,
            self._mode.value,
            get_device(),
            key_mode=self._key_mode(),
            left_id_name="l.__id__",
            right_id_name="r.__id__",
            probe_only_left=self._probe_only_left,
        )

        out_exprs: dict[str, ex.ColumnExpression] = {}
        for a in args:
            if isinstance(a, thisclass.ThisSplat):
                src = left if a.cls in (thisclass.left, thisclass.this) else right
                for n in src._dtypes:
                    if n not in a.exclude:
                        out_exprs[n] = ex.ColumnReference(src, n)
                continue
            a = thisclass.substitute_this(
                ex.wrap_expr(a),
                {thisclass.left: left, thisclass.right: right},
            )
            if not isinstance(a, ex.ColumnReference):
                raise ValueError("positional join select args must be column refs")
            out_exprs[a.name] = a
        for name, e in kwargs.items():
            out_exprs[name] = thisclass.substitute_this(
                ex.wrap_expr(e),
                {thisclass.left: left, thisclass.right: right},
            )

        opt_left = self._mode in (JoinMode.RIGHT, JoinMode.OUTER)
        opt_right = self._mode in (JoinMode.LEFT, JoinMode.OUTER)

        mid_dtypes: dict[str, dt.DType] = {}
        for n, t in left._dtypes.items():
            mid_dtypes[f"l.{n}"] = dt.Optional(t) if opt_left else t
        for n, t in right._dtypes.items():
            mid_dtypes[f"r.{n}"] = dt.Optional(t) if opt_right else t
        mid_dtypes["l.__id__"] = dt.Optional(dt.POINTER) if opt_left else dt.POINTER
        mid_dtypes["r.__id__"] = dt.Optional(dt.POINTER) if opt_right else dt.POINTER

        def rewrite(e: ex.ColumnExpression) -> ex.ColumnExpression:
            if isinstance(e, ex.ColumnReference):
                t = e.table
                if t is left or getattr(t, "_node", None) is left._node:
                    return _raw_ref("l.__id__" if e.name == "id" else f"l.{e.name}")
                if t is right or getattr(t, "_node", None) is right._node:
                    return _raw_ref("r.__id__" if e.name == "id" else f"r.{e.name}")
                if t is None:
                    return e
                if isinstance(t, type) and t is thisclass.this:
                    name = e.name
                    in_l = name in left._dtypes
                    in_r = name in right._dtypes
                    if in_l and in_r and name not in self._unified:
                        raise ValueError(f"column {name!r} ambiguous in join select")
                    return _raw_ref(f"l.{name}" if in_l else f"r.{name}")
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
            return new

        final = {name: rewrite(e) for name, e in out_exprs.items()}
        proj = ExprMapNode(node, final, get_device())
        dtypes = {n: infer_dtype(e, mid_dtypes) for n, e in final.items()}
        return Table(proj, dtypes, Universe())

    def _all_columns_table(self):
        """Materialize the join with every column of both sides (prefixed
        names resolved) — backing for filter/groupby/reduce on JoinResult."""
        left, right = self._left, self._right
        sel = {}
        for n in left._dtypes:
            sel[n if n not in right._dtypes or n in self._unified else f"{n}_left"] = (
                ex.ColumnReference(left, n)
            )
        for n in right._dtypes:
            if n in self._unified:
                continue
            sel[n if n not in left._dtypes else f"{n}_right"] = ex.ColumnReference(
                right, n
            )
        return self.select(**sel)

    def filter(self, expression):
        return self._all_columns_table().filter(
            _remap_joined(expression, self._left, self._right, self._unified)
        )

    def groupby(self, *args, **kwargs):
        t = self._all_columns_table()
        args = [_remap_joined(a, self._left, self._right, self._unified) for a in args]
        return t.groupby(*args, **kwargs)

    def reduce(self, *args, **kwargs):
        t = self._all_columns_table()
        args = [_remap_joined(a, self._left, self._right, self._unified) for a in args]
        kwargs = {
            k: _remap_joined(v, self._left, self._right, self._unified)
            for k, v in kwargs.items()
        }
        return t.reduce(*args, **kwargs)


def _remap_joined(e, left, right, unified):
    """Rebind left/right-table refs to the materialized joined table's
    column names (suffix disambiguation as in the reference)."""
    from pathway_amd.internals import expression as exm

    e = thisclass.substitute_this(
        exm.wrap_expr(e), {thisclass.left: left, thisclass.right: right}
    )

    def rec(x):
        if isinstance(x, exm.ColumnReference):
            t = x.table
            if t is left or getattr(t, "_node", None) is getattr(left, "_node", None):
                name = x.name
                if name in right._dtypes and name not in unified:
                    name = f"{name}_left"
                return exm.ColumnReference(None, name)
            if t is right or getattr(t, "_node", None) is getattr(right, "_node", None):
                name = x.name
                if name in left._dtypes and name not in unified:
                    name = f"{name}_right"
                return exm.ColumnReference(None, name)
            return x
        if not isinstance(x, exm.ColumnExpression):
            return x
        new = object.__new__(type(x))
        new.__dict__.update(x.__dict__)
        for attr, val in list(x.__dict__.items()):
            if isinstance(val, exm.ColumnExpression):
                new.__dict__[attr] = rec(val)
            elif isinstance(val, tuple) and any(
                isinstance(v, exm.ColumnExpression) for v in val
            ):
                new.__dict__[attr] = tuple(
                    rec(v) if isinstance(v, exm.ColumnExpression) else v for v in val
                )
        return new

    return rec(e)


def make_ix_table(query, source, pexpr, optional: bool):
    """t.ix(expr): row of `source` addressed by pointer per `query` row."""
    from pathway_amd.engine.nodes_join import JoinNode
    from pathway_amd.internals.table import Table

    rmap = {n: n for n in source._dtypes}
    node = JoinNode(
        query._node,
        source._node,
        [pexpr],
        [ex.ColumnReference(source, "id")],
        {},
        rmap,
        "left",
        get_device(),
        key_mode="left",
    )
    dtypes = {
        n: dt.Optional(t) if optional else t for n, t in source._dtypes.items()
    }
    return Table(node, dtypes, query._universe)
