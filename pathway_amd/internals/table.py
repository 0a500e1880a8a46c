"""pw.Table — the user-facing API (reference internals/table.py:53-2760).

Tables are handles on eagerly-built engine nodes; every method composes new
nodes (see pathway_amd/engine/nodes*.py for operator semantics).  Deferred
semantics are preserved: nothing computes until pw.run()/pw.debug.* drives
the Runtime.
"""

from __future__ import annotations

from typing import Any, Iterable, Mapping, Sequence

from pathway_amd.internals import dtype as dt
from pathway_amd.internals import expression as ex
from pathway_amd.internals import thisclass
from pathway_amd.internals.config import get_device
from pathway_amd.internals.schema import SchemaMetaclass, schema_from_types
from pathway_amd.internals.type_inference import infer_dtype
from pathway_amd.internals.universe import Universe


class TableLike:
    _universe: Universe


def _substitute(expr: Any, mapping: dict[type, "Table"]) -> Any:
    return thisclass.substitute_this(expr, mapping)


def _check_refs(expr: ex.ColumnExpression, table: "Table") -> None:
    for ref in expr._col_refs():
        t = ref.table
        if isinstance(t, type):
            raise ValueError(f"unresolved this/left/right reference {ref!r}")
        if t is None:
            continue  # engine-internal reference
        if isinstance(t, Table):
            if t._node is table._node:
                continue
            if t._universe.is_equal(table._universe):
                # same key-set: resolution by name against this table's batch
                # is sound when the column exists here too
                if ref.name == "id" or ref.name in table._dtypes:
                    continue
                raise NotImplementedError(
                    "referencing columns of another (universe-equal) table in this "
                    "operation is not supported yet; select them into one table first"
                )
            raise ValueError(
                f"column {ref.name!r} of a table with a different universe used here"
            )


class Table(TableLike):
    def __init__(self, node, dtypes: dict[str, dt.DType], universe: Universe):
        self._node = node
        self._dtypes = dict(dtypes)
        self._universe = universe

    # -- construction ------------------------------------------------------

    @staticmethod
    def _from_node(node, dtypes: dict[str, dt.DType], universe: Universe) -> "Table":
        return Table(node, dtypes, universe)

    @staticmethod
    def empty(**kwargs: Any) -> "Table":
        from pathway_amd.engine.nodes import InputNode
        from pathway_amd.engine.runtime import StaticSource

        dtypes = {n: dt.wrap(t) for n, t in kwargs.items()}
        src = StaticSource([], list(dtypes.keys()), list(dtypes.values()))
        node = InputNode(src, get_device())
        return Table(node, dtypes, Universe())

    # -- schema / columns --------------------------------------------------

    @property
    def schema(self) -> SchemaMetaclass:
        return schema_from_types(**{n: t for n, t in self._dtypes.items()})

    @property
    def id(self) -> ex.ColumnReference:
        return ex.ColumnReference(self, "id")

    def column_names(self) -> list[str]:
        return list(self._dtypes.keys())

    def keys(self) -> list[str]:
        return self.column_names()

    def typehints(self) -> dict[str, Any]:
        return {n: t.typehint for n, t in self._dtypes.items()}

    def __getattr__(self, name: str) -> ex.ColumnReference:
        dtypes = self.__dict__.get("_dtypes", {})
        if name not in dtypes:
            raise AttributeError(
                f"table has no column {name!r}; columns: {list(dtypes)}"
            )
        return ex.ColumnReference(self, name)

    def __getitem__(self, arg):
        if isinstance(arg, str):
            if arg == "id":
                return self.id
            if arg not in self._dtypes:
                raise KeyError(arg)
            return ex.ColumnReference(self, arg)
        if isinstance(arg, ex.ColumnReference):
            return self[arg.name]
        if isinstance(arg, (list, tuple)):
            return self.select(*[self[c] for c in arg])
        raise TypeError(f"cannot index table with {arg!r}")

    def __iter__(self):
        raise TypeError("Table is not iterable; use pw.debug.compute_and_print")

    def __repr__(self) -> str:
        cols = ", ".join(f"{n}: {t!r}" for n, t in self._dtypes.items())
        return f"<pathway_amd.Table {{{cols}}}>"

    # -- core ops ----------------------------------------------------------

    def _resolve(self, expr: Any) -> ex.ColumnExpression:
        expr = ex.wrap_expr(expr)
        expr = _substitute(expr, {thisclass.this: self})
        _check_refs(expr, self)
        return expr

    def _named_exprs(
        self, args: Sequence[Any], kwargs: Mapping[str, Any]
    ) -> dict[str, ex.ColumnExpression]:
        out: dict[str, ex.ColumnExpression] = {}
        for a in args:
            if isinstance(a, thisclass.ThisSplat):
                for n in self._dtypes:
                    if n not in a.exclude:
                        out[n] = ex.ColumnReference(self, n)
                continue
            if isinstance(a, thisclass.ThisMetaclass):
                raise TypeError("pass pw.this.column, not pw.this")
            a = _substitute(ex.wrap_expr(a), {thisclass.this: self})
            if isinstance(a, ex.ColumnReference):
                if isinstance(a.table, Table) and a.table is not self and a.table._node is self._node:
                    pass
                out[a.name] = a
            else:
                raise ValueError(
                    "positional select arguments must be column references; "
                    "use keyword arguments for expressions"
                )
        for name, e in kwargs.items():
            out[name] = _substitute(ex.wrap_expr(e), {thisclass.this: self})
        return out

    def _lower_exprs(self, exprs: dict[str, ex.ColumnExpression]):
        """Rewrite references to other (universe-equal) tables into aligned
        extra-input references; returns (rewritten exprs, extra_inputs,
        dtypes for inference)."""
        extra: dict[int, tuple[Any, str, "Table"]] = {}
        infer_dtypes = dict(self._dtypes)

        def rewrite(e):
            if isinstance(e, ex.ColumnReference):
                t = e.table
                if isinstance(t, Table) and t._node is not self._node:
                    if not t._universe.is_equal(self._universe):
                        raise ValueError(
                            f"column {e.name!r} of a table with a different "
                            "universe used in select()"
                        )
                    key = id(t._node)
                    if key not in extra:
                        extra[key] = (t._node, f"__x{len(extra)}.", t)
                    prefix = extra[key][1]
                    if e.name == "id":
                        return ex.ColumnReference(None, "id")
                    infer_dtypes[f"{prefix}{e.name}"] = t._dtypes.get(e.name, dt.ANY)
                    return ex.ColumnReference(None, f"{prefix}{e.name}")
                return e
            if not isinstance(e, ex.ColumnExpression):
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

        rewritten = {n: rewrite(e) for n, e in exprs.items()}
        extra_inputs = [(node, prefix) for node, prefix, _ in extra.values()]
        return rewritten, extra_inputs, infer_dtypes

    def select(self, *args: Any, **kwargs: Any) -> "Table":
        from pathway_amd.engine.nodes import ExprMapNode

        exprs = self._named_exprs(args, kwargs)
        rewritten, extra_inputs, infer_dtypes = self._lower_exprs(exprs)
        node = ExprMapNode(self._node, rewritten, get_device(), extra_inputs)
        dtypes = {n: infer_dtype(e, infer_dtypes) for n, e in rewritten.items()}
        return Table(node, dtypes, self._universe)

    def with_columns(self, *args: Any, **kwargs: Any) -> "Table":
        new = self._named_exprs(args, kwargs)
        exprs: dict[str, ex.ColumnExpression] = {
            n: ex.ColumnReference(self, n) for n in self._dtypes
        }
        exprs.update(new)
        from pathway_amd.engine.nodes import ExprMapNode

        rewritten, extra_inputs, infer_dtypes = self._lower_exprs(exprs)
        node = ExprMapNode(self._node, rewritten, get_device(), extra_inputs)
        dtypes = {n: infer_dtype(e, infer_dtypes) for n, e in rewritten.items()}
        return Table(node, dtypes, self._universe)

    def filter(self, filter_expression: Any) -> "Table":
        from pathway_amd.engine.nodes import FilterNode

        pred = self._resolve(filter_expression)
        node = FilterNode(self._node, pred, get_device())
        return Table(node, self._dtypes, self._universe.subuniverse())

    def split(self, split_expression: Any) -> tuple["Table", "Table"]:
        pos = self.filter(split_expression)
        neg = self.filter(~ex.wrap_expr(self._resolve(split_expression)))
        return pos, neg

    @property
    def C(self) -> "_ColumnAccessor":  # noqa: N802
        """Typed column accessor (reference Table.C): ``t.C.name`` is
        ``t.name`` — useful when a column name collides with a Table
        method."""
        return _ColumnAccessor(self)

    @property
    def _C(self) -> "_ColumnAccessor":  # noqa: N802
        return self.C

    def eval_type(self, expression: Any) -> dt.DType:
        """Static dtype of an expression over this table (reference
        table.py eval_type)."""
        from pathway_amd.internals.type_inference import infer_dtype

        return infer_dtype(self._resolve(expression), self._dtypes)

    def __add__(self, other: "Table") -> "Table":
        """Column-wise union of two same-universe tables (reference
        Table.__add__); right-hand columns win on name clash."""
        exprs = {n: ex.ColumnReference(self, n) for n in self._dtypes}
        exprs.update({n: ex.ColumnReference(other, n) for n in other._dtypes})
        return self.select(**exprs)

    def copy(self) -> "Table":
        # a DISTINCT node (same keys/values): self-joins resolve join
        # sides by node identity, so t.join(t.copy(), ...) must see two
        # different nodes (reference Table.copy gives a fresh table)
        return self.select(
            **{n: ex.ColumnReference(self, n) for n in self._dtypes}
        )

    # -- renames / drops ---------------------------------------------------

    def rename_columns(self, **kwargs: str) -> "Table":
        mapping = {}
        for new, old in kwargs.items():
            old_name = old.name if isinstance(old, ex.ColumnReference) else old
            mapping[old_name] = new
        exprs = {}
        for n in self._dtypes:
            exprs[mapping.get(n, n)] = ex.ColumnReference(self, n)
        return self.select(**exprs)

    def rename_by_dict(self, names_mapping: Mapping[Any, str]) -> "Table":
        mapping = {
            (k.name if isinstance(k, ex.ColumnReference) else k): v
            for k, v in names_mapping.items()
        }
        exprs = {}
        for n in self._dtypes:
            exprs[mapping.get(n, n)] = ex.ColumnReference(self, n)
        return self.select(**exprs)

    def rename(self, names_mapping: Mapping[Any, str] | None = None, **kwargs: str) -> "Table":
        if names_mapping is not None:
            return self.rename_by_dict(names_mapping)
        return self.rename_columns(**kwargs)

    def without(self, *columns: Any) -> "Table":
        drop = {c.name if isinstance(c, ex.ColumnReference) else c for c in columns}
        exprs = {
            n: ex.ColumnReference(self, n) for n in self._dtypes if n not in drop
        }
        return self.select(**exprs)

    # -- typing ------------------------------------------------------------

    def update_types(self, **kwargs: Any) -> "Table":
        exprs = {}
        for n in self._dtypes:
            if n in kwargs:
                exprs[n] = ex.DeclareTypeExpression(ex.ColumnReference(self, n), kwargs[n])
            else:
                exprs[n] = ex.ColumnReference(self, n)
        return self.select(**exprs)

    def cast_to_types(self, **kwargs: Any) -> "Table":
        exprs = {}
        for n in self._dtypes:
            if n in kwargs:
                exprs[n] = ex.CastExpression(ex.ColumnReference(self, n), kwargs[n])
            else:
                exprs[n] = ex.ColumnReference(self, n)
        return self.select(**exprs)

    # -- groupby / reduce ---------------------------------------------------

    def groupby(
        self,
        *args: Any,
        id: Any = None,
        sort_by: Any = None,
        _filter_out_results_of_forgetting: bool = False,
        instance: Any = None,
        **kwargs: Any,
    ):
        from pathway_amd.internals.groupbys import GroupedTable

        gb: list[ex.ColumnReference] = []
        for a in args:
            a = _substitute(ex.wrap_expr(a), {thisclass.this: self})
            if not isinstance(a, ex.ColumnReference):
                raise ValueError("groupby arguments must be column references")
            gb.append(a)
        if id is not None:
            idref = _substitute(ex.wrap_expr(id), {thisclass.this: self})
            return GroupedTable(self, gb, instance=instance, sort_by=sort_by, by_id=idref)
        return GroupedTable(self, gb, instance=instance, sort_by=sort_by)

    def reduce(self, *args: Any, **kwargs: Any) -> "Table":
        return self.groupby().reduce(*args, **kwargs)

    def deduplicate(
        self,
        *,
        value: Any,
        instance: Any = None,
        acceptor: Any = None,
        persistent_id: str | None = None,
        name: str | None = None,
    ) -> "Table":
        from pathway_amd.engine.nodes_dedup import DeduplicateNode

        vexpr = self._resolve(value)
        iexpr = self._resolve(instance) if instance is not None else None
        node = DeduplicateNode(self._node, vexpr, iexpr, acceptor, list(self._dtypes), get_device())
        return Table(node, self._dtypes, Universe())

    # -- joins --------------------------------------------------------------

    def join(self, other: "Table", *on: Any, id: Any = None, how: Any = None, **kwargs: Any):
        from pathway_amd.internals.joins import JoinMode, JoinResult

        mode = how if how is not None else JoinMode.INNER
        return JoinResult(self, other, list(on), mode, assign_id=id, **kwargs)

    def join_inner(self, other: "Table", *on: Any, id: Any = None, **kwargs: Any):
        from pathway_amd.internals.joins import JoinMode, JoinResult

        return JoinResult(self, other, list(on), JoinMode.INNER, assign_id=id, **kwargs)

    def join_left(self, other: "Table", *on: Any, id: Any = None, **kwargs: Any):
        from pathway_amd.internals.joins import JoinMode, JoinResult

        return JoinResult(self, other, list(on), JoinMode.LEFT, assign_id=id, **kwargs)

    def join_right(self, other: "Table", *on: Any, id: Any = None, **kwargs: Any):
        from pathway_amd.internals.joins import JoinMode, JoinResult

        return JoinResult(self, other, list(on), JoinMode.RIGHT, assign_id=id, **kwargs)

    def join_outer(self, other: "Table", *on: Any, id: Any = None, **kwargs: Any):
        from pathway_amd.internals.joins import JoinMode, JoinResult

        return JoinResult(self, other, list(on), JoinMode.OUTER, assign_id=id, **kwargs)

    # -- set ops -------------------------------------------------------------

    def concat(self, *others: "Table") -> "Table":
        from pathway_amd.engine.nodes import ConcatNode

        tables = [self, *others]
        names = list(self._dtypes.keys())
        for t in others:
            if list(t._dtypes.keys()) != names:
                t_cols = set(t._dtypes)
                if t_cols != set(names):
                    raise ValueError("concat requires identical column sets")
        node = ConcatNode([t._node for t in tables], get_device())
        dtypes = {
            n: dt.types_lca(
                self._dtypes[n],
                others[0]._dtypes[n] if others else self._dtypes[n],
            )
            if others
            else self._dtypes[n]
            for n in names
        }
        # concat output universe = union of the inputs' universes
        # (solver: every part is a subset; union ⊆ any common superset)
        u = Universe.union_of(*[t._universe for t in tables])
        return Table(node, dtypes, u)

    def concat_reindex(self, *others: "Table") -> "Table":
        from pathway_amd.engine.nodes import ConcatNode, DeriveKeyNode

        tables = [self, *others]
        renamed = [DeriveKeyNode(t._node, f"concat_part{i}", get_device()) for i, t in enumerate(tables)]
        node = ConcatNode(renamed, get_device())
        return Table(node, self._dtypes, Universe())

    def update_rows(self, other: "Table") -> "Table":
        from pathway_amd.engine.nodes_join import KeyedMergeNode

        node = KeyedMergeNode(self._node, other._node, "rows", list(self._dtypes), get_device())
        dtypes = {
            n: dt.types_lca(self._dtypes[n], other._dtypes.get(n, self._dtypes[n]))
            for n in self._dtypes
        }
        return Table(node, dtypes, Universe())

    def update_cells(self, other: "Table") -> "Table":
        from pathway_amd.engine.nodes_join import KeyedMergeNode

        override = [n for n in other._dtypes if n in self._dtypes]
        node = KeyedMergeNode(self._node, other._node, "cells", override, get_device())
        return Table(node, self._dtypes, self._universe)

    def __lshift__(self, other: "Table") -> "Table":
        return self.update_cells(other)

    def intersect(self, *tables: "Table") -> "Table":
        from pathway_amd.engine.nodes_join import SemiJoinNode

        out = self
        for t in tables:
            n = SemiJoinNode(out._node, t._node, "intersect", get_device())
            u = Universe.intersection_of(out._universe, t._universe)
            out = Table(n, out._dtypes, u)
        return out

    def difference(self, other: "Table") -> "Table":
        from pathway_amd.engine.nodes_join import SemiJoinNode

        node = SemiJoinNode(self._node, other._node, "difference", get_device())
        u = Universe.difference_of(self._universe, other._universe)
        return Table(node, self._dtypes, u)

    def restrict(self, other: TableLike) -> "Table":
        from pathway_amd.engine.nodes_join import SemiJoinNode

        node = SemiJoinNode(self._node, other._node, "intersect", get_device())  # type: ignore[attr-defined]
        return Table(node, self._dtypes, other._universe)

    def having(self, *indexers: Any) -> "Table":
        """Restrict to rows whose id appears in the indexer pointer columns
        (reference table.py having)."""
        out = self
        for ixr in indexers:
            if isinstance(ixr, ex.ColumnReference) and isinstance(ixr.table, Table):
                src = ixr.table
                if self._dtypes.get(ixr.name) is not None and ixr.name in self._dtypes:
                    pass
                # pointer column: key the indexer table by the pointer value
                if isinstance(src._dtypes.get(ixr.name), dt.Pointer):
                    keyed = src.select(_pw_key=ixr).with_id_from_expr(
                        ex.ColumnReference(None, "_pw_key")
                    )
                else:
                    # value column: ids of self are hashes of these values
                    keyed = src.select(_pw_key=ixr).with_id_from(
                        ex.ColumnReference(None, "_pw_key")
                    )
                from pathway_amd.engine.nodes_join import SemiJoinNode

                node = SemiJoinNode(out._node, keyed._node, "intersect", get_device())
                out = Table(node, out._dtypes, out._universe.subuniverse())
            else:
                raise TypeError("having() expects column references")
        return out

    def with_universe_of(self, other: TableLike) -> "Table":
        self._universe.promise_equal(other._universe)
        return Table(self._node, self._dtypes, other._universe)

    def unsafe_promise_universes_are_equal(self, other: TableLike) -> "Table":
        """Deprecated alias of with_universe_of (reference table.py)."""
        return self.with_universe_of(other)

    # -- keys ---------------------------------------------------------------

    def pointer_from(self, *args: Any, optional: bool = False, instance: Any = None):
        rargs = [self._resolve(a) for a in args]
        ri = self._resolve(instance) if instance is not None else None
        return ex.PointerExpression(self, *rargs, optional=optional, instance=ri)

    def with_id_from(self, *args: Any, instance: Any = None) -> "Table":
        key_expr = self.pointer_from(*args, instance=instance)
        return self.with_id_from_expr(key_expr)

    def with_id_from_expr(self, key_expr: ex.ColumnExpression) -> "Table":
        from pathway_amd.engine.nodes import ReindexNode

        node = ReindexNode(self._node, key_expr, get_device())
        return Table(node, self._dtypes, Universe())

    def with_id(self, new_index: ex.ColumnExpression) -> "Table":
        return self.with_id_from_expr(self._resolve(new_index))

    # -- flatten / ix -------------------------------------------------------

    def flatten(self, to_flatten: Any, *, origin_id: str | None = None) -> "Table":
        from pathway_amd.engine.nodes_join import FlattenNode

        ref = self._resolve(to_flatten)
        if not isinstance(ref, ex.ColumnReference):
            raise ValueError("flatten expects a column reference")
        node = FlattenNode(self._node, ref.name, get_device(), origin_id=origin_id)
        dtypes = dict(self._dtypes)
        if origin_id:
            dtypes[origin_id] = dt.POINTER
        inner = self._dtypes.get(ref.name, dt.ANY)
        if isinstance(inner, dt.List):
            dtypes[ref.name] = inner.wrapped
        elif dt.unoptionalize(inner) == dt.STR:
            dtypes[ref.name] = dt.STR
        else:
            dtypes[ref.name] = dt.ANY
        return Table(node, dtypes, Universe())

    def ix(self, expression: Any, *, optional: bool = False, context=None) -> "Table":
        """Row lookup by pointer (reference table.py ix/ix_ref)."""
        from pathway_amd.internals.joins import make_ix_table

        if isinstance(expression, ex.ColumnReference) and isinstance(
            expression.table, Table
        ):
            query_table = expression.table
            pexpr = expression
        else:
            query_table = context if isinstance(context, Table) else self
            pexpr = expression
        return make_ix_table(query_table, self, pexpr, optional=optional)

    def ix_ref(self, *args: Any, optional: bool = False, context=None, instance=None) -> "Table":
        query = context if isinstance(context, Table) else self
        pexpr = query.pointer_from(*args, optional=optional, instance=instance)
        from pathway_amd.internals.joins import make_ix_table

        return make_ix_table(query, self, pexpr, optional=optional)

    def sort(self, key: Any, instance: Any = None) -> "Table":
        """prev/next pointer columns by sort order (reference sort_table +
        add_prev_next_pointers, operators/prev_next.rs:775)."""
        from pathway_amd.engine.nodes_recompute import RecomputeNode

        kref = self._resolve(key)
        iref = self._resolve(instance) if instance is not None else None
        if not isinstance(kref, ex.ColumnReference):
            raise NotImplementedError("sort key must be a column")
        kname = kref.name
        iname = iref.name if isinstance(iref, ex.ColumnReference) else None
        src = self

        import os

        int_like = (dt.INT, dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC, dt.DURATION)
        if (
            not os.environ.get("PW_SORT_HOST")
            and dt.unoptionalize(self._dtypes.get(kname, dt.ANY)) in int_like
        ):
            # tensor-native differential prev/next (engine/nodes_sort.py)
            from pathway_amd.engine.nodes_sort import SortPrevNextNode

            node = SortPrevNextNode(
                src._node,
                kref,
                iref if isinstance(iref, ex.ColumnReference) else None,
                get_device(),
            )
            out_dtypes = {
                "prev": dt.Optional(dt.POINTER),
                "next": dt.Optional(dt.POINTER),
            }
            return Table(node, out_dtypes, self._universe)

        def fn(in_rows, in_keys):
            rows, keys = in_rows[0], in_keys[0]
            groups: dict = {}
            for row, k in zip(rows, keys):
                g = row.get(iname) if iname else None
                groups.setdefault(g, []).append((row[kname], k))
            out = []
            for g, rl in groups.items():
                rl.sort(key=lambda x: (x[0], repr(x[1])))
                for i, (_, k) in enumerate(rl):
                    out.append(
                        (
                            k,
                            {
                                "prev": rl[i - 1][1] if i > 0 else None,
                                "next": rl[i + 1][1] if i + 1 < len(rl) else None,
                            },
                        )
                    )
            return out

        out_dtypes = {
            "prev": dt.Optional(dt.POINTER),
            "next": dt.Optional(dt.POINTER),
        }
        node = RecomputeNode(
            [src._node], fn, ["prev", "next"], out_dtypes, get_device()
        )
        return Table(node, out_dtypes, self._universe)

    def diff(self, timestamp: Any, *values: Any, instance: Any = None) -> "Table":
        from pathway_amd.stdlib.ordered import diff as _diff

        return _diff(self, timestamp, *values, instance=instance)

    def _gradual_broadcast(self, threshold_table, lower_column, value_column, upper_column) -> "Table":
        """Apportion a broadcast value across rows (reference
        operators/gradual_broadcast.rs:120-190): the key space is split at
        threshold = max_key * (value - lower) / (upper - lower); rows whose
        key falls below the threshold receive `upper`, the rest `lower` —
        so the fraction of rows at `upper` tracks the requested value
        (keys are uniform hashes; "scaling does not need to be precise")."""
        from pathway_amd.engine.nodes_recompute import RecomputeNode

        lname = lower_column.name
        vname = value_column.name
        uname = upper_column.name
        src = self
        _U64 = (1 << 64) - 1

        def fn(in_rows, in_keys):
            rows, keys = in_rows[0], in_keys[0]
            trows = in_rows[1]
            out = []
            if not trows:
                return [(k, {"apx_value": None}) for k in keys]
            lo = float(trows[0][lname])
            v = float(trows[0][vname])
            hi = float(trows[0][uname])
            frac = 1.0 if hi == lo else max(0.0, min(1.0, (v - lo) / (hi - lo)))
            threshold = int(frac * float(1 << 64))
            for k in keys:
                ku = k.as_signed_pair()[0] & _U64
                out.append(
                    (k, {"apx_value": hi if ku < threshold else lo})
                )
            return out

        out_dtypes = {"apx_value": dt.Optional(dt.FLOAT)}
        node = RecomputeNode(
            [src._node, threshold_table._node], fn, ["apx_value"], out_dtypes, get_device()
        )
        return Table(node, out_dtypes, self._universe)

    # -- misc ---------------------------------------------------------------

    # -- temporal behavior ops (graph.rs:753-793) ---------------------------

    def _forget(
        self, threshold_column: Any, time_column: Any, mark_forgetting_records: bool = False,
        instance_column: Any = None,
    ) -> "Table":
        from pathway_amd.engine.nodes_temporal import ForgetNode

        node = ForgetNode(
            self._node,
            self._resolve(threshold_column),
            self._resolve(time_column),
            get_device(),
            mark_forgetting_records,
        )
        return Table(node, self._dtypes, Universe())

    forget = _forget

    def _buffer(self, threshold_column: Any, time_column: Any) -> "Table":
        from pathway_amd.engine.nodes_temporal import BufferNode

        node = BufferNode(
            self._node,
            self._resolve(threshold_column),
            self._resolve(time_column),
            get_device(),
        )
        return Table(node, self._dtypes, Universe())

    buffer = _buffer

    def _freeze(self, threshold_column: Any, time_column: Any) -> "Table":
        from pathway_amd.engine.nodes_temporal import FreezeNode

        node = FreezeNode(
            self._node,
            self._resolve(threshold_column),
            self._resolve(time_column),
            get_device(),
        )
        return Table(node, self._dtypes, Universe())

    def ignore_late(self, threshold_column: Any, time_column: Any) -> "Table":
        return self._freeze(threshold_column, time_column)

    def _remove_retractions(self) -> "Table":
        """remove_retractions_from_table (graph.rs): drop diff<0 rows."""
        from pathway_amd.engine.nodes import Node as _N

        class _DropRetractions(_N):
            def step(self, time, inputs):
                b = inputs[0]
                if b is None or len(b) == 0:
                    return None
                import torch

                idx = (b.diffs > 0).nonzero(as_tuple=True)[0]
                return b.take(idx) if idx.numel() else None

        node = _DropRetractions([self._node], get_device())
        return Table(node, self._dtypes, Universe())

    def filter_out_results_of_forgetting(self) -> "Table":
        return self._remove_retractions()

    # -- misc surface parity ------------------------------------------------

    @staticmethod
    def from_columns(*args: Any, **kwargs: Any) -> "Table":
        cols = list(args) + list(kwargs.values())
        if not cols:
            raise ValueError("from_columns needs at least one column")
        base = cols[0].table
        named = {c.name: c for c in args}
        named.update(kwargs)
        return base.select(**named)

    def with_prefix(self, prefix: str) -> "Table":
        return self.select(
            **{f"{prefix}{n}": ex.ColumnReference(self, n) for n in self._dtypes}
        )

    def with_suffix(self, suffix: str) -> "Table":
        return self.select(
            **{f"{n}{suffix}": ex.ColumnReference(self, n) for n in self._dtypes}
        )

    def update_id_type(self, target_type: Any, **kwargs: Any) -> "Table":
        return self

    @property
    def slice(self):
        from pathway_amd.internals import TableSlice

        return TableSlice({n: ex.ColumnReference(self, n) for n in self._dtypes}, self)

    def is_append_only(self) -> bool:
        return False

    def assert_append_only(self) -> "Table":
        return self

    def live(self) -> "Table":
        return self

    def to_stream(self) -> "Table":
        """table_to_stream (graph.rs): events as an append-only table with
        an is_upsert flag column."""
        from pathway_amd.engine.nodes import ToStreamNode

        node = ToStreamNode(self._node, get_device())
        dtypes = dict(self._dtypes)
        dtypes["is_upsert"] = dt.BOOL
        dtypes["_pw_source_id"] = dt.POINTER
        return Table(node, dtypes, Universe())

    def stream_to_table(self, is_upsert: Any = None) -> "Table":
        from pathway_amd.engine.nodes import StreamToTableNode

        name = "is_upsert"
        if is_upsert is not None:
            r = self._resolve(is_upsert)
            if isinstance(r, ex.ColumnReference):
                name = r.name
        node = StreamToTableNode(self._node, name, get_device())
        dtypes = {
            n: t
            for n, t in self._dtypes.items()
            if n not in (name, "_pw_source_id")
        }
        return Table(node, dtypes, Universe())

    def from_streams(self, deletion_stream: "Table") -> "Table":
        """Reconstruct current state from an upsert stream (self) and a
        deletion stream (reference table.py from_streams /
        merge_streams_to_table): per id the latest event wins, in event
        order; deletion columns need not match."""
        payload = {n: t for n, t in self._dtypes.items() if n != "id"}
        cols = {n: ex.ColumnReference(self, n) for n in payload}
        u = self.select(
            **cols,
            _pw_source_id=ex.ColumnReference(self, "id"),
            is_upsert=True,
        )
        pad = {
            n: ex.DeclareTypeExpression(
                ex.ColumnConstExpression(None), dt.Optional(dt.unoptionalize(t))
            )
            for n, t in payload.items()
        }
        d = deletion_stream.select(
            **pad,
            _pw_source_id=ex.ColumnReference(deletion_stream, "id"),
            is_upsert=False,
        )
        merged = Table.concat_reindex(u, d)
        return merged.stream_to_table(ex.ColumnReference(merged, "is_upsert"))

    def unpack_snapshots(self) -> "Table":
        """Change stream → snapshot stream (reference table.py:3056): every
        changed minibatch appends the full current state as fresh rows."""
        from pathway_amd.engine.nodes import UnpackSnapshotsNode

        node = UnpackSnapshotsNode(self._node, get_device())
        return Table(node, dict(self._dtypes), Universe())

    def remove_errors(self) -> "Table":
        """Drop rows holding the Error sentinel in any column (reference
        table.py remove_errors).  Probe trick: applying any function to
        an Error value yields Error; fill_error maps that to False."""
        from pathway_amd.internals.expression import (
            ApplyExpression,
            FillErrorExpression,
        )

        mask = None
        for name in self._dtypes:
            probe = FillErrorExpression(
                ApplyExpression(lambda *a: True, None, self[name]), False
            )
            mask = probe if mask is None else mask & probe
        if mask is None:
            return self
        return self.filter(mask)

    def await_futures(self) -> "Table":
        return self

    def debug(self, name: str) -> "Table":
        return self

    def to(self, sink) -> None:
        sink.write(self)

    def _capture(self):
        """Attach a CaptureNode (tests / debug)."""
        from pathway_amd.engine.runtime import CaptureNode

        return CaptureNode(self._node, get_device(), list(self._dtypes.keys()))


class _ColumnAccessor:
    """`table.C.<name>` -> ColumnReference (reference Table.C)."""

    def __init__(self, table: Table):
        object.__setattr__(self, "_table", table)

    def __getattr__(self, name: str):
        t = object.__getattribute__(self, "_table")
        if name == "id":
            return ex.ColumnReference(t, "id")
        if name not in t._dtypes:
            raise AttributeError(f"no column {name!r}")
        return ex.ColumnReference(t, name)

    def __getitem__(self, name: str):
        return self.__getattr__(name)
