"""Row transformers / complex columns (reference internals/row_transformer.py
+ src/engine/dataflow/complex_columns.rs:59-110 Computer machinery).

@pw.transformer classes declare per-row computed attributes with
demand-driven recursion: an output attribute may dereference pointers into
any of the transformer's tables (self.transformer.<table>[ptr].<attr>),
including other computed attributes — evaluated with memoization.
"""

from __future__ import annotations

from typing import Any, Callable


class _InputAttribute:
    """Descriptor: instance access routes through the evaluation context."""

    def __init__(self, name: str | None = None):
        self.name = name

    def __set_name__(self, owner, name):
        self.name = name

    def __get__(self, obj, objtype=None):
        if obj is None:
            return self
        return obj._context.get(obj._table_name, obj._pointer, self.name)


class _OutputAttribute:
    def __init__(self, fn: Callable):
        self.fn = fn
        self.name = fn.__name__

    def __set_name__(self, owner, name):
        self.name = name

    def __get__(self, obj, objtype=None):
        if obj is None:
            return self
        return obj._context.get(obj._table_name, obj._pointer, self.name)


class _Method(_OutputAttribute):
    pass


def input_attribute(type: Any = None):  # noqa: A002
    return _InputAttribute()


def input_method(type: Any = None):  # noqa: A002
    return _InputAttribute()


def output_attribute(fn: Callable = None, **kwargs):
    if fn is None:
        return lambda f: _OutputAttribute(f)
    return _OutputAttribute(fn)


def method(fn: Callable = None, **kwargs):
    if fn is None:
        return lambda f: _Method(f)
    return _Method(fn)


class ClassArg:
    """Base for transformer table classes (reference internals ClassArg)."""

    def __init__(self, context, pointer):
        self._context = context
        self._pointer = pointer

    @property
    def id(self):
        return self._pointer

    @property
    def transformer(self):
        return self._context.namespace

    def pointer_from(self, *args):
        from pathway_amd.internals.api import Pointer, hash_values

        lo, hi = hash_values(list(args))
        return Pointer(lo, hi)

    def __getattr__(self, name: str):
        return self._context.get(self._table_name, self._pointer, name)


class _RowHandle:
    def __init__(self, context, table_name, pointer):
        self._context = context
        self._table_name = table_name
        self._pointer = pointer

    @property
    def id(self):
        return self._pointer

    def __getattr__(self, name: str):
        return self._context.get(self._table_name, self._pointer, name)


class _TableNamespaceHandle:
    def __init__(self, context, table_name):
        self._context = context
        self._table_name = table_name

    def __getitem__(self, pointer):
        return _RowHandle(self._context, self._table_name, pointer)


class _Namespace:
    def __init__(self, context):
        self._context = context

    def __getattr__(self, table_name: str):
        return _TableNamespaceHandle(self._context, table_name)


class _EvalContext:
    """Demand-driven per-run evaluation state (Computer/Context::get)."""

    def __init__(self, spec, rows_by_table):
        self.spec = spec
        self.rows = rows_by_table  # table -> {pointer_repr: {attr: value}}
        self.memo: dict = {}
        self.namespace = _Namespace(self)
        self.in_progress: set = set()

    def get(self, table_name: str, pointer, attr: str):
        key = (table_name, repr(pointer), attr)
        if key in self.memo:
            return self.memo[key]
        trow = self.rows[table_name].get(repr(pointer))
        if trow is None:
            raise KeyError(f"no row {pointer!r} in transformer table {table_name}")
        if attr in trow:
            return trow[attr]
        out_attrs = self.spec[table_name]["outputs"]
        if attr not in out_attrs:
            raise AttributeError(f"{table_name}.{attr}")
        if key in self.in_progress:
            raise RecursionError(f"cyclic attribute {table_name}.{attr}")
        self.in_progress.add(key)
        try:
            cls = self.spec[table_name]["cls"]
            inst = cls.__new__(cls)
            ClassArg.__init__(inst, self, pointer)
            inst._table_name = table_name
            fn = out_attrs[attr].fn
            value = fn(inst)
        finally:
            self.in_progress.discard(key)
        self.memo[key] = value
        return value


class _TransformerResultNamespace:
    pass


def transformer(cls):
    """@pw.transformer decorator."""
    tables_spec: dict[str, dict] = {}
    for tname, tcls in vars(cls).items():
        if isinstance(tcls, type) and issubclass(tcls, ClassArg):
            inputs = {}
            outputs = {}
            for aname, aval in vars(tcls).items():
                if isinstance(aval, _InputAttribute):
                    inputs[aname] = aval
                elif isinstance(aval, _OutputAttribute):
                    outputs[aname] = aval
            tables_spec[tname] = {"cls": tcls, "inputs": inputs, "outputs": outputs}

    def build(**tables):
        from pathway_amd.engine.nodes_recompute import RecomputeNode
        from pathway_amd.internals import dtype as dt
        from pathway_amd.internals.config import get_device
        from pathway_amd.internals.table import Table
        from pathway_amd.internals.universe import Universe

        order = list(tables_spec.keys())
        input_nodes = [tables[t]._node for t in order]
        result = _TransformerResultNamespace()

        for out_ti, tname in enumerate(order):
            spec_t = tables_spec[tname]
            out_attrs = list(spec_t["outputs"].keys())
            if not out_attrs:
                setattr(result, tname, tables[tname])
                continue

            def fn(in_rows, in_keys, _tname=tname, _out=tuple(out_attrs)):
                rows_by_table = {}
                for ti, tn in enumerate(order):
                    rows_by_table[tn] = {
                        repr(k): dict(r) for r, k in zip(in_rows[ti], in_keys[ti])
                    }
                ctx = _EvalContext(tables_spec, rows_by_table)
                ti = order.index(_tname)
                out = []
                for r, k in zip(in_rows[ti], in_keys[ti]):
                    vals = {a: ctx.get(_tname, k, a) for a in _out}
                    out.append((k, vals))
                return out

            out_dtypes = {a: dt.ANY for a in out_attrs}
            node = RecomputeNode(
                input_nodes, fn, out_attrs, out_dtypes, get_device()
            )
            setattr(
                result, tname, Table(node, out_dtypes, tables[tname]._universe)
            )
        return result

    build.__name__ = getattr(cls, "__name__", "transformer")
    return build
