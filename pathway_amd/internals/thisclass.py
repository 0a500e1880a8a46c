"""pw.this / pw.left / pw.right deferred references (reference thisclass.py)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals.expression import ColumnExpression, ColumnReference


class ThisMetaclass(type):
    def __getattr__(cls, name: str) -> ColumnReference:
        if name.startswith("__") and name.endswith("__"):
            raise AttributeError(name)
        if name == "without":
            # pw.this.without(pw.this.a, ...): deferred splat-with-exclusions
            def _without(*cols):
                excl = {
                    c.name if isinstance(c, ColumnReference) else str(c)
                    for c in cols
                }
                return _SplatIter(ThisSplat(cls, exclude=excl))

            return _without
        return ColumnReference(cls, name)

    def __getitem__(cls, name: str) -> ColumnReference:
        if isinstance(name, ColumnReference):
            name = name.name
        return ColumnReference(cls, name)

    def __iter__(cls):
        # `select(*pw.this)` / `select(*pw.left)`: defer the expansion to
        # resolution time (reference table.py supports splatting this)
        yield ThisSplat(cls)

    def __repr__(cls) -> str:
        return cls._repr


class this(metaclass=ThisMetaclass):
    """Placeholder for 'the table this expression is used with'."""

    _repr = "<this>"


class left(metaclass=ThisMetaclass):
    _repr = "<left>"


class right(metaclass=ThisMetaclass):
    _repr = "<right>"


def substitute_this(expr: Any, mapping: dict[type, Any]) -> Any:
    """Replace this/left/right markers in an expression tree with real tables."""
    from pathway_amd.internals import expression as expr_mod

    if not isinstance(expr, ColumnExpression):
        return expr

    def sub(e: ColumnExpression) -> ColumnExpression:
        if isinstance(e, ColumnReference):
            tbl = e.table
            if isinstance(tbl, type) and tbl in mapping:
                target = mapping[tbl]
                return target[e.name] if e.name != "id" else target.id
            return e
        # rebuild with substituted children, preserving structure
        new = object.__new__(type(e))
        new.__dict__.update(e.__dict__)
        for attr, val in list(e.__dict__.items()):
            if isinstance(val, ColumnExpression):
                new.__dict__[attr] = sub(val)
            elif isinstance(val, tuple) and any(
                isinstance(v, ColumnExpression) for v in val
            ):
                new.__dict__[attr] = tuple(
                    sub(v) if isinstance(v, ColumnExpression) else v for v in val
                )
            elif isinstance(val, dict) and any(
                isinstance(v, ColumnExpression) for v in val.values()
            ):
                new.__dict__[attr] = {
                    k: sub(v) if isinstance(v, ColumnExpression) else v
                    for k, v in val.items()
                }
        return new

    _ = expr_mod  # keep import for clarity
    return sub(expr)


class ThisSplat:
    """Deferred `*pw.this` marker: expands to every column of the resolved
    table inside select()/reduce() argument handling."""

    def __init__(self, cls, exclude=None):
        self.cls = cls
        self.exclude = set(exclude or ())


class _SplatIter:
    """Wrapper so `*pw.this.without(...)` splats to one ThisSplat marker."""

    def __init__(self, splat):
        self._splat = splat

    def __iter__(self):
        yield self._splat
