"""pw.sql — SQL to dataflow translation (reference internals/sql/, ~1500 LoC).

Covers SELECT / WHERE / GROUP BY / HAVING / JOIN (inner and LEFT) ON /
UNION ALL / WITH ctes / FROM-subqueries, and in scalar expressions:
arithmetic, comparisons, AND/OR/NOT, IN (...) / NOT IN, BETWEEN,
LIKE (SQL wildcards), CASE WHEN ... THEN ... [ELSE ...] END, and the
count/sum/avg/min/max aggregates.
"""

from __future__ import annotations

import re
from typing import Any

from pathway_amd.internals import expression as ex
from pathway_amd.internals.table import Table


class _SqlError(ValueError):
    pass


_AGGS = {"count": "count", "sum": "sum", "avg": "avg", "min": "min", "max": "max"}


def _tokenize_expr(s: str):
    return re.findall(r"[A-Za-z_][A-Za-z0-9_.]*|\d+\.\d+|\d+|<>|<=|>=|!=|=|<|>|[(),*+\-/%]|'[^']*'", s)


def _parse_scalar(sql_expr: str, table: Table) -> ex.ColumnExpression:
    """Tiny recursive-descent expression parser over a single table scope."""
    toks = _tokenize_expr(sql_expr)
    pos = [0]

    def peek():
        return toks[pos[0]] if pos[0] < len(toks) else None

    def eat(t=None):
        tok = peek()
        if t is not None and (tok is None or tok.upper() != t.upper()):
            raise _SqlError(f"expected {t}, got {tok}")
        pos[0] += 1
        return tok

    def atom():
        tok = peek()
        if tok is None:
            raise _SqlError("unexpected end of expression")
        if tok == "(":
            eat()
            e = or_expr()
            eat(")")
            return e
        if re.fullmatch(r"\d+", tok):
            eat()
            return ex.ColumnConstExpression(int(tok))
        if re.fullmatch(r"\d+\.\d+", tok):
            eat()
            return ex.ColumnConstExpression(float(tok))
        if tok.startswith("'"):
            eat()
            return ex.ColumnConstExpression(tok[1:-1])
        if tok.upper() == "CASE":
            eat()
            whens = []
            else_e = ex.ColumnConstExpression(None)
            while peek() is not None and peek().upper() == "WHEN":
                eat()
                cond = or_expr()
                eat("THEN")
                val = or_expr()
                whens.append((cond, val))
            if peek() is not None and peek().upper() == "ELSE":
                eat()
                else_e = or_expr()
            eat("END")
            out = else_e
            for cond, val in reversed(whens):
                out = ex.IfElseExpression(cond, val, out)
            return out
        name = eat()
        lname = name.lower()
        if lname in _AGGS and peek() == "(":
            eat("(")
            if peek() == "*":
                eat()
                arg = None
            else:
                arg = or_expr()
            eat(")")
            from pathway_amd.internals.expression import ReducerExpression

            if arg is None:
                return ReducerExpression("count")
            return ReducerExpression(_AGGS[lname], arg)
        if lname in ("true", "false"):
            return ex.ColumnConstExpression(lname == "true")
        if lname == "null":
            return ex.ColumnConstExpression(None)
        col = name.split(".")[-1]
        return ex.ColumnReference(table, col)

    def mul_expr():
        e = atom()
        while peek() in ("*", "/", "%"):
            op = eat()
            r = atom()
            e = {"*": e.__mul__, "/": e.__truediv__, "%": e.__mod__}[op](r)
        return e

    def add_expr():
        e = mul_expr()
        while peek() in ("+", "-"):
            op = eat()
            r = mul_expr()
            e = (e + r) if op == "+" else (e - r)
        return e

    def cmp_expr():
        e = add_expr()
        negate = False
        if peek() is not None and peek().upper() == "NOT":
            nxt = toks[pos[0] + 1] if pos[0] + 1 < len(toks) else ""
            if nxt.upper() in ("IN", "LIKE", "BETWEEN"):
                eat()
                negate = True
        if peek() is not None and peek().upper() == "IN":
            eat()
            eat("(")
            vals = [or_expr()]
            while peek() == ",":
                eat()
                vals.append(or_expr())
            eat(")")
            out = None
            for v in vals:
                term = e == v
                out = term if out is None else (out | term)
            return ~out if negate else out
        if peek() is not None and peek().upper() == "BETWEEN":
            eat()
            lo = add_expr()
            eat("AND")
            hi = add_expr()
            out = (e >= lo) & (e <= hi)
            return ~out if negate else out
        if peek() is not None and peek().upper() == "LIKE":
            eat()
            pat = or_expr()
            if not isinstance(pat, ex.ColumnConstExpression):
                raise _SqlError("LIKE needs a literal pattern")
            import fnmatch as _fn
            import re as _re

            rx = _re.compile(
                "^" + _re.escape(str(pat._value)).replace("%", ".*").replace(
                    "_", "."
                ).replace("\\.\\*", "%") + "$",
                _re.S,
            )
            out = ex.ApplyExpression(
                lambda s, _rx=rx: bool(s is not None and _rx.match(str(s))),
                bool,
                e,
            )
            return ~out if negate else out
        if peek() in ("=", "!=", "<>", "<", "<=", ">", ">="):
            op = eat()
            r = add_expr()
            return {
                "=": e.__eq__,
                "!=": e.__ne__,
                "<>": e.__ne__,
                "<": e.__lt__,
                "<=": e.__le__,
                ">": e.__gt__,
                ">=": e.__ge__,
            }[op](r)
        return e

    def not_expr():
        if peek() is not None and peek().upper() == "NOT":
            eat()
            return ~not_expr()
        return cmp_expr()

    def and_expr():
        e = not_expr()
        while peek() is not None and peek().upper() == "AND":
            eat()
            e = e & not_expr()
        return e

    def or_expr():
        e = and_expr()
        while peek() is not None and peek().upper() == "OR":
            eat()
            e = e | and_expr()
        return e

    return or_expr()


def _split_balanced(s: str, sep: str = ",") -> list[str]:
    out, depth, cur = [], 0, []
    for ch in s:
        if ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
        if ch == sep and depth == 0:
            out.append("".join(cur).strip())
            cur = []
        else:
            cur.append(ch)
    if cur:
        out.append("".join(cur).strip())
    return out


def sql(query: str, **tables: Table) -> Table:
    q = query.strip().rstrip(";")
    mw = re.match(r"WITH\s+(.*?)\s+(SELECT\b.*)$", q, re.I | re.S)
    if mw:
        # WITH a AS (...), b AS (...) SELECT ... — split on balanced commas
        ctes = _split_balanced(mw.group(1))
        tables = dict(tables)
        for cte in ctes:
            mc = re.match(r"([A-Za-z_]\w*)\s+AS\s*\((.*)\)\s*$", cte, re.I | re.S)
            if not mc:
                raise _SqlError(f"unsupported WITH clause {cte!r}")
            tables[mc.group(1)] = sql(mc.group(2), **tables)
        return sql(mw.group(2), **tables)
    mf = re.match(r"(SELECT\s+.*?\s+FROM)\s*\(", q, re.I | re.S)
    if mf:
        # FROM (subquery) [alias] — find the matching close paren by scan
        start = mf.end()
        depth, i = 1, start
        while i < len(q) and depth:
            if q[i] == "(":
                depth += 1
            elif q[i] == ")":
                depth -= 1
            i += 1
        inner = q[start : i - 1]
        rest = q[i:]
        ma = re.match(r"\s*(?:AS\s+)?([A-Za-z_]\w*)?(.*)$", rest, re.I | re.S)
        tables = dict(tables)
        alias = ma.group(1) or "_pw_sub"
        tables[alias] = sql(inner, **tables)
        q = f"{mf.group(1)} {alias}{ma.group(2) or ''}"
    if re.search(r"\bUNION\s+ALL\b", q, re.I):
        parts = re.split(r"\bUNION\s+ALL\b", q, flags=re.I)
        result = sql(parts[0], **tables)
        for p in parts[1:]:
            result = result.concat_reindex(sql(p, **tables))
        return result
    m = re.match(
        r"SELECT\s+(?P<sel>.*?)\s+FROM\s+(?P<from>[A-Za-z_][A-Za-z0-9_]*)"
        r"(?:\s+(?P<jkind>INNER\s+|LEFT\s+(?:OUTER\s+)?)?JOIN\s+(?P<jt>[A-Za-z_][A-Za-z0-9_]*)\s+ON\s+(?P<on>.*?))?"
        r"(?:\s+WHERE\s+(?P<where>.*?))?"
        r"(?:\s+GROUP\s+BY\s+(?P<gb>.*?))?"
        r"(?:\s+HAVING\s+(?P<having>.*?))?$",
        q,
        re.I | re.S,
    )
    if not m:
        raise _SqlError(f"unsupported SQL: {query!r}")
    tname = m.group("from")
    if tname not in tables:
        raise _SqlError(f"unknown table {tname!r}")
    t = tables[tname]
    if m.group("jt"):
        jt = m.group("jt")
        if jt not in tables:
            raise _SqlError(f"unknown table {jt!r}")
        right = tables[jt]
        on = m.group("on")
        mo = re.match(
            r"\s*([A-Za-z_][\w.]*)\s*=\s*([A-Za-z_][\w.]*)\s*$", on
        )
        if not mo:
            raise _SqlError(f"unsupported JOIN condition {on!r}")
        lcol = mo.group(1).split(".")[-1]
        rcol = mo.group(2).split(".")[-1]
        if lcol not in t._dtypes:
            lcol, rcol = rcol, lcol
        jkind = (m.group("jkind") or "").strip().upper()
        if jkind.startswith("LEFT"):
            j = t.join_left(right, t[lcol] == right[rcol])
        else:
            j = t.join(right, t[lcol] == right[rcol])
        # materialize all columns of both sides (suffix disambiguation)
        t = j._all_columns_table()
    if m.group("where"):
        t = t.filter(_parse_scalar(m.group("where"), t))
    sel = m.group("sel").strip()
    gb = m.group("gb")

    def split_select(s: str) -> list[str]:
        out, depth, cur = [], 0, []
        for ch in s:
            if ch == "(":
                depth += 1
            elif ch == ")":
                depth -= 1
            if ch == "," and depth == 0:
                out.append("".join(cur).strip())
                cur = []
            else:
                cur.append(ch)
        if cur:
            out.append("".join(cur).strip())
        return out

    items = split_select(sel)
    named: dict[str, Any] = {}
    for it in items:
        if it == "*":
            for c in t.column_names():
                named[c] = ex.ColumnReference(t, c)
            continue
        mas = re.match(r"(.*?)\s+AS\s+([A-Za-z_][A-Za-z0-9_]*)$", it, re.I)
        if mas:
            expr_s, name = mas.group(1), mas.group(2)
        else:
            expr_s = it
            name = re.sub(r"\W+", "_", it.split(".")[-1]).strip("_") or "col"
        named[name] = _parse_scalar(expr_s, t)
    if gb:
        gcols = [c.strip().split(".")[-1] for c in gb.split(",")]
        grouped = t.groupby(*[ex.ColumnReference(t, c) for c in gcols])
        having_expr = None
        tmp_names: list[str] = []
        if m.group("having"):
            # aggregates inside HAVING are computed in the reduce under
            # temporary names, filtered on, then dropped (SQL semantics)
            having_expr = _parse_scalar(m.group("having"), t)
            counter = [0]

            def hoist(e):
                if isinstance(e, ex.ReducerExpression):
                    nm = f"_pw_having_{counter[0]}"
                    counter[0] += 1
                    named[nm] = e
                    tmp_names.append(nm)
                    return ex.ColumnReference(None, nm)
                if not isinstance(e, ex.ColumnExpression):
                    return e
                new = object.__new__(type(e))
                new.__dict__.update(e.__dict__)
                for attr, val in list(e.__dict__.items()):
                    if isinstance(val, ex.ColumnExpression):
                        new.__dict__[attr] = hoist(val)
                    elif isinstance(val, tuple) and any(
                        isinstance(v, ex.ColumnExpression) for v in val
                    ):
                        new.__dict__[attr] = tuple(
                            hoist(v) if isinstance(v, ex.ColumnExpression) else v
                            for v in val
                        )
                return new

            having_expr = hoist(having_expr)
        result = grouped.reduce(**named)
        if having_expr is not None:
            rebound = _rebind(having_expr, result)
            result = result.filter(rebound)
            if tmp_names:
                result = result.without(*tmp_names)
        return result
    has_agg = any(isinstance(e, ex.ReducerExpression) for e in named.values())
    if has_agg:
        return t.groupby().reduce(**named)
    return t.select(**named)


def _rebind(e, table):
    """Re-point bare column refs at `table` (post-reduce scope)."""
    if isinstance(e, ex.ColumnReference):
        return ex.ColumnReference(table, e.name)
    if not isinstance(e, ex.ColumnExpression):
        return e
    new = object.__new__(type(e))
    new.__dict__.update(e.__dict__)
    for attr, val in list(e.__dict__.items()):
        if isinstance(val, ex.ColumnExpression):
            new.__dict__[attr] = _rebind(val, table)
        elif isinstance(val, tuple) and any(
            isinstance(v, ex.ColumnExpression) for v in val
        ):
            new.__dict__[attr] = tuple(
                _rebind(v, table) if isinstance(v, ex.ColumnExpression) else v
                for v in val
            )
    return new


def _balanced(s: str) -> bool:
    depth = 0
    for ch in s:
        if ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
            if depth < 0:
                return False
    return depth == 0
