"""Metadata filter expressions (reference: JMESPath engine,
src/external_integration/mod.rs:41-49).

Supports the subset the reference's DocumentStore tests exercise:
  field == 'v'   field != 'v'   contains(field, 'v')
  globmatch('pat', field)  /  globmatch(field, 'pat')
  expr && expr   expr || expr   !expr   parentheses
Field access with dots: metadata.path etc. (evaluated against the payload
dict; a leading `metadata.` is stripped).
"""

from __future__ import annotations

import fnmatch
import re
from typing import Any


def _get_field(data: Any, path: str) -> Any:
    if data is None:
        return None
    path = path.strip()
    if path.startswith("metadata."):
        path = path[len("metadata."):]
    cur = data
    if hasattr(cur, "value"):
        cur = cur.value
    for part in path.split("."):
        if isinstance(cur, dict):
            cur = cur.get(part)
        else:
            cur = getattr(cur, part, None)
        if cur is None:
            return None
    return cur


_TOKEN = re.compile(
    r"\s*(&&|\|\||==|!=|>=|<=|>|<|!|\(|\)|,|`[^`]*`|'[^']*'|\"[^\"]*\"|[A-Za-z_][\w.]*|\d+\.\d+|\d+)"
)


def _tokenize(s: str) -> list[str]:
    out = []
    i = 0
    while i < len(s):
        m = _TOKEN.match(s, i)
        if not m:
            if s[i].isspace():
                i += 1
                continue
            raise ValueError(f"bad filter syntax at {s[i:]!r}")
        out.append(m.group(1))
        i = m.end()
    return out


def eval_jmespath_filter(expr: str, data: Any) -> bool:
    toks = _tokenize(expr)
    pos = [0]

    def peek():
        return toks[pos[0]] if pos[0] < len(toks) else None

    def eat():
        t = peek()
        pos[0] += 1
        return t

    def literal(tok: str) -> Any:
        if tok.startswith("`") or tok.startswith("'") or tok.startswith('"'):
            inner = tok[1:-1]
            if tok.startswith("`"):
                import json

                try:
                    return json.loads(inner)
                except Exception:
                    return inner.strip('"')
            return inner
        try:
            return int(tok)
        except ValueError:
            pass
        try:
            return float(tok)
        except ValueError:
            pass
        if tok in ("true", "false"):
            return tok == "true"
        if tok == "null":
            return None
        return _get_field(data, tok)

    def atom() -> Any:
        tok = peek()
        if tok == "!":
            eat()
            return not atom()
        if tok == "(":
            eat()
            v = or_expr()
            eat()  # )
            return v
        if tok in ("contains", "globmatch", "starts_with", "ends_with"):
            fn = eat()
            eat()  # (
            a = or_expr()
            eat()  # ,
            b = or_expr()
            eat()  # )
            if fn == "contains":
                try:
                    return b in a if a is not None else False
                except TypeError:
                    return False
            if fn == "globmatch":
                # JMESPath-ext: globmatch(pattern, path)
                pat, val = (a, b)
                if isinstance(val, str) and isinstance(pat, str):
                    return fnmatch.fnmatch(val, pat) or fnmatch.fnmatch(pat, val)
                return False
            if fn == "starts_with":
                return isinstance(a, str) and a.startswith(b)
            if fn == "ends_with":
                return isinstance(a, str) and a.endswith(b)
        return literal(eat())

    def cmp_expr() -> Any:
        a = atom()
        tok = peek()
        if tok in ("==", "!=", ">", "<", ">=", "<="):
            op = eat()
            b = atom()
            try:
                if op == "==":
                    return a == b
                if op == "!=":
                    return a != b
                if op == ">":
                    return a > b
                if op == "<":
                    return a < b
                if op == ">=":
                    return a >= b
                if op == "<=":
                    return a <= b
            except TypeError:
                return False
        return a

    def and_expr() -> Any:
        v = cmp_expr()
        while peek() == "&&":
            eat()
            v = bool(v) and bool(cmp_expr())
        return v

    def or_expr() -> Any:
        v = and_expr()
        while peek() == "||":
            eat()
            v = bool(v) or bool(and_expr())
        return v

    return bool(or_expr())
