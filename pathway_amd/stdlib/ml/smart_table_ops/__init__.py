"""Fuzzy join (reference stdlib/ml/smart_table_ops/_fuzzy_join.py surface)."""
from __future__ import annotations

import enum
from typing import Any

import pathway_amd.internals.common as common
from pathway_amd.internals import dtype as dt
from pathway_amd.internals import thisclass

this = thisclass.this
left = thisclass.left
right = thisclass.right


class JoinType(enum.Enum):
    FULL = 0
    LEFT = 1
    RIGHT = 2


def smart_fuzzy_join(left_table, left_col, right_table, right_col, reserved_scores=None, **kwargs):
    """Token-overlap fuzzy match: pairs rows whose normalized token sets
    overlap best (batch host computation via RecomputeNode)."""
    from pathway_amd.engine.nodes_recompute import RecomputeNode
    from pathway_amd.internals.api import Pointer, hash_values
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    lref = left_table._resolve(left_col)
    rref = right_table._resolve(right_col)
    lname, rname = lref.name, rref.name

    def toks(s):
        return set(str(s).lower().split())

    def fn(in_rows, in_keys):
        lrows, rrows = in_rows
        lkeys, rkeys = in_keys
        out = []
        used = set()
        scored = []
        for lrow, lk in zip(lrows, lkeys):
            lt = toks(lrow[lname])
            for rrow, rk in zip(rrows, rkeys):
                rt = toks(rrow[rname])
                inter = len(lt & rt)
                if inter:
                    score = inter / max(len(lt | rt), 1)
                    scored.append((score, repr(lk), repr(rk), lrow, rrow, lk, rk))
        scored.sort(key=lambda x: -x[0])
        used_l, used_r = set(), set()
        for score, lrepr, rrepr, lrow, rrow, lk, rk in scored:
            if lrepr in used_l or rrepr in used_r:
                continue
            used_l.add(lrepr)
            used_r.add(rrepr)
            lo, hi = hash_values([lrepr, rrepr, "fuzzy"])
            out.append(
                (
                    Pointer(lo, hi),
                    {"left_value": lrow[lname], "right_value": rrow[rname], "score": score},
                )
            )
        return out

    out_dtypes = {"left_value": dt.ANY, "right_value": dt.ANY, "score": dt.FLOAT}
    node = RecomputeNode(
        [left_table._node, right_table._node],
        fn,
        ["left_value", "right_value", "score"],
        out_dtypes,
        get_device(),
    )
    return Table(node, out_dtypes, Universe())


fuzzy_match_tables = smart_fuzzy_join
