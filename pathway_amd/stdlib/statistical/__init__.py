"""Statistical ops (reference stdlib/statistical): interpolate."""
from __future__ import annotations

from enum import Enum
from typing import Any


class InterpolateMode(Enum):
    LINEAR = "linear"


def interpolate(self, timestamp: Any, *values: Any, mode: InterpolateMode = InterpolateMode.LINEAR):
    from pathway_amd.engine.nodes_recompute import RecomputeNode
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals import expression as ex
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table

    ts = self._resolve(timestamp)
    vnames = []
    for v in values:
        rv = self._resolve(v)
        vnames.append(rv.name)
    tname = ts.name

    def fn(in_rows, in_keys):
        rows, keys = in_rows[0], in_keys[0]
        rl = sorted(zip(rows, keys), key=lambda x: x[0][tname])
        out = []
        for n in vnames:
            known = [(r[tname], r[n]) for r, _ in rl if r[n] is not None]
            for i, (r, key) in enumerate(rl):
                pass
        for r, key in rl:
            vals = {}
            for n in vnames:
                if r[n] is not None:
                    vals[n] = float(r[n])
                    continue
                known = [(x[tname], x[n]) for x, _ in rl if x[n] is not None]
                before = [(t, v) for t, v in known if t <= r[tname]]
                after = [(t, v) for t, v in known if t >= r[tname]]
                if before and after:
                    (t0, v0), (t1, v1) = before[-1], after[0]
                    vals[n] = (
                        float(v0)
                        if t1 == t0
                        else v0 + (v1 - v0) * (r[tname] - t0) / (t1 - t0)
                    )
                elif before:
                    vals[n] = float(before[-1][1])
                elif after:
                    vals[n] = float(after[0][1])
                else:
                    vals[n] = None
            vals[tname] = r[tname]
            out.append((key, vals))
        return out

    out_columns = [tname] + vnames
    out_dtypes = {tname: self._dtypes[tname]}
    for n in vnames:
        out_dtypes[n] = dt.Optional(dt.FLOAT)
    node = RecomputeNode([self._node], fn, out_columns, out_dtypes, get_device())
    return Table(node, out_dtypes, self._universe)


__all__ = ["interpolate", "InterpolateMode"]
