"""Argmin/argmax row filtering helpers (reference stdlib/utils/filtering.py)."""
from __future__ import annotations

import pathway_amd.reducers as reducers
from pathway_amd.internals import thisclass


def argmax_rows(table, *on, what=None):
    r = table.groupby(*on).reduce(_pw_argmax_id=reducers.argmax(what))
    return table.ix(r._pw_argmax_id, context=r)


def argmin_rows(table, *on, what=None):
    r = table.groupby(*on).reduce(_pw_argmin_id=reducers.argmin(what))
    return table.ix(r._pw_argmin_id, context=r)
