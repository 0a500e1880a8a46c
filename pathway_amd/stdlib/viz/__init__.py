"""Live visualization (reference stdlib/viz — panel/bokeh plots).

Plot servers need an interactive notebook; here we expose a table_to_pandas
based snapshot plotting hook.
"""
from __future__ import annotations


def plot(table, plotting_function=None, sorting_col=None):
    raise NotImplementedError(
        "live plots require an interactive frontend; use pw.debug.table_to_pandas"
    )


def table_viz(table):
    from pathway_amd.debug import table_to_pandas

    return table_to_pandas(table)


__all__ = ["plot", "table_viz"]
