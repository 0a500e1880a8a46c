"""pw.io.null sink."""
from __future__ import annotations


def write(table, *, name: str | None = None):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    node = OutputNode(table._node, lambda batch: None, get_device())
    G.add_sink(node)
    return node
