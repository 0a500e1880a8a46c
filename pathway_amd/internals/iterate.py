"""pw.iterate — fixpoint iteration (reference dataflow.rs:5060-5190).

MI355X-native design: instead of differential's product timestamps
(outer, round) with non-total order, the engine runs the inner subgraph to
fixpoint synchronously per outer timestamp: each round pushes the previous
round's output deltas back into the iteration ports until no deltas remain
(or iteration_limit).  Incremental across rounds (inner operator state
persists), recomputed per outer time for non-monotone logic.
"""

from __future__ import annotations

from typing import Any


from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.nodes import InputNode, Node, consolidate_batch
from pathway_amd.engine.runtime import topo_order
from pathway_amd.internals.config import get_device


class _PortSource:
    """Source for an iteration port; fed programmatically each round."""

    def __init__(self):
        self.batch: DeltaBatch | None = None

    def next_time(self):
        return None

    def pull(self, time, device):
        b = self.batch
        self.batch = None
        return b

    def reset(self):
        self.batch = None


class IterateNode(Node):
    def __init__(self, outer_nodes: list[Node], names: list[str], build_inner, device, limit=None):
        super().__init__(outer_nodes, device)
        self.names = names
        self.limit = limit
        # build inner subgraph with port tables
        self.ports: dict[str, InputNode] = {}
        self.port_sources: dict[str, _PortSource] = {}
        inner_tables = {}
        from pathway_amd.internals.table import Table
        from pathway_amd.internals.universe import Universe

        self._outer_tables = build_inner["outer_tables"]
        for name in names:
            src = _PortSource()
            port = InputNode(src, device)
            self.ports[name] = port
            self.port_sources[name] = src
            outer_t = build_inner["outer_tables"][name]
            inner_tables[name] = Table(port, outer_t._dtypes, Universe())
        results = build_inner["func"](**inner_tables)
        if isinstance(results, dict):
            self.result_tables = results
        elif hasattr(results, "_asdict"):
            self.result_tables = results._asdict()
        elif isinstance(results, tuple):
            self.result_tables = dict(zip(names, results))
        else:
            self.result_tables = {names[0]: results}
        self.inner_nodes = topo_order([t._node for t in self.result_tables.values()])
        self.last_outputs: dict[str, DeltaBatch | None] = {}

    def reset(self):
        self.last_outputs = {}
        for n in self.inner_nodes:
            if n not in (list(self.ports.values())):
                r = getattr(n, "reset", None)
                if r:
                    r()
        for s in self.port_sources.values():
            s.reset()

    def step(self, time, inputs):
        device = self.device
        feeds: dict[str, DeltaBatch | None] = {
            name: inputs[i] for i, name in enumerate(self.names)
        }
        # deltas injected from OUTSIDE the feedback loop this round — only
        # these get subtracted in the feedback formula δ_{r+1} = O_r − ext_r
        external: dict[str, DeltaBatch | None] = dict(feeds)
        accum: dict[str, list[DeltaBatch]] = {name: [] for name in self.result_tables}
        rounds = 0
        while any(b is not None and len(b) for b in feeds.values()):
            for name, b in feeds.items():
                if name in self.port_sources:
                    self.port_sources[name].batch = b
            outputs: dict[int, DeltaBatch | None] = {}
            for node in self.inner_nodes:
                if isinstance(node, InputNode):
                    out = node.step(time, [])
                else:
                    ins = [outputs.get(id(i)) for i in node.inputs]
                    out = None if all(x is None for x in ins) else node.step(time, ins)
                outputs[id(node)] = out
            new_feeds: dict[str, DeltaBatch | None] = {}
            for name, t in self.result_tables.items():
                ob = outputs.get(id(t._node))
                ob = consolidate_batch(ob) if ob is not None else None
                if ob is not None:
                    accum[name].append(ob)
                if name in self.port_sources:
                    # feedback delta: X_{r+1} = F(X_r) ⇒ δ_{r+1} = O_r minus
                    # any delta injected from OUTSIDE the loop this round
                    # (the outer input on round 1); feedback deltas themselves
                    # are already accounted for in the derivation.
                    ext = external.get(name)
                    parts = []
                    if ob is not None:
                        parts.append(ob)
                    if ext is not None and len(ext):
                        parts.append(DeltaBatch(ext.keys, ext.columns, -ext.diffs, time))
                    new_feeds[name] = (
                        consolidate_batch(DeltaBatch.concat(parts)) if parts else None
                    )
            # ports not in results get no further feed
            feeds = new_feeds
            external = {}
            rounds += 1
            if self.limit is not None and rounds >= self.limit:
                break
        self.last_outputs = {
            name: consolidate_batch(DeltaBatch.concat(batches)) if batches else None
            for name, batches in accum.items()
        }
        return None


class IterateOutNode(Node):
    def __init__(self, iterate_node: IterateNode, name: str, device):
        super().__init__([iterate_node], device)
        self.iterate_node = iterate_node
        self.out_name = name

    def wants_frontier(self) -> bool:
        # IterateNode returns None from step(); outputs are read from
        # last_outputs, so this node must run even with all-None inputs.
        return True

    def step(self, time, inputs):
        return self.iterate_node.last_outputs.pop(self.out_name, None)


def run_iterate(func, iteration_limit: int | None = None, **kwargs: Any):
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    names = list(kwargs.keys())
    outer_tables: dict[str, Table] = dict(kwargs)
    device = get_device()
    node = IterateNode(
        [t._node for t in outer_tables.values()],
        names,
        {"func": func, "outer_tables": outer_tables},
        device,
        limit=iteration_limit,
    )
    outs = {}
    for name, rt in node.result_tables.items():
        out_node = IterateOutNode(node, name, device)
        outs[name] = Table(out_node, rt._dtypes, Universe())
    if len(outs) == 1:
        return next(iter(outs.values()))
    import collections

    Result = collections.namedtuple("IterateResult", list(outs.keys()))  # type: ignore[misc]
    return Result(**outs)
