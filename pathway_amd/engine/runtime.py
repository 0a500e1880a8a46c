"""Engine runtime: per-timestamp synchronous micro-batch execution.

The MI355X-native replacement of the timely worker loop
(reference dataflow.rs:7306-7500 run_with_new_dataflow_graph): instead of
cooperative operator scheduling with progress gossip, the runtime advances
one global timestamp at a time and pushes that timestamp's delta batches
through the node DAG in topological order — on a GPU the entire wave is a
sequence of device-wide kernels, so operator-level interleaving buys
nothing; cross-worker progress is a tiny min-allreduce per step
(progress is control-plane, O(operators) sized).
"""

from __future__ import annotations

from typing import Any, Callable, Iterable

import torch

from pathway_amd.engine.batch import DeltaBatch
from pathway_amd.engine.column import Column, column_from_pylist
from pathway_amd.engine.nodes import InputNode, Node, consolidate_batch
from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import BasePointer, DataRow


class Source:
    """Produces delta batches at nondecreasing times."""

    def next_time(self) -> int | None:
        """Smallest time with pending data; None when exhausted."""
        raise NotImplementedError

    def pull(self, time: int, device) -> DeltaBatch | None:
        raise NotImplementedError


class StaticSource(Source):
    """Fixed set of rows with explicit times (debug tables, static io)."""

    def __init__(
        self,
        rows: list[tuple[BasePointer, list[Any], int, int]],
        column_names: list[str],
        dtypes: list[dt.DType],
    ):
        self.column_names = column_names
        self.dtypes = dtypes
        by_time: dict[int, list] = {}
        for key, values, time, diff in rows:
            by_time.setdefault(time, []).append((key, values, diff))
        self.by_time = dict(sorted(by_time.items()))
        self._pending = sorted(self.by_time.keys())

    def reset(self) -> None:
        self._pending = sorted(self.by_time.keys())

    def seek(self, threshold_time: int) -> None:
        """Skip data at times ≤ threshold (recovery rewind, mod.rs:567)."""
        self._pending = [t for t in self._pending if t > threshold_time]

    def next_time(self) -> int | None:
        return self._pending[0] if self._pending else None

    def pull(self, time: int, device) -> DeltaBatch | None:
        if not self._pending or self._pending[0] != time:
            return None
        self._pending.pop(0)
        rows = self.by_time[time]
        keys = torch.tensor(
            [list(k.as_signed_pair()) for k, _, _ in rows],
            dtype=torch.int64,
            device=device,
        ).reshape(len(rows), 2)
        diffs = torch.tensor([d for _, _, d in rows], dtype=torch.int64, device=device)
        cols: dict[str, Column] = {}
        for j, name in enumerate(self.column_names):
            vals = [v[j] for _, v, _ in rows]
            cols[name] = column_from_pylist(vals, self.dtypes[j], device)
        return DeltaBatch(keys, cols, diffs, time)


class CallbackSource(Source):
    """Streaming source driven by a generator of (time, rows) batches."""

    def __init__(self, gen: Iterable, column_names: list[str], dtypes: list[dt.DType]):
        self.gen = iter(gen)
        self.column_names = column_names
        self.dtypes = dtypes
        self._buffered: tuple[int, Any] | None = None
        self._done = False
        self._advance()

    def _advance(self):
        if self._done:
            return
        try:
            self._buffered = next(self.gen)
        except StopIteration:
            self._buffered = None
            self._done = True

    def next_time(self) -> int | None:
        return None if self._buffered is None else self._buffered[0]

    def pull(self, time: int, device) -> DeltaBatch | None:
        if self._buffered is None or self._buffered[0] != time:
            return None
        _, payload = self._buffered
        self._advance()
        if isinstance(payload, DeltaBatch):
            return payload.to_device(device)
        rows = payload
        keys = torch.tensor(
            [list(k.as_signed_pair()) for k, _, _ in rows],
            dtype=torch.int64,
            device=device,
        ).reshape(len(rows), 2)
        diffs = torch.tensor([d for _, _, d in rows], dtype=torch.int64, device=device)
        cols: dict[str, Column] = {}
        for j, name in enumerate(self.column_names):
            vals = [v[j] for _, v, _ in rows]
            cols[name] = column_from_pylist(vals, self.dtypes[j], device)
        return DeltaBatch(keys, cols, diffs, time)


class PushSource(Source):
    """Rows pushed programmatically at runtime (REST servers, interactive
    sessions, python ConnectorSubjects in streaming mode)."""

    def __init__(self, column_names: list[str], dtypes: list[dt.DType]):
        self.column_names = column_names
        self.dtypes = dtypes
        self.pending: dict[int, list] = {}

    def push(self, key: BasePointer, values: list[Any], time: int, diff: int = 1):
        self.pending.setdefault(time, []).append((key, values, diff))

    def next_time(self) -> int | None:
        return min(self.pending) if self.pending else None

    def pull(self, time: int, device) -> DeltaBatch | None:
        rows = self.pending.pop(time, None)
        if not rows:
            return None
        keys = torch.tensor(
            [list(k.as_signed_pair()) for k, _, _ in rows],
            dtype=torch.int64,
            device=device,
        ).reshape(len(rows), 2)
        diffs = torch.tensor([d for _, _, d in rows], dtype=torch.int64, device=device)
        cols: dict[str, Column] = {}
        for j, name in enumerate(self.column_names):
            vals = [v[j] for _, v, _ in rows]
            cols[name] = column_from_pylist(vals, self.dtypes[j], device)
        return DeltaBatch(keys, cols, diffs, time)

    def reset(self) -> None:
        pass  # pushed-but-unconsumed rows survive a reset

    def seek(self, threshold_time: int) -> None:
        self.pending = {t: r for t, r in self.pending.items() if t > threshold_time}


#: process-wide "replay in progress" flag — sinks suppress re-emission of
#: already-delivered outputs during recovery replay
REPLAY_ACTIVE = [False]

#: streaming-source next_time() sentinels
STREAM_READY = object()   # has data now, any commit time works
STREAM_WAITING = object()  # alive but idle — the runtime should wait


def _check_errors(batch) -> None:
    """Raise on Value::Error reaching a sink when terminate_on_error is set
    (reference parse_graph.py:182-201 / table.py:2753 semantics)."""
    from pathway_amd.internals.config import pathway_config

    if not pathway_config.terminate_on_error:
        return
    from pathway_amd.engine.column import ObjectColumn
    from pathway_amd.internals.api import ERROR

    for name, col in batch.columns.items():
        if isinstance(col, ObjectColumn):
            for v in col.values:
                if v is ERROR:
                    raise RuntimeError(
                        f"Error value in column {name!r} reached an output; "
                        "use pw.fill_error(...) or set terminate_on_error=False"
                    )


class CaptureNode(Node):
    """Collects the full update stream of a table (debug / tests / sinks)."""

    def __init__(self, input_node: Node, device, column_names: list[str] | None = None):
        super().__init__([input_node], device)
        self.rows: list[DataRow] = []
        self.column_names = column_names

    def reset(self) -> None:
        self.rows = []

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0 or REPLAY_ACTIVE[0]:
            return None
        b = consolidate_batch(b)
        if b is None:
            return None
        _check_errors(b)
        for key, values, t, diff in b.rows():
            step = 1 if diff > 0 else -1
            for _ in range(abs(diff)):
                self.rows.append(DataRow(key=key, values=values, time=t, diff=step))
        return None


class OutputNode(Node):
    """Delivers consolidated batches to a writer callback (sinks)."""

    def __init__(self, input_node: Node, writer: Callable[[DeltaBatch], None], device):
        super().__init__([input_node], device)
        self.writer = writer

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0 or REPLAY_ACTIVE[0]:
            return None
        b = consolidate_batch(b)
        if b is not None:
            self.writer(b)
        return None

    def flush(self, time: int):
        fl = getattr(self.writer, "flush", None)
        if fl is not None:
            fl(time)


class SubscribeNode(Node):
    """io.subscribe: per-row on_change callback + on_time_end/on_end."""

    def __init__(
        self,
        input_node: Node,
        device,
        on_change: Callable,
        on_time_end: Callable | None = None,
        on_end: Callable | None = None,
    ):
        super().__init__([input_node], device)
        self.on_change = on_change
        self.on_time_end = on_time_end
        self.on_end = on_end

    def step(self, time, inputs):
        b = inputs[0]
        if b is None or len(b) == 0 or REPLAY_ACTIVE[0]:
            return None
        b = consolidate_batch(b)
        if b is None:
            return None
        names = list(b.columns.keys())
        for key, values, t, diff in b.rows():
            row = dict(zip(names, values))
            self.on_change(key=key, row=row, time=t, is_addition=diff > 0)
        return None

    def finish_time(self, time: int):
        if self.on_time_end is not None:
            self.on_time_end(time)

    def finish(self):
        if self.on_end is not None:
            self.on_end()


def topo_order(sinks: list[Node]) -> list[Node]:
    seen: dict[int, Node] = {}
    order: list[Node] = []

    def visit(n: Node, stack: set[int]):
        if id(n) in seen:
            return
        if id(n) in stack:
            raise RuntimeError("cycle in dataflow graph")
        stack.add(id(n))
        for i in n.inputs:
            visit(i, stack)
        stack.discard(id(n))
        seen[id(n)] = n
        order.append(n)

    for s in sinks:
        visit(s, set())
    return order


class Runtime:
    """Single-worker synchronous runtime; multi-GPU coordination lives in
    parallel/exchange.py (each rank runs its own Runtime in lockstep)."""

    def __init__(self, sinks: list[Node], device=None, comm=None, persistence=None):
        if device is None:
            # follow the device the graph was built on (PW_DEVICE)
            from pathway_amd.internals.config import get_device

            device = get_device()
        self.device = torch.device(device)
        self.nodes = topo_order(sinks)
        self.sources = [n for n in self.nodes if isinstance(n, InputNode)]
        self.sinks = sinks
        self.comm = comm  # parallel context or None
        self.persistence = persistence
        self._clock = 0
        from pathway_amd.engine.monitoring import RunStats

        self.stats = RunStats()
        self.monitor = None
        self._metrics_recorder = None
        from pathway_amd.internals.config import pathway_config

        mdir = getattr(pathway_config, "detailed_metrics_dir", None)
        if mdir:
            from pathway_amd.engine.monitoring import DetailedMetricsRecorder

            self._metrics_recorder = DetailedMetricsRecorder(mdir)
        for i, src in enumerate(self.sources):
            if not getattr(src, "persistent_id", None):
                src.persistent_id = getattr(src.source, "name", None) or f"src{i}"
        if persistence is not None:
            # connector offsets (e.g. Kafka (topic, partition) -> next
            # offset) ride the AdvanceTime events so recovery can seek
            # readers (reference OffsetAntichain, connectors/mod.rs:319)
            def _collect():
                out = {}
                for s in self.sources:
                    reader = getattr(s.source, "reader", None)
                    offs = getattr(reader, "offsets", None)
                    if offs:
                        out[s.persistent_id] = [
                            (str(k), str(v)) for k, v in offs.items()
                        ]
                return out

            persistence._offsets_fn = _collect

    def _next_time(self):
        """Returns (time, waiting): time=None & waiting=False means done."""
        BIG = 2**62
        raw = [s.source.next_time() for s in self.sources]
        numeric = [t for t in raw if isinstance(t, int)]
        ready = any(t is STREAM_READY for t in raw)
        waiting = any(t is STREAM_WAITING for t in raw)
        if numeric:
            local = min(numeric)
        elif ready:
            local = self._clock
        elif waiting:
            local = BIG - 1
        else:
            local = BIG
        if self.comm is not None:
            enc = self.comm.allreduce_min_time(local)
            local = enc if enc is not None else BIG
        if local >= BIG:
            return None, False
        if local == BIG - 1:
            return None, True
        return local, False

    def step_once(self, time: int, injected: dict[int, DeltaBatch | None] | None = None):
        outputs: dict[int, DeltaBatch | None] = {}
        n_ingested = 0
        n_output = 0
        pm = self.persistence
        import time as _t

        from pathway_amd.internals import roctx

        trace = roctx.enabled()
        ops_stats = self.stats.operators
        for node in self.nodes:
            _s0 = _t.perf_counter()
            if trace:
                roctx.range_push(f"pw::{type(node).__name__}@{time}")
            if isinstance(node, InputNode):
                if injected is not None:
                    out = injected.get(id(node))
                else:
                    out = node.step(time, [])
                    if out is not None:
                        n_ingested += len(out)
                    if (
                        pm is not None
                        and out is not None
                        and not getattr(pm, "operator_persisting", False)
                    ):
                        pm.record(node.persistent_id, time, out)
            else:
                ins = [outputs.get(id(i)) for i in node.inputs]
                if all(b is None for b in ins) and not node.wants_frontier():
                    out = None
                else:
                    out = node.step(time, ins)
                    st = ops_stats[type(node).__name__]
                    st.steps += 1
                    st.total_time_s += _t.perf_counter() - _s0
                    st.rows_in += sum(len(b) for b in ins if b is not None)
                    if out is not None:
                        st.rows_out += len(out)
            if trace:
                roctx.range_pop()
            outputs[id(node)] = out
        for node in self.nodes:
            if isinstance(node, (SubscribeNode, OutputNode, CaptureNode)):
                b = outputs.get(id(node.inputs[0]))
                if b is not None:
                    n_output += len(b)
            if isinstance(node, SubscribeNode):
                node.finish_time(time)
            if isinstance(node, OutputNode):
                node.flush(time)
        return n_ingested, n_output

    def _stateful_node_ids(self):
        out = []
        for i, n in enumerate(self.nodes):
            out.append((f"{i}:{type(n).__name__}", n))
        return out

    def save_operator_snapshot(self, time: int) -> None:
        from pathway_amd.persistence.operator_snapshot import node_state_save

        states = {}
        for nid, node in self._stateful_node_ids():
            st = node_state_save(node)
            if st is not None:
                states[nid] = st
        self.persistence.op_store.save(time, states)

    def load_operator_snapshot(self) -> bool:
        from pathway_amd.persistence.operator_snapshot import node_state_load

        snap = self.persistence.op_store.load()
        if snap is None:
            return False
        time, states = snap
        by_id = dict(self._stateful_node_ids())
        for nid, st in states.items():
            node = by_id.get(nid)
            if node is not None:
                node_state_load(node, st, self.device)
        for src in self.sources:
            seek = getattr(src.source, "seek", None)
            if seek is not None:
                seek(time)
        return True

    def replay_persisted(self) -> None:
        """Recovery: push snapshotted input batches through the graph
        (sinks suppressed), then seek live sources past the threshold."""
        pm = self.persistence
        if pm is None or pm.threshold_time < 0:
            return
        from pathway_amd.engine.column import infer_and_build_column
        from pathway_amd.internals.api import BasePointer

        by_time: dict[int, dict[int, DeltaBatch]] = {}
        for src in self.sources:
            for time, block in pm.replay_blocks(src.persistent_id):
                names = block["names"]
                rows = block["rows"]
                keys = torch.tensor(
                    [
                        list(BasePointer(lo, hi).as_signed_pair())
                        for (lo, hi), _, _ in rows
                    ],
                    dtype=torch.int64,
                    device=self.device,
                ).reshape(len(rows), 2)
                diffs = torch.tensor(
                    [d for _, _, d in rows], dtype=torch.int64, device=self.device
                )
                cols = {}
                for j, n in enumerate(names):
                    vals = [v[j] for _, v, _ in rows]
                    cols[n], _ = infer_and_build_column(vals, self.device)
                by_time.setdefault(time, {})[id(src)] = DeltaBatch(
                    keys, cols, diffs, time
                )
        REPLAY_ACTIVE[0] = True
        try:
            for t in sorted(by_time):
                self.step_once(t, injected=by_time[t])
        finally:
            REPLAY_ACTIVE[0] = False
        for src in self.sources:
            seek = getattr(src.source, "seek", None)
            if seek is not None:
                seek(pm.threshold_time)

    def run(self, max_steps: int | None = None) -> None:
        steps = 0
        if self.persistence is not None and not getattr(self, "_replayed", False):
            if getattr(self.persistence, "operator_persisting", False):
                self.load_operator_snapshot()
            else:
                self.replay_persisted()
            self._replayed = True
        import os
        import time as _time

        elastic = None
        if os.environ.get("PATHWAY_ELASTIC") and self.persistence is not None:
            from pathway_amd.engine.monitoring import (
                EXIT_CODE_DOWNSCALE,
                EXIT_CODE_UPSCALE,
                WorkloadTracker,
            )

            elastic = WorkloadTracker()
        _last_loop = _time.perf_counter()
        while True:
            t, waiting = self._next_time()
            if t is None:
                if not waiting:
                    break
                _time.sleep(0.005)
                continue
            _s0 = _time.perf_counter()
            ingested, output = self.step_once(t)
            _now = _time.perf_counter()
            self.stats.record_step(t, _now - _s0, ingested, output)
            if self._metrics_recorder is not None and (
                _now - getattr(self, "_last_metrics_flush", 0.0) >= 1.0
            ):
                self._metrics_recorder.record(self.stats)
                self._last_metrics_flush = _now
            if elastic is not None:
                busy = (_now - _s0) / max(_now - _last_loop, 1e-9)
                advice = elastic.add_point(min(busy, 1.0))
                _last_loop = _now
                if advice == "up":
                    import sys

                    from pathway_amd.engine.monitoring import EXIT_CODE_UPSCALE

                    sys.exit(EXIT_CODE_UPSCALE)
                if advice == "down":
                    import sys

                    from pathway_amd.engine.monitoring import EXIT_CODE_DOWNSCALE

                    sys.exit(EXIT_CODE_DOWNSCALE)
            if self.monitor is not None:
                self.monitor.maybe_report()
            self._clock = max(self._clock, t + 2)
            if self.persistence is not None:
                if getattr(self.persistence, "operator_persisting", False):
                    # honor snapshot_interval_ms (reference saves operator
                    # state at interval boundaries, not every step)
                    interval = getattr(
                        self.persistence, "snapshot_interval_ms", 0
                    )
                    now = _time.monotonic()
                    last = getattr(self, "_last_opsnap", 0.0)
                    if interval <= 0 or (now - last) * 1000.0 >= interval:
                        self.save_operator_snapshot(t)
                        self._last_opsnap = now
                self.persistence.commit(t)
            steps += 1
            if max_steps is not None and steps >= max_steps:
                break
        if self._metrics_recorder is not None:
            self._metrics_recorder.record(self.stats)
        for node in self.nodes:
            if isinstance(node, SubscribeNode):
                node.finish()
