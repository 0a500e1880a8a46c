"""Persistence format round 2: bincode-compatible event codec, LZ4
chunks, metadata keys, backends, restricted operator-snapshot loading.

Reference formats: src/persistence/{input_snapshot,state}.rs, backends/.
"""

import numpy as np
import pytest

import pathway_amd as pw
from pathway_amd.internals.api import Pointer
from pathway_amd.internals.datetime_types import DateTimeNaive, Duration
from pathway_amd.internals.json import Json
from pathway_amd.persistence import codec
from pathway_amd.persistence.backends import FileStore, MockStore, S3Store


def test_value_roundtrip():
    vals = [
        None, True, False, 42, -1, 2**62, 3.5, float("inf"),
        "żółć", b"\x00\xff", (1, "two", None, (3.0,)),
        Pointer(123, 456),
        np.arange(6, dtype=np.int64).reshape(2, 3),
        np.array([1.5, -2.5]),
        DateTimeNaive.from_ns(1694512345123456789),
        Duration.from_ns(-42),
        Json({"a": [1, {"b": None}]}),
    ]
    for v in vals:
        out = bytearray()
        codec.encode_value(out, v)
        r = codec._Reader(bytes(out))
        back = codec.decode_value(r)
        if isinstance(v, np.ndarray):
            assert np.array_equal(back, v) and back.shape == v.shape
        elif isinstance(v, Json):
            assert back.value == v.value
        else:
            assert back == v, v
        assert r.i == len(out)


def test_event_byte_layout_pinned():
    """Format-stability: the exact bytes are pinned (bincode v1 rules —
    u32 variant tags, u64 lengths, LE fixed-width)."""
    ev = codec.encode_event(
        codec.E_INSERT, key=Pointer(1, 2), values=[7, "hi"]
    )
    expect = (
        b"\x00\x00\x00\x00"              # Event::Insert
        b"\x01" + b"\x00" * 7            # key.lo = 1
        + b"\x02" + b"\x00" * 7          # key.hi = 2
        + b"\x02" + b"\x00" * 7          # vec len = 2
        + b"\x02\x00\x00\x00"            # Value::Int
        + b"\x07" + b"\x00" * 7          # 7
        + b"\x05\x00\x00\x00"            # Value::String
        + b"\x02" + b"\x00" * 7          # len 2
        + b"hi"
    )
    assert ev == expect

    adv = codec.encode_event(codec.E_ADVANCE_TIME, time=4,
                             offsets=[("t:0", "5")])
    assert adv[:4] == b"\x02\x00\x00\x00"
    kind, (t, offs), _ = codec.decode_event(adv)
    assert (kind, t, offs) == (codec.E_ADVANCE_TIME, 4, [("t:0", "5")])


def test_event_roundtrip_delete_finished():
    ev = codec.encode_event(codec.E_DELETE, key=Pointer(9, 8), values=[None])
    kind, (key, values), n = codec.decode_event(ev)
    assert kind == codec.E_DELETE and key == Pointer(9, 8) and values == [None]
    fin = codec.encode_event(codec.E_FINISHED)
    assert codec.decode_event(fin)[0] == codec.E_FINISHED


def test_codec_rejects_pyobjects():
    class Weird:
        pass

    with pytest.raises(codec.CodecError):
        codec.encode_value(bytearray(), Weird())


def _run_persisted(backend, tmp_path, rows):
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types

    G.clear()
    schema = schema_from_types(v=int)
    t = table_from_rows(schema, rows, is_stream=True)
    res = t.reduce(s=pw.reducers.sum(pw.this.v))
    out = str(tmp_path / "out.csv")
    pw.io.csv.write(res, out)
    cfg = pw.persistence.Config(backend=backend)
    pw.run(persistence_config=cfg)


def test_persistence_over_mock_backend(tmp_path):
    backend = pw.persistence.Backend.mock()
    _run_persisted(backend, tmp_path, [(1, 0, 1), (2, 2, 1)])
    store = backend._store
    keys = store.list("")
    assert any(k.startswith("metadata/") for k in keys)
    assert any(k.startswith("snapshots/0/") for k in keys)
    # chunks decode with the binary codec (no pickle)
    from pathway_amd.persistence.engine import SnapshotReader

    src_prefix = sorted(
        {k.rsplit("/", 1)[0] for k in keys if k.startswith("snapshots/0/")}
    )[0]
    frames = list(SnapshotReader(store, src_prefix).frames())
    assert frames
    assert frames[0][:4] == b"\x00\x00\x00\x00"  # data frame kind


def test_persistence_over_s3_backend(tmp_path):
    from tests.fakes.fake_s3 import FakeS3

    srv = FakeS3().start()
    try:
        backend = pw.persistence.Backend.s3(
            "persist/root",
            bucket_settings={"bucket_name": "pb", "endpoint": srv.endpoint,
                             "access_key": "a", "secret_access_key": "s"},
        )
        _run_persisted(backend, tmp_path, [(5, 0, 1)])
        keys = [k for (b, k) in srv.objects if b == "pb"]
        assert any("metadata/" in k for k in keys)
        assert any("snapshots/0/" in k for k in keys)
    finally:
        srv.stop()


def test_recovery_threshold_multi_worker_min():
    from pathway_amd.persistence.engine import PersistenceManager

    class Cfg:
        backend = pw.persistence.Backend.mock()
        persistence_mode = None
        snapshot_interval_ms = 0

    store = None
    pm0 = PersistenceManager(Cfg(), worker=0)
    store = pm0.store
    # simulate two workers at different committed times
    import json as _json

    store.put("metadata/2-0-0", _json.dumps({"threshold_time": 10}).encode())
    store.put("metadata/2-1-0", _json.dumps({"threshold_time": 6}).encode())
    store.put("metadata/2-1-1", _json.dumps({"threshold_time": 8}).encode())

    class Cfg2:
        backend = Cfg.backend
        persistence_mode = None
        snapshot_interval_ms = 0

    Cfg2.backend._store = store
    pm = PersistenceManager(Cfg2(), worker=0)
    # worker 0 latest = 10, worker 1 latest = 8 -> min = 8
    assert pm.threshold_time == 8


def test_restricted_unpickler_blocks_code():
    import pickle

    from pathway_amd.persistence.operator_snapshot import _safe_loads

    class Evil:
        def __reduce__(self):
            return (__import__("os").system, ("echo pwned",))

    payload = pickle.dumps(Evil())
    with pytest.raises(Exception):
        _safe_loads(payload)
    # benign numpy payloads still load
    ok = pickle.dumps({"a": np.arange(3), "b": [1, "x"]})
    back = _safe_loads(ok)
    assert back["b"] == [1, "x"]


def test_lz4_chunk_rotation(tmp_path):
    from pathway_amd.persistence.engine import SnapshotReader, SnapshotWriter

    store = FileStore(str(tmp_path))
    w = SnapshotWriter(store, "snapshots/0/src", max_chunk_bytes=200)
    for i in range(10):
        w.write_block(b"payload-%03d" % i + b"x" * 50, 1)
    w.close()
    chunks = store.list("snapshots/0/src/")
    assert len(chunks) > 1  # rotated by byte budget
    frames = list(SnapshotReader(store, "snapshots/0/src").frames())
    assert len(frames) == 10
    assert frames[3].startswith(b"payload-003")


def test_kafka_offsets_ride_persistence(tmp_path):
    """Connector offsets land in AdvanceTime events; source_offsets()
    recovers them for seek (reference OffsetAntichain round trip)."""
    import json as _json

    import pathway_amd as pw
    from pathway_amd.internals.rungraph import G
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.persistence.engine import PersistenceManager
    from tests.fakes.fake_kafka import FakeKafkaBroker

    broker = FakeKafkaBroker(num_partitions=1).start()
    try:
        for i in range(3):
            broker.seed("pt", 0, [(None, _json.dumps({"k": i}).encode())])
        G.clear()
        t = pw.io.kafka.read(
            rdkafka_settings={"bootstrap.servers": broker.bootstrap},
            topic="pt", schema=schema_from_types(k=int), format="json",
            mode="static", name="ksrc",
        )
        out = str(tmp_path / "o.csv")
        pw.io.csv.write(t, out)
        backend = pw.persistence.Backend.filesystem(str(tmp_path / "p"))
        cfg = pw.persistence.Config(backend=backend)
        pw.run(persistence_config=cfg)

        class Cfg:
            pass

        Cfg.backend = backend
        Cfg.persistence_mode = None
        Cfg.snapshot_interval_ms = 0
        pm = PersistenceManager(Cfg(), worker=0)
        offs = pm.source_offsets("ksrc")
        assert offs.get("('pt', 0)") == "3"
    finally:
        broker.stop()
