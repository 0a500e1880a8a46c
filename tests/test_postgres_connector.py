"""Postgres connector: wire-protocol v3 client, simple query, logical
replication CDC — against the in-process fake server."""

import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.io._pg_protocol import PgClient
from tests.fakes.fake_postgres import FakePostgres


@pytest.fixture()
def pg():
    s = FakePostgres().start()
    yield s
    s.stop()


def _settings(pg):
    return {"host": "127.0.0.1", "port": pg.port, "user": "u", "dbname": "d"}


def test_wire_client_query(pg):
    c = PgClient(port=pg.port, user="u", database="d")
    c.query("CREATE TABLE t1 (a, b)")
    c.query("INSERT INTO t1 (a, b) VALUES (1, 'x'), (2, 'y''z')")
    cols, rows = c.query("SELECT * FROM t1")
    assert cols == ["a", "b"]
    assert rows == [["1", "x"], ["2", "y'z"]]
    cols, rows = c.query("SELECT * FROM t1 WHERE a = 2")
    assert rows == [["2", "y'z"]]
    c.query("DELETE FROM t1 WHERE a = 1")
    _, rows = c.query("SELECT * FROM t1")
    assert len(rows) == 1
    c.close()


def test_wire_client_error(pg):
    c = PgClient(port=pg.port)
    with pytest.raises(Exception) as e:
        c.query("GRANT NONSENSE")
    assert "unsupported" in str(e.value)
    # connection still usable after error
    c.query("CREATE TABLE ok (x)")
    c.close()


def test_postgres_write_and_static_read(pg):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    pw.io.postgres.write(t, _settings(pg), "out",
                         init_mode="create_if_not_exists")
    pw.run()

    G.clear()
    back = pw.io.postgres.read(
        _settings(pg), "out",
        schema=schema_from_types(a=int, b=str), mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(back)
    assert sorted((cols["a"][k], cols["b"][k]) for k in keys) == [
        (1, "x"), (2, "y")
    ]


def test_postgres_write_retraction_deletes(pg):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b | __time__ | __diff__
        1 | x | 2        | 1
        2 | y | 2        | 1
        2 | y | 4        | -1
        """
    )
    pw.io.postgres.write(t, _settings(pg), "snap",
                         init_mode="create_if_not_exists")
    pw.run()
    c = PgClient(port=pg.port)
    _, rows = c.query("SELECT * FROM snap")
    assert [(r[0], r[1]) for r in rows] == [("1", "x")]
    c.close()


def test_postgres_cdc_streaming(pg):
    """Logical replication: inserts and deletes stream as CDC events."""
    c = PgClient(port=pg.port)
    c.query("CREATE TABLE live (k, v)")
    c.query("INSERT INTO live (k, v) VALUES (1, 'a')")

    def later():
        time.sleep(0.3)
        c2 = PgClient(port=pg.port)
        c2.query("INSERT INTO live (k, v) VALUES (2, 'b'), (3, 'c')")
        c2.query("DELETE FROM live WHERE k = 1")
        c2.close()

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    t = pw.io.postgres.read(
        _settings(pg), "live",
        schema=schema_from_types(k=int, v=str), mode="streaming",
        _max_changes=4,
    )
    keys, cols = pw.debug.table_to_dicts(t)
    th.join()
    rows = sorted((cols["k"][k], cols["v"][k]) for k in keys)
    # k=1 inserted then deleted; k=2,3 remain
    assert rows == [(2, "b"), (3, "c")]
    c.close()


def test_replication_protocol_frames(pg):
    """The raw START_REPLICATION stream yields XLogData payloads."""
    c = PgClient(port=pg.port)
    c.query("CREATE TABLE w (x)")
    c.query("INSERT INTO w (x) VALUES (42)")
    c.close()

    rc = PgClient(port=pg.port, replication=True)
    gen = rc.start_replication("s1")
    lsn, payload = next(gen)
    assert lsn > 0
    import json

    change = json.loads(payload)["change"][0]
    assert change["kind"] == "insert"
    assert change["table"] == "w"
    assert change["columnvalues"] == [42]
    rc.close()
