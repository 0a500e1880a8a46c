"""MySQL connector: v10 handshake + COM_QUERY wire client against the
in-process fake server (real packet framing, lenenc resultsets)."""

import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.io._mysql_protocol import MySQLClient, native_password_auth
from tests.fakes.fake_mysql import FakeMySQL


@pytest.fixture()
def my():
    s = FakeMySQL().start()
    yield s
    s.stop()


def _settings(my):
    return {"host": "127.0.0.1", "port": my.port, "user": "u",
            "password": "pw", "database": "d"}


def test_native_password_scramble():
    # deterministic: SHA1(pass) ^ SHA1(nonce + SHA1(SHA1(pass)))
    out = native_password_auth("secret", b"0" * 20)
    assert len(out) == 20
    assert native_password_auth("", b"0" * 20) == b""
    assert out != native_password_auth("secret", b"1" * 20)


def test_wire_client_roundtrip(my):
    c = MySQLClient(port=my.port, user="u", password="p", database="d")
    c.query("CREATE TABLE t1 (a, b)")
    c.query("INSERT INTO t1 (a, b) VALUES (1, 'x'), (2, 'y\\'z')")
    cols, rows = c.query("SELECT * FROM t1")
    assert cols == ["a", "b"]
    assert rows == [["1", "x"], ["2", "y'z"]]
    c.query("DELETE FROM t1 WHERE a = 1")
    _, rows = c.query("SELECT * FROM t1")
    assert len(rows) == 1
    c.close()


def test_mysql_write_and_static_read(my):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    pw.io.mysql.write(t, _settings(my), "out",
                      init_mode="create_if_not_exists")
    pw.run()

    G.clear()
    back = pw.io.mysql.read(
        _settings(my), "out",
        schema=schema_from_types(a=int, b=str), mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(back)
    assert sorted((cols["a"][k], cols["b"][k]) for k in keys) == [
        (1, "x"), (2, "y")
    ]


def test_mysql_streaming_watermark_tail(my):
    c = MySQLClient(port=my.port)
    c.query("CREATE TABLE live (seq, v)")
    c.query("INSERT INTO live (seq, v) VALUES (1, 'a')")

    def later():
        time.sleep(0.3)
        c2 = MySQLClient(port=my.port)
        c2.query("INSERT INTO live (seq, v) VALUES (2, 'b'), (3, 'c')")
        c2.close()

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    t = pw.io.mysql.read(
        _settings(my), "live",
        schema=schema_from_types(seq=int, v=str), mode="streaming",
        watermark_column="seq", refresh_interval=0.1, _max_polls=10,
    )
    res = t.groupby().reduce(n=pw.reducers.count(),
                             s=pw.reducers.sum(pw.this.seq))
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    assert list(cols["n"].values()) == [3]
    assert list(cols["s"].values()) == [6]
    c.close()
