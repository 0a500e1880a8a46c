"""MSSQL connector: TDS client (prelogin/login7/SQLBatch token streams)
against the in-process fake server."""

import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.io._tds_protocol import TdsClient, TdsError, encode_password
from tests.fakes.fake_mssql import FakeMSSQL


@pytest.fixture()
def ms():
    s = FakeMSSQL().start()
    yield s
    s.stop()


def _settings(ms):
    return {"host": "127.0.0.1", "port": ms.port, "user": "sa",
            "password": "pw", "database": "d"}


def test_password_obfuscation():
    # [MS-TDS] 2.2.6.4: nibble swap then XOR 0xA5 over UCS-2 bytes
    out = encode_password("A")  # 'A' = 0x41 0x00 UCS-2
    assert out == bytes([((0x41 << 4) & 0xF0 | 0x41 >> 4) ^ 0xA5,
                         ((0x00 << 4) & 0xF0 | 0x00 >> 4) ^ 0xA5])


def test_tds_client_roundtrip(ms):
    c = TdsClient(port=ms.port, user="sa", password="p", database="d")
    c.query("CREATE TABLE t1 (a, b)")
    c.query("INSERT INTO t1 (a, b) VALUES (1, N'x'), (2, N'y''z')")
    cols, rows = c.query("SELECT * FROM t1")
    assert cols == ["a", "b"]
    assert rows == [["1", "x"], ["2", "y'z"]]
    c.query("DELETE FROM t1 WHERE a = 1")
    _, rows = c.query("SELECT * FROM t1")
    assert len(rows) == 1
    with pytest.raises(TdsError):
        c.query("GRANT NONSENSE")
    c.close()


def test_mssql_write_and_read(ms):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    pw.io.mssql.write(t, _settings(ms), "out",
                      init_mode="create_if_not_exists")
    pw.run()

    G.clear()
    back = pw.io.mssql.read(
        _settings(ms), "out",
        schema=schema_from_types(a=int, b=str), mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(back)
    assert sorted((cols["a"][k], cols["b"][k]) for k in keys) == [
        (1, "x"), (2, "y")
    ]


def test_mssql_streaming_watermark(ms):
    c = TdsClient(port=ms.port)
    c.query("CREATE TABLE live (seq, v)")
    c.query("INSERT INTO live (seq, v) VALUES (1, N'a')")

    def later():
        time.sleep(0.3)
        c2 = TdsClient(port=ms.port)
        c2.query("INSERT INTO live (seq, v) VALUES (2, N'b')")
        c2.close()

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    t = pw.io.mssql.read(
        _settings(ms), "live",
        schema=schema_from_types(seq=int, v=str), mode="streaming",
        watermark_column="seq", refresh_interval=0.1, _max_polls=10,
    )
    res = t.groupby().reduce(n=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    assert list(cols["n"].values()) == [2]
    c.close()
