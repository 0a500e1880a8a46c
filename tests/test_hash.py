"""Hashing: python host impl vs vectorized torch impl (and vs known vectors)."""

import struct

import torch

from pathway_amd.internals import api
from pathway_amd.engine import hashing


def test_xxh64_known_vectors():
    # canonical xxh64 test vectors
    assert api.xxh64(b"", 0) == 0xEF46DB3751D8E999
    assert api.xxh64(b"a", 0) == 0xD24EC4F1A98C6E5B
    assert api.xxh64(b"abc", 0) == 0x44BC2CF5AD770999
    assert (
        api.xxh64(b"abcdefghijklmnopqrstuvwxyz0123456789", 0) == 0x64F23ECF1609B766
    )


def test_torch_words_match_python():
    torch.manual_seed(0)
    for nwords in (1, 2, 3, 4, 5, 8, 9):
        vals = torch.randint(-(2**62), 2**62, (17, nwords), dtype=torch.int64)
        got = hashing.xxh64_words([vals[:, j] for j in range(nwords)], seed=7)
        for i in range(vals.shape[0]):
            data = b"".join(
                struct.pack("<q", int(vals[i, j])) for j in range(nwords)
            )
            expect = api.xxh64(data, 7)
            g = int(got[i]) & ((1 << 64) - 1)
            assert g == expect, (nwords, i)


def test_value_hash_matches_serialize():
    t = torch.tensor([0, 1, -5, 2**40], dtype=torch.int64)
    lo, hi = hashing.column_value_hash(t, "int")
    for i, v in enumerate([0, 1, -5, 2**40]):
        elo, ehi = api.hash128(api.serialize_value(v))
        assert (int(lo[i]) & api.MASK64) == elo
        assert (int(hi[i]) & api.MASK64) == ehi


def test_float_and_bool_hash():
    t = torch.tensor([0.5, -1.25, 3.0], dtype=torch.float64)
    lo, hi = hashing.column_value_hash(t, "float")
    for i, v in enumerate([0.5, -1.25, 3.0]):
        elo, _ = api.hash128(api.serialize_value(v))
        assert (int(lo[i]) & api.MASK64) == elo

    b = torch.tensor([True, False])
    lo, hi = hashing.column_value_hash(b, "bool")
    for i, v in enumerate([True, False]):
        elo, _ = api.hash128(api.serialize_value(v))
        assert (int(lo[i]) & api.MASK64) == elo


def test_combined_row_hash_matches_hash_values():
    a = torch.tensor([3, 4], dtype=torch.int64)
    b = torch.tensor([10, 20], dtype=torch.int64)
    pa = hashing.column_value_hash(a, "int")
    pb = hashing.column_value_hash(b, "int")
    lo, hi = hashing.combine_value_hashes([pa, pb])
    for i, (x, y) in enumerate([(3, 10), (4, 20)]):
        elo, ehi = api.hash_values([x, y])
        assert (int(lo[i]) & api.MASK64) == elo
        assert (int(hi[i]) & api.MASK64) == ehi


def test_datetime_ns_precision_host_device_agree():
    # ns-precision datetimes must hash identically via the
    # host serialize path and the device int64-ns column path.
    import pandas as pd

    ns = 1694512345123456789
    ts = pd.Timestamp(ns, unit="ns")
    elo, ehi = api.hash128(api.serialize_value(ts))
    t = torch.tensor([ns], dtype=torch.int64)
    lo, hi = hashing.column_value_hash(t, "datetime_naive")
    assert (int(lo[0]) & api.MASK64) == elo
    assert (int(hi[0]) & api.MASK64) == ehi

    # duration too
    import datetime

    td = datetime.timedelta(microseconds=123456789123)
    expected_ns = int(pd.Timedelta(td).value)
    elo2, _ = api.hash128(api.serialize_value(td))
    lo2, _ = hashing.column_value_hash(
        torch.tensor([expected_ns], dtype=torch.int64), "duration"
    )
    assert (int(lo2[0]) & api.MASK64) == elo2


def test_float_zero_and_nan_normalization():
    # -0.0 and 0.0 hash identically; all NaNs hash to one canonical pattern.
    assert api.serialize_value(0.0) == api.serialize_value(-0.0)
    nan1 = float("nan")
    nan2 = struct.unpack("<d", struct.pack("<Q", 0x7FF8000000000001))[0]
    assert api.serialize_value(nan1) == api.serialize_value(nan2)

    t = torch.tensor([0.0, -0.0, nan1, nan2], dtype=torch.float64)
    lo, _ = hashing.column_value_hash(t, "float")
    assert int(lo[0]) == int(lo[1])
    assert int(lo[2]) == int(lo[3])
    elo, _ = api.hash128(api.serialize_value(0.0))
    assert (int(lo[0]) & api.MASK64) == elo
    enan, _ = api.hash128(api.serialize_value(nan1))
    assert (int(lo[2]) & api.MASK64) == enan


def test_ndarray_key_includes_shape_and_dtype():
    import numpy as np

    a = np.arange(4, dtype=np.int64).reshape(2, 2)
    b = np.arange(4, dtype=np.int64)
    assert api.serialize_value(a) != api.serialize_value(b)
    c = b.view(np.float64)
    assert api.serialize_value(b) != api.serialize_value(c)


def test_wrap_datetime_public_types():
    import pandas as pd

    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.datetime_types import (
        DateTimeNaive,
        DateTimeUtc,
        Duration,
    )

    assert dt.wrap(DateTimeNaive) == dt.DATE_TIME_NAIVE
    assert dt.wrap(DateTimeUtc) == dt.DATE_TIME_UTC
    assert dt.wrap(Duration) == dt.DURATION
    assert dt.wrap(pd.Timestamp) == dt.DATE_TIME_NAIVE
    assert dt.wrap(pd.Timedelta) == dt.DURATION

    class MyTs(pd.Timestamp):
        pass

    assert dt.wrap(MyTs) == dt.DATE_TIME_NAIVE
