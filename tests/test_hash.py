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
