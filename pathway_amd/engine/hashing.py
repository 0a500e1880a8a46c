"""Vectorized xxh64/128 over word-aligned rows — torch implementation.

Bit-exact with pathway_amd.internals.api.xxh64 for inputs whose length is a
multiple of 8 bytes.  Runs on CPU and ROCm tensors alike; on the GPU the hot
string/varlen path is replaced by the hand-written HIP kernel
(ops/csrc/hip/hash_kernels.hip) — this module doubles as its fp-exact
reference for numerics tests.

int64 arithmetic note: torch int64 add/mul wrap mod 2^64 in two's complement,
which is bit-identical to unsigned mod-2^64 arithmetic; logical shifts are
emulated by arithmetic shift + mask.
"""

from __future__ import annotations

import torch

from pathway_amd.internals.api import (
    SEED_HI,
    SEED_LO,
    TAG_BOOL,
    TAG_DT_NAIVE,
    TAG_DT_UTC,
    TAG_DURATION,
    TAG_FLOAT,
    TAG_INT,
    TAG_NONE,
    TAG_POINTER,
)

_P1 = 0x9E3779B185EBCA87
_P2 = 0xC2B2AE3D27D4EB4F
_P3 = 0x165667B19E3779F9
_P4 = 0x85EBCA77C2B2AE63
_P5 = 0x27D4EB2F165667C5


def _c(x: int) -> int:
    """Constant as signed int64 two's-complement."""
    x &= (1 << 64) - 1
    return x - (1 << 64) if x >= (1 << 63) else x


def _lshr(x: torch.Tensor, s: int) -> torch.Tensor:
    # logical right shift for int64 tensors
    return (x >> s) & ((1 << (64 - s)) - 1)


def _rotl(x: torch.Tensor, r: int) -> torch.Tensor:
    return (x << r) | _lshr(x, 64 - r)


def _round(acc: torch.Tensor, inp: torch.Tensor) -> torch.Tensor:
    acc = acc + inp * _c(_P2)
    acc = _rotl(acc, 31)
    return acc * _c(_P1)


def _merge_round(acc: torch.Tensor, val: torch.Tensor) -> torch.Tensor:
    val = _round(torch.zeros_like(val), val)
    acc = acc ^ val
    return acc * _c(_P1) + _c(_P4)


def _avalanche(h: torch.Tensor) -> torch.Tensor:
    h = h ^ _lshr(h, 33)
    h = h * _c(_P2)
    h = h ^ _lshr(h, 29)
    h = h * _c(_P3)
    h = h ^ _lshr(h, 32)
    return h


def xxh64_words(words: list[torch.Tensor], seed: int) -> torch.Tensor:
    """xxh64 of rows made of len(words) 8-byte words; vectorized over rows.

    words: list of int64 tensors, all same shape (n,).  Returns (n,) int64.
    """
    nwords = len(words)
    nbytes = nwords * 8
    seed = _c(seed)
    if nwords == 0:
        raise ValueError("empty rows")
    proto = words[0]
    i = 0
    if nbytes >= 32:
        v1 = torch.full_like(proto, _c(seed + _P1 + _P2))
        v2 = torch.full_like(proto, _c(seed + _P2))
        v3 = torch.full_like(proto, seed)
        v4 = torch.full_like(proto, _c(seed - _P1))
        while (i + 4) * 8 <= nbytes:
            v1 = _round(v1, words[i])
            v2 = _round(v2, words[i + 1])
            v3 = _round(v3, words[i + 2])
            v4 = _round(v4, words[i + 3])
            i += 4
        h = _rotl(v1, 1) + _rotl(v2, 7) + _rotl(v3, 12) + _rotl(v4, 18)
        h = _merge_round(h, v1)
        h = _merge_round(h, v2)
        h = _merge_round(h, v3)
        h = _merge_round(h, v4)
    else:
        h = torch.full_like(proto, _c(seed + _P5))
    h = h + nbytes
    while i < nwords:
        k = _round(torch.zeros_like(proto), words[i])
        h = h ^ k
        h = _rotl(h, 27) * _c(_P1) + _c(_P4)
        i += 1
    return _avalanche(h)


def hash128_words(words: list[torch.Tensor]) -> tuple[torch.Tensor, torch.Tensor]:
    if words[0].is_cuda and len(words) <= 20:
        # GPU path MUST go through the fused HIP kernel (ops raises loudly
        # if libpwhip.so is missing on a GPU host); >20 words (very wide
        # rows) use the torch formulation, which also runs on device
        from pathway_amd import ops

        return ops.hash128_words_gpu([w.contiguous() for w in words])
    return xxh64_words(words, SEED_LO), xxh64_words(words, SEED_HI)


def _float_bits(x: torch.Tensor) -> torch.Tensor:
    # Normalize -0.0 -> 0.0 and NaN -> canonical 0x7FF8... so values that
    # compare equal hash equal, matching api.serialize_value.
    x = x.to(torch.float64)
    x = torch.where(x == 0.0, torch.zeros_like(x), x)
    bits = x.view(torch.int64)
    canon_nan = torch.full_like(bits, 0x7FF8000000000000)
    return torch.where(torch.isnan(x), canon_nan, bits)


def value_hash_words(col_words: torch.Tensor, tag: int) -> tuple[torch.Tensor, torch.Tensor]:
    """128-bit per-value hash of a fixed-width column: hash of [tag, payload]."""
    if col_words.is_cuda:
        from pathway_amd import ops

        return ops.value_hash_gpu(col_words, tag)
    tags = torch.full_like(col_words, tag)
    return hash128_words([tags, col_words])


def column_value_hash(
    tensor: torch.Tensor, dtype_kind: str
) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-value 128-bit hashes of a device column, matching serialize_value.

    dtype_kind: one of int/float/bool/datetime_naive/datetime_utc/duration.
    """
    tag = {
        "int": TAG_INT,
        "float": TAG_FLOAT,
        "bool": TAG_BOOL,
        "datetime_naive": TAG_DT_NAIVE,
        "datetime_utc": TAG_DT_UTC,
        "duration": TAG_DURATION,
    }[dtype_kind]
    if dtype_kind == "float":
        words = _float_bits(tensor)
    elif dtype_kind == "bool":
        words = tensor.to(torch.int64)
    else:
        words = tensor.to(torch.int64)
    return value_hash_words(words, tag)


def pointer_value_hash(
    keys: torch.Tensor,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-value hash of a pointer column stored as (n,2) int64."""
    tags = torch.full_like(keys[:, 0], TAG_POINTER)
    return hash128_words([tags, keys[:, 0], keys[:, 1]])


def combine_value_hashes(
    parts: list[tuple[torch.Tensor, torch.Tensor]]
) -> tuple[torch.Tensor, torch.Tensor]:
    """hash_values(): 128-bit hash of concatenated per-value (lo,hi) pairs."""
    words: list[torch.Tensor] = []
    for lo, hi in parts:
        words.append(lo)
        words.append(hi)
    return hash128_words(words)


def derive_key_words(
    salt: int, parts: list[tuple[torch.Tensor, torch.Tensor]]
) -> tuple[torch.Tensor, torch.Tensor]:
    """Vectorized api.derive_key: salted hash of input key pairs."""
    proto = parts[0][0]
    words: list[torch.Tensor] = [torch.full_like(proto, _c(salt))]
    for lo, hi in parts:
        words.append(lo)
        words.append(hi)
    return hash128_words(words)


_NONE_HASH_CACHE: dict[str, tuple[torch.Tensor, torch.Tensor]] = {}


def none_value_hash_scalar(device) -> tuple[torch.Tensor, torch.Tensor]:
    """0-dim broadcastable (lo, hi) hash of None, cached per device."""
    key = str(device)
    cached = _NONE_HASH_CACHE.get(key)
    if cached is None:
        from pathway_amd.internals.api import hash128, serialize_value

        lo, hi = hash128(serialize_value(None))
        cached = (
            torch.tensor(_c(lo), dtype=torch.int64, device=device),
            torch.tensor(_c(hi), dtype=torch.int64, device=device),
        )
        _NONE_HASH_CACHE[key] = cached
    return cached


def none_value_hash(n: int, device) -> tuple[torch.Tensor, torch.Tensor]:
    lo, hi = none_value_hash_scalar(device)
    _ = TAG_NONE
    return lo.expand(n), hi.expand(n)
