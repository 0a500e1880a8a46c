"""ctypes bindings for the native IO scanner (libpwio.so).

The data plane of the file connectors (reference: Rust data_storage /
data_format): mmap + single-pass scans produce flat offset arrays; Python
only slices the decoded buffer per field — no per-row interpreter work in
the scan itself.
"""

from __future__ import annotations

import ctypes
import os

import numpy as np

_THIS = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_THIS, "libpwio.so")
_lib = None
_load_error: Exception | None = None


def _try_load():
    global _lib, _load_error
    if _lib is not None or _load_error is not None:
        return _lib
    try:
        if not os.path.exists(_LIB_PATH):
            from pathway_amd.ops.build import build_io

            build_io(verbose=False)
        lib = ctypes.CDLL(_LIB_PATH)
        lib.pw_count_lines.restype = ctypes.c_int64
        lib.pw_count_lines.argtypes = [ctypes.c_char_p]
        lib.pw_scan_lines.restype = ctypes.c_int64
        lib.pw_scan_lines.argtypes = [
            ctypes.c_char_p,
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int64,
        ]
        lib.pw_csv_shape.restype = ctypes.c_int
        lib.pw_csv_shape.argtypes = [
            ctypes.c_char_p,
            ctypes.c_char,
            ctypes.c_void_p,
            ctypes.c_void_p,
        ]
        lib.pw_csv_normalize.restype = ctypes.c_int64
        lib.pw_csv_normalize.argtypes = [
            ctypes.c_char_p,
            ctypes.c_char,
            ctypes.c_int64,
            ctypes.c_void_p,
            ctypes.c_int64,
            ctypes.c_void_p,
            ctypes.c_void_p,
        ]
        lib.pw_scan_csv.restype = ctypes.c_int64
        lib.pw_scan_csv.argtypes = [
            ctypes.c_char_p,
            ctypes.c_char,
            ctypes.c_int64,
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int64,
            ctypes.c_void_p,
        ]
        _lib = lib
    except Exception as e:  # noqa: BLE001
        _load_error = e
        _lib = None
    return _lib


def available() -> bool:
    return _try_load() is not None


def _ptr(a: np.ndarray):
    return ctypes.c_void_p(a.ctypes.data)


def read_lines(path: str) -> list[str]:
    """All lines of a text file (no trailing newline / CR)."""
    lib = _try_load()
    if lib is None:
        raise RuntimeError(f"libpwio unavailable: {_load_error}")
    p = path.encode()
    n = lib.pw_count_lines(p)
    if n < 0:
        raise OSError(f"cannot read {path}")
    if n == 0:
        return []
    starts = np.empty(n, dtype=np.int64)
    ends = np.empty(n, dtype=np.int64)
    got = lib.pw_scan_lines(p, _ptr(starts), _ptr(ends), n)
    if got < 0:
        raise OSError(f"line scan failed on {path} ({got})")
    with open(path, "rb") as fh:
        buf = fh.read()
    return [
        buf[s:e].decode("utf-8", "replace")
        for s, e in zip(starts[:got].tolist(), ends[:got].tolist())
    ]


def _decode_field(buf: bytes, s: int, e: int, quoted: bool) -> str:
    raw = buf[s:e]
    if quoted:
        raw = raw.strip()
        if len(raw) >= 2 and raw[:1] == b'"' and raw[-1:] == b'"':
            raw = raw[1:-1]
        raw = raw.replace(b'""', b'"')
    return raw.decode("utf-8", "replace")


def read_csv(path: str, delimiter: str = ",") -> tuple[list[str], list[list[str]]]:
    """(header, rows) of a CSV file; RFC-4180 quoting, rows with a
    mismatched field count are dropped (as the reference's dsv parser
    reports and skips malformed lines)."""
    lib = _try_load()
    if lib is None:
        raise RuntimeError(f"libpwio unavailable: {_load_error}")
    p = path.encode()
    d = ctypes.c_char(delimiter.encode())
    rows_c = ctypes.c_int64()
    cols_c = ctypes.c_int64()
    rc = lib.pw_csv_shape(p, d, ctypes.byref(rows_c), ctypes.byref(cols_c))
    if rc != 0:
        raise OSError(f"cannot read {path}")
    nrows, ncols = rows_c.value, cols_c.value
    if nrows == 0 or ncols == 0:
        return [], []
    fsize = os.path.getsize(path)
    cap = fsize + nrows * ncols + 16
    outbuf = ctypes.create_string_buffer(cap)
    out_len = ctypes.c_int64()
    skipped = ctypes.c_int64()
    got = lib.pw_csv_normalize(
        p, d, ncols, outbuf, cap, ctypes.byref(out_len), ctypes.byref(skipped)
    )
    if got < 0:
        raise OSError(f"csv scan failed on {path} ({got})")
    # ONE decode + ONE split: all per-field work stays in C
    text = outbuf.raw[: max(out_len.value - 1, 0)].decode("utf-8", "replace")
    fields = text.split("\0") if text else []
    rows = [fields[i : i + ncols] for i in range(0, got * ncols, ncols)]
    header = rows[0] if rows else []
    return header, rows[1:]


def _ensure_lz4(lib):
    if not hasattr(lib, "_lz4_ready"):
        lib.pw_lz4_compress_bound.restype = ctypes.c_int64
        lib.pw_lz4_compress_bound.argtypes = [ctypes.c_int64]
        lib.pw_lz4_compress.restype = ctypes.c_int64
        lib.pw_lz4_compress.argtypes = [
            ctypes.c_char_p, ctypes.c_int64, ctypes.c_void_p, ctypes.c_int64,
        ]
        lib.pw_lz4_decompress.restype = ctypes.c_int64
        lib.pw_lz4_decompress.argtypes = [
            ctypes.c_char_p, ctypes.c_int64, ctypes.c_void_p, ctypes.c_int64,
        ]
        lib._lz4_ready = True
    return lib


def lz4_compress(data: bytes) -> bytes:
    """LZ4 block compression via the native codec (pw_io.cpp)."""
    lib = _try_load()
    if lib is None:
        raise RuntimeError(f"libpwio unavailable: {_load_error}")
    _ensure_lz4(lib)
    n = len(data)
    cap = lib.pw_lz4_compress_bound(n)
    out = ctypes.create_string_buffer(cap)
    got = lib.pw_lz4_compress(data, n, out, cap)
    if got < 0:
        raise RuntimeError("lz4 compress overflow")
    return out.raw[:got]


def lz4_decompress(data: bytes, uncompressed_size: int) -> bytes:
    lib = _try_load()
    if lib is None:
        raise RuntimeError(f"libpwio unavailable: {_load_error}")
    _ensure_lz4(lib)
    out = ctypes.create_string_buffer(max(uncompressed_size, 1))
    got = lib.pw_lz4_decompress(data, len(data), out, uncompressed_size)
    if got != uncompressed_size:
        raise RuntimeError(f"lz4 decompress: got {got}, want {uncompressed_size}")
    return out.raw[:uncompressed_size]
