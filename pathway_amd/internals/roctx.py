"""rocTX per-operator ranges (SURVEY §5.1 tracing).

The reference exports OpenTelemetry spans per operator; the MI355X-
native equivalent is rocTX ranges, which rocprofv3 `--marker-trace`
correlates with kernel dispatches — one range per (operator, step)
around each node's step() (enable with PW_ROCTX=1).

Binds libroctx64.so from the ROCm install via ctypes; all calls are
no-ops when the library is absent or the flag is off.
"""

from __future__ import annotations

import ctypes
import os

_lib = None
_tried = False


def _load():
    global _lib, _tried
    if _tried:
        return _lib
    _tried = True
    if not os.environ.get("PW_ROCTX"):
        return None
    for cand in (
        "libroctx64.so",
        "/opt/rocm/lib/libroctx64.so",
        "/opt/rocm/lib/librocprofiler-sdk-roctx.so",
    ):
        try:
            lib = ctypes.CDLL(cand)
            lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
            lib.roctxRangePushA.restype = ctypes.c_int
            lib.roctxRangePop.restype = ctypes.c_int
            _lib = lib
            return _lib
        except OSError:
            continue
    return None


def enabled() -> bool:
    return _load() is not None


def range_push(name: str) -> None:
    lib = _load()
    if lib is not None:
        lib.roctxRangePushA(name.encode())


def range_pop() -> None:
    lib = _load()
    if lib is not None:
        lib.roctxRangePop()


class range_ctx:
    __slots__ = ("name",)

    def __init__(self, name: str):
        self.name = name

    def __enter__(self):
        range_push(self.name)

    def __exit__(self, *exc):
        range_pop()
