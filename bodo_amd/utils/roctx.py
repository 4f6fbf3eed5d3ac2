"""roctx marker ranges for rocprofv3 correlation (reference/SURVEY §5.1:
"add rocprof counter capture hooks per pipeline and per HIP kernel").

With BODO_AMD_ROCTX=1, every executor operator pushes a roctx range, so
``rocprofv3 --marker-trace`` (NOT combined with --pmc — see the gpurun
rules) attributes kernel time to logical-plan operators by name."""

from __future__ import annotations

import ctypes
import os

_LIB = None
_TRIED = False


def _lib():
    global _LIB, _TRIED
    if _TRIED:
        return _LIB
    _TRIED = True
    if os.environ.get("BODO_AMD_ROCTX", "0") not in ("1", "true"):
        return None
    for name in ("librocprofiler-sdk-roctx.so", "libroctx64.so"):
        try:
            lib = ctypes.CDLL(name)
            lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
            lib.roctxRangePop.argtypes = []
            _LIB = lib
            break
        except OSError:
            continue
    return _LIB


def range_push(name: str) -> None:
    lib = _lib()
    if lib is not None:
        lib.roctxRangePushA(name.encode())


def range_pop() -> None:
    lib = _lib()
    if lib is not None:
        lib.roctxRangePop()


class Range:
    __slots__ = ("name",)

    def __init__(self, name: str):
        self.name = name

    def __enter__(self):
        range_push(self.name)
        return self

    def __exit__(self, *exc):
        range_pop()
        return False
