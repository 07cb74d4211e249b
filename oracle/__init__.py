"""ctypes wrapper for the CPU oracle (TEST INFRASTRUCTURE ONLY).

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this package. The product path (dbeel_amd) never does.
"""
from __future__ import annotations

import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_DIR, "liboracle.so")


class RunView(ctypes.Structure):
    _fields_ = [
        ("data", ctypes.POINTER(ctypes.c_uint8)),
        ("data_len", ctypes.c_size_t),
        ("index", ctypes.POINTER(ctypes.c_uint8)),
        ("index_len", ctypes.c_size_t),
    ]


class CompactResult(ctypes.Structure):
    _fields_ = [
        ("data", ctypes.POINTER(ctypes.c_uint8)),
        ("data_len", ctypes.c_size_t),
        ("index", ctypes.POINTER(ctypes.c_uint8)),
        ("index_len", ctypes.c_size_t),
        ("entries_written", ctypes.c_uint64),
    ]


def build(force: bool = False) -> str:
    if force or not os.path.exists(_LIB):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)
    return _LIB


def _ptr_bytes(ptr, n: int) -> bytes:
    """Copy n bytes from a ctypes uint8 pointer. ctypes.string_at truncates
    its size argument to 32 bits on this interpreter (CPython 3.10), which
    silently corrupts >4 GB results — use a numpy view instead."""
    if not n:
        return b""
    return np.ctypeslib.as_array(ptr, shape=(int(n),)).tobytes()


_lib = None


def _load():
    global _lib
    if _lib is None:
        lib = ctypes.CDLL(build())
        lib.dbeel_oracle_compact.restype = ctypes.c_int
        lib.dbeel_oracle_compact.argtypes = [
            ctypes.POINTER(RunView),
            ctypes.c_size_t,
            ctypes.c_int,
            ctypes.POINTER(CompactResult),
        ]
        lib.dbeel_oracle_result_free.argtypes = [ctypes.POINTER(CompactResult)]
        lib.dbeel_oracle_last_error.restype = ctypes.c_char_p
        _lib = lib
    return _lib


def _as_u8(buf) -> np.ndarray:
    a = np.frombuffer(buf, dtype=np.uint8) if not isinstance(buf, np.ndarray) else buf
    return np.ascontiguousarray(a, dtype=np.uint8)


def make_run_views(runs):
    """runs: list of (data, index) bytes/arrays -> (RunView array, keepalive)."""
    keep = []
    views = (RunView * len(runs))()
    for i, (d, x) in enumerate(runs):
        d = _as_u8(d)
        x = _as_u8(x)
        keep += [d, x]
        views[i].data = d.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))
        views[i].data_len = d.nbytes
        views[i].index = x.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))
        views[i].index_len = x.nbytes
    return views, keep


def compact(runs, keep_tombstones: bool) -> tuple[bytes, bytes, int]:
    """Returns (data_bytes, index_bytes, entries_written). Raises on error."""
    lib = _load()
    views, keep = make_run_views(runs)
    res = CompactResult()
    rc = lib.dbeel_oracle_compact(
        views, len(runs), int(keep_tombstones), ctypes.byref(res)
    )
    if rc != 0:
        raise RuntimeError(
            f"oracle error {rc}: {lib.dbeel_oracle_last_error().decode()}"
        )
    try:
        data = _ptr_bytes(res.data, res.data_len)
        index = _ptr_bytes(res.index, res.index_len)
        n = int(res.entries_written)
    finally:
        lib.dbeel_oracle_result_free(ctypes.byref(res))
    del keep
    return data, index, n
