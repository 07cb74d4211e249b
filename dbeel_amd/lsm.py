"""ctypes bindings for the file-level host layer (include/dbeel_lsm.h):
compact-with-journal, crash-recovery replay, the compaction trigger
policy, and the behavioral bloom check."""
from __future__ import annotations

import ctypes
import os

from .engine import DbeelGpuError, load as _load_gpu

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        lib = _load_gpu()  # same .so
        lib.dbeel_lsm_compact.restype = ctypes.c_int
        lib.dbeel_lsm_compact.argtypes = [
            ctypes.c_char_p, ctypes.POINTER(ctypes.c_uint64), ctypes.c_size_t,
            ctypes.c_uint64, ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
            ctypes.POINTER(ctypes.c_uint64),
        ]
        lib.dbeel_lsm_replay.restype = ctypes.c_int
        lib.dbeel_lsm_replay.argtypes = [
            ctypes.c_char_p, ctypes.POINTER(ctypes.c_uint32)
        ]
        lib.dbeel_lsm_compact_tree.restype = ctypes.c_int
        lib.dbeel_lsm_compact_tree.argtypes = [
            ctypes.c_char_p, ctypes.c_uint64, ctypes.c_int, ctypes.c_uint64,
            ctypes.POINTER(ctypes.c_uint32),
        ]
        lib.dbeel_bloom_contains.restype = ctypes.c_int
        lib.dbeel_bloom_contains.argtypes = [
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_int),
        ]
        _lib = lib
    return _lib


def _check(rc, lib):
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())


DEFAULT_BLOOM_MIN_SIZE = 1_048_576  # mod.rs:19


def compact(dir: str, indices, output_index: int, keep_tombstones: bool,
            device: int = 0,
            bloom_min_size: int = DEFAULT_BLOOM_MIN_SIZE) -> int:
    lib = load()
    arr = (ctypes.c_uint64 * len(indices))(*indices)
    out = ctypes.c_uint64()
    rc = lib.dbeel_lsm_compact(
        dir.encode(), arr, len(indices), output_index,
        int(keep_tombstones), device, bloom_min_size, ctypes.byref(out),
    )
    _check(rc, lib)
    return int(out.value)


def replay(dir: str) -> int:
    lib = load()
    n = ctypes.c_uint32()
    rc = lib.dbeel_lsm_replay(dir.encode(), ctypes.byref(n))
    _check(rc, lib)
    return int(n.value)


def compact_tree(dir: str, compaction_factor: int = 2, device: int = 0,
                 bloom_min_size: int = DEFAULT_BLOOM_MIN_SIZE) -> int:
    lib = load()
    n = ctypes.c_uint32()
    rc = lib.dbeel_lsm_compact_tree(
        dir.encode(), compaction_factor, device, bloom_min_size,
        ctypes.byref(n),
    )
    _check(rc, lib)
    return int(n.value)


def major_compact(dir: str, device: int = 0,
                  bloom_min_size: int = DEFAULT_BLOOM_MIN_SIZE) -> int:
    """Merge EVERY live sstable into one run, dropping tombstones (safe:
    the group covers everything). Bounds tombstone buildup under the
    conservative full-coverage rule (include/dbeel_lsm.h)."""
    lib = load()
    if not hasattr(lib, "_major_ready"):
        lib.dbeel_lsm_major_compact.restype = ctypes.c_int
        lib.dbeel_lsm_major_compact.argtypes = [
            ctypes.c_char_p, ctypes.c_int, ctypes.c_uint64,
            ctypes.POINTER(ctypes.c_uint64),
        ]
        lib._major_ready = True
    n = ctypes.c_uint64()
    rc = lib.dbeel_lsm_major_compact(dir.encode(), device, bloom_min_size,
                                     ctypes.byref(n))
    _check(rc, lib)
    return int(n.value)


def bloom_contains(bloom_bytes: bytes, key: bytes) -> bool:
    lib = load()
    bb = (ctypes.c_uint8 * len(bloom_bytes)).from_buffer_copy(bloom_bytes)
    kb = (ctypes.c_uint8 * max(len(key), 1)).from_buffer_copy(key or b"\0")
    out = ctypes.c_int()
    rc = lib.dbeel_bloom_contains(bb, len(bloom_bytes), kb, len(key),
                                  ctypes.byref(out))
    _check(rc, lib)
    return bool(out.value)


def write_run_files(dir: str, index: int, data: bytes, idx: bytes) -> None:
    """Write {index:020}.data/.index (test/bench helper)."""
    os.makedirs(dir, exist_ok=True)
    with open(os.path.join(dir, f"{index:020d}.data"), "wb") as f:
        f.write(data)
    with open(os.path.join(dir, f"{index:020d}.index"), "wb") as f:
        f.write(idx)


def read_run_files(dir: str, index: int) -> tuple[bytes, bytes]:
    with open(os.path.join(dir, f"{index:020d}.data"), "rb") as f:
        data = f.read()
    with open(os.path.join(dir, f"{index:020d}.index"), "rb") as f:
        idx = f.read()
    return data, idx
