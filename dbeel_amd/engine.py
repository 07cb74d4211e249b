"""ctypes bindings for libdbeel_gpu.so — the product compaction engine.

The HIP library is the product path: if it is missing or no GPU is present,
every call raises loudly (no CPU fallback — the CPU restatement in oracle/
is test infrastructure only).
"""
from __future__ import annotations

import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_DIR, "libdbeel_gpu.so")

ERROR_NAMES = {
    0: "OK",
    1: "INVALID_ARG",
    2: "CORRUPT",
    3: "ITEM_TOO_LARGE",
    4: "HIP",
    5: "NO_GPU",
    6: "OOM",
    7: "IO",
}


class DbeelGpuError(RuntimeError):
    def __init__(self, code: int, msg: str):
        super().__init__(f"dbeel_gpu error {ERROR_NAMES.get(code, code)}: {msg}")
        self.code = code


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


class CompactTimings(ctypes.Structure):
    _fields_ = [
        ("h2d_ms", ctypes.c_double),
        ("prep_ms", ctypes.c_double),
        ("rank_ms", ctypes.c_double),
        ("scan_ms", ctypes.c_double),
        ("emit_ms", ctypes.c_double),
        ("copy_ms", ctypes.c_double),
        ("kernel_ms", ctypes.c_double),
        ("d2h_ms", ctypes.c_double),
    ]

    def as_dict(self):
        return {k: getattr(self, k) for k, _ in self._fields_}


def _ptr_bytes(ptr, n: int) -> bytes:
    """Copy n bytes from a ctypes uint8 pointer. ctypes.string_at truncates
    its size argument to 32 bits on this interpreter (CPython 3.10), which
    silently corrupts >4 GB results — use a numpy view instead."""
    if not n:
        return b""
    return np.ctypeslib.as_array(ptr, shape=(int(n),)).tobytes()


_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        if not os.path.exists(LIB_PATH):
            raise FileNotFoundError(
                f"{LIB_PATH} not built — run __graft_entry__.build() "
                "(hipcc --offload-arch=gfx950). The product path has no "
                "CPU fallback."
            )
        lib = ctypes.CDLL(LIB_PATH)
        lib.dbeel_gpu_compact_timed.restype = ctypes.c_int
        lib.dbeel_gpu_compact_timed.argtypes = [
            ctypes.POINTER(RunView), ctypes.c_size_t, ctypes.c_int,
            ctypes.c_int, ctypes.POINTER(CompactResult),
            ctypes.POINTER(CompactTimings),
        ]
        lib.dbeel_gpu_result_free.argtypes = [ctypes.POINTER(CompactResult)]
        lib.dbeel_gpu_last_error.restype = ctypes.c_char_p
        lib.dbeel_gpu_job_create.restype = ctypes.c_int
        lib.dbeel_gpu_job_create.argtypes = [
            ctypes.POINTER(RunView), ctypes.c_size_t, ctypes.c_int,
            ctypes.POINTER(ctypes.c_void_p),
        ]
        lib.dbeel_gpu_job_run.restype = ctypes.c_int
        lib.dbeel_gpu_job_run.argtypes = [
            ctypes.c_void_p, ctypes.c_int,
            ctypes.POINTER(ctypes.c_uint64), ctypes.POINTER(ctypes.c_uint64),
            ctypes.POINTER(CompactTimings),
        ]
        lib.dbeel_gpu_job_fetch.restype = ctypes.c_int
        lib.dbeel_gpu_job_fetch.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(CompactResult)
        ]
        lib.dbeel_gpu_job_destroy.argtypes = [ctypes.c_void_p]
        lib.dbeel_gpu_encode_run.restype = ctypes.c_int
        lib.dbeel_gpu_encode_run.argtypes = [
            ctypes.c_uint64,
            ctypes.POINTER(ctypes.c_uint8), ctypes.POINTER(ctypes.c_uint64),
            ctypes.POINTER(ctypes.c_uint8), ctypes.POINTER(ctypes.c_uint64),
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_int,
            ctypes.POINTER(CompactResult),
        ]
        lib.dbeel_gpu_pin_host.restype = ctypes.c_int
        lib.dbeel_gpu_pin_host.argtypes = [ctypes.c_void_p, ctypes.c_size_t]
        lib.dbeel_gpu_unpin_host.restype = ctypes.c_int
        lib.dbeel_gpu_unpin_host.argtypes = [ctypes.c_void_p]
        lib.dbeel_gpu_job_ingest.restype = ctypes.c_int
        lib.dbeel_gpu_job_ingest.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(RunView), ctypes.c_size_t,
            ctypes.POINTER(IngestStats),
        ]
        lib.dbeel_gpu_compact_sliced.restype = ctypes.c_int
        lib.dbeel_gpu_compact_sliced.argtypes = [
            ctypes.POINTER(RunView), ctypes.c_size_t, ctypes.c_int,
            ctypes.c_int, ctypes.c_uint64, ctypes.POINTER(CompactResult),
        ]
        lib.dbeel_gpu_scan.restype = ctypes.c_int
        lib.dbeel_gpu_scan.argtypes = [
            ctypes.POINTER(RunView), ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_uint32), ctypes.POINTER(ctypes.c_uint32),
            ctypes.c_size_t, ctypes.c_int, ctypes.POINTER(CompactResult),
        ]
        _lib = lib
    return _lib


def _as_u8(buf) -> np.ndarray:
    if isinstance(buf, np.ndarray):
        return np.ascontiguousarray(buf, dtype=np.uint8)
    return np.frombuffer(buf, dtype=np.uint8)


def _views(runs):
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


def compact(runs, keep_tombstones: bool, device: int = 0,
            want_timings: bool = False):
    """Compact runs [(data, index), ...] on `device`.

    Returns (data_bytes, index_bytes, entries_written[, timings_dict]).
    """
    lib = load()
    views, keepalive = _views(runs)
    res = CompactResult()
    tim = CompactTimings()
    rc = lib.dbeel_gpu_compact_timed(
        views, len(runs), int(keep_tombstones), device,
        ctypes.byref(res), ctypes.byref(tim),
    )
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())
    try:
        data = _ptr_bytes(res.data, res.data_len)
        index = _ptr_bytes(res.index, res.index_len)
        n = int(res.entries_written)
    finally:
        lib.dbeel_gpu_result_free(ctypes.byref(res))
    del keepalive
    if want_timings:
        return data, index, n, tim.as_dict()
    return data, index, n


def scan(runs, start_key: bytes | None = None,
         end_key: bytes | None = None, hash_ranges=None, device: int = 0):
    """Migration/iteration scan (AsyncIter analogue, lsm_tree.rs:141-282 +
    tasks/migration.rs:62-131): yields every entry in the reference's
    order — runs ascending, entries in key order within each — with no
    dedup and no tombstone filter, restricted by an optional key range
    [start_key, end_key) and/or murmur3_32 hash ranges [(start, end), ...]
    (hash_bytes shards.rs:99-101, between_cmp migration.rs:54-60).
    Returns (data_bytes, index_bytes, n) shaped like a run file."""
    lib = load()
    views, keepalive = _views(runs)

    def _kb(b):
        if b is None:
            return None, 0, None
        a = np.frombuffer(b or b"\0", dtype=np.uint8)
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)), len(b), a

    sp, sl, sa = _kb(start_key)
    ep, el, ea = _kb(end_key)
    n_ranges = len(hash_ranges) if hash_ranges else 0
    if n_ranges:
        rs = np.array([r[0] for r in hash_ranges], dtype=np.uint32)
        re_ = np.array([r[1] for r in hash_ranges], dtype=np.uint32)
        rsp = rs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))
        rep = re_.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))
    else:
        rsp = rep = None
    res = CompactResult()
    rc = lib.dbeel_gpu_scan(views, len(runs), sp, sl, ep, el, rsp, rep,
                            n_ranges, device, ctypes.byref(res))
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())
    try:
        data = _ptr_bytes(res.data, res.data_len)
        index = _ptr_bytes(res.index, res.index_len)
        n = int(res.entries_written)
    finally:
        lib.dbeel_gpu_result_free(ctypes.byref(res))
    del keepalive
    return data, index, n


class IngestStats(ctypes.Structure):
    _fields_ = [
        ("ingest_ms", ctypes.c_double),
        ("copy_ms", ctypes.c_double),
        ("prep_ms", ctypes.c_double),
        ("bytes", ctypes.c_uint64),
        ("chunks", ctypes.c_uint64),
    ]

    def as_dict(self):
        return {k: getattr(self, k) for k, _ in self._fields_}


def pin_host(arr: np.ndarray) -> None:
    """Page-lock a numpy buffer (hipHostRegister) so streamed ingest runs
    as true async DMA. Pin once, outside the hot loop."""
    lib = load()
    rc = lib.dbeel_gpu_pin_host(arr.ctypes.data, arr.nbytes)
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())


def unpin_host(arr: np.ndarray) -> None:
    lib = load()
    rc = lib.dbeel_gpu_unpin_host(arr.ctypes.data)
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())


def compact_sliced(runs, keep_tombstones: bool, device: int = 0,
                   max_resident_bytes: int = 0):
    """Sliced compaction for inputs larger than HBM: key space cut at
    pivots, each slice compacted on-device, outputs concatenated —
    byte-identical to one whole compaction (include/dbeel_gpu.h)."""
    lib = load()
    views, keepalive = _views(runs)
    res = CompactResult()
    rc = lib.dbeel_gpu_compact_sliced(
        views, len(runs), int(keep_tombstones), device,
        max_resident_bytes, ctypes.byref(res),
    )
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())
    try:
        data = _ptr_bytes(res.data, res.data_len)
        index = _ptr_bytes(res.index, res.index_len)
        n = int(res.entries_written)
    finally:
        lib.dbeel_gpu_result_free(ctypes.byref(res))
    del keepalive
    return data, index, n


class LookupHit(ctypes.Structure):
    _fields_ = [
        ("run", ctypes.c_int32),
        ("is_tombstone", ctypes.c_uint32),
        ("value_offset", ctypes.c_uint64),
        ("value_len", ctypes.c_uint64),
    ]


def lookup(runs, keys, device: int = 0):
    """Batched point lookup (LSMTree::get over sstables,
    lsm_tree.rs:605-723): for each key returns the value bytes of the
    first match scanning runs newest-index-first (the reference's
    `sstables.iter().rev()`, lsm_tree.rs:692-696 — highest run index
    wins), b"" for a deleted key (tombstone), or None if absent."""
    lib = load()
    if not hasattr(lib, "_lookup_ready"):
        lib.dbeel_gpu_lookup.restype = ctypes.c_int
        lib.dbeel_gpu_lookup.argtypes = [
            ctypes.POINTER(RunView), ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_uint8), ctypes.POINTER(ctypes.c_uint64),
            ctypes.c_uint64, ctypes.c_int, ctypes.POINTER(LookupHit),
        ]
        lib._lookup_ready = True
    views, keepalive = _views(runs)
    blob = b"".join(keys)
    koff = np.zeros(len(keys) + 1, dtype=np.uint64)
    np.cumsum([len(k) for k in keys], out=koff[1:])
    ka = np.frombuffer(blob or b"\0", dtype=np.uint8)
    hits = (LookupHit * len(keys))()
    rc = lib.dbeel_gpu_lookup(
        views, len(runs),
        ka.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        koff.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        len(keys), device, hits,
    )
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())
    out = []
    datas = [bytes(_as_u8(d)) if not isinstance(d, bytes) else d
             for d, _ in runs]
    for h in hits:
        if h.run < 0:
            out.append(None)
        elif h.is_tombstone:
            out.append(b"")
        else:
            out.append(
                datas[h.run][h.value_offset : h.value_offset + h.value_len]
            )
    del keepalive
    return out


def encode_run(entries, device: int = 0):
    """GPU run encoder (the memtable-flush path, lsm_tree.rs:925-946):
    entries = iterable of (key: bytes, data: bytes, timestamp: int),
    ALREADY sorted ascending by key (memtable order). Returns
    (data_bytes, index_bytes, n)."""
    lib = load()
    keys = b"".join(e[0] for e in entries)
    vals = b"".join(e[1] for e in entries)
    n = len(entries)
    koff = np.zeros(n + 1, dtype=np.uint64)
    voff = np.zeros(n + 1, dtype=np.uint64)
    ts = np.zeros(16 * n, dtype=np.uint8)
    for i, (k, v, t) in enumerate(entries):
        koff[i + 1] = koff[i] + len(k)
        voff[i + 1] = voff[i] + len(v)
        ts[16 * i : 16 * (i + 1)] = np.frombuffer(
            int(t).to_bytes(16, "little", signed=True), dtype=np.uint8
        )
    ka = np.frombuffer(keys or b"\0", dtype=np.uint8)
    va = np.frombuffer(vals or b"\0", dtype=np.uint8)
    res = CompactResult()
    rc = lib.dbeel_gpu_encode_run(
        n,
        ka.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        koff.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        va.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        voff.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ts.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        device, ctypes.byref(res),
    )
    if rc != 0:
        raise DbeelGpuError(rc, lib.dbeel_gpu_last_error().decode())
    try:
        data = _ptr_bytes(res.data, res.data_len)
        index = _ptr_bytes(res.index, res.index_len)
    finally:
        lib.dbeel_gpu_result_free(ctypes.byref(res))
    return data, index, n


class Job:
    """Resident compaction job: inputs uploaded once, repeated runs timed
    with inputs already in HBM (the BASELINE measurement mode)."""

    def __init__(self, runs, device: int = 0):
        self._lib = load()
        views, self._keepalive = _views(runs)
        h = ctypes.c_void_p()
        rc = self._lib.dbeel_gpu_job_create(
            views, len(runs), device, ctypes.byref(h)
        )
        if rc != 0:
            raise DbeelGpuError(rc, self._lib.dbeel_gpu_last_error().decode())
        self._h = h
        self.input_bytes = sum(v.data_len + v.index_len for v in views)

    def run(self, keep_tombstones: bool):
        dl = ctypes.c_uint64()
        ne = ctypes.c_uint64()
        tim = CompactTimings()
        rc = self._lib.dbeel_gpu_job_run(
            self._h, int(keep_tombstones), ctypes.byref(dl), ctypes.byref(ne),
            ctypes.byref(tim),
        )
        if rc != 0:
            raise DbeelGpuError(rc, self._lib.dbeel_gpu_last_error().decode())
        return int(dl.value), int(ne.value), tim.as_dict()

    def ingest(self, runs):
        """Streamed pinned ingest (north_star): re-upload fresh run
        contents into this job's resident slab, chunked hipMemcpyAsync on
        a copy stream overlapped with the prepare kernel on the compute
        stream. Shapes must match the job's. The next run() skips the
        prepare stage (already done, hidden behind the transfer).
        Returns the ingest stats dict."""
        views, keepalive = _views(runs)
        st = IngestStats()
        rc = self._lib.dbeel_gpu_job_ingest(
            self._h, views, len(views), ctypes.byref(st)
        )
        if rc != 0:
            raise DbeelGpuError(rc, self._lib.dbeel_gpu_last_error().decode())
        del keepalive
        return st.as_dict()

    def fetch(self):
        res = CompactResult()
        rc = self._lib.dbeel_gpu_job_fetch(self._h, ctypes.byref(res))
        if rc != 0:
            raise DbeelGpuError(rc, self._lib.dbeel_gpu_last_error().decode())
        try:
            data = _ptr_bytes(res.data, res.data_len)
            index = _ptr_bytes(res.index, res.index_len)
            n = int(res.entries_written)
        finally:
            self._lib.dbeel_gpu_result_free(ctypes.byref(res))
        return data, index, n

    def close(self):
        if getattr(self, "_h", None):
            self._lib.dbeel_gpu_job_destroy(self._h)
            self._h = None
            self._keepalive = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class BatchJob(Job):
    """Batched INDEPENDENT jobs in one launch set (one dbeel shard each,
    BASELINE configs[3]): jobs = [runs, runs, ...]. run() executes every
    job's merge in one kernel pipeline; fetch_job(j) returns job j's
    (data, index, n) with offsets rebased to its own run file."""

    def __init__(self, jobs, device: int = 0):
        self._lib = load()
        if not hasattr(self._lib, "_batch_ready"):
            self._lib.dbeel_gpu_job_create_batched.restype = ctypes.c_int
            self._lib.dbeel_gpu_job_create_batched.argtypes = [
                ctypes.POINTER(RunView), ctypes.c_size_t,
                ctypes.POINTER(ctypes.c_uint32), ctypes.c_size_t,
                ctypes.c_int, ctypes.POINTER(ctypes.c_void_p),
            ]
            self._lib.dbeel_gpu_job_fetch_job.restype = ctypes.c_int
            self._lib.dbeel_gpu_job_fetch_job.argtypes = [
                ctypes.c_void_p, ctypes.c_size_t,
                ctypes.POINTER(CompactResult),
            ]
            self._lib._batch_ready = True
        flat = [rv for runs in jobs for rv in runs]
        views, self._keepalive = _views(flat)
        rpj = (ctypes.c_uint32 * len(jobs))(*[len(r) for r in jobs])
        h = ctypes.c_void_p()
        rc = self._lib.dbeel_gpu_job_create_batched(
            views, len(flat), rpj, len(jobs), device, ctypes.byref(h)
        )
        if rc != 0:
            raise DbeelGpuError(rc, self._lib.dbeel_gpu_last_error().decode())
        self._h = h
        self.n_jobs = len(jobs)
        self.input_bytes = sum(v.data_len + v.index_len for v in views)

    def fetch_job(self, j: int):
        res = CompactResult()
        rc = self._lib.dbeel_gpu_job_fetch_job(self._h, j,
                                               ctypes.byref(res))
        if rc != 0:
            raise DbeelGpuError(rc, self._lib.dbeel_gpu_last_error().decode())
        try:
            data = _ptr_bytes(res.data, res.data_len)
            index = _ptr_bytes(res.index, res.index_len)
            n = int(res.entries_written)
        finally:
            self._lib.dbeel_gpu_result_free(ctypes.byref(res))
        return data, index, n

