"""CPU-side ABI checks: the product library builds for gfx950, loads, and
exports every symbol include/dbeel_gpu.h declares. No compute without a GPU.
"""
import ctypes
import os

import pytest

from conftest import REPO_ROOT

LIB = os.path.join(REPO_ROOT, "dbeel_amd", "libdbeel_gpu.so")

HEADER_SYMBOLS = [
    "dbeel_gpu_compact",
    "dbeel_gpu_compact_timed",
    "dbeel_gpu_compact_sliced",
    "dbeel_gpu_result_free",
    "dbeel_gpu_last_error",
    "dbeel_gpu_encode_run",
    "dbeel_gpu_job_create",
    "dbeel_gpu_job_create_batched",
    "dbeel_gpu_job_run",
    "dbeel_gpu_job_fetch",
    "dbeel_gpu_job_fetch_job",
    "dbeel_gpu_job_ingest",
    "dbeel_gpu_job_destroy",
    "dbeel_gpu_scan",
    "dbeel_gpu_pin_host",
    "dbeel_gpu_unpin_host",
]

LSM_SYMBOLS = [
    "dbeel_lsm_compact",
    "dbeel_lsm_replay",
    "dbeel_lsm_compact_tree",
    "dbeel_lsm_major_compact",
    "dbeel_bloom_contains",
]


def _built():
    if not os.path.exists(LIB):
        import __graft_entry__

        __graft_entry__.build()
    return LIB


def test_product_lib_exports_header_symbols():
    lib = ctypes.CDLL(_built())
    for sym in HEADER_SYMBOLS:
        assert getattr(lib, sym, None) is not None, sym


def test_header_declares_every_symbol():
    hdr = open(os.path.join(REPO_ROOT, "include", "dbeel_gpu.h")).read()
    for sym in HEADER_SYMBOLS:
        assert sym in hdr, sym


def test_lsm_header_and_exports():
    hdr = open(os.path.join(REPO_ROOT, "include", "dbeel_lsm.h")).read()
    lib = ctypes.CDLL(_built())
    for sym in LSM_SYMBOLS:
        assert sym in hdr, sym
        assert getattr(lib, sym, None) is not None, sym


def test_device_minus_one_rejected():
    """device=-1 must NOT fall back to a CPU path (product/oracle
    separation); it is an invalid argument by design."""
    _built()
    import dbeel_amd
    from dbeel_amd.engine import DbeelGpuError
    from dbeel_amd.format import Entry, build_run

    runs = [build_run([Entry(b"a", b"b", 1)])]
    with pytest.raises(DbeelGpuError) as ei:
        dbeel_amd.compact(runs, keep_tombstones=True, device=-1)
    assert ei.value.code == 1  # INVALID_ARG


def test_entry_count_bounds_rejected():
    """Crossranks carry a 31-bit payload (CR_LOSER flag in bit 31), so a
    run with >= 2^31 entries must be rejected up front —
    validation runs before any device access, so this tests on CPU. The
    fake index pointer is never dereferenced: the length checks fire
    first."""
    _built()
    import numpy as np

    from dbeel_amd.engine import RunView, load

    lib = load()
    buf = np.zeros(16, dtype=np.uint8)
    views = (RunView * 1)()
    views[0].data = buf.ctypes.data_as(
        ctypes.POINTER(ctypes.c_uint8))
    views[0].data_len = 16
    views[0].index = buf.ctypes.data_as(
        ctypes.POINTER(ctypes.c_uint8))
    views[0].index_len = (1 << 31) * 16  # 2^31 entries
    out = ctypes.c_void_p()
    rc = lib.dbeel_gpu_job_create(views, 1, 0, ctypes.byref(out))
    assert rc == 3, rc  # ITEM_TOO_LARGE

    # per-run counts under 2^31 but job total >= 2^32 entries: the u32
    # survivor positions (d_pos, win_p0) bound the JOB, also rejected
    views2 = (RunView * 3)()
    for v in views2:
        v.data = buf.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))
        v.data_len = 16
        v.index = buf.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))
        v.index_len = ((1 << 31) - 1) * 16  # sums past 2^32
    rc = lib.dbeel_gpu_job_create(views2, 3, 0, ctypes.byref(out))
    assert rc == 3, rc


def test_no_gpu_errors_loudly():
    """On a GPU-less host, using a device ordinal must raise, never silently
    fall back."""
    _built()
    import dbeel_amd
    from dbeel_amd.engine import DbeelGpuError
    from dbeel_amd.format import Entry, build_run

    runs = [build_run([Entry(b"a", b"b", 1)])]
    try:
        dbeel_amd.compact(runs, keep_tombstones=True, device=0)
    except DbeelGpuError as e:
        assert e.code in (4, 5)  # HIP / NO_GPU
    else:
        # a GPU is actually present (gpurun box) — fine
        pass


def _build_abi_demo(tmp_path):
    import subprocess

    out = tmp_path / "abi_demo"
    subprocess.run(
        ["gcc", "-O2", "-I", os.path.join(REPO_ROOT, "include"),
         os.path.join(REPO_ROOT, "tools", "abi_demo.c"),
         "-L", os.path.join(REPO_ROOT, "dbeel_amd"), "-ldbeel_gpu",
         "-Wl,-rpath," + os.path.join(REPO_ROOT, "dbeel_amd"),
         "-o", str(out)],
        check=True,
    )
    return out


def test_abi_demo_compiles_with_plain_gcc(tmp_path):
    """The drop-in boundary is host-language-free: a plain-C consumer
    (tools/abi_demo.c) compiles with gcc against include/dbeel_gpu.h and
    links libdbeel_gpu.so — the same shape as dbeel's Rust FFI stub.
    (Execution needs a GPU; test_abi_demo_runs in the gpu suite.)"""
    _built()
    _build_abi_demo(tmp_path)


@pytest.mark.gpu
def test_abi_demo_runs(tmp_path):
    """The plain-C ABI consumer executes end to end on the GPU: compact
    (newest-wins + tombstone drop), newest-index-first lookup, full
    iteration scan."""
    import subprocess

    _built()
    out = _build_abi_demo(tmp_path)
    r = subprocess.run([str(out)], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr + r.stdout
    assert "abi_demo OK" in r.stdout


def test_oracle_lib_exports():
    import oracle

    lib = ctypes.CDLL(oracle.build())
    for sym in ("dbeel_oracle_compact", "dbeel_oracle_result_free",
                "dbeel_oracle_last_error"):
        assert getattr(lib, sym, None) is not None, sym
