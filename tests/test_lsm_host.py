"""File-level host layer tests.

CPU side: journal replay against hand-built reference-layout journals
(bincode CompactionAction — lsm_tree.rs:73-77; deletes-then-renames replay
— lsm_tree.rs:576-590) and bloom-format error handling.
GPU side (test_lsm_gpu.py): end-to-end compact-with-journal, crash
recovery, trigger policy.
"""
import os
import struct

import pytest

from dbeel_amd import lsm


def _bincode_journal(renames, deletes) -> bytes:
    """Reference-layout CompactionAction bytes (bincode fixint LE:
    Vec = u64 count; PathBuf = u64 len + utf8)."""
    out = struct.pack("<Q", len(renames))
    for a, b in renames:
        out += struct.pack("<Q", len(a)) + a.encode()
        out += struct.pack("<Q", len(b)) + b.encode()
    out += struct.pack("<Q", len(deletes))
    for d in deletes:
        out += struct.pack("<Q", len(d)) + d.encode()
    return out


def _skip_unless_built():
    try:
        lsm.load()
    except FileNotFoundError:
        pytest.skip("libdbeel_gpu.so not built")


def test_replay_deletes_then_renames(tmp_path):
    _skip_unless_built()
    d = str(tmp_path)
    # staged outputs + stale inputs, like a crash between journal write and
    # the renames (lsm_tree.rs:1101-1111)
    open(f"{d}/{5:020d}.compact_data", "wb").write(b"NEWDATA")
    open(f"{d}/{5:020d}.compact_index", "wb").write(b"NEWINDEX")
    open(f"{d}/{0:020d}.data", "wb").write(b"old0")
    open(f"{d}/{0:020d}.index", "wb").write(b"old0i")
    open(f"{d}/{2:020d}.data", "wb").write(b"old2")
    renames = [
        (f"{d}/{5:020d}.compact_data", f"{d}/{5:020d}.data"),
        (f"{d}/{5:020d}.compact_index", f"{d}/{5:020d}.index"),
        (f"{d}/{5:020d}.compact_bloom", f"{d}/{5:020d}.bloom"),  # absent ok
    ]
    deletes = [
        f"{d}/{0:020d}.data", f"{d}/{0:020d}.index", f"{d}/{0:020d}.bloom",
        f"{d}/{2:020d}.data", f"{d}/{2:020d}.index", f"{d}/{2:020d}.bloom",
    ]
    jpath = f"{d}/{5:020d}.compact_action"
    open(jpath, "wb").write(_bincode_journal(renames, deletes))

    n = lsm.replay(d)
    assert n == 1
    assert open(f"{d}/{5:020d}.data", "rb").read() == b"NEWDATA"
    assert open(f"{d}/{5:020d}.index", "rb").read() == b"NEWINDEX"
    assert not os.path.exists(f"{d}/{0:020d}.data")
    assert not os.path.exists(f"{d}/{2:020d}.data")
    assert not os.path.exists(jpath)
    # idempotent: nothing left to do
    assert lsm.replay(d) == 0


def test_replay_already_completed_crash(tmp_path):
    """Crash AFTER renames+deletes but before the journal unlink: replay
    must be a no-op except removing the journal (sources absent)."""
    _skip_unless_built()
    d = str(tmp_path)
    open(f"{d}/{3:020d}.data", "wb").write(b"FINAL")
    renames = [(f"{d}/{3:020d}.compact_data", f"{d}/{3:020d}.data")]
    jpath = f"{d}/{3:020d}.compact_action"
    open(jpath, "wb").write(_bincode_journal(renames, []))
    assert lsm.replay(d) == 1
    assert open(f"{d}/{3:020d}.data", "rb").read() == b"FINAL"
    assert not os.path.exists(jpath)


def test_replay_ignores_trailing_garbage(tmp_path):
    """`while let Ok` parse loop (lsm_tree.rs:432-436): stop at the first
    malformed action, still remove the journal."""
    _skip_unless_built()
    d = str(tmp_path)
    open(f"{d}/{1:020d}.compact_data", "wb").write(b"X")
    j = _bincode_journal(
        [(f"{d}/{1:020d}.compact_data", f"{d}/{1:020d}.data")], []
    ) + b"\xff\xff\xff"
    open(f"{d}/{1:020d}.compact_action", "wb").write(j)
    assert lsm.replay(d) == 1
    assert os.path.exists(f"{d}/{1:020d}.data")


def test_replay_multiple_actions_per_journal(tmp_path):
    """One journal file may hold several concatenated CompactionActions
    (`while let Ok(action) = deserialize_from`, lsm_tree.rs:432-436):
    all are applied in order."""
    _skip_unless_built()
    d = str(tmp_path)
    open(f"{d}/{1:020d}.compact_data", "wb").write(b"A")
    open(f"{d}/{3:020d}.compact_data", "wb").write(b"B")
    j = _bincode_journal(
        [(f"{d}/{1:020d}.compact_data", f"{d}/{1:020d}.data")], []
    ) + _bincode_journal(
        [(f"{d}/{3:020d}.compact_data", f"{d}/{3:020d}.data")], []
    )
    open(f"{d}/{3:020d}.compact_action", "wb").write(j)
    assert lsm.replay(d) == 1
    assert open(f"{d}/{1:020d}.data", "rb").read() == b"A"
    assert open(f"{d}/{3:020d}.data", "rb").read() == b"B"


def test_missing_input_files_are_io_errors(tmp_path):
    """Host filesystem failures carry DBEEL_ERR_IO (7), distinct from
    device failures (DBEEL_ERR_HIP) — an FFI caller mapping to the
    reference's error.rs taxonomy must be able to tell them apart
    (ADVICE r01). Reading absent run files fails before any device
    access, so this tests on CPU."""
    _skip_unless_built()
    from dbeel_amd.engine import DbeelGpuError

    with pytest.raises(DbeelGpuError) as ei:
        lsm.compact(str(tmp_path), [0, 2], 1, keep_tombstones=False)
    assert ei.value.code == 7  # IO
    assert "IO" in str(ei.value)


def test_bloom_contains_rejects_garbage():
    _skip_unless_built()
    from dbeel_amd.engine import DbeelGpuError

    with pytest.raises(DbeelGpuError):
        lsm.bloom_contains(b"NOTABLOOM" + b"\0" * 64, b"key")
