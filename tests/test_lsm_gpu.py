"""GPU end-to-end tests of the file-level host layer: compact with the
full artifact discipline, crash recovery via the journal, the trigger
policy, and behavioral bloom (SURVEY.md §8f rows 1-2 follow-ons)."""
import os

import numpy as np
import pytest

import oracle
from dbeel_amd import lsm
from dbeel_amd.format import parse_run
from dbeel_amd.genruns import make_runs

pytestmark = pytest.mark.gpu


def _write_runs(d, indices, runs):
    for idx, (data, index) in zip(indices, runs):
        lsm.write_run_files(d, idx, bytes(data), bytes(index))


def test_compact_files_end_to_end(tmp_path):
    d = str(tmp_path)
    runs = make_runs(3, 5000, 16, 128, overlap_frac=0.4, tombstone_frac=0.1,
                     seed=42)
    _write_runs(d, [0, 2, 4], runs)
    n = lsm.compact(d, [0, 2, 4], 5, keep_tombstones=False, device=0,
                    bloom_min_size=1)
    od, oi, on = oracle.compact(runs, keep_tombstones=False)
    assert n == on
    data, index = lsm.read_run_files(d, 5)
    assert data == od and index == oi
    # inputs gone, no journal left, bloom present (input > min size)
    for idx in (0, 2, 4):
        assert not os.path.exists(f"{d}/{idx:020d}.data")
        assert not os.path.exists(f"{d}/{idx:020d}.index")
    assert not os.path.exists(f"{d}/{5:020d}.compact_action")
    assert os.path.exists(f"{d}/{5:020d}.bloom")

    # behavioral bloom: zero false negatives on written keys, fp <= ~2%
    # on absent keys (BLOOM_MAX_ALLOWED_ERROR=0.01, lsm_tree.rs:48)
    bloom = open(f"{d}/{5:020d}.bloom", "rb").read()
    ents = parse_run(data, index)
    for e in ents[::7]:
        assert lsm.bloom_contains(bloom, e.key)
    rng = np.random.default_rng(1)
    fp = sum(
        lsm.bloom_contains(bloom, bytes(rng.integers(0, 256, 16,
                                                     dtype=np.uint8)))
        for _ in range(2000)
    )
    assert fp < 60  # ~1% expected, generous bound


def test_major_compact_bounds_tombstones(tmp_path):
    """dbeel_lsm_major_compact merges every live sstable and drops
    tombstones (safe: full coverage) — the operator tool bounding the
    buildup the conservative compact_tree rule can leave."""
    d = str(tmp_path)
    runs = make_runs(4, 3000, 16, 64, overlap_frac=0.5, tombstone_frac=0.3,
                     seed=7)
    _write_runs(d, [0, 2, 4, 6], runs)
    n = lsm.major_compact(d, device=0, bloom_min_size=1 << 30)
    od, oi, on = oracle.compact(runs, keep_tombstones=False)
    assert n == on
    # single surviving run at the next odd index (no odd sstables existed
    # -> index 1, tasks/compaction.rs:38-43), tombstones gone
    import glob

    left = sorted(glob.glob(f"{d}/*.index"))
    assert len(left) == 1 and left[0].endswith(f"{1:020d}.index"), left
    data, index = lsm.read_run_files(d, 1)
    assert data == od and index == oi
    assert all(not e.is_tombstone for e in parse_run(data, index))
    # idempotent-ish: a second call with one sstable is a no-op
    assert lsm.major_compact(d, device=0) == 0


def test_compact_files_no_bloom_below_threshold(tmp_path):
    d = str(tmp_path)
    runs = make_runs(2, 200, 16, 32, seed=7)
    _write_runs(d, [0, 2], runs)
    lsm.compact(d, [0, 2], 3, keep_tombstones=True, device=0,
                bloom_min_size=1 << 30)
    assert not os.path.exists(f"{d}/{3:020d}.bloom")
    od, oi, _ = oracle.compact(runs, keep_tombstones=True)
    data, index = lsm.read_run_files(d, 3)
    assert data == od and index == oi


def test_crash_recovery_roundtrip(tmp_path):
    """Crash after the journal is durable (the flow_events-style hook):
    staging files + journal on disk, inputs intact; replay completes the
    compaction idempotently (recovery path §3.3)."""
    d = str(tmp_path)
    runs = make_runs(3, 3000, 16, 64, overlap_frac=0.5, tombstone_frac=0.1,
                     seed=11)
    _write_runs(d, [0, 2, 4], runs)

    os.environ["DBEEL_LSM_CRASH_AFTER_JOURNAL"] = "1"
    try:
        lsm.compact(d, [0, 2, 4], 5, keep_tombstones=False, device=0,
                    bloom_min_size=1)
    finally:
        del os.environ["DBEEL_LSM_CRASH_AFTER_JOURNAL"]

    # crashed state: staging + journal + old inputs all present
    assert os.path.exists(f"{d}/{5:020d}.compact_data")
    assert os.path.exists(f"{d}/{5:020d}.compact_action")
    assert os.path.exists(f"{d}/{0:020d}.data")

    assert lsm.replay(d) == 1

    od, oi, _ = oracle.compact(runs, keep_tombstones=False)
    data, index = lsm.read_run_files(d, 5)
    assert data == od and index == oi
    assert not os.path.exists(f"{d}/{0:020d}.data")
    assert not os.path.exists(f"{d}/{5:020d}.compact_action")
    assert not os.path.exists(f"{d}/{5:020d}.compact_data")


def test_compact_tree_trigger_policy(tmp_path):
    """Three equal-size-class runs at even indices (the flush layout of
    get_after_compaction, lsm_tree.rs:1425-1434): one compaction into the
    next odd index, tombstones dropped (single group = final level)."""
    d = str(tmp_path)
    runs = make_runs(3, 1000, 16, 64, overlap_frac=0.5, tombstone_frac=0.2,
                     seed=13)
    _write_runs(d, [0, 2, 4], runs)

    n = lsm.compact_tree(d, compaction_factor=2, device=0, bloom_min_size=1)
    assert n == 1
    od, oi, _ = oracle.compact(runs, keep_tombstones=False)
    data, index = lsm.read_run_files(d, 1)  # next odd output index
    assert data == od and index == oi
    ents = parse_run(data, index)
    assert not any(e.is_tombstone for e in ents)

    # nothing left to merge
    assert lsm.compact_tree(d, compaction_factor=2, device=0) == 0


def test_compact_tree_respects_factor(tmp_path):
    d = str(tmp_path)
    runs = make_runs(2, 1000, 16, 64, seed=17)
    _write_runs(d, [0, 2], runs)
    assert lsm.compact_tree(d, compaction_factor=4, device=0) == 0
    assert os.path.exists(f"{d}/{0:020d}.data")


def test_compact_tree_size_classes(tmp_path):
    """Runs in different size classes are not merged together; each class
    with >= factor members compacts separately."""
    d = str(tmp_path)
    small = make_runs(2, 500, 16, 64, seed=19)
    big = make_runs(2, 40_000, 16, 64, seed=23)
    _write_runs(d, [0, 2], small)
    _write_runs(d, [4, 6], big)
    n = lsm.compact_tree(d, compaction_factor=2, device=0, bloom_min_size=1)
    assert n == 2
    # neither group covers every sstable, so BOTH keep tombstones (the
    # deterministic no-resurrection rule — DESIGN.md divergence note);
    # big class into 1, small class into 3
    bd, bi, _ = oracle.compact(big, keep_tombstones=True)
    sd, si, _ = oracle.compact(small, keep_tombstones=True)
    data1, index1 = lsm.read_run_files(d, 1)
    data3, index3 = lsm.read_run_files(d, 3)
    assert (data1, index1) == (bd, bi)
    assert (data3, index3) == (sd, si)
