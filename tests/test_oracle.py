"""CPU oracle tests: golden-vector parity + the reference's own semantic
known-answer scenario (`get_after_compaction`, lsm_tree.rs:1328-1451)."""
import numpy as np
import pytest

import oracle
from conftest import GOLDEN_CASES, load_golden
from dbeel_amd.format import Entry, build_run, parse_run
from pymerge import merge as pymerge_merge


@pytest.mark.parametrize("name", GOLDEN_CASES)
@pytest.mark.parametrize("keep", [True, False])
def test_oracle_matches_golden(name, keep):
    runs, exp_keep, exp_drop = load_golden(name)
    exp = exp_keep if keep else exp_drop
    data, index, n = oracle.compact(runs, keep_tombstones=keep)
    assert index == exp[1]
    assert data == exp[0]
    assert n == len(exp[1]) // 16


def test_get_after_compaction_scenario():
    """Byte-level restatement of the reference's semantic known-answer test
    (lsm_tree.rs:1328-1451): 94 inserts across 3 flushed runs of capacity 32,
    2 deletes (tombstones), compact([0,2,4], 5, keep_tombstones=false),
    3*32-4 = 92 survivors, deleted keys absent, order preserved."""
    cap = 32

    def key(n):
        return bytes([n & 0xFF, n >> 8])  # u16 LE (lsm_tree.rs:1408-1411)

    ts = 0

    def ent(k, v):
        nonlocal ts
        ts += 1
        return Entry(k, v, ts)

    # run 0: keys 0..31, run 2: keys 32..63 (sets in key order; memtable
    # sorts by key bytes — for n < 256 LE 2-byte keys lex order == n order)
    run0 = build_run([ent(key(n), key(n)) for n in range(cap)])
    run2 = build_run([ent(key(n), key(n)) for n in range(cap, 2 * cap)])
    # run 4: keys 64..93 plus tombstones for [1,0] and [4,0]; memtable order
    # is lexicographic: [1,0] and [4,0] sort before [64,0]..[93,0]
    last = [ent(key(n), key(n)) for n in range(2 * cap, 3 * cap - 2)]
    last += [ent(key(1), b""), ent(key(4), b"")]
    last.sort(key=lambda e: e.key)
    run4 = build_run(last)

    data, index, n = oracle.compact([run0, run2, run4], keep_tombstones=False)
    assert n == 3 * cap - 4  # lsm_tree.rs:1381-1383
    entries = parse_run(data, index)
    keys = [e.key for e in entries]
    assert key(1) not in keys and key(4) not in keys  # lsm_tree.rs:1389-1390
    assert keys == sorted(keys)
    assert len(set(keys)) == len(keys)
    for e in entries:
        assert e.data == e.key  # surviving values intact
    # range scan [1,0]..[5,0] yields values [2,0],[3,0] (lsm_tree.rs:1391-1397)
    in_range = [e.data for e in entries if key(1) <= e.key < key(5)]
    assert in_range == [key(2), key(3)]


def test_oracle_agrees_with_pymerge_random():
    """Randomized cross-check of the two independent restatements."""
    rng = np.random.default_rng(7)
    for trial in range(5):
        runs = []
        for r in range(int(rng.integers(1, 6))):
            keys = sorted(
                {bytes(rng.integers(0, 256, int(rng.integers(1, 12)), dtype=np.uint8))
                 for _ in range(40)}
            )
            ents = []
            for i, k in enumerate(keys):
                dlen = int(rng.integers(0, 4)) * 33
                data = bytes(rng.integers(0, 256, dlen, dtype=np.uint8))
                # random timestamps (collisions across runs exercise the
                # run-index tie-break; newest-in-older-run also covered)
                ents.append(Entry(k, data, int(rng.integers(-50, 50))))
            runs.append(build_run(ents))
        for keep in (True, False):
            exp = pymerge_merge(runs, keep)
            data, index, _ = oracle.compact(runs, keep)
            assert (data, index) == exp, f"trial {trial} keep={keep}"


def test_oracle_corrupt_index_rejected():
    runs = [(b"\x00" * 64, b"\x01" * 15)]  # index not 16-byte records
    with pytest.raises(RuntimeError, match="2"):
        oracle.compact(runs, True)


def test_oracle_corrupt_entry_rejected():
    # index claims an entry beyond data_len
    import struct

    idx = struct.pack("<QII", 0, 9, 64)
    runs = [(b"\x00" * 32, idx)]
    with pytest.raises(RuntimeError):
        oracle.compact(runs, True)
