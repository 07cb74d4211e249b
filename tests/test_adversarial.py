"""Adversarial parity hardening (CPU side).

The oracle's pinning chain bottoms out in two builder-written restatements
agreeing — the residual risk is a SHARED misreading of
lsm_tree.rs. These tests attack that risk three ways:

1. targeted worst cases at the implementation's own seams
   (tests/adversarial.py) checked oracle-vs-pymerge bit-exact — pymerge
   orders via Python's heapq on (key, ts, run) tuples, a third-party total
   order independent of both the oracle's hand-written heap and the GPU's
   merge-path corank;
2. a randomized heap-order audit: every pop sequence the oracle implies
   (reconstructed from its output) is verified against heapq directly;
3. mutation sensitivity: a mutated oracle (tie-break flipped) must FAIL
   the named cases — proving the cases can actually detect a
   tie-break-direction misreading.
"""
import ctypes
import os
import subprocess

import numpy as np
import pytest

import oracle
from adversarial import all_cases, case_all_runs_same_keys
from pymerge import merge as pymerge_merge

CASES = all_cases()


@pytest.mark.parametrize("name,runs", CASES, ids=[c[0] for c in CASES])
@pytest.mark.parametrize("keep", [False, True], ids=["drop", "keep"])
def test_oracle_matches_heapq_restatement(name, runs, keep):
    od, oi, on = oracle.compact(runs, keep_tombstones=keep)
    pd, pi = pymerge_merge(runs, keep_tombstones=keep)
    assert oi == pi, f"{name}: index bytes diverge"
    assert od == pd, f"{name}: data bytes diverge"
    assert on == len(pi) // 16


def test_randomized_heap_pop_audit():
    """Third-party check of the oracle's pop order: for random inputs,
    reconstruct the full merged order with heapq (no dedup) and verify the
    oracle's survivors are exactly the last pop of each equal-key group,
    in pop order (lsm_tree.rs:1038-1066)."""
    import heapq

    from dbeel_amd.format import Entry, build_run, parse_run

    rng = np.random.default_rng(0xD17)
    for trial in range(30):
        n_runs = int(rng.integers(2, 7))
        pool = [bytes(rng.integers(97, 123, int(rng.integers(1, 12)),
                                   dtype=np.uint8))
                for _ in range(40)]
        runs, all_items = [], []
        for r in range(n_runs):
            picks = sorted(
                {pool[int(i)] for i in rng.integers(0, len(pool), 25)}
            )
            ents = []
            for k in picks:
                ts = int(rng.integers(-5, 5))  # dense ts -> many collisions
                data = b"" if rng.random() < 0.3 else bytes(
                    rng.integers(0, 256, int(rng.integers(1, 9)),
                                 dtype=np.uint8))
                ents.append(Entry(k, data, ts))
                all_items.append((k, ts, r, data))
            runs.append(build_run(ents))

        # heapq-defined total pop order (key, ts, run index)
        heapq.heapify(all_items)
        pops = [heapq.heappop(all_items) for _ in range(len(all_items))]
        survivors = []
        for i, (k, ts, r, data) in enumerate(pops):
            last_of_group = i + 1 == len(pops) or pops[i + 1][0] != k
            if last_of_group and data != b"":
                survivors.append((k, data, ts))

        od, oi, on = oracle.compact(runs, keep_tombstones=False)
        got = [(e.key, e.data, e.timestamp) for e in parse_run(od, oi)]
        assert got == survivors, f"trial {trial}"


def _build_mutant(tmp_path, old, new, tag):
    """Compile a mutated oracle from a patched copy of compact_oracle.c
    (the shipped source stays pristine)."""
    src = open(os.path.join(os.path.dirname(oracle.__file__),
                            "compact_oracle.c")).read()
    assert src.count(old) == 1, "mutation anchor not found exactly once"
    msrc = src.replace(old, new)
    cpath = tmp_path / f"mut_{tag}.c"
    sopath = tmp_path / f"libmut_{tag}.so"
    cpath.write_text(msrc)
    subprocess.run(
        ["gcc", "-O2", "-shared", "-fPIC", str(cpath), "-o", str(sopath)],
        check=True,
    )
    lib = ctypes.CDLL(str(sopath))
    lib.dbeel_oracle_compact.restype = ctypes.c_int
    lib.dbeel_oracle_compact.argtypes = [
        ctypes.POINTER(oracle.RunView), ctypes.c_size_t, ctypes.c_int,
        ctypes.POINTER(oracle.CompactResult),
    ]
    lib.dbeel_oracle_result_free.argtypes = [
        ctypes.POINTER(oracle.CompactResult)
    ]
    return lib


def _mutant_compact(lib, runs, keep):
    views, keepalive = oracle.make_run_views(runs)
    res = oracle.CompactResult()
    rc = lib.dbeel_oracle_compact(views, len(runs), int(keep),
                                  ctypes.byref(res))
    assert rc == 0
    try:
        data = oracle._ptr_bytes(res.data, res.data_len)
        index = oracle._ptr_bytes(res.index, res.index_len)
    finally:
        lib.dbeel_oracle_result_free(ctypes.byref(res))
    del keepalive
    return data, index


def test_mutation_tiebreak_detected(tmp_path):
    """Flip the run-index tie-break in the oracle -> the ts-collision case
    MUST diverge from the heapq restatement (the named cases are sensitive
    to a tie-break misreading, not just volume)."""
    lib = _build_mutant(
        tmp_path,
        "if (ra != rb) return ra < rb ? -1 : 1;",
        "if (ra != rb) return ra < rb ? 1 : -1;",
        "tiebreak",
    )
    runs = case_all_runs_same_keys(8, "equal")
    md, mi = _mutant_compact(lib, runs, keep=True)
    pd, pi = pymerge_merge(runs, keep_tombstones=True)
    assert (md, mi) != (pd, pi), (
        "tie-break mutation not detected by the ts-collision case"
    )


def test_mutation_ts_order_detected(tmp_path):
    """Flip the timestamp comparison direction (low half) -> the
    ts-descending case MUST diverge (winner changes from the ts-max to the
    ts-min entry)."""
    lib = _build_mutant(
        tmp_path,
        "if (a->ts_lo != b->ts_lo) return a->ts_lo < b->ts_lo ? -1 : 1;",
        "if (a->ts_lo != b->ts_lo) return a->ts_lo < b->ts_lo ? 1 : -1;",
        "tsorder",
    )
    runs = case_all_runs_same_keys(8, "desc")
    md, mi = _mutant_compact(lib, runs, keep=True)
    pd, pi = pymerge_merge(runs, keep_tombstones=True)
    assert (md, mi) != (pd, pi), (
        "timestamp-order mutation not detected by the ts-descending case"
    )


def test_mutation_ts_high_half_detected(tmp_path):
    """Flip the i128 HIGH-half comparison (sign-carrying) -> the
    ts-extremes case MUST diverge (negative vs positive timestamps order
    through ts_hi)."""
    lib = _build_mutant(
        tmp_path,
        "if (a->ts_hi != b->ts_hi) return a->ts_hi < b->ts_hi ? -1 : 1;",
        "if (a->ts_hi != b->ts_hi) return a->ts_hi < b->ts_hi ? 1 : -1;",
        "tshi",
    )
    from adversarial import case_ts_extremes

    runs = case_ts_extremes()
    md, mi = _mutant_compact(lib, runs, keep=True)
    pd, pi = pymerge_merge(runs, keep_tombstones=True)
    assert (md, mi) != (pd, pi), (
        "ts high-half mutation not detected by the ts-extremes case"
    )


def test_mutation_dedup_detected(tmp_path):
    """Invert the peek-dedup rule (first pop of an equal-key group wins
    instead of last) -> the window-straddle case MUST diverge."""
    lib = _build_mutant(
        tmp_path,
        "write = !(nx->e.key_len == cur.e.key_len &&",
        "write = (nx->e.key_len == cur.e.key_len &&",
        "dedup",
    )
    from adversarial import case_window_straddle

    runs = case_window_straddle()
    md, mi = _mutant_compact(lib, runs, keep=True)
    pd, pi = pymerge_merge(runs, keep_tombstones=True)
    assert (md, mi) != (pd, pi), (
        "dedup-rule mutation not detected by the window-straddle case"
    )
