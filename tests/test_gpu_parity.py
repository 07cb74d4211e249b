"""GPU parity tests (marked gpu; run on a real MI355X via gpurun).

The bar (SURVEY.md §8c / north_star): output `.data`/`.index` bytes from the
HIP engine are BIT-IDENTICAL to the CPU oracle on the same inputs — golden
fixtures, seeded BASELINE shapes, and edge cases. Plus size-independent
property checks at larger scale.
"""
import numpy as np
import pytest

import oracle
from conftest import GOLDEN_CASES, load_golden
from dbeel_amd.format import Entry, build_run, parse_run
from dbeel_amd.genruns import make_runs

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine():
    import dbeel_amd

    return dbeel_amd


@pytest.mark.parametrize("name", GOLDEN_CASES)
@pytest.mark.parametrize("keep", [True, False])
def test_golden_parity(engine, name, keep):
    runs, exp_keep, exp_drop = load_golden(name)
    exp = exp_keep if keep else exp_drop
    data, index, n = engine.compact(runs, keep_tombstones=keep, device=0)
    assert index == exp[1]
    assert data == exp[0]
    assert n == len(exp[1]) // 16


@pytest.mark.parametrize(
    "shape",
    [
        # (n_runs, entries, ksize, vsize, overlap, tomb)
        (2, 10_000, 16, 64, 0.0, 0.0),       # cfg1 exact
        (4, 50_000, 16, 256, 0.3, 0.0),      # cfg2 scaled
        (8, 30_000, 32, 1024, 0.5, 0.05),    # cfg3 scaled
        (16, 8_000, 16, 256, 0.5, 0.2),      # 16-way merge
        (1, 5_000, 16, 64, 0.0, 0.3),        # single run
        (3, 2_000, 8, 32, 1.0, 0.5),         # full overlap, many tombstones
    ],
)
@pytest.mark.parametrize("keep", [True, False])
def test_seeded_shape_parity(engine, shape, keep):
    n_runs, n, ks, vs, ov, tb = shape
    runs = make_runs(n_runs, n, ks, vs, overlap_frac=ov, tombstone_frac=tb,
                     seed=0xDBEE1 + n_runs)
    gd, gi, gn = engine.compact(runs, keep_tombstones=keep, device=0)
    od, oi, on = oracle.compact(runs, keep_tombstones=keep)
    assert gn == on
    assert gi == oi
    assert gd == od


def test_ragged_keys_parity(engine):
    """Variable-length keys 1..128 B incl. prefix pairs and empty keys,
    mixed value sizes (cfg5 precursor)."""
    rng = np.random.default_rng(99)
    runs = []
    for r in range(6):
        keys = {b"", b"\x00", b"\x00\x00"}
        for _ in range(800):
            keys.add(bytes(rng.integers(0, 256, int(rng.integers(1, 129)),
                                        dtype=np.uint8)))
        ents = []
        for i, k in enumerate(sorted(keys)):
            dlen = int(rng.integers(0, 6)) * 111
            ents.append(Entry(k, bytes(rng.integers(0, 256, dlen,
                                                    dtype=np.uint8)),
                              int(rng.integers(-100, 100))))
        runs.append(build_run(ents))
    for keep in (True, False):
        gd, gi, gn = engine.compact(runs, keep_tombstones=keep, device=0)
        od, oi, on = oracle.compact(runs, keep_tombstones=keep)
        assert (gd, gi, gn) == (od, oi, on)


def test_cfg5_varkey_16run_parity(engine):
    """cfg5 shape (BASELINE configs[4]): 16-run merge of variable-length
    zipf msgpack str keys + 4 KiB values, scaled to parity-test size."""
    from dbeel_amd.genruns import make_runs_varkey

    runs = make_runs_varkey(16, 800, value_size=4096, overlap_frac=0.3,
                            tombstone_frac=0.05, seed=0x5E5)
    for keep in (True, False):
        gd, gi, gn = engine.compact(runs, keep_tombstones=keep, device=0)
        od, oi, on = oracle.compact(runs, keep_tombstones=keep)
        assert gn == on
        assert gi == oi
        assert gd == od


def test_empty_inputs(engine):
    runs = [(b"", b""), (b"", b"")]
    gd, gi, gn = engine.compact(runs, keep_tombstones=True, device=0)
    assert (gd, gi, gn) == (b"", b"", 0)


def test_corrupt_input_rejected(engine):
    import struct

    from dbeel_amd.engine import DbeelGpuError

    # unsorted run (flush invariant violated) must error loudly
    r1 = build_run([Entry(b"b", b"1", 1)])
    r2 = build_run([Entry(b"a", b"2", 2)])
    bad_data = r1[0] + r2[0]
    bad_index = r1[1] + struct.pack("<QII", len(r1[0]), 9, 33)
    with pytest.raises(DbeelGpuError) as ei:
        engine.compact([(bad_data, bad_index)], keep_tombstones=True, device=0)
    assert ei.value.code == 2  # CORRUPT

    # index record pointing past data_len
    idx = struct.pack("<QII", 1000, 9, 64)
    with pytest.raises(DbeelGpuError) as ei:
        engine.compact([(b"\x00" * 64, idx)], keep_tombstones=True, device=0)
    assert ei.value.code == 2


def test_full_cfg3_exact_parity(engine):
    """FULL BASELINE config 3 (8 x 1 GiB runs, 50% overlap, 5% tombstones):
    exact byte parity of the 6.1 GB output vs the CPU oracle, plus the
    size-independent properties (sortedness enforced by feeding the output
    back through the engine, which validates the flush invariant)."""
    from dbeel_amd.genruns import make_config

    runs = make_config("cfg3")
    gd, gi, gn = engine.compact(runs, keep_tombstones=False, device=0)
    od, oi, on = oracle.compact(runs, keep_tombstones=False)
    assert gn == on
    assert gi == oi
    assert gd == od
    # output is itself a valid sorted unique-key run: recompacting the
    # single run must be byte-idempotent (k_rankreduce would error on any
    # sortedness violation)
    rd, ri, rn = engine.compact([(gd, gi)], keep_tombstones=True, device=0)
    assert (rd, ri, rn) == (gd, gi, gn)


def test_sixtyfour_runs_bound(engine):
    """MAX_RUNS=64 merge (the ABI bound); 65 runs must be rejected."""
    from dbeel_amd.engine import DbeelGpuError
    from dbeel_amd.format import Entry, build_run

    rng = np.random.default_rng(77)
    runs = []
    for r in range(64):
        keys = sorted({bytes(rng.integers(0, 256, 6, dtype=np.uint8))
                       for _ in range(30)})
        runs.append(build_run(
            [Entry(k, bytes([r]), (r << 20) + i)
             for i, k in enumerate(keys)]
        ))
    gd, gi, gn = engine.compact(runs, keep_tombstones=True, device=0)
    od, oi, on = oracle.compact(runs, keep_tombstones=True)
    assert (gd, gi, gn) == (od, oi, on)

    with pytest.raises(DbeelGpuError) as ei:
        engine.compact(runs + [runs[0]], keep_tombstones=True, device=0)
    assert ei.value.code == 1


def test_large_scale_properties(engine):
    """Mid-scale run (~600 MB input) — byte parity vs oracle plus
    size-independent properties: sortedness, unique keys, verbatim bytes,
    count."""
    runs = make_runs(8, 120_000, 32, 1024, overlap_frac=0.5,
                     tombstone_frac=0.05, seed=0xC0FFEE)
    gd, gi, gn = engine.compact(runs, keep_tombstones=False, device=0)
    od, oi, on = oracle.compact(runs, keep_tombstones=False)
    assert gn == on and gi == oi and gd == od

    ents = parse_run(gd, gi)
    keys = [e.key for e in ents]
    assert keys == sorted(keys)
    assert len(set(keys)) == len(keys)
    assert not any(e.is_tombstone for e in ents)


def test_encode_run_matches_entry_writer_layout(engine):
    """GPU flush encoder: byte parity with the
    EntryWriter layout (entry_writer.rs:71-98) as pinned by
    dbeel_amd.format.build_run, incl. tombstones, empty keys, negative
    timestamps and ragged sizes."""
    from dbeel_amd.engine import encode_run
    from dbeel_amd.format import Entry, build_run

    rng = np.random.default_rng(31)
    ents = []
    keys = {b""} | {
        bytes(rng.integers(0, 256, int(rng.integers(1, 100)), dtype=np.uint8))
        for _ in range(3000)
    }
    for i, k in enumerate(sorted(keys)):
        dlen = int(rng.integers(0, 5)) * 211
        data = bytes(rng.integers(0, 256, dlen, dtype=np.uint8))
        ents.append((k, data, int(rng.integers(-(10**18), 10**18))))

    gd, gi, gn = encode_run(ents, device=0)
    ed, ei = build_run([Entry(*e) for e in ents])
    assert gn == len(ents)
    assert gi == ei
    assert gd == ed

    # flush -> compact composition: an encoded run feeds compaction
    data2, index2, n2 = engine.compact([(gd, gi)], keep_tombstones=True,
                                       device=0)
    assert (data2, index2) == (gd, gi)  # single sorted run is idempotent


def test_batched_lookup(engine):
    """GPU-batched point lookup vs a host-side model of
    the reference read path: LSMTree::get scans sstables newest-INDEX-first
    and returns the first key match (lsm_tree.rs:692-696), so the highest
    run index wins regardless of timestamp. Covers present, overwritten,
    deleted and absent keys, plus the delete->get->KeyNotFound distinction
    (tests/db_server.rs:183-213 semantics)."""
    from dbeel_amd.engine import lookup
    from dbeel_amd.format import parse_run

    runs = make_runs(4, 3000, 16, 64, overlap_frac=0.5, tombstone_frac=0.15,
                     seed=55)
    runs = [(bytes(d), bytes(i)) for d, i in runs]

    # host model: highest run index wins (reference read-path rule)
    model = {}
    for r, (d, i) in enumerate(runs):
        for e in parse_run(d, i):
            cur = model.get(e.key)
            if cur is None or r > cur[0]:
                model[e.key] = (r, e.data)

    rng = np.random.default_rng(3)
    present = list(model.keys())
    queries = [present[int(rng.integers(0, len(present)))]
               for _ in range(500)]
    absent = [bytes(rng.integers(0, 256, 16, dtype=np.uint8))
              for _ in range(100)]
    queries += absent

    got = lookup(runs, queries, device=0)
    for k, g in zip(queries, got):
        if k in model:
            exp = model[k][1]  # b"" for tombstone-winner (deleted)
            assert g == exp, k.hex()
        else:
            assert g is None, k.hex()


def _adversarial_cases():
    from adversarial import all_cases

    return all_cases()


@pytest.mark.parametrize("case", _adversarial_cases(),
                         ids=lambda c: c[0])
@pytest.mark.parametrize("keep", [True, False], ids=["keep", "drop"])
def test_adversarial_cases_gpu(engine, case, keep):
    """The adversarial seam cases (tests/adversarial.py — corank-window
    straddles, 64-run identical keys, staged-prefix boundaries, long-key
    blob-fallback ties, zero-padding families, i128 extremes) bit-exact
    GPU vs oracle."""
    name, runs = case
    gd, gi, gn = engine.compact(runs, keep_tombstones=keep, device=0)
    od, oi, on = oracle.compact(runs, keep_tombstones=keep)
    assert gn == on, name
    assert gi == oi, name
    assert gd == od, name


def test_lookup_newest_index_first(engine):
    """The reference read path returns the match from the newest sstable
    INDEX even when it carries an OLDER timestamp (set_with_timestamp via
    replication can write one) — lsm_tree.rs:692-696 `.rev()` + first
    match. The compaction winner rule (max timestamp) differs here; the
    lookup must follow the read path, not the compaction rule."""
    from dbeel_amd.engine import lookup
    from dbeel_amd.format import Entry, build_run

    k = b"duplicated-key!!"
    run0 = build_run([Entry(k, b"newer-ts-old-run", 500)])
    run1 = build_run([Entry(k, b"older-ts-new-run", 100)])
    got = lookup([run0, run1], [k], device=0)
    assert got == [b"older-ts-new-run"]
    # tombstone in the newer run hides the older value (delete semantics)
    run1t = build_run([Entry(k, b"", 100)])
    got = lookup([run0, run1t], [k], device=0)
    assert got == [b""]


def test_migration_scan_parity(engine):
    """Migration/iteration scan: GPU dbeel_gpu_scan vs
    the host AsyncIter model — every entry in run-ascending order, no
    dedup/tombstone filtering, with key-range and murmur3 hash-range
    filters (incl. the reference's everything-matches wrapped range)."""
    from dbeel_amd.engine import scan
    from pymm3 import scan_model

    runs = make_runs(5, 8_000, 16, 128, overlap_frac=0.4,
                     tombstone_frac=0.1, seed=0x5CA7)

    cases = [
        dict(),                                     # full iteration
        dict(start_key=b"\x40" * 4),                # lower bound
        dict(end_key=b"\xc0" * 4),                  # upper bound
        dict(start_key=b"\x20", end_key=b"\xa0"),   # both bounds
        dict(hash_ranges=[(0, 2**31)]),             # half the ring
        dict(hash_ranges=[(2**31, 0)]),             # wrapped: everything
        dict(hash_ranges=[(0, 1 << 28), (3 << 30, 2**32 - 1)]),
        dict(start_key=b"\x10", end_key=b"\xf0",
             hash_ranges=[(1 << 30, 3 << 30)]),     # combined
    ]
    for kw in cases:
        gd, gi, gn = scan(runs, device=0, **kw)
        md, mi, mn = scan_model(runs, **kw)
        assert gn == mn, kw
        assert gi == mi, kw
        assert gd == md, kw


def test_migration_scan_at_scale(engine):
    """Scan parity at a cfg3-shaped scale (scaled entries): ordered
    range export stays bit-exact when windows/chunking kick in."""
    from dbeel_amd.engine import scan
    from pymm3 import scan_model

    runs = make_runs(8, 60_000, 32, 1024, overlap_frac=0.5,
                     tombstone_frac=0.05, seed=0xDBEE1)
    kw = dict(hash_ranges=[(0, 1 << 30), (3 << 30, 1 << 31)])
    gd, gi, gn = scan(runs, device=0, **kw)
    md, mi, mn = scan_model(runs, **kw)
    assert (gn, gi, gd) == (mn, mi, md)


def test_scan_modes_agree(engine):
    """The packed single survivor scan (bytes|count in one u64) and the
    two-scan fallback must produce identical results, including batched
    per-job slicing whose boundary reads decode the packed values."""
    import os

    from dbeel_amd.engine import BatchJob

    jobs = [
        make_runs(4, 8_000, 16, 256, overlap_frac=0.4, tombstone_frac=0.1,
                  seed=301),
        make_runs(3, 6_000, 32, 512, overlap_frac=0.6, tombstone_frac=0.2,
                  seed=302),
    ]
    results = {}
    for mode in ("packed", "two"):
        os.environ["DBEEL_SCAN_MODE"] = mode
        try:
            with BatchJob(jobs, device=0) as bj:
                tot = bj.run(False)[:2]
                results[mode] = (tot, [bj.fetch_job(j)
                                       for j in range(len(jobs))])
        finally:
            os.environ.pop("DBEEL_SCAN_MODE", None)
    assert results["packed"] == results["two"]
    # and both match the oracle
    for j, runs in enumerate(jobs):
        od, oi, on = oracle.compact(runs, keep_tombstones=False)
        gd, gi, gn = results["packed"][1][j]
        assert (gn, gi, gd) == (on, oi, od)


def test_batched_jobs_parity(engine):
    """Batched independent jobs (one launch set, BASELINE configs[3]'s
    8-jobs-per-GPU shape): each job's sliced-out result must be
    bit-identical to compacting that job alone with the oracle — ranks,
    crossrank slots and winner flags are job-local, and a key duplicated
    ACROSS jobs must survive in every job (no cross-job dedup)."""
    from dbeel_amd.engine import BatchJob

    jobs = [
        make_runs(4, 9_000, 16, 256, overlap_frac=0.4, tombstone_frac=0.1,
                  seed=201),
        make_runs(2, 5_000, 16, 64, overlap_frac=1.0, tombstone_frac=0.3,
                  seed=202),
        make_runs(1, 3_000, 16, 128, seed=203),   # single-run job
        make_runs(8, 2_000, 32, 512, overlap_frac=0.5, tombstone_frac=0.05,
                  seed=204),
    ]
    # duplicate one job entirely: same keys in two different jobs —
    # cross-job isolation means both jobs keep their own winners
    jobs.append(jobs[1])

    for keep in (False, True):
        with BatchJob(jobs, device=0) as bj:
            out_bytes, out_entries, _ = bj.run(keep)
            total_d = total_n = 0
            for j, runs in enumerate(jobs):
                gd, gi, gn = bj.fetch_job(j)
                od, oi, on = oracle.compact(runs, keep_tombstones=keep)
                assert gn == on, f"job {j}"
                assert gi == oi, f"job {j}"
                assert gd == od, f"job {j}"
                total_d += len(gd)
                total_n += gn
            assert out_bytes == total_d
            assert out_entries == total_n
            # jobs 1 and 4 are identical inputs -> identical outputs
            assert bj.fetch_job(1) == bj.fetch_job(4)


def test_streamed_ingest_parity(engine):
    """Streamed pinned ingest (north_star: pinned host DRAM, chunked
    hipMemcpyAsync, prepare overlapped on the compute stream): a job
    created from one data set, re-ingested with fresh contents of the
    same shape, must produce bit-exact oracle output for the NEW
    contents — including with sub-MiB chunks forcing many
    chunk/prepare interleavings — and repeated runs after one ingest
    stay identical."""
    import os

    from dbeel_amd.engine import pin_host, unpin_host

    # tombstone_frac 0 so both seeds give identical run byte sizes
    # (ingest requires matching shapes; dedup still exercised via overlap)
    runs_a = make_runs(4, 30_000, 16, 256, overlap_frac=0.4, seed=71)
    runs_b = make_runs(4, 30_000, 16, 256, overlap_frac=0.4, seed=72)
    # same shape guaranteed: same entry counts and fixed sizes
    for a, b in zip(runs_a, runs_b):
        assert a[0].nbytes == b[0].nbytes and a[1].nbytes == b[1].nbytes

    for arr_pair in runs_b:
        pin_host(arr_pair[0])
        pin_host(arr_pair[1])
    try:
        os.environ["DBEEL_STREAM_CHUNK_MB"] = "1"  # force many chunks
        with engine.Job(runs_a, device=0) as job:
            st = job.ingest(runs_b)
            assert st["chunks"] > 8, st
            assert st["bytes"] == sum(d.nbytes + i.nbytes
                                      for d, i in runs_b)
            d1, n1, t1 = job.run(keep_tombstones=False)
            gd, gi, gn = job.fetch()
            od, oi, on = oracle.compact(runs_b, keep_tombstones=False)
            assert (gn, gi, gd) == (on, oi, od)
            # prepare was overlapped into the ingest; the run skipped it
            assert t1["prep_ms"] < 0.05, t1
            # repeated run after one ingest: identical result
            d2, n2, _ = job.run(keep_tombstones=False)
            assert (d2, n2) == (d1, n1)
    finally:
        os.environ.pop("DBEEL_STREAM_CHUNK_MB", None)
        for arr_pair in runs_b:
            unpin_host(arr_pair[0])
            unpin_host(arr_pair[1])


def test_batched_job_streamed_ingest(engine):
    """Streamed ingest composes with batched jobs: fresh contents of the
    same shapes streamed into a 3-job batch, each job's sliced result
    bit-exact vs the oracle on the new contents."""
    import os

    from dbeel_amd.engine import BatchJob

    mk = lambda seed: [
        make_runs(3, 6_000, 16, 128, overlap_frac=0.5, seed=seed),
        make_runs(2, 4_000, 16, 64, overlap_frac=0.2, seed=seed + 1),
        make_runs(4, 3_000, 16, 256, seed=seed + 2),
    ]
    jobs_a, jobs_b = mk(500), mk(900)
    os.environ["DBEEL_STREAM_CHUNK_MB"] = "1"
    try:
        with BatchJob(jobs_a, device=0) as bj:
            flat_b = [rv for runs in jobs_b for rv in runs]
            st = bj.ingest(flat_b)
            assert st["chunks"] > 4
            bj.run(False)
            for j, runs in enumerate(jobs_b):
                gd, gi, gn = bj.fetch_job(j)
                od, oi, on = oracle.compact(runs, keep_tombstones=False)
                assert (gn, gi, gd) == (on, oi, od), f"job {j}"
    finally:
        os.environ.pop("DBEEL_STREAM_CHUNK_MB", None)


def test_streamed_ingest_shape_mismatch_rejected(engine):
    runs_a = make_runs(2, 5_000, 16, 64, seed=1)
    runs_b = make_runs(2, 6_000, 16, 64, seed=2)
    from dbeel_amd.engine import DbeelGpuError

    with engine.Job(runs_a, device=0) as job:
        with pytest.raises(DbeelGpuError) as ei:
            job.ingest(runs_b)
        assert ei.value.code == 1  # INVALID_ARG


def test_sliced_compaction_parity(engine):
    """Sliced compaction (runs larger than HBM, here forced with a tiny
    resident budget): byte-identical to the whole-job compaction and to
    the oracle, including equal-key groups that must never split across
    slice pivots."""
    from dbeel_amd.engine import compact_sliced

    runs = make_runs(6, 40_000, 16, 128, overlap_frac=0.6,
                     tombstone_frac=0.1, seed=99)
    total = sum(d.nbytes + i.nbytes for d, i in runs)
    for keep in (True, False):
        od, oi, on = oracle.compact(runs, keep_tombstones=keep)
        # budget forces ~7 slices
        gd, gi, gn = compact_sliced(runs, keep_tombstones=keep, device=0,
                                    max_resident_bytes=total // 7)
        assert gn == on
        assert gi == oi
        assert gd == od


def test_zero_data_run_with_entries_rejected(engine):
    """A run whose index claims entries but whose data file is empty is
    corrupt; both the normal pipeline and streamed ingest (which launches
    the ranged prepare separately for zero-data runs) must flag it."""
    from dbeel_amd.engine import DbeelGpuError

    good = make_runs(2, 1_000, 16, 64, seed=3)
    bad = [good[0], (np.zeros(0, dtype=np.uint8), good[1][1])]
    with pytest.raises(DbeelGpuError) as ei:
        engine.compact(bad, keep_tombstones=True, device=0)
    assert ei.value.code == 2  # CORRUPT

    with engine.Job(bad, device=0) as job:
        with pytest.raises(DbeelGpuError) as ei:
            job.ingest(bad)
        assert ei.value.code == 2


def test_ingest_rejects_corrupt_offsets(engine):
    """Streamed ingest with NON-MONOTONE index offsets (corrupt input)
    must report CORRUPT, never wrap a chunk copy or accept the data."""
    import os

    from dbeel_amd.engine import DbeelGpuError

    runs = make_runs(2, 2_000, 16, 64, seed=11)
    bad = [(d.copy(), i.copy()) for d, i in runs]
    # swap two offset fields in run 0's index -> offsets not monotone
    idx = bad[0][1]
    rec0 = idx[0:8].copy()
    idx[0:8] = idx[160:168]
    idx[160:168] = rec0
    os.environ["DBEEL_STREAM_CHUNK_MB"] = "1"
    try:
        with engine.Job(runs, device=0) as job:
            with pytest.raises(DbeelGpuError) as ei:
                job.ingest(bad)
            assert ei.value.code == 2  # CORRUPT
            # the job stays usable: re-ingest good data, run, verify
            job.ingest(runs)
            job.run(False)
            gd, gi, gn = job.fetch()
            od, oi, on = oracle.compact(runs, keep_tombstones=False)
            assert (gn, gi, gd) == (on, oi, od)
    finally:
        os.environ.pop("DBEEL_STREAM_CHUNK_MB", None)


def test_sliced_rejects_corrupt_index(engine):
    """Sliced compaction reads host key bytes during pivot selection —
    a record pointing past the run's data must yield CORRUPT, not an
    out-of-bounds host read."""
    import struct

    from dbeel_amd.engine import DbeelGpuError, compact_sliced

    runs = [list(r) for r in make_runs(3, 4_000, 16, 64, seed=12)]
    bad_idx = runs[0][1].copy()
    # entry 2000 is the FIRST binary-search probe (mid of 0..4000) and a
    # candidate pivot — point its offset far beyond data_len
    bad_idx[2000 * 16 : 2000 * 16 + 8] = np.frombuffer(
        struct.pack("<Q", 1 << 40), dtype=np.uint8)
    runs[0][1] = bad_idx
    total = sum(d.nbytes + i.nbytes for d, i in runs)
    with pytest.raises(DbeelGpuError) as ei:
        compact_sliced([tuple(r) for r in runs], keep_tombstones=False,
                       device=0, max_resident_bytes=total // 4)
    assert ei.value.code == 2  # CORRUPT


def test_sliced_compaction_midscale_parity(engine):
    """Sliced compaction at ~0.9 GB (cfg3-shaped, forced 4 slices): the
    pivot/binary-search machinery and cross-slice offset rebasing stay
    bit-exact when slices span many copy windows."""
    from dbeel_amd.engine import compact_sliced

    runs = make_runs(8, 100_000, 32, 1024, overlap_frac=0.5,
                     tombstone_frac=0.05, seed=0xDBEE1)
    total = sum(d.nbytes + i.nbytes for d, i in runs)
    od, oi, on = oracle.compact(runs, keep_tombstones=False)
    gd, gi, gn = compact_sliced(runs, keep_tombstones=False, device=0,
                                max_resident_bytes=total // 4)
    assert (gn, gi, gd) == (on, oi, od)


def test_sliced_compaction_varkey_parity(engine):
    """Sliced compaction with VARIABLE-length msgpack keys: host pivot
    selection and per-run binary search must respect raw-byte key order
    for ragged keys too (cfg5-style runs, forced multi-slice)."""
    from dbeel_amd.engine import compact_sliced
    from dbeel_amd.genruns import make_runs_varkey

    runs = make_runs_varkey(6, 5_000, value_size=512, overlap_frac=0.4,
                            tombstone_frac=0.1, seed=0xBEEF)
    total = sum(d.nbytes + i.nbytes for d, i in runs)
    od, oi, on = oracle.compact(runs, keep_tombstones=False)
    gd, gi, gn = compact_sliced(runs, keep_tombstones=False, device=0,
                                max_resident_bytes=total // 5)
    assert (gn, gi, gd) == (on, oi, od)


def test_resident_job_repeatable(engine):
    """Job API: repeated runs on resident inputs give identical results and
    both keep_tombstones settings work on one upload."""
    runs = make_runs(4, 20_000, 16, 256, overlap_frac=0.4,
                     tombstone_frac=0.1, seed=5)
    with engine.Job(runs, device=0) as job:
        d1, n1, t1 = job.run(keep_tombstones=False)
        data1, index1, _ = job.fetch()
        d2, n2, t2 = job.run(keep_tombstones=False)
        data2, index2, _ = job.fetch()
        assert (d1, n1) == (d2, n2)
        assert data1 == data2 and index1 == index2
        assert t2["kernel_ms"] > 0
        od, oi, on = oracle.compact(runs, keep_tombstones=False)
        assert (data1, index1, n1) == (od, oi, on)
