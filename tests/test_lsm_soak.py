"""LSM lifecycle soak (GPU): repeated flush -> compact_tree cycles with a
host-side model, mirroring dbeel's flush/compaction loop
(run_compaction_loop tasks/compaction.rs:104-137 over flush outputs at
even indices, lsm_tree.rs:901-915), plus a mid-soak crash/replay.

The model is a plain dict (newest write wins, empty value = delete) — the
semantics dbeel's get() exposes after any sequence of flushes and
compactions."""
import os

import numpy as np
import pytest

from dbeel_amd import lsm
from dbeel_amd.engine import encode_run
from dbeel_amd.format import parse_run

pytestmark = pytest.mark.gpu


def discover(d):
    idxs = sorted(
        int(f.split(".")[0]) for f in os.listdir(d) if f.endswith(".index")
    )
    return idxs


def visible_state(d):
    """Newest-wins view over all runs: higher sstable index wins for even/
    odd mix the timestamp decides; our soak uses strictly increasing
    timestamps so (ts) ordering is the ground truth."""
    best = {}
    for idx in discover(d):
        data, index = lsm.read_run_files(d, idx)
        for e in parse_run(data, index):
            cur = best.get(e.key)
            if cur is None or e.timestamp > cur[0]:
                best[e.key] = (e.timestamp, e.data)
    return {k: v for k, (ts, v) in best.items() if v != b""}


def test_flush_compact_soak(tmp_path):
    d = str(tmp_path)
    rng = np.random.default_rng(2024)
    model = {}
    ts = [0]

    def flush(batch, index):
        ents = []
        for k, v in sorted(batch.items()):
            ts[0] += 1
            ents.append((k, v, ts[0]))
            if v == b"":
                model.pop(k, None)
            else:
                model[k] = v
        data, idx, _ = encode_run(ents, device=0)
        lsm.write_run_files(d, index, data, idx)

    key_pool = [bytes(rng.integers(97, 123, 8, dtype=np.uint8))
                for _ in range(400)]

    flush_idx = 0
    for cycle in range(4):
        # several flushes at even indices (flush outputs, lsm_tree.rs:914)
        for _ in range(3):
            batch = {}
            for _ in range(120):
                k = key_pool[int(rng.integers(0, len(key_pool)))]
                batch[k] = (b"" if rng.random() < 0.15 else
                            bytes(rng.integers(0, 256, 64, dtype=np.uint8)))
            flush(batch, flush_idx)
            flush_idx += 2

        if cycle == 2:
            # crash mid-compaction, then recover (recovery path §3.3)
            os.environ["DBEEL_LSM_CRASH_AFTER_JOURNAL"] = "1"
            try:
                idxs = discover(d)
                out = max(idxs) + 1
                out += out % 2 == 0  # compaction outputs are odd
                                     # (tasks/compaction.rs:38-43)
                lsm.compact(d, idxs, out, keep_tombstones=True,
                            device=0, bloom_min_size=1 << 40)
            finally:
                del os.environ["DBEEL_LSM_CRASH_AFTER_JOURNAL"]
            assert lsm.replay(d) == 1
        else:
            lsm.compact_tree(d, compaction_factor=2, device=0,
                             bloom_min_size=1 << 40)

        assert visible_state(d) == model, f"cycle {cycle}"

    # final major compaction drops every tombstone
    idxs = discover(d)
    if len(idxs) > 1:
        out = max(idxs) + 1
        out += out % 2 == 0
        lsm.compact(d, idxs, out, keep_tombstones=False, device=0,
                    bloom_min_size=1 << 40)
    idxs = discover(d)
    assert len(idxs) == 1
    data, index = lsm.read_run_files(d, idxs[0])
    ents = parse_run(data, index)
    assert {e.key: e.data for e in ents} == model
    assert not any(e.is_tombstone for e in ents)
    keys = [e.key for e in ents]
    assert keys == sorted(keys)
