"""Seeded synthetic SSTable run generator for the BASELINE.json configs.

Shapes (BASELINE.json `configs`):
  cfg1: 2 runs x 10k entries, 16 B keys / 64 B values
  cfg2: 4 runs x 1M  entries, 16 B keys / 256 B values
  cfg3: 8 runs x ~987k entries (1 GiB data each), 32 B keys / 1 KiB values,
        50% key overlap (half the keys appear in exactly 2 runs),
        5% tombstones
  cfg4: 64 jobs of (4 runs x 256 MiB), cfg2 shapes — built per-job via
        make_job(..., job_seed)
  cfg5: 16 runs, zipf msgpack str keys (8-128 B), 4 KiB values

Determinism: default seed 0xDBEE1; timestamps strictly increase with run
index (ts = run_index * 2^40 + position) so merge winners are deterministic
(BASELINE.md). Keys are uniformly random bytes, sorted ascending per run,
unique within a run (dbeel flush invariant, lsm_tree.rs:925-946).
"""
from __future__ import annotations

import numpy as np

from .format import build_run_fixed_key

DEFAULT_SEED = 0xDBEE1
TS_RUN_STRIDE = 1 << 40


def _sort_fixed_keys(keys: np.ndarray) -> np.ndarray:
    """Lexicographic ascending sort of (N, K) u8 keys, K multiple of 8."""
    N, K = keys.shape
    cols = keys.view(">u8").reshape(N, K // 8)  # big-endian u64 words
    order = np.lexsort(tuple(cols[:, j] for j in range(K // 8 - 1, -1, -1)))
    return keys[order]


def _unique_keys(rng: np.random.Generator, n: int, K: int) -> np.ndarray:
    """n distinct random-looking K-byte keys (u8 matrix), not sorted.

    Uniqueness by construction: the first 8 bytes are a Weyl sequence
    (i * odd-constant mod 2^64 — a bijection on u64, so no two keys
    collide) xored with a seeded offset; remaining bytes fully random.
    This replaces an np.unique over the whole pool (the dominant
    generation cost at GiB scale) with O(n) work, and keeps the
    big-endian prefix distribution uniform for merge-path purposes."""
    GOLDEN = np.uint64(0x9E3779B97F4A7C15)
    base = rng.integers(0, 2**63, dtype=np.uint64)
    first = (np.arange(n, dtype=np.uint64) * GOLDEN) ^ base
    keys = np.empty((n, K), dtype=np.uint8)
    keys[:, :8] = first.view(np.uint8).reshape(n, 8)
    if K > 8:
        keys[:, 8:] = rng.integers(0, 256, size=(n, K - 8), dtype=np.uint8)
    return rng.permutation(keys, axis=0)


def make_runs(
    n_runs: int,
    entries_per_run: int,
    key_size: int,
    value_size: int,
    overlap_frac: float = 0.0,
    tombstone_frac: float = 0.0,
    seed: int = DEFAULT_SEED,
) -> list[tuple[np.ndarray, np.ndarray]]:
    """Build n_runs (data, index) byte-array pairs.

    overlap_frac o: a fraction o of each run's entries use keys shared with
    exactly one other run (paired runs 2i/2i+1), so those keys appear in
    exactly 2 runs; the rest are unique to the run.
    tombstone_frac t: fraction of each run's entries with empty values.
    """
    rng = np.random.default_rng(seed)
    n_shared = int(entries_per_run * overlap_frac)
    n_own = entries_per_run - n_shared

    # Draw one global pool of distinct keys, then partition: per-pair shared
    # sets and per-run unique sets.
    n_pairs = n_runs // 2
    total = n_pairs * n_shared + n_runs * n_own + (n_runs % 2) * n_shared
    pool = _unique_keys(rng, total, key_size)
    pos = 0
    shared_sets = []
    for p in range(n_pairs):
        shared_sets.append(pool[pos : pos + n_shared])
        pos += n_shared
    odd_shared = None
    if n_runs % 2 and n_shared:
        odd_shared = pool[pos : pos + n_shared]  # unpaired run: keys used once
        pos += n_shared

    out = []
    for r in range(n_runs):
        own = pool[pos : pos + n_own]
        pos += n_own
        if n_shared:
            if r // 2 < n_pairs:
                sh = shared_sets[r // 2]
            else:
                sh = odd_shared
            keys = np.concatenate([own, sh], axis=0)
        else:
            keys = own
        keys = _sort_fixed_keys(keys)
        n = keys.shape[0]

        vsizes = np.full(n, value_size, dtype=np.uint64)
        if tombstone_frac > 0:
            tomb = rng.random(n) < tombstone_frac
            vsizes[tomb] = 0
        n_val_bytes = int(vsizes.sum())
        # tiled random pool: value CONTENT is never compared or branched
        # on (verbatim byte copies), so a 32 MiB seeded pool tiled to
        # size is as good as fully fresh bytes and ~20x faster
        pool_sz = min(n_val_bytes, 32 << 20) or 1
        vpool = rng.integers(0, 256, size=pool_sz, dtype=np.uint8)
        reps = -(-n_val_bytes // pool_sz) if n_val_bytes else 0
        vfill = np.tile(vpool, reps)[:n_val_bytes]
        ts = np.uint64(r * TS_RUN_STRIDE) + np.arange(n, dtype=np.uint64)
        data, index = build_run_fixed_key(keys, vsizes, vfill, ts)
        out.append((data, index))
    return out


CONFIGS = {
    "cfg1": dict(n_runs=2, entries_per_run=10_000, key_size=16, value_size=64),
    "cfg2": dict(n_runs=4, entries_per_run=1_000_000, key_size=16, value_size=256),
    "cfg3": dict(
        n_runs=8,
        entries_per_run=987_000,
        key_size=32,
        value_size=1024,
        overlap_frac=0.5,
        tombstone_frac=0.05,
    ),
    # cfg4 = 64 jobs of this shape (per-job seed via make_job)
    "cfg4_job": dict(
        n_runs=4, entries_per_run=883_000, key_size=16, value_size=256
    ),
    # cfg5: built via make_runs_varkey (variable-length msgpack keys)
    "cfg5": dict(
        n_runs=16, entries_per_run=200_000, value_size=4096,
        overlap_frac=0.3, tombstone_frac=0.02
    ),
}


def make_config(name: str, seed: int = DEFAULT_SEED, scale: float = 1.0):
    cfg = dict(CONFIGS[name])
    if scale != 1.0:
        cfg["entries_per_run"] = max(16, int(cfg["entries_per_run"] * scale))
    return make_runs(seed=seed, **cfg)


def input_bytes(runs) -> int:
    return sum(len(d) + len(i) for d, i in runs)


def make_runs_varkey(
    n_runs: int,
    entries_per_run: int,
    value_size: int = 4096,
    overlap_frac: float = 0.3,
    tombstone_frac: float = 0.02,
    seed: int = DEFAULT_SEED,
):
    """cfg5-shaped runs: variable-length msgpack str keys, zipf(1.1)
    lengths clipped to 8..128 B total, 4 KiB values (BASELINE.json
    configs[4]). Keys are raw msgpack `str8` encodings (0xd9 | len |
    bytes) compared as raw bytes, exactly as dbeel compares them
    (Entry::cmp on Vec<u8>, mod.rs:75-81)."""
    rng = np.random.default_rng(seed)

    def draw_keys(n):
        keys = set()
        while len(keys) < n:
            need = n - len(keys) + 16
            Ls = np.clip(rng.zipf(1.1, need), 6, 126).astype(np.int64)
            body = rng.integers(32, 127, int(Ls.sum()), dtype=np.uint8)
            pos = 0
            for L in Ls:
                keys.add(bytes([0xD9, L]) + body[pos : pos + L].tobytes())
                pos += L
                if len(keys) >= n:
                    break
        # set iteration order is hash-salted per process; sort for
        # determinism, then rng-permute so the shared-key slices below
        # remain randomly selected (not contiguous key ranges)
        lst = sorted(keys)
        return [lst[i] for i in rng.permutation(len(lst))]

    n_shared = int(entries_per_run * overlap_frac)
    shared = draw_keys(n_shared * (n_runs // 2)) if n_shared else []
    out = []
    spos = 0
    for r in range(n_runs):
        own = draw_keys(entries_per_run - n_shared)
        ks = own
        if n_shared:
            if r % 2 == 0 and spos + n_shared <= len(shared):
                sh = shared[spos : spos + n_shared]
            else:
                sh = shared[max(0, spos - n_shared) : spos]
            if r % 2 == 1:
                spos += n_shared
            ks = own + sh[: n_shared]
        ks = sorted(set(ks))
        n = len(ks)
        tomb = rng.random(n) < tombstone_frac
        n_norm = int(n - tomb.sum())
        vblob = rng.integers(0, 256, n_norm * value_size,
                             dtype=np.uint8).tobytes()
        out.append(_build_run_var(ks, tomb, vblob, value_size, r << 40))
    return out


def _build_run_var(ks, tomb, vblob, V, ts_base):
    """Fast var-key run builder: one join over per-field pieces
    (memoryview value slices — no per-entry intermediate copies)."""
    import struct

    from .format import INDEX_DTYPE

    n = len(ks)
    idx = np.zeros(n, dtype=INDEX_DTYPE)
    parts = []
    mv = memoryview(vblob)
    dlenV = struct.pack("<Q", V)
    dlen0 = struct.pack("<Q", 0)
    klen_cache = {}
    off = 0
    vpos = 0
    offs = idx["offset"]
    kss = idx["key_size"]
    fss = idx["full_size"]
    for i, k in enumerate(ks):
        kl = len(k)
        hdr = klen_cache.get(kl)
        if hdr is None:
            hdr = struct.pack("<Q", kl)
            klen_cache[kl] = hdr
        parts.append(hdr)
        parts.append(k)
        if tomb[i]:
            parts.append(dlen0)
            dl = 0
        else:
            parts.append(dlenV)
            parts.append(mv[vpos : vpos + V])
            vpos += V
            dl = V
        parts.append(int(ts_base + i).to_bytes(16, "little"))
        offs[i] = off
        kss[i] = 8 + kl
        fss[i] = 32 + kl + dl
        off += 32 + kl + dl
    data = b"".join(parts)
    return (np.frombuffer(data, dtype=np.uint8),
            idx.view(np.uint8).reshape(-1))
