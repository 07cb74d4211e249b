"""Independent pure-Python restatement of MurmurHash3 x86_32 and the
reference's migration scan semantics (TEST INFRASTRUCTURE).

The reference hashes keys with murmur3_32 seed 0 (hash_bytes,
shards.rs:96-101; murmur3 crate v0.5.2 — public Appleby algorithm) and
filters migration iteration by between_cmp hash ranges
(tasks/migration.rs:54-60). AsyncIter yields every entry, sstables
ascending, entries in index order, no dedup, no tombstone filter
(lsm_tree.rs:210-276).
"""
from __future__ import annotations

M = 0xFFFFFFFF


def murmur3_32(key: bytes, seed: int = 0) -> int:
    c1, c2 = 0xCC9E2D51, 0x1B873593
    h = seed & M
    nblocks = len(key) // 4
    for i in range(nblocks):
        k = int.from_bytes(key[i * 4 : i * 4 + 4], "little")
        k = (k * c1) & M
        k = ((k << 15) | (k >> 17)) & M
        k = (k * c2) & M
        h ^= k
        h = ((h << 13) | (h >> 19)) & M
        h = (h * 5 + 0xE6546B64) & M
    tail = key[nblocks * 4 :]
    k1 = 0
    if len(tail) >= 3:
        k1 ^= tail[2] << 16
    if len(tail) >= 2:
        k1 ^= tail[1] << 8
    if len(tail) >= 1:
        k1 ^= tail[0]
        k1 = (k1 * c1) & M
        k1 = ((k1 << 15) | (k1 >> 17)) & M
        k1 = (k1 * c2) & M
        h ^= k1
    h ^= len(key)
    h ^= h >> 16
    h = (h * 0x85EBCA6B) & M
    h ^= h >> 13
    h = (h * 0xC2B2AE35) & M
    h ^= h >> 16
    return h


def between_cmp(h: int, start: int, end: int) -> bool:
    """EXACT restatement of tasks/migration.rs:54-60 (note: the wrapped
    case end < start evaluates true for every hash in the reference —
    restated verbatim)."""
    if end < start:
        return h < start or h >= end
    return h >= start and h < end


def scan_model(runs, start_key=None, end_key=None, hash_ranges=None):
    """Host model of dbeel_gpu_scan: the AsyncIter yield order with the
    combined filters. Returns (data_bytes, index_bytes, n)."""
    import struct

    from dbeel_amd.format import parse_run

    out_data, out_index = [], []
    off = 0
    n = 0
    for d, i in runs:
        d = bytes(d)
        for e in parse_run(d, bytes(i)):
            if start_key is not None and e.key < start_key:
                continue
            if end_key is not None and e.key >= end_key:
                continue
            if hash_ranges:
                h = murmur3_32(e.key, 0)
                if not any(between_cmp(h, s, t) for s, t in hash_ranges):
                    continue
            from dbeel_amd.format import encode_entry

            raw = encode_entry(e)
            out_index.append(
                struct.pack("<QII", off, 8 + len(e.key), len(raw))
            )
            out_data.append(raw)
            off += len(raw)
            n += 1
    return b"".join(out_data), b"".join(out_index), n
