"""Independent pure-Python restatement of dbeel's compaction merge.

Used ONLY to generate/check golden fixtures (tests/golden) — a second,
independent restatement so the C oracle is pinned by agreement of two
implementations written separately from the same reference code
(lsm_tree.rs:950-1170). Mirrors the reference exactly:

- BinaryHeap of CompactionItem with reversed (Entry, index) order
  (lsm_tree.rs:52-71): pops ascending (key, timestamp, run index).
- Write an entry only if the next heap min has a different key -> last pop
  of an equal-key group wins (lsm_tree.rs:1041-1046).
- Drop tombstones (empty data) unless keep_tombstones (lsm_tree.rs:1045).
- Survivor bytes copied verbatim; index offsets recomputed from 0
  (entry_writer.rs:71-98).
"""
from __future__ import annotations

import heapq
import struct

import numpy as np

INDEX_DTYPE = np.dtype(
    [("offset", "<u8"), ("key_size", "<u4"), ("full_size", "<u4")]
)


def merge(runs, keep_tombstones: bool) -> tuple[bytes, bytes]:
    """runs: list of (data_bytes, index_bytes). Returns (data, index)."""
    views = []
    for data, index in runs:
        d = bytes(data)
        recs = np.frombuffer(bytes(index), dtype=INDEX_DTYPE)
        views.append((d, recs))

    def entry_at(r, i):
        d, recs = views[r]
        off = int(recs["offset"][i])
        ks = int(recs["key_size"][i])
        fs = int(recs["full_size"][i])
        raw = d[off : off + fs]
        key = raw[8 : ks]  # key_size = 8 + key_len
        (dlen,) = struct.unpack_from("<Q", raw, ks)
        ts = int.from_bytes(raw[fs - 16 : fs], "little", signed=True)
        return key, ts, dlen, raw

    heap = []
    for r in range(len(views)):
        if len(views[r][1]):
            key, ts, dlen, raw = entry_at(r, 0)
            # heap key replicates CompactionItem: ascending (key, ts, run)
            heapq.heappush(heap, (key, ts, r, 0, dlen, raw))

    out_data = []
    out_index = []
    off = 0
    while heap:
        key, ts, r, i, dlen, raw = heapq.heappop(heap)
        write = True
        if heap:
            write = heap[0][0] != key
        if write and not keep_tombstones and dlen == 0:
            write = False
        if write:
            out_index.append(struct.pack("<QII", off, 8 + len(key), len(raw)))
            out_data.append(raw)
            off += len(raw)
        if i + 1 < len(views[r][1]):
            k2, t2, dl2, raw2 = entry_at(r, i + 1)
            heapq.heappush(heap, (k2, t2, r, i + 1, dl2, raw2))
    return b"".join(out_data), b"".join(out_index)
