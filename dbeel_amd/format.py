"""On-disk SSTable format helpers (bincode fixint LE restatement).

Wire format (reference cites):
  entry  = key_len:u64 | key | data_len:u64 | data | timestamp:i128
           (mod.rs:68-73 Entry{key,value}, EntryValue{data,timestamp};
            bincode fixint LE — utils/bincode.rs:9-16; timestamp serialized
            as unix nanos i128 — utils/timestamp_nanos.rs:6-11)
  index  = offset:u64 | key_size:u32 | full_size:u32  (16 B — mod.rs:33,45-50)
  key_size = 8 + len(key); full_size = 32 + len(key) + len(data)

These helpers are shared by tests, the synthetic-run generator and the
golden-fixture scripts. They are NOT the product path (that is
libdbeel_gpu.so); pure-Python encode/decode is used only at test scale.
"""
from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Iterable

import numpy as np

INDEX_ENTRY_SIZE = 16
ENTRY_OVERHEAD = 32  # 8 (key_len) + 8 (data_len) + 16 (timestamp)
TOMBSTONE = b""

INDEX_DTYPE = np.dtype(
    [("offset", "<u8"), ("key_size", "<u4"), ("full_size", "<u4")]
)


@dataclass
class Entry:
    key: bytes
    data: bytes
    timestamp: int  # i128 unix nanos

    @property
    def is_tombstone(self) -> bool:
        return len(self.data) == 0


def encode_entry(e: Entry) -> bytes:
    ts = int(e.timestamp)
    return (
        struct.pack("<Q", len(e.key))
        + e.key
        + struct.pack("<Q", len(e.data))
        + e.data
        + ts.to_bytes(16, "little", signed=True)
    )


def decode_entry(buf: bytes | memoryview) -> Entry:
    mv = memoryview(buf)
    (klen,) = struct.unpack_from("<Q", mv, 0)
    key = bytes(mv[8 : 8 + klen])
    (dlen,) = struct.unpack_from("<Q", mv, 8 + klen)
    data = bytes(mv[16 + klen : 16 + klen + dlen])
    ts = int.from_bytes(mv[16 + klen + dlen : 32 + klen + dlen], "little", signed=True)
    if 32 + klen + dlen != len(mv):
        raise ValueError("trailing bytes in entry (bincode RejectTrailing)")
    return Entry(key, data, ts)


def build_run(entries: Iterable[Entry]) -> tuple[bytes, bytes]:
    """Encode entries (already in key order) into (data_bytes, index_bytes).

    Slow generic path — test scale only.
    """
    data_parts: list[bytes] = []
    idx_parts: list[bytes] = []
    off = 0
    for e in entries:
        enc = encode_entry(e)
        idx_parts.append(
            struct.pack("<QII", off, 8 + len(e.key), len(enc))
        )
        data_parts.append(enc)
        off += len(enc)
    return b"".join(data_parts), b"".join(idx_parts)


def parse_run(data: bytes, index: bytes) -> list[Entry]:
    recs = np.frombuffer(index, dtype=INDEX_DTYPE)
    out = []
    mv = memoryview(data)
    for off, ks, fs in recs:
        out.append(decode_entry(mv[int(off) : int(off) + int(fs)]))
    return out


def build_run_fixed_key(
    keys: np.ndarray,
    value_sizes: np.ndarray,
    value_fill: np.ndarray,
    timestamps_lo: np.ndarray,
) -> tuple[np.ndarray, np.ndarray]:
    """Vectorized run builder for fixed-width keys and two-class values.

    keys:          (N, K) u8, already sorted ascending (unique within run).
    value_sizes:   (N,) u64 — per-entry data_len (0 = tombstone). All nonzero
                   sizes must be equal (V).
    value_fill:    (sum(value_sizes),) u8 random bytes for the value payloads.
    timestamps_lo: (N,) u64 — timestamp nanos (high 64 bits zero).

    Returns (data u8 array, index u8 array). Uses the "fixed max-width matrix
    then boolean-compact" trick so generation stays numpy-speed at GiB scale.
    """
    N, K = keys.shape
    value_sizes = value_sizes.astype(np.uint64)
    nz = value_sizes[value_sizes != 0]
    V = int(nz[0]) if nz.size else 0
    if nz.size and not np.all(nz == V):
        raise ValueError("all non-tombstone values must share one size")

    full = (ENTRY_OVERHEAD + K + value_sizes).astype(np.uint64)
    maxw = ENTRY_OVERHEAD + K + V
    uniform = bool(np.all(value_sizes == V))
    if uniform:
        # no tombstones: every entry is maxw wide, so the (N, maxw)
        # matrix IS the data file — skip the boolean-mask compact (the
        # dominant cost at GiB scale)
        flat = np.zeros(N * maxw, dtype=np.uint8)
        M = flat.reshape(N, maxw)
    else:
        M = np.zeros((N, maxw), dtype=np.uint8)

    # key_len:u64 LE (constant row prefix)
    M[:, 0:8] = np.frombuffer(
        int(K).to_bytes(8, "little"), dtype=np.uint8
    )[None, :]
    # key bytes
    M[:, 8 : 8 + K] = keys
    # data_len:u64 LE at col 8+K (V everywhere, 0 on tombstone rows)
    is_tomb = value_sizes == 0
    it = np.flatnonzero(is_tomb)
    M[:, 8 + K : 16 + K] = value_sizes.astype("<u8").view(
        np.uint8).reshape(N, 8)
    ts_bytes = timestamps_lo.astype("<u8").view(np.uint8).reshape(N, 8)
    if V:
        # values for non-tombstone rows; timestamp written unconditionally
        # at the full-width position (tombstone rows' copy lands past
        # their entry end and is masked out of the compact below)
        inorm = np.flatnonzero(~is_tomb)
        M[inorm, 16 + K : 16 + K + V] = value_fill.reshape(-1, V)
        M[:, 16 + K + V : 24 + K + V] = ts_bytes
    if it.size or not V:
        rows = it if V else np.arange(N)
        M[rows, 16 + K : 24 + K] = ts_bytes[rows]
    # (timestamp high 8 bytes already zero)

    if uniform:
        data = flat
    else:
        # compact out the unused tail of tombstone rows by concatenating
        # contiguous full-width segments + truncated tombstone rows —
        # one copy pass, no (N, maxw) boolean mask (which alone costs
        # ~maxw bytes/entry of extra memory and two passes)
        pieces = []
        prev = 0
        for t in it:
            if t > prev:
                pieces.append(M[prev:t].reshape(-1))
            pieces.append(M[t, : ENTRY_OVERHEAD + K])
            prev = t + 1
        if prev < N:
            pieces.append(M[prev:].reshape(-1))
        data = np.concatenate(pieces) if pieces else M[:0].reshape(-1)

    offsets = np.zeros(N, dtype=np.uint64)
    np.cumsum(full[:-1], out=offsets[1:])
    idx = np.zeros(N, dtype=INDEX_DTYPE)
    idx["offset"] = offsets
    idx["key_size"] = 8 + K
    idx["full_size"] = full
    return data, idx.view(np.uint8).reshape(-1)
