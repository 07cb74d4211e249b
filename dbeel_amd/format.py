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
    M = np.zeros((N, maxw), dtype=np.uint8)

    # key_len:u64 LE
    M[:, 0] = K & 0xFF
    for j in range(1, 8):
        M[:, j] = (K >> (8 * j)) & 0xFF
    # key bytes
    M[:, 8 : 8 + K] = keys
    # data_len:u64 LE at col 8+K
    dl = value_sizes
    for j in range(8):
        M[:, 8 + K + j] = ((dl >> np.uint64(8 * j)) & np.uint64(0xFF)).astype(np.uint8)
    # values + timestamp: two classes by tombstone-ness
    is_tomb = value_sizes == 0
    norm = ~is_tomb
    if V:
        M[norm, 16 + K : 16 + K + V] = value_fill.reshape(-1, V)
    ts = timestamps_lo.astype(np.uint64)
    # timestamp i128 LE: low u64 then high u64 (= 0 here)
    for j in range(8):
        tb = ((ts >> np.uint64(8 * j)) & np.uint64(0xFF)).astype(np.uint8)
        if V:
            M[norm, 16 + K + V + j] = tb[norm]
        M[is_tomb, 16 + K + j] = tb[is_tomb]
    # (high 8 bytes already zero)

    mask = np.arange(maxw, dtype=np.uint64)[None, :] < full[:, None]
    data = M[mask]

    offsets = np.zeros(N, dtype=np.uint64)
    np.cumsum(full[:-1], out=offsets[1:])
    idx = np.zeros(N, dtype=INDEX_DTYPE)
    idx["offset"] = offsets
    idx["key_size"] = 8 + K
    idx["full_size"] = full
    return data, idx.view(np.uint8).reshape(-1)
