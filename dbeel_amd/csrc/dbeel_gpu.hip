/* dbeel_gpu.hip — MI355X-native (gfx950/CDNA4) SSTable compaction engine.
 *
 * Implements include/dbeel_gpu.h: the drop-in replacement for dbeel's
 * LSMTree::compact hot loop (reference lsm_tree.rs:950-1156; semantics
 * restated in the header). This is a from-scratch GPU design, not a port of
 * the reference's heap loop:
 *
 *   k_prepare    : validates every entry (bounds + bincode field
 *                  cross-check + EntryWriter density; corrupt input
 *                  errors loudly where the reference silently truncates)
 *                  and extracts the dense key-prefix (8 B) and tiered
 *                  suffix-aux arrays (16/48/64 B: key bytes 8.., klen,
 *                  staged i128 ts in the duplicate-prone tiers) the
 *                  merge runs on. Rangeable for streamed ingest.
 *   k_corank     : block-cooperative merge-path co-ranking — for every
 *                  run pair WITHIN a job, diagonal-partitioned
 *                  4096-position windows; blocks stage both prefix
 *                  segments into LDS coalesced and walk CORANK_STEPS
 *                  merged positions per thread, recording crossranks
 *                  (order = key bytes asc, timestamp i128 asc, run index
 *                  asc — lsm_tree.rs:52-71 + mod.rs:75-81) and
 *                  newest-wins supersession flags (lsm_tree.rs:1041-1044).
 *   k_rankreduce : crossranks -> job-local rank (+ job entry base);
 *                  winner/tombstone rules; strict-sortedness check
 *                  (flush invariant, lsm_tree.rs:925-946); one 16-B
 *                  rank-indexed record (keep flag in src bit 63).
 *   scans        : two rocPRIM exclusive scans (transform iterators,
 *                  plain u64/u32 + plus to stay on the decoupled-
 *                  lookback fast path) -> survivor byte offsets +
 *                  positions.
 *   k_emit       : output .index records (offset/key_size/full_size,
 *                  entry_writer.rs:79-87) + compacted source map.
 *   k_winmap/k_copy : verbatim survivor copy, balanced by DESTINATION
 *                  granule (16-B granules per lane, window/grid geometry
 *                  picked per survivor size from interleaved A/Bs, LDS
 *                  granule->entry map) so throughput is independent of
 *                  entry size; full windows gather granule values into
 *                  registers before storing; aligned non-temporal 16-B
 *                  stores, unaligned 16-B loads, boundary granules via
 *                  16-B two-load blends.
 *   k_scanflag   : the AsyncIter migration scan filter (murmur3 hash
 *                  ranges + key bounds — lsm_tree.rs:141-282,
 *                  tasks/migration.rs:54-60).
 *   k_encode_*   : the memtable-flush run encoder (lsm_tree.rs:925-946).
 *   host side    : resident jobs, batched independent jobs (one launch
 *                  set, per-job output slicing), streamed pinned ingest
 *                  with copy/compute overlap, sliced compaction for
 *                  inputs larger than HBM, device bloom build.
 *
 * All integer/byte work, HBM-bandwidth bound; MFMA unused by design
 * (BASELINE.json north_star).
 *
 * Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC dbeel_gpu.hip
 *        -o libdbeel_gpu.so
 */
#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <new>
#include <vector>

#include <hip/hip_runtime.h>
#include <rocprim/rocprim.hpp>

#include "../../include/dbeel_gpu.h"

#define MAX_RUNS 64

static thread_local char g_err[512];

static void set_err(const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(g_err, sizeof g_err, fmt, ap);
    va_end(ap);
}

extern "C" const char* dbeel_gpu_last_error(void) { return g_err; }

#define HIP_CHECK(call)                                                     \
    do {                                                                    \
        hipError_t _e = (call);                                             \
        if (_e != hipSuccess) {                                             \
            set_err("%s failed: %s (%s:%d)", #call, hipGetErrorString(_e),  \
                    __FILE__, __LINE__);                                    \
            return (_e == hipErrorOutOfMemory) ? DBEEL_ERR_OOM              \
                                               : DBEEL_ERR_HIP;             \
        }                                                                   \
    } while (0)

/* ------------------------------------------------------------------ */
/* Device-side format view                                            */
/* ------------------------------------------------------------------ */

struct RunsDesc {
    const uint8_t* data[MAX_RUNS];
    const uint8_t* index[MAX_RUNS];
    uint64_t data_len[MAX_RUNS];
    uint64_t count[MAX_RUNS];       /* entries per run                  */
    uint64_t entry_base[MAX_RUNS];  /* exclusive prefix sum of count    */
    uint8_t job_lo[MAX_RUNS];       /* run-range [lo, hi) of the        */
    uint8_t job_hi[MAX_RUNS];       /* INDEPENDENT job owning run r     */
    int n_runs;
    uint64_t total;                 /* sum of count                     */
};

struct EView {
    const uint8_t* raw;
    const uint8_t* key;
    uint64_t klen;
    uint64_t off;
    uint32_t key_size, full_size;
};

__device__ __forceinline__ uint64_t ld_u64(const uint8_t* p) {
    uint64_t v;
    __builtin_memcpy(&v, p, 8);
    return v;
}
__device__ __forceinline__ uint32_t ld_u32(const uint8_t* p) {
    uint32_t v;
    __builtin_memcpy(&v, p, 4);
    return v;
}

/* Loads index record + key view. Returns false on corrupt record. */
__device__ __forceinline__ bool load_entry(const RunsDesc& R, int r,
                                           uint64_t i, EView& e) {
    const uint8_t* rec = R.index[r] + i * 16;
    e.off = ld_u64(rec);
    e.key_size = ld_u32(rec + 8);
    e.full_size = ld_u32(rec + 12);
    if (e.key_size < 8 || e.full_size < 32 ||
        (uint64_t)e.key_size + 24 > e.full_size ||
        e.off + e.full_size > R.data_len[r])
        return false;
    e.raw = R.data[r] + e.off;
    e.klen = e.key_size - 8;
    e.key = e.raw + 8;
    return true;
}

/* Lexicographic compare of key byte strings (Vec<u8> cmp, mod.rs:75-81). */
__device__ __forceinline__ int cmp_keys(const uint8_t* a, uint64_t la,
                                        const uint8_t* b, uint64_t lb) {
    uint64_t n = la < lb ? la : lb;
    uint64_t i = 0;
    for (; i + 8 <= n; i += 8) {
        uint64_t va = ld_u64(a + i), vb = ld_u64(b + i);
        if (va != vb) {
            va = __builtin_bswap64(va);
            vb = __builtin_bswap64(vb);
            return va < vb ? -1 : 1;
        }
    }
    if (i < n) {
        uint64_t rem = n - i;
        uint64_t va = 0, vb = 0;
        for (uint64_t j = 0; j < rem; j++) {  /* <= 7 bytes */
            va |= (uint64_t)a[i + j] << (8 * j);
            vb |= (uint64_t)b[i + j] << (8 * j);
        }
        if (va != vb) {
            va = __builtin_bswap64(va);
            vb = __builtin_bswap64(vb);
            return va < vb ? -1 : 1;
        }
    }
    return la < lb ? -1 : (la > lb ? 1 : 0);
}

/* Timestamp: trailing i128 LE (utils/timestamp_nanos.rs:6-11). */
__device__ __forceinline__ void load_ts(const EView& e, uint64_t& lo,
                                        int64_t& hi) {
    lo = ld_u64(e.raw + e.full_size - 16);
    int64_t h;
    __builtin_memcpy(&h, e.raw + e.full_size - 8, 8);
    hi = h;
}

/* Full total order: (key, timestamp, run index) — lsm_tree.rs:52-71. */
__device__ __forceinline__ int cmp_full(const EView& a, int ra,
                                        const EView& b, int rb) {
    int c = cmp_keys(a.key, a.klen, b.key, b.klen);
    if (c) return c;
    uint64_t alo, blo;
    int64_t ahi, bhi;
    load_ts(a, alo, ahi);
    load_ts(b, blo, bhi);
    if (ahi != bhi) return ahi < bhi ? -1 : 1;
    if (alo != blo) return alo < blo ? -1 : 1;
    return ra < rb ? -1 : (ra > rb ? 1 : 0);
}

/* ------------------------------------------------------------------ */
/* Kernels                                                            */
/* ------------------------------------------------------------------ */

/* err codes stored in the device flag */
#define DERR_CORRUPT 1u
#define DERR_UNSORTED 2u

/* Big-endian key prefix: the first min(8, klen) key bytes, zero-padded,
 * byteswapped so u64 comparison == lexicographic byte comparison of those
 * bytes. pfx(a) != pfx(b) implies sign(u64 cmp) == sign(key cmp); equality
 * requires the full comparator (length / tail bytes / timestamp). */
__device__ __forceinline__ uint64_t key_prefix(const uint8_t* key,
                                               uint64_t klen) {
    uint64_t v = 0;
    if (klen >= 8) {
        v = ld_u64(key);
    } else {
        for (uint64_t j = 0; j < klen; j++) v |= (uint64_t)key[j] << (8 * j);
    }
    return __builtin_bswap64(v);
}

/* Dense per-entry search record: the key's SUFFIX bytes (8..8+KB,
 * zero-padded — the first 8 live in the pfx array, never duplicated here)
 * plus the key length. Prefix-tied comparisons read THIS instead of the
 * scattered multi-GiB data blob. Timestamps are NOT staged: the (ts, run)
 * tie-break only fires on fully EQUAL keys (true cross-run duplicates),
 * which read the two entries' trailing i128 from the blob — trading a
 * rare scattered read for a dense 16-byte write saved on EVERY entry
 * (cfg3: 1.0 GB of prep traffic). The tier (KB = 12/28/60 -> 16/32/64-B
 * records) is picked per job from the max key_size in the uploaded
 * indexes; longer keys stay CORRECT in any tier (blob fallback when two
 * keys tie through 8+KB bytes), smaller tiers are purely a traffic
 * optimization. */
template <int KB, bool TS> struct AuxT;
template <int KB> struct AuxT<KB, false> {
    uint8_t key[KB]; /* key bytes 8..8+KB, zero-padded */
    uint32_t klen;
};
template <int KB> struct AuxT<KB, true> {
    uint8_t key[KB];
    uint32_t klen;
    uint32_t rsv;
    uint64_t ts_lo; /* timestamp i128 LE halves — staged for tiers where
                       cross-run duplicates are expected so the (ts, run)
                       tie-break stays dense instead of 2 scattered blob
                       lines per comparison (PMC r02: blob ts reads cost
                       corank +1.3 GB on cfg3) */
    int64_t ts_hi;
};
static_assert(sizeof(AuxT<12, false>) == 16, "tier-0 aux must be 16B");
static_assert(sizeof(AuxT<24, true>) == 48, "tier-1 aux must be 48B");
static_assert(sizeof(AuxT<40, true>) == 64, "tier-2 aux must be 64B");

/* Key-only compare of entries A=(rA,iA), B=(rB,iB) via suffix records.
 * PRECONDITION: pfx(A) == pfx(B) (every caller compares pfx first), so
 * the first min(8, klen) key bytes already agree and zero-padding in pfx
 * is consistent: equality there with different klen means one key is a
 * zero-extension of the other within 8 bytes, which the klen comparison
 * below orders correctly (a strict prefix sorts first). */
template <int KB, bool TS>
__device__ __forceinline__ int cmp_keys_sfx(const RunsDesc& R,
                                            const AuxT<KB, TS>* aux, int rA,
                                            uint64_t iA, int rB,
                                            uint64_t iB) {
    const AuxT<KB, TS>* a = aux + R.entry_base[rA] + iA;
    const AuxT<KB, TS>* b = aux + R.entry_base[rB] + iB;
    uint32_t la = a->klen, lb = b->klen;
    uint32_t n = la < lb ? la : lb;
    uint32_t ns = n > 8 ? n - 8 : 0; /* suffix bytes to compare */
    if (ns > KB) ns = KB;
    #pragma unroll
    for (uint32_t i = 0; i < KB; i += 8) {
        if (i >= ns) break;
        uint64_t va, vb;
        __builtin_memcpy(&va, a->key + i, 8);
        __builtin_memcpy(&vb, b->key + i, 8);
        if (i + 8 > ns) { /* mask the tail; reading into klen is masked
                             off (and pad bytes are zero anyway) */
            uint64_t mask = (~0ull) >> (8 * (i + 8 - ns));
            va &= mask;
            vb &= mask;
        }
        if (va != vb) {
            va = __builtin_bswap64(va);
            vb = __builtin_bswap64(vb);
            return va < vb ? -1 : 1;
        }
    }
    if (la > (uint32_t)(8 + KB) && lb > (uint32_t)(8 + KB)) {
        /* rare: long keys tied through the staged 8+KB bytes */
        EView ea, eb;
        if (!load_entry(R, rA, iA, ea) || !load_entry(R, rB, iB, eb))
            return -1; /* corrupt input flagged elsewhere; value discarded */
        return cmp_keys(ea.key, ea.klen, eb.key, eb.klen);
    }
    return la < lb ? -1 : (la > lb ? 1 : 0);
}

/* (timestamp, run index) tie-break from the blob — only reached on fully
 * equal keys (lsm_tree.rs:58-65 + mod.rs:75-81: ts i128 asc, then run
 * index asc). */
__device__ __forceinline__ int cmp_ts_run_blob(const RunsDesc& R, int rA,
                                               uint64_t iA, int rB,
                                               uint64_t iB) {
    EView a, b;
    if (!load_entry(R, rA, iA, a) || !load_entry(R, rB, iB, b))
        return rA < rB ? -1 : 1; /* corrupt flagged elsewhere; stable */
    uint64_t alo, blo;
    int64_t ahi, bhi;
    load_ts(a, alo, ahi);
    load_ts(b, blo, bhi);
    if (ahi != bhi) return ahi < bhi ? -1 : 1;
    if (alo != blo) return alo < blo ? -1 : 1;
    return rA < rB ? -1 : (rA > rB ? 1 : 0);
}

/* Full-order compare (key, timestamp, run index — lsm_tree.rs:52-71). */
template <int KB, bool TS>
__device__ __forceinline__ int cmp_sfx_full(const RunsDesc& R,
                                            const AuxT<KB, TS>* aux, int rA,
                                            uint64_t iA, int rB,
                                            uint64_t iB) {
    int c = cmp_keys_sfx<KB, TS>(R, aux, rA, iA, rB, iB);
    if (c) return c;
    if constexpr (TS) {
        const AuxT<KB, TS>* a = aux + R.entry_base[rA] + iA;
        const AuxT<KB, TS>* b = aux + R.entry_base[rB] + iB;
        if (a->ts_hi != b->ts_hi) return a->ts_hi < b->ts_hi ? -1 : 1;
        if (a->ts_lo != b->ts_lo) return a->ts_lo < b->ts_lo ? -1 : 1;
        return rA < rB ? -1 : (rA > rB ? 1 : 0);
    } else {
        return cmp_ts_run_blob(R, rA, iA, rB, iB);
    }
}

/* Validates every entry (bounds + bincode field cross-check,
 * read_next_entry lsm_tree.rs:1158-70) and extracts the dense key-prefix
 * and aux arrays the merge runs on. Run sortedness is checked in
 * k_rankreduce (on the dense pfx/aux arrays). */
template <int KB, bool TS>
__global__ void k_prepare(RunsDesc R, uint64_t g0, uint64_t g1,
                          uint64_t* pfx, AuxT<KB, TS>* aux, uint32_t* err) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = g0 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < g1; g += stride) {
        /* locate run (<= 64 runs: linear scan on cached desc) */
        int r = 0;
        while (r + 1 < R.n_runs && g >= R.entry_base[r + 1]) r++;
        uint64_t i = g - R.entry_base[r];
        EView e;
        if (!load_entry(R, r, i, e)) {
            atomicOr(err, DERR_CORRUPT);
            pfx[g] = 0;
            AuxT<KB, TS> z = {};
            aux[g] = z;
            continue;
        }
        pfx[g] = key_prefix(e.key, e.klen);
        AuxT<KB, TS> a;
        a.klen = (uint32_t)e.klen;
        if constexpr (TS) {
            a.rsv = 0;
            load_ts(e, a.ts_lo, a.ts_hi);
        }
        /* stage key SUFFIX bytes 8..8+KB, zero-padded. 8-byte chunked
         * copy; reading up to 7 bytes past the key is safe (the 8-byte
         * data_len field follows it inside the entry) and the over-read
         * is masked off */
        uint32_t nk = e.klen > 8 ? (uint32_t)e.klen - 8 : 0;
        if (nk > KB) nk = KB;
        if (nk >= KB) {
            /* full staged suffix (uniform-key workloads: cfg3's 32-B
             * keys fill the 24-B tier exactly) — no masking, straight
             * word copies (prepare measured 40% issue-stall-bound) */
            #pragma unroll
            for (uint32_t j = 0; j < KB; j += 8) {
                uint64_t v = ld_u64(e.key + 8 + j);
                uint32_t nb = (KB - j) < 8 ? (KB - j) : 8;
                __builtin_memcpy(a.key + j, &v, nb);
            }
        } else {
            #pragma unroll
            for (uint32_t j = 0; j < KB; j += 8) {
                uint64_t v = 0;
                if (j < nk) {
                    v = ld_u64(e.key + 8 + j);
                    if (j + 8 > nk)
                        v &= (~0ull) >> (8 * (j + 8 - nk));
                }
                uint32_t nb = (KB - j) < 8 ? (KB - j) : 8;
                __builtin_memcpy(a.key + j, &v, nb);
            }
        }
        aux[g] = a;
        /* bincode field cross-check */
        if (ld_u64(e.raw) != e.klen ||
            ld_u64(e.raw + 8 + e.klen) != (uint64_t)e.full_size - 32 - e.klen)
            atomicOr(err, DERR_CORRUPT);
        /* entries within a run must not overlap and offsets must be
         * monotone (EntryWriter appends — entry_writer.rs:71-98; the
         * streamed-ingest chunking relies on this invariant) */
        if (i + 1 < R.count[r]) {
            uint64_t noff = ld_u64(R.index[r] + (i + 1) * 16);
            if (e.off + e.full_size > noff)
                atomicOr(err, DERR_CORRUPT);
        }
    }
}

/* ------------------------------------------------------------------ */
/* Crossranks by block-cooperative merge-path co-ranking               */
/*                                                                     */
/* For every run pair (a, b), the merged sequence (full order: key,    */
/* timestamp, run index — a strict total order, so no merge ties) is   */
/* partitioned into CORANK_BLOCK_POS-wide windows by diagonal binary   */
/* search. Each 256-thread block stages the two pfx segments of its    */
/* window into LDS with coalesced loads, sub-partitions the window     */
/* CORANK_STEPS merged positions per thread (diagonal search in LDS),  */
/* walks linearly, recording for every consumed entry its crossrank    */
/* into the opposite run (= the opposite cursor) and whether the       */
/* opposite run's next entry carries the same key (then it is later    */
/* in the order and supersedes it — the newest-wins dedup of           */
/* lsm_tree.rs:1041-1046). Work is (k-1)*N sequential LDS compares;    */
/* global traffic is one coalesced pass over pfx plus rare aux reads   */
/* on prefix ties (duplicate keys).                                    */
/*                                                                     */
/* cr layout: column-major, cr[s * total + g] = crossrank of entry g   */
/* into the s-th OTHER run of g's run (s = r2 < r ? r2 : r2-1), with   */
/* bit 31 = superseded flag. Every slot is written exactly once.       */
/* ------------------------------------------------------------------ */

#define CORANK_BLOCK 256
#define CORANK_STEPS 8
#define CORANK_BLOCK_POS (CORANK_BLOCK * CORANK_STEPS) /* 4096 */
#define CR_LOSER 0x80000000u
#define CR_MASK 0x7FFFFFFFu

struct PairDesc {
    uint32_t a, b;
    uint64_t chunk_base; /* exclusive prefix sum of per-pair windows */
};

/* Window-corner diagonal searches, one THREAD per chunk — fully
 * parallel. Previously each corank block ran its start/end corner
 * searches on 2 threads while the other 254 waited at the barrier
 * (corank measured 83% wave-parked); precomputing corners removes the
 * serial section. d_corners[t] = the merged-path `ia` at chunk t's
 * starting diagonal; a chunk's END corner is the next chunk's start
 * (same pair) or the pair's (Na, Nb) terminus. */
template <int KB, bool TS>
__global__ void k_corners(RunsDesc R, const uint64_t* pfx,
                          const AuxT<KB, TS>* aux, const PairDesc* pairs,
                          uint32_t n_pairs, uint64_t total_chunks,
                          uint64_t* corners) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         t < total_chunks; t += stride) {
        uint32_t plo = 0, phi = n_pairs;
        while (plo < phi) {
            uint32_t mid = (plo + phi) >> 1;
            if (pairs[mid].chunk_base <= t)
                plo = mid + 1;
            else
                phi = mid;
        }
        const PairDesc P = pairs[plo - 1];
        const int a = (int)P.a, b = (int)P.b;
        const uint64_t Na = R.count[a], Nb = R.count[b];
        const uint64_t* gpa = pfx + R.entry_base[a];
        const uint64_t* gpb = pfx + R.entry_base[b];
        uint64_t diag = (t - P.chunk_base) * CORANK_BLOCK_POS;
        uint64_t slo = diag > Nb ? diag - Nb : 0;
        uint64_t shi = diag < Na ? diag : Na;
        while (slo < shi) {
            uint64_t mid = (slo + shi) >> 1;
            uint64_t pa = gpa[mid], pb = gpb[diag - mid - 1];
            int c = (pa != pb)
                        ? (pa < pb ? -1 : 1)
                        : cmp_sfx_full<KB, TS>(R, aux, a, mid, b,
                                               diag - mid - 1);
            if (c < 0)
                slo = mid + 1;
            else
                shi = mid;
        }
        corners[t] = slo;
    }
}

template <int KB, bool TS, int BLK = CORANK_BLOCK>
__global__ __launch_bounds__(BLK) void k_corank(
    RunsDesc R, const uint64_t* pfx, const AuxT<KB, TS>* aux,
    const PairDesc* pairs, uint32_t n_pairs, uint64_t total_chunks,
    const uint64_t* corners, uint32_t* cr) {
    constexpr int STEPS = CORANK_BLOCK_POS / BLK; /* positions/thread */
    __shared__ uint64_t s_pfx[CORANK_BLOCK_POS + 2];
    __shared__ uint32_t s_cr[CORANK_BLOCK_POS]; /* staged crossranks:
        [0..lenA) for run a, [lenA..lenA+lenB) for run b — each block's
        output ranges are contiguous, so results are staged here and
        written back coalesced (direct interleaved 4-B stores measured
        3x write amplification from line RMW thrash) */
    for (uint64_t t = blockIdx.x; t < total_chunks; t += gridDim.x) {
        /* largest pair with chunk_base <= t (uniform across the block) */
        uint32_t plo = 0, phi = n_pairs;
        while (plo < phi) {
            uint32_t mid = (plo + phi) >> 1;
            if (pairs[mid].chunk_base <= t)
                plo = mid + 1;
            else
                phi = mid;
        }
        /* next pair's base bounds this pair's chunk range */
        uint64_t pair_end_chunk =
            (plo < n_pairs) ? pairs[plo].chunk_base : total_chunks;
        const PairDesc P = pairs[plo - 1];
        const int a = (int)P.a, b = (int)P.b;
        const uint64_t Na = R.count[a], Nb = R.count[b];
        const uint64_t* gpa = pfx + R.entry_base[a];
        const uint64_t* gpb = pfx + R.entry_base[b];
        uint64_t diag0 = (t - P.chunk_base) * CORANK_BLOCK_POS;
        uint64_t diag1 = diag0 + CORANK_BLOCK_POS;
        if (diag1 > Na + Nb) diag1 = Na + Nb;

        /* window corners precomputed by k_corners */
        const uint64_t iaS = corners[t];
        const uint64_t ibS = diag0 - iaS;
        const uint64_t iaE = (t + 1 < pair_end_chunk) ? corners[t + 1] : Na;
        const uint64_t ibE = diag1 - iaE;
        const uint32_t lenA = (uint32_t)(iaE - iaS);
        const uint32_t lenB = (uint32_t)(ibE - ibS);

        /* stage both pfx segments (coalesced); sA = s_pfx[0..lenA),
         * sB = s_pfx[lenA..lenA+lenB) */
        for (uint32_t u = threadIdx.x; u < lenA; u += BLK)
            s_pfx[u] = gpa[iaS + u];
        for (uint32_t u = threadIdx.x; u < lenB; u += BLK)
            s_pfx[lenA + u] = gpb[ibS + u];
        __syncthreads();
        const uint64_t* sA = s_pfx;
        const uint64_t* sB = s_pfx + lenA;

        /* per-thread sub-window: local diagonal search inside LDS */
        uint32_t L = lenA + lenB;
        uint32_t d = threadIdx.x * STEPS;
        if (d < L) {
            uint32_t slo = d > lenB ? d - lenB : 0;
            uint32_t shi = d < lenA ? d : lenA;
            while (slo < shi) {
                uint32_t mid = (slo + shi) >> 1;
                uint64_t pa = sA[mid], pb = sB[d - mid - 1];
                int c = (pa != pb) ? (pa < pb ? -1 : 1)
                                   : cmp_sfx_full<KB, TS>(R, aux, a, iaS + mid,
                                                          b, ibS + d - mid - 1);
                if (c < 0)
                    slo = mid + 1;
                else
                    shi = mid;
            }
            uint32_t ja = slo, jb = d - slo;
            uint32_t dend = d + STEPS;
            if (dend > L) dend = L;

            for (uint32_t pos = d; pos < dend; pos++) {
                bool take_a;
                if (ja >= lenA)
                    take_a = false;
                else if (jb >= lenB)
                    take_a = true;
                else {
                    uint64_t pa = sA[ja], pb = sB[jb];
                    take_a = (pa != pb)
                                 ? (pa < pb)
                                 : (cmp_sfx_full<KB, TS>(R, aux, a, iaS + ja, b,
                                                         ibS + jb) < 0);
                }
                if (take_a) {
                    uint64_t gib = ibS + jb;
                    uint32_t v = (uint32_t)gib;
                    /* next b entry: staged, or first beyond the window */
                    if (gib < Nb) {
                        uint64_t pb =
                            (jb < lenB) ? sB[jb] : gpb[gib];
                        if (pb == sA[ja] &&
                            cmp_keys_sfx<KB, TS>(R, aux, a, iaS + ja, b,
                                                 gib) == 0)
                            v |= CR_LOSER;
                    }
                    s_cr[ja] = v;
                    ja++;
                } else {
                    uint64_t gia = iaS + ja;
                    uint32_t v = (uint32_t)gia;
                    if (gia < Na) {
                        uint64_t pa =
                            (ja < lenA) ? sA[ja] : gpa[gia];
                        if (pa == sB[jb] &&
                            cmp_keys_sfx<KB, TS>(R, aux, b, ibS + jb, a,
                                                 gia) == 0)
                            v |= CR_LOSER;
                    }
                    s_cr[lenA + jb] = v;
                    jb++;
                }
            }
        }
        __syncthreads();
        /* coalesced writeback of both contiguous output ranges.
         * Crossrank slots are JOB-local (a batched job's runs only pair
         * within their job) */
        {
            uint32_t al = (uint32_t)a - R.job_lo[a];
            uint32_t bl = (uint32_t)b - R.job_lo[b];
            uint32_t sa = bl < al ? bl : bl - 1; /* slot of b in a */
            uint32_t sb = al < bl ? al : al - 1; /* slot of a in b */
            uint32_t* cra = cr + (uint64_t)sa * R.total + R.entry_base[a] +
                            iaS;
            uint32_t* crb = cr + (uint64_t)sb * R.total + R.entry_base[b] +
                            ibS;
            for (uint32_t u = threadIdx.x; u < lenA; u += BLK)
                cra[u] = s_cr[u];
            for (uint32_t u = threadIdx.x; u < lenB; u += BLK)
                crb[u] = s_cr[lenA + u];
        }
        __syncthreads();
    }
}

/* Rank-indexed scratch record: one 16-B scattered write per entry.
 * src packs keep flag (bit 63) | run (bits 48..62) | data offset (48).
 * The survivor scan reads it through a transform iterator. */
struct RankRec {
    uint64_t src;
    uint32_t key_size; /* 8 + key_len (index record field) */
    uint32_t full_size;
};
static_assert(sizeof(RankRec) == 16, "rank record must be 16B");
#define RR_KEEP (1ull << 63)
#define RR_OFF_MASK 0xFFFFFFFFFFFFull

/* Reduce per-pair crossranks to the global rank, apply the winner /
 * tombstone rules, and emit the rank-indexed scratch records. Also checks
 * each run is strictly sorted by key (flush invariant,
 * lsm_tree.rs:925-946) on the dense pfx/aux arrays. */
template <int KB, bool TS>
__global__ void k_rankreduce(RunsDesc R, const uint64_t* pfx,
                             const AuxT<KB, TS>* aux, const uint32_t* cr,
                             RankRec* rrec, int keep_tombstones,
                             uint32_t* err) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < R.total; g += stride) {
        int r = 0;
        while (r + 1 < R.n_runs && g >= R.entry_base[r + 1]) r++;
        uint64_t i = g - R.entry_base[r];
        EView e;
        if (!load_entry(R, r, i, e)) {
            /* k_prepare has flagged this input; keep memory safe and park
             * the entry at its local slot (results will be discarded) */
            RankRec z = {0, 8, 32};
            rrec[g] = z;
            continue;
        }
        if (i + 1 < R.count[r]) {
            uint64_t p0 = pfx[g], p1 = pfx[g + 1];
            if (p0 > p1 ||
                (p0 == p1 &&
                 cmp_keys_sfx<KB, TS>(R, aux, r, i, r, i + 1) >= 0))
                atomicOr(err, DERR_UNSORTED);
        }

        /* rank is local to the entry's JOB (independent jobs share one
         * launch); the job's entry base turns it into the global rrec
         * slot, so job outputs concatenate in job order */
        uint64_t rank = i + R.entry_base[R.job_lo[r]];
        bool winner = true;
        int nslots = (int)R.job_hi[r] - (int)R.job_lo[r] - 1;
        for (int s = 0; s < nslots; s++) {
            uint32_t v = cr[(uint64_t)s * R.total + g];
            rank += v & CR_MASK;
            winner &= !(v & CR_LOSER);
        }
        uint64_t dlen = (uint64_t)e.full_size - 32 - e.klen;
        bool keep = winner && (keep_tombstones || dlen != 0);
        RankRec m;
        m.src = ((uint64_t)r << 48) | e.off | (keep ? RR_KEEP : 0);
        m.key_size = e.key_size;
        m.full_size = e.full_size;
        rrec[rank] = m;
    }
}

/* transform-iterator functors for the survivor scans (plain arithmetic
 * types + rocprim::plus keep rocPRIM on its decoupled-lookback fast
 * path — a fused custom-STRUCT scan measured 10x slower). When the job
 * fits the packing bounds (bytes < 2^38 = 256 GB, entries < 2^26 = 67M
 * — every BASELINE shape does), ONE u64 scan carries BOTH running sums:
 * survivor bytes in the low 38 bits, survivor count in the high 26
 * (sums stay in-field, so lane-wise u64 addition never carries across);
 * that halves the scan's passes over rrec. Larger jobs fall back to two
 * scans. */
#define PK_SHIFT 38
#define PK_MASK ((1ull << PK_SHIFT) - 1)
struct RankRecSize {
    __device__ uint64_t operator()(const RankRec& r) const {
        return (r.src & RR_KEEP) ? (uint64_t)r.full_size : 0ull;
    }
};
struct RankRecFlag {
    __device__ uint32_t operator()(const RankRec& r) const {
        return (r.src & RR_KEEP) ? 1u : 0u;
    }
};
struct RankRecPacked {
    __device__ uint64_t operator()(const RankRec& r) const {
        return (r.src & RR_KEEP)
                   ? ((uint64_t)r.full_size | (1ull << PK_SHIFT))
                   : 0ull;
    }
};

/* Output index records are the input format: offset u64 | key_size u32 |
 * full_size u32 (entry_writer.rs:79-87, offsets recomputed from 0).
 * src_map gets the ABSOLUTE device address of each survivor's bytes. */
__global__ void k_emit(RunsDesc R, const RankRec* rrec,
                       const uint64_t* dst_off, const uint32_t* pos,
                       uint64_t total, uint8_t* out_index,
                       uint64_t* src_map) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < total; g += stride) {
        RankRec m = rrec[g];
        if (!(m.src & RR_KEEP)) continue;
        uint64_t p = pos[g];
        uint8_t* rec = out_index + p * 16;
        uint64_t off = dst_off[g];
        __builtin_memcpy(rec, &off, 8);
        __builtin_memcpy(rec + 8, &m.key_size, 4);
        __builtin_memcpy(rec + 12, &m.full_size, 4);
        src_map[p] = (uint64_t)(R.data[(m.src >> 48) & 0x7FFF] +
                                (m.src & RR_OFF_MASK));
    }
}

/* k_emit for the packed single-scan mode: offsets and positions decode
 * from one u64 (low 38 bits = byte offset, high 26 = survivor index). */
__global__ void k_emit_packed(RunsDesc R, const RankRec* rrec,
                              const uint64_t* packed, uint64_t total,
                              uint8_t* out_index, uint64_t* src_map) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < total; g += stride) {
        RankRec m = rrec[g];
        if (!(m.src & RR_KEEP)) continue;
        uint64_t v = packed[g];
        uint64_t p = v >> PK_SHIFT;
        uint8_t* rec = out_index + p * 16;
        uint64_t off = v & PK_MASK;
        __builtin_memcpy(rec, &off, 8);
        __builtin_memcpy(rec + 8, &m.key_size, 4);
        __builtin_memcpy(rec + 12, &m.full_size, 4);
        src_map[p] = (uint64_t)(R.data[(m.src >> 48) & 0x7FFF] +
                                (m.src & RR_OFF_MASK));
    }
}

/* Balanced verbatim copy: a 16-KiB destination window per 256-thread
 * block, 4 x 16-B granules per thread (independent loads/stores for ILP;
 * granule passes are thread-contiguous so stores coalesce). Entries are
 * >= 32 B (validated), so a window intersects < 16384/32 + 2 entries and
 * SPAN=516 staged offsets always cover it; a 16-B granule straddles at
 * most one entry boundary. */
#define COPY_BLOCK 256
#define COPY_WINDOW 16384
#define COPY_SPAN 516

#define COPY_GRANULES (COPY_WINDOW / 16)
/* geometry variants for within-probe A/B (env DBEEL_COPY_VARIANT):
 * 0 = 256 threads x 16 KiB (default), 1 = 256 x 8 KiB,
 * 2 = 512 x 16 KiB, 3 = 512 x 32 KiB */

/* Per destination window, the largest survivor p with offset(p) <= window
 * start — one thread per window (parallel), consumed by k_copy. */
__global__ void k_winmap(const uint8_t* out_index, uint64_t n_surv,
                         uint64_t total_bytes, uint32_t win,
                         uint32_t* win_p0) {
    uint64_t n_windows = (total_bytes + win - 1) / win;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t w = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         w < n_windows; w += stride) {
        uint64_t wstart = (uint64_t)w * win;
        uint64_t lo = 0, hi = n_surv; /* first offset > wstart, minus 1 */
        while (lo < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (ld_u64(out_index + mid * 16) <= wstart)
                lo = mid + 1;
            else
                hi = mid;
        }
        win_p0[w] = (uint32_t)(lo - 1); /* offset(0)=0 -> lo >= 1 */
    }
}

template <int BLK, int WIN>
__global__ __launch_bounds__(BLK) void k_copy(
    const uint8_t* out_index, const uint64_t* src_map,
    const uint32_t* win_p0, uint64_t n_surv, uint64_t total_bytes,
    uint8_t* out_data) {
    constexpr int SPAN = WIN / 32 + 4;
    constexpr int GRAN = WIN / 16;
    __shared__ uint64_t s_off[SPAN + 1];
    __shared__ uint64_t s_src[SPAN];
    __shared__ uint16_t s_gid[GRAN]; /* granule -> entry slot */

    uint64_t n_windows = (total_bytes + WIN - 1) / WIN;
    for (uint64_t w = blockIdx.x; w < n_windows; w += gridDim.x) {
        uint64_t wstart = w * WIN;
        uint64_t p0 = win_p0[w];
        /* entries intersecting this window are p0 .. win_p0[w+1]; staging
         * only those (not the worst-case SPAN) saves ~30x index re-reads
         * on KiB-sized entries */
        uint64_t p_end = (w + 1 < n_windows) ? (uint64_t)win_p0[w + 1] + 1
                                             : n_surv;
        uint32_t cnt = (uint32_t)(p_end - p0);
        if (cnt > SPAN) cnt = SPAN;
        if (cnt > n_surv - p0) cnt = (uint32_t)(n_surv - p0);
        for (uint32_t u = threadIdx.x; u <= cnt; u += BLK) {
            uint64_t p = p0 + u;
            if (u == cnt)
                s_off[cnt] = (p < n_surv) ? ld_u64(out_index + p * 16)
                                          : total_bytes;
            else {
                s_off[u] = ld_u64(out_index + p * 16);
                s_src[u] = src_map[p];
            }
        }
        __syncthreads();
        /* fill the granule -> entry map: entry slot u owns granules whose
         * START byte lies in [s_off[u], s_off[u+1]). Two fills: an
         * entry-sweep when entries are small (each writes few granules),
         * a granule-parallel binary search when few LARGE entries cover
         * the window (the sweep would serialize hundreds of LDS writes
         * on a handful of threads — measured 2x on 4 KiB values). */
        if (cnt >= 128) {
            for (uint32_t u = threadIdx.x; u < cnt; u += BLK) {
                uint64_t b0 = s_off[u], b1 = s_off[u + 1];
                uint64_t g0 = (b0 <= wstart) ? 0 : ((b0 - wstart + 15) >> 4);
                uint64_t g1 = (b1 - wstart + 15) >> 4; /* exclusive */
                if (g1 > GRAN) g1 = GRAN;
                for (uint64_t g = g0; g < g1; g++) s_gid[g] = (uint16_t)u;
            }
        } else {
            #pragma unroll
            for (int q = 0; q < GRAN / BLK; q++) {
                uint32_t g = q * BLK + threadIdx.x;
                uint64_t gpos = wstart + (uint64_t)g * 16;
                uint32_t lo = 0, hi = cnt; /* largest u: s_off[u] <= gpos */
                while (lo < hi) {
                    uint32_t mid = (lo + hi) >> 1;
                    if (s_off[mid] <= gpos)
                        lo = mid + 1;
                    else
                        hi = mid;
                }
                s_gid[g] = (uint16_t)(lo ? lo - 1 : 0);
            }
        }
        __syncthreads();

        typedef unsigned int v4u __attribute__((ext_vector_type(4)));
        if (wstart + WIN <= total_bytes) {
            /* full interior window: every granule is a whole 16 B. Two
             * phases — gather all GRAN/BLK granule values into registers
             * first (independent load chains, keeps several loads in
             * flight per wave), then store. The single-phase form left
             * most waves parked on one load at a time (93% parked on the
             * 4 KiB-value shape, r01 profile). */
            v4u vv[GRAN / BLK];
            #pragma unroll
            for (int q = 0; q < GRAN / BLK; q++) {
                uint32_t gl = q * BLK + threadIdx.x; /* window-local */
                uint64_t gpos = wstart + (uint64_t)gl * 16;
                uint32_t j = s_gid[gl];
                uint64_t e_end = s_off[j + 1];
                const uint8_t* src = (const uint8_t*)s_src[j] +
                                     (gpos - s_off[j]);
                if (gpos + 16 <= e_end) {
                    /* plain (cached) loads: a nontemporal-load variant
                     * measured 15-18% SLOWER (copy 2.15 -> 2.52 ms on
                     * cfg3) — adjacent granules of the same entry share
                     * lines through L2 and NT hints forfeit that */
                    __builtin_memcpy(&vv[q], src, 16);
                } else {
                    /* one entry boundary inside the granule: 16-B blend
                     * of the entry tail and the next survivor's head.
                     * Byte loops here cost a whole wave ~3k serial
                     * cycles, and odd entry sizes put one straddle in
                     * nearly every wave (measured 2.4x on odd-size
                     * values). Reading past either entry stays inside
                     * the padded input slab. */
                    uint32_t c1 = (uint32_t)(e_end - gpos); /* 1..15 */
                    const uint8_t* src2 = (const uint8_t*)s_src[j + 1];
                    uint64_t l0, h0, l2, h2;
                    __builtin_memcpy(&l0, src, 8);
                    __builtin_memcpy(&h0, src + 8, 8);
                    __builtin_memcpy(&l2, src2, 8);
                    __builtin_memcpy(&h2, src2 + 8, 8);
                    uint64_t lo, hi;
                    if (c1 < 8) {
                        uint32_t sh = 8 * c1;
                        uint64_t m = (~0ull) >> (64 - sh);
                        lo = (l0 & m) | (l2 << sh);
                        hi = (h2 << sh) | (l2 >> (64 - sh));
                    } else {
                        uint32_t c = c1 - 8; /* 0..7 */
                        lo = l0;
                        if (c == 0) {
                            hi = l2;
                        } else {
                            uint64_t m = (~0ull) >> (64 - 8 * c);
                            hi = (h0 & m) | (l2 << (8 * c));
                        }
                    }
                    __builtin_memcpy(&vv[q], &lo, 8);
                    __builtin_memcpy(
                        reinterpret_cast<uint8_t*>(&vv[q]) + 8, &hi, 8);
                }
            }
            /* streamed once, never re-read: keep L2 for sources */
            #pragma unroll
            for (int q = 0; q < GRAN / BLK; q++) {
                uint32_t gl = q * BLK + threadIdx.x;
                __builtin_nontemporal_store(
                    vv[q], reinterpret_cast<v4u*>(out_data + wstart +
                                                  (uint64_t)gl * 16));
            }
        } else {
            /* the job's final (partial) window */
            #pragma unroll
            for (int q = 0; q < GRAN / BLK; q++) {
                uint32_t gl = q * BLK + threadIdx.x;
                uint64_t gpos = wstart + (uint64_t)gl * 16;
                if (gpos >= total_bytes) break;
                uint32_t j = s_gid[gl];
                uint64_t e_end = s_off[j + 1];
                const uint8_t* src = (const uint8_t*)s_src[j] +
                                     (gpos - s_off[j]);
                uint32_t nbytes = (uint32_t)(
                    (total_bytes - gpos) < 16 ? (total_bytes - gpos) : 16);
                uint8_t* dst = out_data + gpos;
                uint32_t c1 = (uint32_t)(gpos + nbytes <= e_end
                                             ? nbytes
                                             : e_end - gpos);
                for (uint32_t b = 0; b < c1; b++) dst[b] = src[b];
                if (c1 < nbytes) {
                    const uint8_t* src2 = (const uint8_t*)s_src[j + 1];
                    for (uint32_t b = c1; b < nbytes; b++)
                        dst[b] = src2[b - c1];
                }
            }
        }
        __syncthreads();
    }
}

/* ------------------------------------------------------------------ */
/* Run encoder (the memtable-flush path)                              */
/*                                                                    */
/* Encodes an already-sorted (key, value, timestamp) stream into a    */
/* run: the same bincode-fixint entry layout + 16-B index records as  */
/* flush_memtable_to_disk / EntryWriter (lsm_tree.rs:925-946,         */
/* entry_writer.rs:71-98). Wave-per-entry copy: lanes move 8-byte     */
/* chunks of key/value bytes; the header fields and timestamp are     */
/* written by lane 0.                                                 */
/* ------------------------------------------------------------------ */

static uint32_t pick_grid(uint64_t work_items, uint32_t block);

__global__ void k_encode_sizes(uint64_t n, const uint64_t* key_off,
                               const uint64_t* val_off, uint64_t* sizes) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        sizes[i] = 32 + (key_off[i + 1] - key_off[i]) +
                   (val_off[i + 1] - val_off[i]);
    }
}

__global__ void k_encode_index(uint64_t n, const uint64_t* key_off,
                               const uint64_t* dst_off,
                               const uint64_t* sizes, uint8_t* out_index) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        uint8_t* rec = out_index + i * 16;
        uint64_t off = dst_off[i];
        uint32_t key_size = (uint32_t)(8 + key_off[i + 1] - key_off[i]);
        uint32_t full_size = (uint32_t)sizes[i];
        __builtin_memcpy(rec, &off, 8);
        __builtin_memcpy(rec + 8, &key_size, 4);
        __builtin_memcpy(rec + 12, &full_size, 4);
    }
}

__device__ __forceinline__ void wave_copy_bytes(uint8_t* dst,
                                                const uint8_t* src,
                                                uint64_t n, uint32_t lane) {
    /* 8-byte chunks per lane, byte tail by lane 0 */
    uint64_t chunks = n >> 3;
    for (uint64_t c = lane; c < chunks; c += 64) {
        uint64_t v;
        __builtin_memcpy(&v, src + c * 8, 8);
        __builtin_memcpy(dst + c * 8, &v, 8);
    }
    if (lane == 0)
        for (uint64_t b = chunks * 8; b < n; b++) dst[b] = src[b];
}

__global__ void k_encode_data(uint64_t n, const uint8_t* keys,
                              const uint64_t* key_off, const uint8_t* vals,
                              const uint64_t* val_off, const uint8_t* ts,
                              const uint64_t* dst_off, uint8_t* out_data) {
    uint32_t lane = threadIdx.x & 63;
    uint64_t wave = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    uint64_t n_waves = ((uint64_t)gridDim.x * blockDim.x) >> 6;
    for (uint64_t i = wave; i < n; i += n_waves) {
        uint64_t klen = key_off[i + 1] - key_off[i];
        uint64_t vlen = val_off[i + 1] - val_off[i];
        uint8_t* dst = out_data + dst_off[i];
        if (lane == 0) {
            __builtin_memcpy(dst, &klen, 8);
            __builtin_memcpy(dst + 8 + klen, &vlen, 8);
            __builtin_memcpy(dst + 16 + klen + vlen, ts + i * 16, 16);
        }
        wave_copy_bytes(dst + 8, keys + key_off[i], klen, lane);
        wave_copy_bytes(dst + 16 + klen, vals + val_off[i], vlen, lane);
    }
}

extern "C" int dbeel_gpu_encode_run(uint64_t n_entries, const uint8_t* keys,
                                    const uint64_t* key_offsets,
                                    const uint8_t* values,
                                    const uint64_t* value_offsets,
                                    const uint8_t* timestamps, int device,
                                    dbeel_compact_result* out) {
    g_err[0] = 0;
    if (!out || (n_entries && (!keys || !key_offsets || !values ||
                               !value_offsets || !timestamps))) {
        set_err("encode_run: null argument");
        return DBEEL_ERR_INVALID_ARG;
    }
    memset(out, 0, sizeof *out);
    if (device < 0) {
        set_err("device must be >= 0 (no CPU fallback)");
        return DBEEL_ERR_INVALID_ARG;
    }
    int ndev = 0;
    hipError_t de = hipGetDeviceCount(&ndev);
    if (de != hipSuccess || device >= ndev) {
        set_err("no usable HIP device %d", device);
        return DBEEL_ERR_NO_GPU;
    }
    HIP_CHECK(hipSetDevice(device));
    if (n_entries == 0) {
        out->data = (uint8_t*)malloc(1);
        out->index = (uint8_t*)malloc(1);
        return DBEEL_OK;
    }
    uint64_t n = n_entries;
    uint64_t kbytes = key_offsets[n], vbytes = value_offsets[n];
    for (uint64_t i = 0; i < n; i++) {
        uint64_t klen = key_offsets[i + 1] - key_offsets[i];
        uint64_t vlen = value_offsets[i + 1] - value_offsets[i];
        if (key_offsets[i + 1] < key_offsets[i] ||
            value_offsets[i + 1] < value_offsets[i] ||
            32 + klen + vlen > 0xFFFFFFFFull) {
            set_err("entry %llu too large or offsets not monotone",
                    (unsigned long long)i);
            return DBEEL_ERR_ITEM_TOO_LARGE; /* error.rs:60-61 analogue */
        }
    }

    uint8_t *d_keys = nullptr, *d_vals = nullptr, *d_ts = nullptr,
            *d_outd = nullptr, *d_outi = nullptr;
    uint64_t *d_koff = nullptr, *d_voff = nullptr, *d_sizes = nullptr,
             *d_doff = nullptr;
    void* d_tmp = nullptr;
    hipStream_t s = nullptr;
    uint64_t total = 0;
    int rc = DBEEL_OK;
    size_t tmp_bytes = 0;

#define ENC_CHECK(call)                                                     \
    do {                                                                    \
        hipError_t _e = (call);                                             \
        if (_e != hipSuccess) {                                             \
            set_err("%s failed: %s", #call, hipGetErrorString(_e));         \
            rc = (_e == hipErrorOutOfMemory) ? DBEEL_ERR_OOM                \
                                             : DBEEL_ERR_HIP;               \
            goto done;                                                      \
        }                                                                   \
    } while (0)

    ENC_CHECK(hipStreamCreate(&s));
    ENC_CHECK(hipMalloc(&d_keys, kbytes ? kbytes : 1));
    ENC_CHECK(hipMalloc(&d_vals, vbytes ? vbytes : 1));
    ENC_CHECK(hipMalloc(&d_ts, n * 16));
    ENC_CHECK(hipMalloc(&d_koff, (n + 1) * 8));
    ENC_CHECK(hipMalloc(&d_voff, (n + 1) * 8));
    ENC_CHECK(hipMalloc(&d_sizes, n * 8));
    ENC_CHECK(hipMalloc(&d_doff, n * 8));
    ENC_CHECK(hipMemcpyAsync(d_keys, keys, kbytes ? kbytes : 1,
                             hipMemcpyHostToDevice, s));
    ENC_CHECK(hipMemcpyAsync(d_vals, values, vbytes ? vbytes : 1,
                             hipMemcpyHostToDevice, s));
    ENC_CHECK(hipMemcpyAsync(d_ts, timestamps, n * 16,
                             hipMemcpyHostToDevice, s));
    ENC_CHECK(hipMemcpyAsync(d_koff, key_offsets, (n + 1) * 8,
                             hipMemcpyHostToDevice, s));
    ENC_CHECK(hipMemcpyAsync(d_voff, value_offsets, (n + 1) * 8,
                             hipMemcpyHostToDevice, s));
    hipLaunchKernelGGL(k_encode_sizes, dim3(pick_grid(n, 256)), dim3(256), 0,
                       s, n, d_koff, d_voff, d_sizes);
    (void)rocprim::exclusive_scan(nullptr, tmp_bytes, d_sizes, d_doff,
                                  (uint64_t)0, n, rocprim::plus<uint64_t>(),
                                  s);
    ENC_CHECK(hipMalloc(&d_tmp, tmp_bytes));
    (void)rocprim::exclusive_scan(d_tmp, tmp_bytes, d_sizes, d_doff,
                                  (uint64_t)0, n, rocprim::plus<uint64_t>(),
                                  s);
    total = 32 * n + kbytes + vbytes; /* 32+klen+vlen per entry */
    ENC_CHECK(hipMalloc(&d_outd, total));
    ENC_CHECK(hipMalloc(&d_outi, n * 16));
    hipLaunchKernelGGL(k_encode_index, dim3(pick_grid(n, 256)), dim3(256), 0,
                       s, n, d_koff, d_doff, d_sizes, d_outi);
    hipLaunchKernelGGL(k_encode_data, dim3(pick_grid(n * 64, 256)), dim3(256),
                       0, s, n, d_keys, d_koff, d_vals, d_voff, d_ts, d_doff,
                       d_outd);
    ENC_CHECK(hipStreamSynchronize(s));
    ENC_CHECK(hipGetLastError());

    out->data = (uint8_t*)malloc(total ? total : 1);
    out->index = (uint8_t*)malloc(n * 16);
    if (!out->data || !out->index) {
        free(out->data);
        free(out->index);
        memset(out, 0, sizeof *out);
        rc = DBEEL_ERR_OOM;
        goto done;
    }
    ENC_CHECK(hipMemcpyAsync(out->data, d_outd, total,
                             hipMemcpyDeviceToHost, s));
    ENC_CHECK(hipMemcpyAsync(out->index, d_outi, n * 16,
                             hipMemcpyDeviceToHost, s));
    ENC_CHECK(hipStreamSynchronize(s));
    out->data_len = total;
    out->index_len = n * 16;
    out->entries_written = n;
done:
    hipFree(d_keys);
    hipFree(d_vals);
    hipFree(d_ts);
    hipFree(d_koff);
    hipFree(d_voff);
    hipFree(d_sizes);
    hipFree(d_doff);
    hipFree(d_tmp);
    hipFree(d_outd);
    hipFree(d_outi);
    if (s) hipStreamDestroy(s);
    if (rc != DBEEL_OK && out->data) {
        free(out->data);
        free(out->index);
        memset(out, 0, sizeof *out);
    }
    return rc;
#undef ENC_CHECK
}

/* ------------------------------------------------------------------ */
/* Host side                                                          */
/* ------------------------------------------------------------------ */

static inline uint64_t ld_u64_host(const uint8_t* p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v;
}

/* Max key_size across all uploaded index slabs — picks the aux tier. */
__global__ void k_maxks(RunsDesc R, uint32_t* out) {
    __shared__ uint32_t smax;
    if (threadIdx.x == 0) smax = 0;
    __syncthreads();
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint32_t m = 0;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < R.total; g += stride) {
        int r = 0;
        while (r + 1 < R.n_runs && g >= R.entry_base[r + 1]) r++;
        uint64_t i = g - R.entry_base[r];
        uint32_t ks = ld_u32(R.index[r] + i * 16 + 8);
        if (ks > m) m = ks;
    }
    atomicMax(&smax, m);
    __syncthreads();
    if (threadIdx.x == 0) atomicMax(out, smax);
}

struct dbeel_gpu_job {
    int device = -1;
    hipStream_t stream = nullptr;
    hipStream_t copy_stream = nullptr; /* streamed-ingest H2D lane */
    hipEvent_t ev[8] = {};
    int prep_valid = 0; /* set by job_ingest: pfx/aux/d_err already built
                           for the current inputs (overlapped with the
                           transfer); job_run skips the prepare stage.
                           job_create leaves it 0 so the resident
                           measurement mode always runs the full
                           pipeline per step. */
    RunsDesc desc{};
    uint8_t* d_input = nullptr; /* one slab: all run data+index            */
    RankRec* d_rank = nullptr;
    uint64_t* d_dstoff = nullptr;
    uint32_t* d_pos = nullptr;
    int aux_kind = 2;           /* 0: 16-B, 1: 32-B, 2: 64-B aux records   */
    uint8_t* d_outindex = nullptr;
    uint64_t* d_srcmap = nullptr;
    uint8_t* d_outdata = nullptr;
    void* d_scantmp = nullptr;
    size_t scantmp_bytes = 0;
    uint32_t* d_err = nullptr;
    uint64_t* d_pfx = nullptr;   /* dense big-endian key prefixes       */
    void* d_aux = nullptr;       /* dense 64B search records            */
    uint32_t* d_cr = nullptr;    /* per-pair crossranks, column-major   */
    void* d_pairs = nullptr;     /* PairDesc table                      */
    uint32_t n_pairs = 0;
    uint64_t total_chunks = 0;
    uint32_t* d_winp0 = nullptr; /* copy window -> first survivor       */
    uint64_t* d_corners = nullptr; /* corank chunk start corners        */
    uint64_t total_entries = 0;
    uint64_t total_data_bytes = 0;
    uint64_t input_bytes = 0;
    std::vector<uint64_t> job_gbase; /* entry base per job (+ total) */
    /* last-run results */
    int last_scan_packed = 0; /* d_dstoff holds packed (pos<<38|off) */
    uint64_t out_data_len = 0;
    uint64_t out_entries = 0;
    bool have_result = false;
    double h2d_ms = 0.0;
};

static int validate_runs(const dbeel_run_view* runs, size_t n_runs) {
    if (!runs || n_runs == 0) {
        set_err("runs is null or empty");
        return DBEEL_ERR_INVALID_ARG;
    }
    if (n_runs > MAX_RUNS) {
        set_err("n_runs %zu exceeds MAX_RUNS %d", n_runs, MAX_RUNS);
        return DBEEL_ERR_INVALID_ARG;
    }
    for (size_t r = 0; r < n_runs; r++) {
        if (runs[r].index_len % 16) {
            set_err("run %zu: index_len %zu not a multiple of 16", r,
                    runs[r].index_len);
            return DBEEL_ERR_CORRUPT;
        }
        if ((runs[r].data_len && !runs[r].data) ||
            (runs[r].index_len && !runs[r].index)) {
            set_err("run %zu: null pointer with nonzero length", r);
            return DBEEL_ERR_INVALID_ARG;
        }
        if (runs[r].data_len >= (1ull << 48)) {
            set_err("run %zu: data_len too large", r);
            return DBEEL_ERR_ITEM_TOO_LARGE;
        }
        if (runs[r].index_len / 16 >= (1ull << 31)) {
            /* crossranks carry a 31-bit payload (CR_LOSER uses bit 31):
             * enforce the representation bound explicitly rather than by
             * incidental OOM */
            set_err("run %zu: 2^31 or more entries unsupported", r);
            return DBEEL_ERR_ITEM_TOO_LARGE;
        }
    }
    uint64_t total = 0;
    for (size_t r = 0; r < n_runs; r++) total += runs[r].index_len / 16;
    if (total >= (1ull << 32)) {
        /* survivor positions (d_pos scan, win_p0) are uint32 */
        set_err("job has 2^32 or more total entries");
        return DBEEL_ERR_ITEM_TOO_LARGE;
    }
    return DBEEL_OK;
}

static int job_create_impl(const dbeel_run_view* runs, size_t n_runs,
                           const uint32_t* runs_per_job, size_t n_jobs,
                           int device, dbeel_gpu_job** out_job) {
    g_err[0] = 0;
    if (!out_job) {
        set_err("out_job is null");
        return DBEEL_ERR_INVALID_ARG;
    }
    *out_job = nullptr;
    int rc = validate_runs(runs, n_runs);
    if (rc) return rc;
    {
        uint64_t sum = 0;
        for (size_t j = 0; j < n_jobs; j++) {
            if (!runs_per_job[j]) {
                set_err("job %zu has zero runs", j);
                return DBEEL_ERR_INVALID_ARG;
            }
            sum += runs_per_job[j];
        }
        if (sum != n_runs) {
            set_err("runs_per_job sums to %llu != n_runs %zu",
                    (unsigned long long)sum, n_runs);
            return DBEEL_ERR_INVALID_ARG;
        }
    }
    if (device < 0) {
        set_err("device must be >= 0 (no CPU fallback in the product "
                "library; the CPU restatement lives in oracle/liboracle.so "
                "and is test infrastructure only)");
        return DBEEL_ERR_INVALID_ARG;
    }
    int ndev = 0;
    hipError_t de = hipGetDeviceCount(&ndev);
    if (de != hipSuccess || device >= ndev) {
        set_err("no usable HIP device %d (count=%d, %s)", device, ndev,
                hipGetErrorString(de));
        return DBEEL_ERR_NO_GPU;
    }
    HIP_CHECK(hipSetDevice(device));

    dbeel_gpu_job* job = new (std::nothrow) dbeel_gpu_job();
    if (!job) return DBEEL_ERR_OOM;
    job->device = device;

    uint64_t input_bytes = 0, total = 0, total_data = 0;
    for (size_t r = 0; r < n_runs; r++) {
        input_bytes += runs[r].data_len + runs[r].index_len;
        total += runs[r].index_len / 16;
        total_data += runs[r].data_len;
    }
    job->total_entries = total;
    job->total_data_bytes = total_data;
    job->input_bytes = input_bytes;

#define JOB_CHECK(call)                                                     \
    do {                                                                    \
        hipError_t _e = (call);                                             \
        if (_e != hipSuccess) {                                             \
            set_err("%s failed: %s", #call, hipGetErrorString(_e));         \
            dbeel_gpu_job_destroy(job);                                     \
            return (_e == hipErrorOutOfMemory) ? DBEEL_ERR_OOM              \
                                               : DBEEL_ERR_HIP;             \
        }                                                                   \
    } while (0)

    JOB_CHECK(hipStreamCreate(&job->stream));
    for (int i = 0; i < 8; i++) JOB_CHECK(hipEventCreate(&job->ev[i]));

    /* +-16 B padding: the copy kernel's aligned-base loads may touch up
     * to 15 B before/after an entry's bytes */
    JOB_CHECK(hipMalloc(&job->d_input, (input_bytes ? input_bytes : 16) + 32));
    uint64_t n = total ? total : 1;
    JOB_CHECK(hipMalloc(&job->d_rank, n * sizeof(RankRec)));
    JOB_CHECK(hipMalloc(&job->d_dstoff, n * sizeof(uint64_t)));
    JOB_CHECK(hipMalloc(&job->d_pos, n * sizeof(uint32_t)));
    JOB_CHECK(hipMalloc(&job->d_outindex, n * 16));
    JOB_CHECK(hipMalloc(&job->d_srcmap, n * sizeof(uint64_t)));
    JOB_CHECK(hipMalloc(&job->d_outdata, total_data ? total_data : 16));
    JOB_CHECK(hipMalloc(&job->d_err, 2 * sizeof(uint32_t)));
    JOB_CHECK(hipMalloc(&job->d_pfx, n * sizeof(uint64_t)));
    uint32_t max_job_runs = 1;
    {
        for (size_t j = 0; j < n_jobs; j++)
            if (runs_per_job[j] > max_job_runs)
                max_job_runs = runs_per_job[j];
    }
    JOB_CHECK(hipMalloc(&job->d_cr,
                        (max_job_runs > 1 ? (max_job_runs - 1) * n : 1) *
                            4));
    /* sized for the smallest copy-window variant (8 KiB) */
    JOB_CHECK(hipMalloc(&job->d_winp0,
                        (total_data / 8192 + 2) * sizeof(uint32_t)));

    size_t t1 = 0, t2 = 0;
    rocprim::exclusive_scan(nullptr, t1,
                            rocprim::make_transform_iterator(job->d_rank,
                                                             RankRecSize{}),
                            job->d_dstoff, (uint64_t)0, n,
                            rocprim::plus<uint64_t>(), job->stream);
    rocprim::exclusive_scan(nullptr, t2,
                            rocprim::make_transform_iterator(job->d_rank,
                                                             RankRecFlag{}),
                            job->d_pos, (uint32_t)0, n,
                            rocprim::plus<uint32_t>(), job->stream);
    job->scantmp_bytes = t1 > t2 ? t1 : t2;
    JOB_CHECK(hipMalloc(&job->d_scantmp, job->scantmp_bytes));

    /* Upload: one slab; record per-run device pointers. */
    JOB_CHECK(hipEventRecord(job->ev[0], job->stream));
    uint8_t* p = job->d_input + 16;
    RunsDesc& D = job->desc;
    memset(&D, 0, sizeof D);
    D.n_runs = (int)n_runs;
    D.total = total;
    {
        uint32_t r0 = 0;
        for (size_t j = 0; j < n_jobs; j++) {
            for (uint32_t r = r0; r < r0 + runs_per_job[j]; r++) {
                D.job_lo[r] = (uint8_t)r0;
                D.job_hi[r] = (uint8_t)(r0 + runs_per_job[j]);
            }
            r0 += runs_per_job[j];
        }
    }
    uint64_t base = 0;
    for (size_t r = 0; r < n_runs; r++) {
        D.data[r] = p;
        D.data_len[r] = runs[r].data_len;
        if (runs[r].data_len) {
            JOB_CHECK(hipMemcpyAsync(p, runs[r].data, runs[r].data_len,
                                     hipMemcpyHostToDevice, job->stream));
            p += runs[r].data_len;
        }
        D.index[r] = p;
        D.count[r] = runs[r].index_len / 16;
        if (runs[r].index_len) {
            JOB_CHECK(hipMemcpyAsync(p, runs[r].index, runs[r].index_len,
                                     hipMemcpyHostToDevice, job->stream));
            p += runs[r].index_len;
        }
        D.entry_base[r] = base;
        base += D.count[r];
    }
    /* merge-path pair table: one entry per unordered run pair with work */
    {
        struct HostPair { uint32_t a, b; uint64_t chunk_base; };
        size_t max_pairs = n_runs * (n_runs - 1) / 2 + 1;
        HostPair* hp = (HostPair*)malloc(max_pairs * sizeof(HostPair));
        if (!hp) {
            dbeel_gpu_job_destroy(job);
            return DBEEL_ERR_OOM;
        }
        uint64_t cbase = 0;
        uint32_t np = 0;
        for (uint32_t a = 0; a < n_runs; a++)
            for (uint32_t b = a + 1; b < D.job_hi[a]; b++) {
                uint64_t len = D.count[a] + D.count[b];
                if (!len) continue;
                hp[np].a = a;
                hp[np].b = b;
                hp[np].chunk_base = cbase;
                cbase += (len + CORANK_BLOCK_POS - 1) / CORANK_BLOCK_POS;
                np++;
            }
        job->n_pairs = np;
        job->total_chunks = cbase;
        {
            hipError_t _e = hipMalloc(&job->d_corners,
                                      (cbase ? cbase : 1) * 8);
            if (_e != hipSuccess) {
                set_err("corner table alloc failed: %s",
                        hipGetErrorString(_e));
                free(hp);
                dbeel_gpu_job_destroy(job);
                return DBEEL_ERR_OOM;
            }
        }
        if (np) {
            hipError_t _e = hipMalloc(&job->d_pairs, np * sizeof(PairDesc));
            if (_e == hipSuccess)
                _e = hipMemcpy(job->d_pairs, hp, np * sizeof(PairDesc),
                               hipMemcpyHostToDevice);
            free(hp);
            if (_e != hipSuccess) {
                set_err("pair table upload failed: %s",
                        hipGetErrorString(_e));
                dbeel_gpu_job_destroy(job);
                return DBEEL_ERR_HIP;
            }
        } else {
            free(hp);
        }
    }
    {
        uint32_t r0 = 0;
        for (size_t j = 0; j < n_jobs; j++) {
            job->job_gbase.push_back(D.entry_base[r0]);
            r0 += runs_per_job[j];
        }
        job->job_gbase.push_back(total);
    }
    JOB_CHECK(hipEventRecord(job->ev[1], job->stream));
    JOB_CHECK(hipStreamSynchronize(job->stream));
    float ms = 0;
    JOB_CHECK(hipEventElapsedTime(&ms, job->ev[0], job->ev[1]));
    job->h2d_ms = ms;

    /* pick the aux tier from the max key_size (index slabs are resident
     * now); the tier is a traffic optimization only — every tier is
     * correct for every key length (blob fallback on deep ties) */
    if (total) {
        JOB_CHECK(hipMemsetAsync(job->d_err, 0, 8, job->stream));
        hipLaunchKernelGGL(k_maxks, dim3(pick_grid(total, 256)), dim3(256),
                           0, job->stream, job->desc, job->d_err + 1);
        uint32_t maxks = 0;
        JOB_CHECK(hipMemcpyAsync(&maxks, job->d_err + 1, 4,
                                 hipMemcpyDeviceToHost, job->stream));
        JOB_CHECK(hipStreamSynchronize(job->stream));
        uint32_t maxklen = maxks > 8 ? maxks - 8 : 0;
        job->aux_kind = maxklen <= 20 ? 0 : (maxklen <= 32 ? 1 : 2);
    } else {
        job->aux_kind = 0;
    }
    uint64_t aux_sz = job->aux_kind == 0 ? 16 : (job->aux_kind == 1 ? 48
                                                                    : 64);
    JOB_CHECK(hipMalloc(&job->d_aux, n * aux_sz));

    *out_job = job;
    return DBEEL_OK;
#undef JOB_CHECK
}

extern "C" int dbeel_gpu_job_create(const dbeel_run_view* runs,
                                    size_t n_runs, int device,
                                    dbeel_gpu_job** out_job) {
    uint32_t one[1] = {(uint32_t)n_runs};
    return job_create_impl(runs, n_runs, one, 1, device, out_job);
}

/* Batched independent jobs in ONE launch set (BASELINE configs[3]'s
 * 8-jobs-per-GPU shape): runs[] holds every job's runs back to back,
 * runs_per_job[] their counts. Jobs never pair with each other (ranks,
 * crossranks and winner flags are job-local); outputs land concatenated
 * in job order and dbeel_gpu_job_fetch_job slices them back out. One
 * kernel pipeline per step replaces 8 per-stream pipelines' launch
 * storms and partial-fill tails. */
extern "C" int dbeel_gpu_job_create_batched(const dbeel_run_view* runs,
                                            size_t n_runs,
                                            const uint32_t* runs_per_job,
                                            size_t n_jobs, int device,
                                            dbeel_gpu_job** out_job) {
    if (!runs_per_job || !n_jobs) {
        set_err("runs_per_job is null/empty");
        return DBEEL_ERR_INVALID_ARG;
    }
    return job_create_impl(runs, n_runs, runs_per_job, n_jobs, device,
                           out_job);
}

/* Per-job slice of a batched job's last result: data/index of job
 * `job_idx`, index offsets rebased to the job's own run file. */
extern "C" int dbeel_gpu_job_fetch_job(dbeel_gpu_job* job, size_t job_idx,
                                       dbeel_compact_result* out) {
    g_err[0] = 0;
    if (!job || !out || !job->have_result ||
        job_idx + 1 >= job->job_gbase.size()) {
        set_err("fetch_job: no result / bad job index");
        return DBEEL_ERR_INVALID_ARG;
    }
    HIP_CHECK(hipSetDevice(job->device));
    memset(out, 0, sizeof *out);
    uint64_t g0 = job->job_gbase[job_idx];
    uint64_t g1 = job->job_gbase[job_idx + 1];
    /* boundary offsets/positions: tiny D2H reads (packed scan mode
     * carries both in one u64 — decode below) */
    uint64_t off0 = 0, off1 = job->out_data_len;
    uint32_t pos0 = 0, pos1 = (uint32_t)job->out_entries;
    hipStream_t s = job->stream;
    int pk = job->last_scan_packed;
    if (g0 > 0) {
        HIP_CHECK(hipMemcpyAsync(&off0, job->d_dstoff + g0, 8,
                                 hipMemcpyDeviceToHost, s));
        if (!pk)
            HIP_CHECK(hipMemcpyAsync(&pos0, job->d_pos + g0, 4,
                                     hipMemcpyDeviceToHost, s));
    }
    if (g1 < job->total_entries) {
        HIP_CHECK(hipMemcpyAsync(&off1, job->d_dstoff + g1, 8,
                                 hipMemcpyDeviceToHost, s));
        if (!pk)
            HIP_CHECK(hipMemcpyAsync(&pos1, job->d_pos + g1, 4,
                                     hipMemcpyDeviceToHost, s));
    }
    HIP_CHECK(hipStreamSynchronize(s));
    if (pk) {
        if (g0 > 0) {
            pos0 = (uint32_t)(off0 >> PK_SHIFT);
            off0 &= PK_MASK;
        }
        if (g1 < job->total_entries) {
            pos1 = (uint32_t)(off1 >> PK_SHIFT);
            off1 &= PK_MASK;
        }
    }
    uint64_t dlen = off1 - off0;
    uint64_t nsurv = (uint64_t)pos1 - pos0;
    out->data = (uint8_t*)malloc(dlen ? dlen : 1);
    out->index = (uint8_t*)malloc(nsurv ? nsurv * 16 : 1);
    if (!out->data || !out->index) {
        free(out->data);
        free(out->index);
        memset(out, 0, sizeof *out);
        set_err("host malloc failed");
        return DBEEL_ERR_OOM;
    }
    hipError_t ce = hipSuccess;
    if (dlen)
        ce = hipMemcpyAsync(out->data, job->d_outdata + off0, dlen,
                            hipMemcpyDeviceToHost, s);
    if (ce == hipSuccess && nsurv)
        ce = hipMemcpyAsync(out->index,
                            job->d_outindex + (uint64_t)pos0 * 16,
                            nsurv * 16, hipMemcpyDeviceToHost, s);
    if (ce == hipSuccess) ce = hipStreamSynchronize(s);
    if (ce != hipSuccess) {
        free(out->data);
        free(out->index);
        memset(out, 0, sizeof *out);
        set_err("fetch_job D2H failed: %s", hipGetErrorString(ce));
        return DBEEL_ERR_HIP;
    }
    /* rebase offsets to the job's own file */
    for (uint64_t i = 0; i < nsurv; i++) {
        uint64_t o = ld_u64_host(out->index + i * 16) - off0;
        memcpy(out->index + i * 16, &o, 8);
    }
    out->data_len = dlen;
    out->index_len = nsurv * 16;
    out->entries_written = nsurv;
    return DBEEL_OK;
}

extern "C" void dbeel_gpu_job_destroy(dbeel_gpu_job* job) {
    if (!job) return;
    if (job->device >= 0) hipSetDevice(job->device);
    hipFree(job->d_input);
    hipFree(job->d_rank);
    hipFree(job->d_dstoff);
    hipFree(job->d_pos);
    hipFree(job->d_outindex);
    hipFree(job->d_srcmap);
    hipFree(job->d_outdata);
    hipFree(job->d_scantmp);
    hipFree(job->d_err);
    hipFree(job->d_pfx);
    hipFree(job->d_aux);
    hipFree(job->d_cr);
    hipFree(job->d_pairs);
    hipFree(job->d_winp0);
    hipFree(job->d_corners);
    for (int i = 0; i < 8; i++)
        if (job->ev[i]) hipEventDestroy(job->ev[i]);
    if (job->stream) hipStreamDestroy(job->stream);
    if (job->copy_stream) hipStreamDestroy(job->copy_stream);
    delete job;
}

/* ------------------------------------------------------------------ */
/* Streamed pinned ingest (north_star: pinned host DRAM, chunked       */
/* hipMemcpyAsync, compute overlap). See include/dbeel_gpu.h.          */
/* ------------------------------------------------------------------ */

extern "C" int dbeel_gpu_pin_host(const void* ptr, size_t len) {
    g_err[0] = 0;
    if (!ptr || !len) {
        set_err("pin_host: null/empty buffer");
        return DBEEL_ERR_INVALID_ARG;
    }
    hipError_t e = hipHostRegister(const_cast<void*>(ptr), len,
                                   hipHostRegisterDefault);
    if (e == hipErrorHostMemoryAlreadyRegistered) return DBEEL_OK;
    if (e != hipSuccess) {
        set_err("hipHostRegister failed: %s", hipGetErrorString(e));
        return DBEEL_ERR_HIP;
    }
    return DBEEL_OK;
}

extern "C" int dbeel_gpu_unpin_host(const void* ptr) {
    g_err[0] = 0;
    hipError_t e = hipHostUnregister(const_cast<void*>(ptr));
    if (e != hipSuccess && e != hipErrorHostMemoryNotRegistered) {
        set_err("hipHostUnregister failed: %s", hipGetErrorString(e));
        return DBEEL_ERR_HIP;
    }
    return DBEEL_OK;
}

extern "C" int dbeel_gpu_job_ingest(dbeel_gpu_job* job,
                                    const dbeel_run_view* runs,
                                    size_t n_runs,
                                    dbeel_ingest_stats* stats) {
    g_err[0] = 0;
    if (!job || !runs) {
        set_err("job_ingest: null argument");
        return DBEEL_ERR_INVALID_ARG;
    }
    if ((int)n_runs != job->desc.n_runs) {
        set_err("job_ingest: run count %zu != job's %d", n_runs,
                job->desc.n_runs);
        return DBEEL_ERR_INVALID_ARG;
    }
    int rc = validate_runs(runs, n_runs);
    if (rc) return rc;
    for (size_t r = 0; r < n_runs; r++) {
        if (runs[r].data_len != job->desc.data_len[r] ||
            runs[r].index_len / 16 != job->desc.count[r]) {
            set_err("job_ingest: run %zu shape differs from the job's", r);
            return DBEEL_ERR_INVALID_ARG;
        }
    }
    HIP_CHECK(hipSetDevice(job->device));
    if (!job->copy_stream) HIP_CHECK(hipStreamCreate(&job->copy_stream));
    hipStream_t cs = job->copy_stream; /* H2D lane */
    hipStream_t ks = job->stream;      /* compute lane */
    job->prep_valid = 0;
    job->have_result = false;

    uint64_t chunk_bytes = 64ull << 20;
    if (const char* c = getenv("DBEEL_STREAM_CHUNK_MB")) {
        long v = atol(c);
        if (v >= 1) chunk_bytes = (uint64_t)v << 20;
    }

    /* chunk events: copies are in-order on cs, so event i implies all
     * earlier chunks arrived */
    std::vector<hipEvent_t> evs;
    hipEvent_t ev_cs0 = nullptr, ev_cs1 = nullptr, ev_ks0 = nullptr,
               ev_ks1 = nullptr;
    HIP_CHECK(hipEventCreate(&ev_cs0));
    HIP_CHECK(hipEventCreate(&ev_cs1));
    HIP_CHECK(hipEventCreate(&ev_ks0));
    HIP_CHECK(hipEventCreate(&ev_ks1));
    HIP_CHECK(hipMemsetAsync(job->d_err, 0, 8, ks));

    auto cleanup = [&]() {
        for (hipEvent_t e : evs) hipEventDestroy(e);
        hipEventDestroy(ev_cs0);
        hipEventDestroy(ev_cs1);
        hipEventDestroy(ev_ks0);
        hipEventDestroy(ev_ks1);
    };
#define ING_CHECK(call)                                                     \
    do {                                                                    \
        hipError_t _e = (call);                                             \
        if (_e != hipSuccess) {                                             \
            set_err("%s failed: %s", #call, hipGetErrorString(_e));         \
            cleanup();                                                      \
            return (_e == hipErrorOutOfMemory) ? DBEEL_ERR_OOM              \
                                               : DBEEL_ERR_HIP;             \
        }                                                                   \
    } while (0)

    ING_CHECK(hipEventRecord(ev_cs0, cs));
    ING_CHECK(hipEventRecord(ev_ks0, ks));
    uint64_t total_chunks = 0;

    /* index slabs first (small; prepare reads them for every entry) */
    for (size_t r = 0; r < n_runs; r++) {
        if (!runs[r].index_len) continue;
        ING_CHECK(hipMemcpyAsync(
            const_cast<uint8_t*>(job->desc.index[r]), runs[r].index,
            runs[r].index_len, hipMemcpyHostToDevice, cs));
    }
    {
        hipEvent_t e;
        ING_CHECK(hipEventCreate(&e));
        evs.push_back(e);
        ING_CHECK(hipEventRecord(e, cs));
        ING_CHECK(hipStreamWaitEvent(ks, e, 0));
    }

    /* stream each run's data in entry-aligned chunks; launch the ranged
     * prepare for a chunk's entries as soon as its bytes are on device */
    for (size_t r = 0; r < n_runs; r++) {
        uint64_t n = job->desc.count[r];
        uint64_t dlen = runs[r].data_len;
        if (!dlen) {
            /* entries with no data bytes are corrupt; the ranged prepare
             * must still run over them so the corruption is FLAGGED (it
             * is normally launched per data chunk) */
            if (n) {
                uint64_t g0 = job->desc.entry_base[r];
                uint64_t g1 = g0 + n;
                uint32_t grid = pick_grid(n, 256);
                switch (job->aux_kind) {
#define PREP_EMPTY(KB, TS)                                                  \
    case (KB == 12 ? 0 : (KB == 24 ? 1 : 2)):                               \
        hipLaunchKernelGGL((k_prepare<KB, TS>), dim3(grid), dim3(256), 0,   \
                           ks, job->desc, g0, g1, job->d_pfx,               \
                           (AuxT<KB, TS>*)job->d_aux, job->d_err);          \
        break;
                    PREP_EMPTY(12, false)
                    PREP_EMPTY(24, true)
                    PREP_EMPTY(40, true)
#undef PREP_EMPTY
                }
            }
            continue;
        }
        const uint8_t* hidx = runs[r].index;
        uint64_t e0 = 0, b0 = 0;
        while (b0 < dlen) {
            uint64_t btarget = b0 + chunk_bytes;
            uint64_t e1, b1;
            if (btarget >= dlen) {
                e1 = n;
                b1 = dlen;
            } else {
                /* first entry with offset >= btarget (offsets monotone,
                 * non-overlapping — verified on-device by k_prepare) */
                uint64_t lo = e0, hi = n;
                while (lo < hi) {
                    uint64_t mid = (lo + hi) >> 1;
                    if (ld_u64_host(hidx + mid * 16) < btarget)
                        lo = mid + 1;
                    else
                        hi = mid;
                }
                e1 = lo;
                b1 = (e1 >= n) ? dlen : ld_u64_host(hidx + e1 * 16);
                if (e1 >= n) e1 = n;
                if (b1 <= b0) { /* one giant entry spans the chunk */
                    e1 = e0 + 1;
                    b1 = (e1 >= n) ? dlen : ld_u64_host(hidx + e1 * 16);
                }
                /* corrupt (non-monotone) offsets could still point
                 * backwards or past the run: clamp to a safe single
                 * tail chunk — the on-device validation then reports
                 * CORRUPT instead of this loop wrapping a copy size */
                if (b1 <= b0 || b1 > dlen) {
                    e1 = n;
                    b1 = dlen;
                }
            }
            ING_CHECK(hipMemcpyAsync(
                const_cast<uint8_t*>(job->desc.data[r]) + b0,
                runs[r].data + b0, b1 - b0, hipMemcpyHostToDevice, cs));
            hipEvent_t e;
            ING_CHECK(hipEventCreate(&e));
            evs.push_back(e);
            ING_CHECK(hipEventRecord(e, cs));
            ING_CHECK(hipStreamWaitEvent(ks, e, 0));
            if (e1 > e0) {
                uint64_t g0 = job->desc.entry_base[r] + e0;
                uint64_t g1 = job->desc.entry_base[r] + e1;
                uint32_t grid = pick_grid(g1 - g0, 256);
                switch (job->aux_kind) {
#define PREP_RANGE(KB, TS)                                                  \
    case (KB == 12 ? 0 : (KB == 24 ? 1 : 2)):                               \
        hipLaunchKernelGGL((k_prepare<KB, TS>), dim3(grid), dim3(256), 0,   \
                           ks, job->desc, g0, g1, job->d_pfx,               \
                           (AuxT<KB, TS>*)job->d_aux, job->d_err);          \
        break;
                    PREP_RANGE(12, false)
                    PREP_RANGE(24, true)
                    PREP_RANGE(40, true)
#undef PREP_RANGE
                }
            }
            total_chunks++;
            e0 = e1;
            b0 = b1;
        }
    }

    ING_CHECK(hipEventRecord(ev_cs1, cs));
    ING_CHECK(hipEventRecord(ev_ks1, ks));
    uint32_t err = 0;
    ING_CHECK(hipMemcpyAsync(&err, job->d_err, 4, hipMemcpyDeviceToHost,
                             ks));
    ING_CHECK(hipStreamSynchronize(cs));
    ING_CHECK(hipStreamSynchronize(ks));
    ING_CHECK(hipGetLastError());
    if (err) {
        cleanup();
        set_err("corrupt entry/index record in streamed ingest");
        return DBEEL_ERR_CORRUPT;
    }

    if (stats) {
        float copy_ms = 0, prep_ms = 0, wall_ms = 0;
        (void)hipEventElapsedTime(&copy_ms, ev_cs0, ev_cs1);
        (void)hipEventElapsedTime(&prep_ms, ev_ks0, ev_ks1);
        (void)hipEventElapsedTime(&wall_ms, ev_cs0, ev_ks1);
        if (prep_ms > wall_ms) wall_ms = prep_ms;
        stats->ingest_ms = wall_ms;
        stats->copy_ms = copy_ms;
        stats->prep_ms = prep_ms;
        stats->bytes = job->input_bytes;
        stats->chunks = total_chunks;
    }
    job->h2d_ms = 0.0; /* transfer accounted in ingest stats */
    job->prep_valid = 1;
    cleanup();
    return DBEEL_OK;
#undef ING_CHECK
}

static uint32_t pick_grid(uint64_t work_items, uint32_t block) {
    uint64_t blocks = (work_items + block - 1) / block;
    /* 256 CUs x 8 workgroups/CU = 2048; cap and grid-stride the rest */
    if (blocks > 2048) blocks = 2048;
    if (blocks == 0) blocks = 1;
    return (uint32_t)blocks;
}


/* Copy-geometry selection measured by interleaved A/B (tools/ab_copy.py,
 * r02): ~1 KiB survivors (cfg3) run 11% faster with 64 KiB windows /
 * 512 threads (fewer window transitions+barriers per block); 4 KiB
 * survivors (cfg5) want MORE blocks (grid 16384) at the default window
 * (the 4096 cap left most waves parked); sub-512-B survivors keep the
 * default. Env DBEEL_COPY_VARIANT / DBEEL_COPY_GRID override for A/Bs. */
static void launch_copy(hipStream_t s, const uint8_t* out_index,
                        const uint64_t* src_map, uint32_t* win_p0,
                        uint64_t n_surv, uint64_t total_out,
                        uint8_t* out_data) {
    int variant = -1;
    if (const char* v = getenv("DBEEL_COPY_VARIANT")) variant = atoi(v);
    uint64_t gcap = 0;
    if (const char* g = getenv("DBEEL_COPY_GRID")) {
        long v = atol(g);
        if (v >= 64) gcap = (uint64_t)v;
    }
    if (variant < 0) {
        uint64_t avg = n_surv ? total_out / n_surv : 0;
        if (avg > 2048) {
            variant = 0; /* 4 KiB-class: more blocks, small windows */
            if (!gcap) gcap = 16384;
        } else if (avg >= 512) {
            variant = 4; /* ~1 KiB-class: 64 KiB windows */
        } else {
            variant = 3; /* sub-512-B: 512thr x 32 KiB + more blocks
                            (cfg2 A/B: 0.516 -> 0.467 ms) */
            if (!gcap) gcap = 16384;
        }
    }
    if (!gcap) gcap = 4096;
    uint32_t win = (variant == 1)   ? 8192
                   : (variant == 3) ? 32768
                   : (variant == 4) ? 65536
                                    : 16384;
    uint32_t blk = (variant >= 2) ? 512 : 256;
    uint64_t windows = (total_out + win - 1) / win;
    uint32_t grid = windows > gcap ? (uint32_t)gcap : (uint32_t)windows;
    hipLaunchKernelGGL(k_winmap, dim3(pick_grid(windows, 256)), dim3(256),
                       0, s, out_index, n_surv, total_out, win, win_p0);
    void (*kc)(const uint8_t*, const uint64_t*, const uint32_t*, uint64_t,
               uint64_t, uint8_t*) = k_copy<256, 16384>;
    if (variant == 1) kc = k_copy<256, 8192>;
    if (variant == 2) kc = k_copy<512, 16384>;
    if (variant == 3) kc = k_copy<512, 32768>;
    if (variant == 4) kc = k_copy<512, 65536>;
    hipLaunchKernelGGL(kc, dim3(grid), dim3(blk), 0, s, out_index, src_map,
                       win_p0, n_surv, total_out, out_data);
}

extern "C" int dbeel_gpu_job_run(dbeel_gpu_job* job, int keep_tombstones,
                                 uint64_t* out_data_len,
                                 uint64_t* out_entries,
                                 dbeel_compact_timings* t) {
    g_err[0] = 0;
    if (!job) {
        set_err("job is null");
        return DBEEL_ERR_INVALID_ARG;
    }
    HIP_CHECK(hipSetDevice(job->device));
    hipStream_t s = job->stream;
    uint64_t n = job->total_entries;
    job->have_result = false;

    int skip_prep = job->prep_valid;
    HIP_CHECK(hipMemsetAsync(job->d_err, 0, 2 * sizeof(uint32_t), s));

    HIP_CHECK(hipEventRecord(job->ev[0], s));
    if (n) {
        uint32_t grid = pick_grid(n, 256);
        uint64_t cgrid = job->total_chunks;
        if (cgrid > 16384) cgrid = 16384;
        int corank_wide = 0; /* A/B: 512-thread corank blocks */
        if (const char* w = getenv("DBEEL_CORANK_BLOCK"))
            corank_wide = atoi(w) >= 512;
        switch (job->aux_kind) {
#define STAGE1(KB, TS)                                                      \
    case (KB == 12 ? 0 : (KB == 24 ? 1 : 2)):                               \
        if (!skip_prep)                                                     \
            hipLaunchKernelGGL((k_prepare<KB, TS>), dim3(grid), dim3(256),  \
                               0, s, job->desc, (uint64_t)0, n, job->d_pfx, \
                               (AuxT<KB, TS>*)job->d_aux, job->d_err);      \
        hipEventRecord(job->ev[6], s);                                      \
        if (job->n_pairs) {                                                 \
            hipLaunchKernelGGL((k_corners<KB, TS>),                         \
                               dim3(pick_grid(job->total_chunks, 256)),     \
                               dim3(256), 0, s, job->desc, job->d_pfx,      \
                               (const AuxT<KB, TS>*)job->d_aux,             \
                               (const PairDesc*)job->d_pairs, job->n_pairs, \
                               job->total_chunks, job->d_corners);          \
            if (corank_wide)                                                \
                hipLaunchKernelGGL((k_corank<KB, TS, 512>),                 \
                                   dim3((uint32_t)cgrid), dim3(512), 0, s,  \
                                   job->desc, job->d_pfx,                   \
                                   (const AuxT<KB, TS>*)job->d_aux,         \
                                   (const PairDesc*)job->d_pairs,           \
                                   job->n_pairs, job->total_chunks,         \
                                   job->d_corners, job->d_cr);              \
            else                                                            \
                hipLaunchKernelGGL((k_corank<KB, TS, 256>),                 \
                                   dim3((uint32_t)cgrid),                   \
                                   dim3(CORANK_BLOCK), 0, s, job->desc,     \
                                   job->d_pfx,                              \
                                   (const AuxT<KB, TS>*)job->d_aux,         \
                                   (const PairDesc*)job->d_pairs,           \
                                   job->n_pairs, job->total_chunks,         \
                                   job->d_corners, job->d_cr);              \
        }                                                                   \
        hipLaunchKernelGGL((k_rankreduce<KB, TS>), dim3(grid), dim3(256),   \
                           0, s, job->desc, job->d_pfx,                     \
                           (const AuxT<KB, TS>*)job->d_aux, job->d_cr,      \
                           job->d_rank, keep_tombstones, job->d_err);       \
        break;
            STAGE1(12, false)
            STAGE1(24, true)
            STAGE1(40, true)
#undef STAGE1
        }
    } else {
        HIP_CHECK(hipEventRecord(job->ev[6], s));
    }
    HIP_CHECK(hipEventRecord(job->ev[1], s));
    bool packed = n < (1ull << 26) &&
                  job->total_data_bytes < (1ull << PK_SHIFT);
    if (const char* m = getenv("DBEEL_SCAN_MODE")) {
        if (!strcmp(m, "two")) packed = false;
    }
    if (n) {
        size_t tmp = job->scantmp_bytes;
        if (packed) {
            (void)rocprim::exclusive_scan(
                job->d_scantmp, tmp,
                rocprim::make_transform_iterator(job->d_rank,
                                                 RankRecPacked{}),
                job->d_dstoff, (uint64_t)0, n, rocprim::plus<uint64_t>(),
                s);
        } else {
            (void)rocprim::exclusive_scan(
                job->d_scantmp, tmp,
                rocprim::make_transform_iterator(job->d_rank,
                                                 RankRecSize{}),
                job->d_dstoff, (uint64_t)0, n, rocprim::plus<uint64_t>(),
                s);
            tmp = job->scantmp_bytes;
            (void)rocprim::exclusive_scan(
                job->d_scantmp, tmp,
                rocprim::make_transform_iterator(job->d_rank,
                                                 RankRecFlag{}),
                job->d_pos, (uint32_t)0, n, rocprim::plus<uint32_t>(), s);
        }
    }
    HIP_CHECK(hipEventRecord(job->ev[2], s));
    if (n) {
        uint32_t grid = pick_grid(n, 256);
        if (packed)
            hipLaunchKernelGGL(k_emit_packed, dim3(grid), dim3(256), 0, s,
                               job->desc, job->d_rank, job->d_dstoff, n,
                               job->d_outindex, job->d_srcmap);
        else
            hipLaunchKernelGGL(k_emit, dim3(grid), dim3(256), 0, s,
                               job->desc, job->d_rank, job->d_dstoff,
                               job->d_pos, n, job->d_outindex,
                               job->d_srcmap);
    }
    HIP_CHECK(hipEventRecord(job->ev[3], s));

    /* Need totals on host to size/launch the copy; one small sync. */
    RankRec last_rec = {};
    uint64_t last_off = 0;
    uint32_t last_pos = 0;
    uint32_t err = 0;
    if (n) {
        HIP_CHECK(hipMemcpyAsync(&last_rec, job->d_rank + (n - 1), 16,
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipMemcpyAsync(&last_off, job->d_dstoff + (n - 1), 8,
                                 hipMemcpyDeviceToHost, s));
        if (!packed)
            HIP_CHECK(hipMemcpyAsync(&last_pos, job->d_pos + (n - 1), 4,
                                     hipMemcpyDeviceToHost, s));
    }
    HIP_CHECK(hipMemcpyAsync(&err, job->d_err, 4, hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    if (err) {
        set_err(err & DERR_UNSORTED
                    ? "input run not strictly sorted by unique keys "
                      "(dbeel flush invariant violated)"
                    : "corrupt entry/index record");
        return DBEEL_ERR_CORRUPT;
    }
    if (packed) {
        last_pos = (uint32_t)(last_off >> PK_SHIFT);
        last_off &= PK_MASK;
    }
    job->last_scan_packed = packed ? 1 : 0;
    uint64_t last_kept = (n && (last_rec.src & RR_KEEP)) ? 1 : 0;
    uint64_t total_out = last_off + (last_kept ? last_rec.full_size : 0);
    uint64_t n_surv = (uint64_t)last_pos + last_kept;

    HIP_CHECK(hipEventRecord(job->ev[4], s));
    if (total_out)
        launch_copy(s, job->d_outindex, job->d_srcmap, job->d_winp0,
                    n_surv, total_out, job->d_outdata);
    HIP_CHECK(hipEventRecord(job->ev[5], s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(hipGetLastError());

    job->out_data_len = total_out;
    job->out_entries = n_surv;
    job->have_result = true;
    if (out_data_len) *out_data_len = total_out;
    if (out_entries) *out_entries = n_surv;
    if (t) {
        float prep_ms = 0, rank_ms = 0, scan_ms = 0, emit_ms = 0,
              copy_ms = 0;
        (void)hipEventElapsedTime(&prep_ms, job->ev[0], job->ev[6]);
        (void)hipEventElapsedTime(&rank_ms, job->ev[6], job->ev[1]);
        (void)hipEventElapsedTime(&scan_ms, job->ev[1], job->ev[2]);
        (void)hipEventElapsedTime(&emit_ms, job->ev[2], job->ev[3]);
        (void)hipEventElapsedTime(&copy_ms, job->ev[4], job->ev[5]);
        t->h2d_ms = job->h2d_ms;
        t->prep_ms = prep_ms;
        t->rank_ms = rank_ms;
        t->scan_ms = scan_ms;
        t->emit_ms = emit_ms;
        t->copy_ms = copy_ms;
        t->kernel_ms = prep_ms + rank_ms + scan_ms + emit_ms + copy_ms;
        t->d2h_ms = 0.0;
    }
    return DBEEL_OK;
}

/* ------------------------------------------------------------------ */
/* Bloom builder (behavioral filter over the surviving keys)          */
/*                                                                    */
/* The reference builds a bloomfilter::Bloom over every written key   */
/* (lsm_tree.rs:1026-1051); its file bytes are unpinnable (random     */
/* SipHash keys), so this engine ships its own "DBLM" format — same   */
/* behavior (zero false negatives, ~1% fp at k=7), built here on the  */
/* device from the job's resident output. Hash = seeded FNV-1a 64     */
/* with a splitmix64 finalizer, identical to dbeel_lsm.cpp's checker. */
/* ------------------------------------------------------------------ */

__device__ __forceinline__ uint64_t d_fnv1a64(const uint8_t* p, uint64_t n,
                                              uint64_t seed) {
    uint64_t h = 1469598103934665603ull ^ seed;
    for (uint64_t i = 0; i < n; i++) {
        h ^= p[i];
        h *= 1099511628211ull;
    }
    h ^= h >> 30;
    h *= 0xbf58476d1ce4e5b9ull;
    h ^= h >> 27;
    h *= 0x94d049bb133111ebull;
    h ^= h >> 31;
    return h;
}

__global__ void k_bloom(const uint8_t* out_data, const uint8_t* out_index,
                        uint64_t n_surv, uint64_t n_bits, uint64_t seed,
                        uint32_t kk, uint32_t* bits) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n_surv; i += stride) {
        const uint8_t* rec = out_index + i * 16;
        uint64_t off = ld_u64(rec);
        uint32_t key_size = ld_u32(rec + 8);
        const uint8_t* key = out_data + off + 8;
        uint64_t klen = key_size - 8;
        uint64_t h1 = d_fnv1a64(key, klen, seed);
        uint64_t h2 =
            d_fnv1a64(key, klen, seed ^ 0x9e3779b97f4a7c15ull) | 1;
        for (uint32_t j = 0; j < kk; j++) {
            uint64_t bit = (h1 + j * h2) % n_bits;
            atomicOr(&bits[bit >> 5], 1u << (bit & 31));
        }
    }
}

extern "C" int dbeel_gpu_job_bloom(dbeel_gpu_job* job, uint8_t** out_bytes,
                                   uint64_t* out_len) {
    g_err[0] = 0;
    if (!job || !out_bytes || !out_len || !job->have_result) {
        set_err("job_bloom: no result available");
        return DBEEL_ERR_INVALID_ARG;
    }
    HIP_CHECK(hipSetDevice(job->device));
    uint64_t n = job->out_entries ? job->out_entries : 1;
    /* sizing for 1% fp (BLOOM_MAX_ALLOWED_ERROR, lsm_tree.rs:48) */
    uint64_t n_bits = (uint64_t)(9.585 * (double)n) + 64;
    uint64_t n_words = (n_bits + 31) / 32;
    uint32_t* d_bits = nullptr;
    HIP_CHECK(hipMalloc(&d_bits, n_words * 4));
    HIP_CHECK(hipMemsetAsync(d_bits, 0, n_words * 4, job->stream));
    if (job->out_entries)
        hipLaunchKernelGGL(k_bloom, dim3(pick_grid(job->out_entries, 256)),
                           dim3(256), 0, job->stream, job->d_outdata,
                           job->d_outindex, job->out_entries, n_bits,
                           (uint64_t)0xDBEE1, 7u, d_bits);
    /* "DBLM" | ver | k | pad | n_bits | seed | bitmap (dbeel_lsm.h) */
    uint64_t blen = 32 + ((n_bits + 7) / 8);
    uint8_t* buf = (uint8_t*)calloc(1, blen + 4);
    if (!buf) {
        hipFree(d_bits);
        set_err("host alloc failed");
        return DBEEL_ERR_OOM;
    }
    memcpy(buf, "DBLM", 4);
    uint32_t ver = 1, kk = 7, pad = 0;
    memcpy(buf + 4, &ver, 4);
    memcpy(buf + 8, &kk, 4);
    memcpy(buf + 12, &pad, 4);
    memcpy(buf + 16, &n_bits, 8);
    uint64_t seed = 0xDBEE1;
    memcpy(buf + 24, &seed, 8);
    hipError_t e = hipMemcpyAsync(buf + 32, d_bits, (n_bits + 7) / 8,
                                  hipMemcpyDeviceToHost, job->stream);
    if (e == hipSuccess) e = hipStreamSynchronize(job->stream);
    hipFree(d_bits);
    if (e != hipSuccess) {
        free(buf);
        set_err("bloom D2H failed: %s", hipGetErrorString(e));
        return DBEEL_ERR_HIP;
    }
    *out_bytes = buf;
    *out_len = blen;
    return DBEEL_OK;
}

extern "C" void dbeel_gpu_bloom_free(uint8_t* p) { free(p); }

extern "C" int dbeel_gpu_job_fetch(dbeel_gpu_job* job,
                                   dbeel_compact_result* out) {
    g_err[0] = 0;
    if (!job || !out || !job->have_result) {
        set_err("job_fetch: no result available");
        return DBEEL_ERR_INVALID_ARG;
    }
    HIP_CHECK(hipSetDevice(job->device));
    memset(out, 0, sizeof *out);
    uint64_t dlen = job->out_data_len, ilen = job->out_entries * 16;
    out->data = (uint8_t*)malloc(dlen ? dlen : 1);
    out->index = (uint8_t*)malloc(ilen ? ilen : 1);
    if (!out->data || !out->index) {
        free(out->data);
        free(out->index);
        memset(out, 0, sizeof *out);
        set_err("host malloc failed");
        return DBEEL_ERR_OOM;
    }
    if (dlen)
        HIP_CHECK(hipMemcpyAsync(out->data, job->d_outdata, dlen,
                                 hipMemcpyDeviceToHost, job->stream));
    if (ilen)
        HIP_CHECK(hipMemcpyAsync(out->index, job->d_outindex, ilen,
                                 hipMemcpyDeviceToHost, job->stream));
    HIP_CHECK(hipStreamSynchronize(job->stream));
    out->data_len = dlen;
    out->index_len = ilen;
    out->entries_written = job->out_entries;
    return DBEEL_OK;
}

/* ------------------------------------------------------------------ */
/* Batched point lookup                                               */
/*                                                                    */
/* GPU analogue of LSMTree::get over the sstables                     */
/* (lsm_tree.rs:605-723, binary search per run lsm_tree.rs:674-723):  */
/* one thread per query key binary-searches the runs newest-INDEX-    */
/* first and returns the first key match, exactly like the reference  */
/* read path (`sstables.iter().rev()`, lsm_tree.rs:692-696) — the     */
/* highest run index wins regardless of timestamp. Tombstones are     */
/* reported as found with value_len 0 so the caller distinguishes     */
/* deleted from absent (tests/db_server.rs delete->get->KeyNotFound   */
/* semantics). Bloom prefiltering stays on the host                   */
/* (dbeel_bloom_contains).                                            */
/* ------------------------------------------------------------------ */

__global__ void k_lookup(RunsDesc R, const uint8_t* keys,
                         const uint64_t* key_off, uint64_t n_keys,
                         dbeel_lookup_hit* out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t q = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         q < n_keys; q += stride) {
        const uint8_t* key = keys + key_off[q];
        uint64_t klen = key_off[q + 1] - key_off[q];
        int best_run = -1;
        uint64_t best_off = 0, best_vlen = 0;
        /* newest-index-first, first match wins (lsm_tree.rs:692-696) */
        for (int r = R.n_runs - 1; r >= 0; r--) {
            uint64_t lo = 0, hi = R.count[r];
            while (lo < hi) {
                uint64_t mid = (lo + hi) >> 1;
                EView m;
                if (!load_entry(R, r, mid, m)) {
                    hi = mid; /* corrupt input; search degrades safely */
                    continue;
                }
                int c = cmp_keys(m.key, m.klen, key, klen);
                if (c < 0)
                    lo = mid + 1;
                else
                    hi = mid;
            }
            if (lo >= R.count[r]) continue;
            EView m;
            if (!load_entry(R, r, lo, m)) continue;
            if (m.klen != klen || cmp_keys(m.key, m.klen, key, klen) != 0)
                continue;
            best_run = r;
            best_off = m.off + 16 + m.klen; /* value bytes start */
            best_vlen = (uint64_t)m.full_size - 32 - m.klen;
            break;
        }
        out[q].run = best_run;
        out[q].is_tombstone = (best_run >= 0 && best_vlen == 0) ? 1 : 0;
        out[q].value_offset = best_off;
        out[q].value_len = best_vlen;
    }
}

extern "C" int dbeel_gpu_lookup(const dbeel_run_view* runs, size_t n_runs,
                                const uint8_t* keys,
                                const uint64_t* key_offsets, uint64_t n_keys,
                                int device, dbeel_lookup_hit* out) {
    g_err[0] = 0;
    if (!out || (n_keys && (!keys || !key_offsets))) {
        set_err("lookup: null argument");
        return DBEEL_ERR_INVALID_ARG;
    }
    int rc = validate_runs(runs, n_runs);
    if (rc) return rc;
    if (device < 0) {
        set_err("device must be >= 0 (no CPU fallback)");
        return DBEEL_ERR_INVALID_ARG;
    }
    int ndev = 0;
    hipError_t de = hipGetDeviceCount(&ndev);
    if (de != hipSuccess || device >= ndev) {
        set_err("no usable HIP device %d", device);
        return DBEEL_ERR_NO_GPU;
    }
    HIP_CHECK(hipSetDevice(device));
    if (n_keys == 0) return DBEEL_OK;

    dbeel_gpu_job* job = nullptr;
    rc = dbeel_gpu_job_create(runs, n_runs, device, &job);
    if (rc) return rc;

    uint64_t kbytes = key_offsets[n_keys];
    uint8_t* d_keys = nullptr;
    uint64_t* d_koff = nullptr;
    dbeel_lookup_hit* d_out = nullptr;
    hipError_t e = hipMalloc(&d_keys, kbytes ? kbytes : 1);
    if (e == hipSuccess) e = hipMalloc(&d_koff, (n_keys + 1) * 8);
    if (e == hipSuccess)
        e = hipMalloc(&d_out, n_keys * sizeof(dbeel_lookup_hit));
    if (e == hipSuccess && kbytes)
        e = hipMemcpyAsync(d_keys, keys, kbytes, hipMemcpyHostToDevice,
                           job->stream);
    if (e == hipSuccess)
        e = hipMemcpyAsync(d_koff, key_offsets, (n_keys + 1) * 8,
                           hipMemcpyHostToDevice, job->stream);
    if (e == hipSuccess) {
        hipLaunchKernelGGL(k_lookup, dim3(pick_grid(n_keys, 256)), dim3(256),
                           0, job->stream, job->desc, d_keys, d_koff, n_keys,
                           d_out);
        e = hipMemcpyAsync(out, d_out, n_keys * sizeof(dbeel_lookup_hit),
                           hipMemcpyDeviceToHost, job->stream);
    }
    if (e == hipSuccess) e = hipStreamSynchronize(job->stream);
    if (e == hipSuccess) e = hipGetLastError();
    hipFree(d_keys);
    hipFree(d_koff);
    hipFree(d_out);
    dbeel_gpu_job_destroy(job);
    if (e != hipSuccess) {
        set_err("lookup failed: %s", hipGetErrorString(e));
        return DBEEL_ERR_HIP;
    }
    return DBEEL_OK;
}

/* ------------------------------------------------------------------ */
/* Migration / iteration scan (AsyncIter analogue)                    */
/*                                                                    */
/* See include/dbeel_gpu.h. The yield order is the reference's:       */
/* sstables ascending, entries in index order within each             */
/* (lsm_tree.rs:210-276) — which is exactly the global entry order g, */
/* so the existing survivor-scan/emit/copy machinery packs the        */
/* filtered entries verbatim. No dedup, no tombstone filter.          */
/* ------------------------------------------------------------------ */

/* MurmurHash3 x86_32 — restatement of the murmur3 crate v0.5.2's      */
/* murmur3_32 (the reference's hash_bytes, shards.rs:96-101, seed 0   */
/* everywhere). Public algorithm (Appleby); pinned by the published   */
/* test vectors in tests/test_scan_host.py.                           */
__device__ __host__ __forceinline__ uint32_t rotl32(uint32_t x, int r) {
    return (x << r) | (x >> (32 - r));
}
__device__ __host__ inline uint32_t murmur3_32(const uint8_t* key,
                                               uint64_t len,
                                               uint32_t seed) {
    const uint32_t c1 = 0xcc9e2d51u, c2 = 0x1b873593u;
    uint32_t h = seed;
    uint64_t nblocks = len / 4;
    for (uint64_t i = 0; i < nblocks; i++) {
        uint32_t k;
        __builtin_memcpy(&k, key + i * 4, 4); /* little-endian */
        k *= c1;
        k = rotl32(k, 15);
        k *= c2;
        h ^= k;
        h = rotl32(h, 13);
        h = h * 5 + 0xe6546b64u;
    }
    uint32_t k1 = 0;
    const uint8_t* tail = key + nblocks * 4;
    switch (len & 3) {
        case 3: k1 ^= (uint32_t)tail[2] << 16; /* fallthrough */
        case 2: k1 ^= (uint32_t)tail[1] << 8;  /* fallthrough */
        case 1:
            k1 ^= tail[0];
            k1 *= c1;
            k1 = rotl32(k1, 15);
            k1 *= c2;
            h ^= k1;
    }
    h ^= (uint32_t)len;
    h ^= h >> 16;
    h *= 0x85ebca6bu;
    h ^= h >> 13;
    h *= 0xc2b2ae35u;
    h ^= h >> 16;
    return h;
}

/* host-callable export so the CPU test suite can pin the restatement
 * against the published MurmurHash3 vectors without a GPU */
extern "C" uint32_t dbeel_murmur3_32(const uint8_t* key, uint64_t len,
                                     uint32_t seed) {
    return murmur3_32(key, len, seed);
}

/* between_cmp restated EXACTLY (tasks/migration.rs:54-60). Note the
 * reference's wrapped case (end < start) evaluates true for every hash —
 * restated verbatim, divergence-free. */
__device__ __forceinline__ bool hash_between(uint32_t h, uint32_t start,
                                             uint32_t end) {
    if (end < start) return h < start || h >= end;
    return h >= start && h < end;
}

__global__ void k_scanflag(RunsDesc R, const uint8_t* kbounds,
                           uint64_t start_len, uint64_t end_len,
                           int have_start, int have_end,
                           const uint32_t* rstart, const uint32_t* rend,
                           uint32_t n_ranges, RankRec* rrec,
                           uint32_t* err) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < R.total; g += stride) {
        int r = 0;
        while (r + 1 < R.n_runs && g >= R.entry_base[r + 1]) r++;
        uint64_t i = g - R.entry_base[r];
        EView e;
        if (!load_entry(R, r, i, e)) {
            atomicOr(err, DERR_CORRUPT);
            RankRec z = {0, 8, 32};
            rrec[g] = z;
            continue;
        }
        bool keep = true;
        if (have_start &&
            cmp_keys(e.key, e.klen, kbounds, start_len) < 0)
            keep = false;
        if (keep && have_end &&
            cmp_keys(e.key, e.klen, kbounds + start_len, end_len) >= 0)
            keep = false;
        if (keep && n_ranges) {
            uint32_t h = murmur3_32(e.key, e.klen, 0);
            bool in_any = false;
            for (uint32_t j = 0; j < n_ranges && !in_any; j++)
                in_any = hash_between(h, rstart[j], rend[j]);
            keep = in_any;
        }
        RankRec m;
        m.src = ((uint64_t)r << 48) | e.off | (keep ? RR_KEEP : 0);
        m.key_size = e.key_size;
        m.full_size = e.full_size;
        rrec[g] = m; /* dense: rank == g (yield order, no reordering) */
    }
}

extern "C" int dbeel_gpu_scan(const dbeel_run_view* runs, size_t n_runs,
                              const uint8_t* start_key,
                              size_t start_key_len, const uint8_t* end_key,
                              size_t end_key_len,
                              const uint32_t* range_starts,
                              const uint32_t* range_ends, size_t n_ranges,
                              int device, dbeel_compact_result* out) {
    g_err[0] = 0;
    if (!out || (n_ranges && (!range_starts || !range_ends)) ||
        (start_key_len && !start_key) || (end_key_len && !end_key)) {
        set_err("scan: null argument");
        return DBEEL_ERR_INVALID_ARG;
    }
    dbeel_gpu_job* job = nullptr;
    int rc = dbeel_gpu_job_create(runs, n_runs, device, &job);
    if (rc) return rc;
    hipStream_t s = job->stream;
    uint64_t n = job->total_entries;

    uint8_t* d_kb = nullptr;
    uint32_t* d_ranges = nullptr;
    hipError_t e = hipSuccess;
    uint64_t kb_len = start_key_len + end_key_len;
    if (kb_len == 0) kb_len = 1;
    e = hipMalloc(&d_kb, kb_len);
    if (e == hipSuccess && start_key_len)
        e = hipMemcpyAsync(d_kb, start_key, start_key_len,
                           hipMemcpyHostToDevice, s);
    if (e == hipSuccess && end_key_len)
        e = hipMemcpyAsync(d_kb + start_key_len, end_key, end_key_len,
                           hipMemcpyHostToDevice, s);
    if (e == hipSuccess)
        e = hipMalloc(&d_ranges, (n_ranges ? 2 * n_ranges : 1) * 4);
    if (e == hipSuccess && n_ranges) {
        e = hipMemcpyAsync(d_ranges, range_starts, n_ranges * 4,
                           hipMemcpyHostToDevice, s);
        if (e == hipSuccess)
            e = hipMemcpyAsync(d_ranges + n_ranges, range_ends,
                               n_ranges * 4, hipMemcpyHostToDevice, s);
    }
    if (e == hipSuccess)
        e = hipMemsetAsync(job->d_err, 0, 8, s);
    if (e == hipSuccess && n) {
        hipLaunchKernelGGL(k_scanflag, dim3(pick_grid(n, 256)), dim3(256),
                           0, s, job->desc, d_kb, (uint64_t)start_key_len,
                           (uint64_t)end_key_len, start_key ? 1 : 0,
                           end_key ? 1 : 0, d_ranges,
                           d_ranges + n_ranges, (uint32_t)n_ranges,
                           job->d_rank, job->d_err);
        size_t tmp = job->scantmp_bytes;
        (void)rocprim::exclusive_scan(
            job->d_scantmp, tmp,
            rocprim::make_transform_iterator(job->d_rank, RankRecSize{}),
            job->d_dstoff, (uint64_t)0, n, rocprim::plus<uint64_t>(), s);
        tmp = job->scantmp_bytes;
        (void)rocprim::exclusive_scan(
            job->d_scantmp, tmp,
            rocprim::make_transform_iterator(job->d_rank, RankRecFlag{}),
            job->d_pos, (uint32_t)0, n, rocprim::plus<uint32_t>(), s);
        hipLaunchKernelGGL(k_emit, dim3(pick_grid(n, 256)), dim3(256), 0,
                           s, job->desc, job->d_rank, job->d_dstoff,
                           job->d_pos, n, job->d_outindex, job->d_srcmap);
    }
    RankRec last_rec = {};
    uint64_t last_off = 0;
    uint32_t last_pos = 0, derr = 0;
    if (e == hipSuccess && n) {
        e = hipMemcpyAsync(&last_rec, job->d_rank + (n - 1), 16,
                           hipMemcpyDeviceToHost, s);
        if (e == hipSuccess)
            e = hipMemcpyAsync(&last_off, job->d_dstoff + (n - 1), 8,
                               hipMemcpyDeviceToHost, s);
        if (e == hipSuccess)
            e = hipMemcpyAsync(&last_pos, job->d_pos + (n - 1), 4,
                               hipMemcpyDeviceToHost, s);
    }
    if (e == hipSuccess)
        e = hipMemcpyAsync(&derr, job->d_err, 4, hipMemcpyDeviceToHost, s);
    if (e == hipSuccess) e = hipStreamSynchronize(s);
    if (e == hipSuccess) e = hipGetLastError();

    job->last_scan_packed = 0;
    int ret = DBEEL_OK;
    if (e != hipSuccess) {
        set_err("scan failed: %s", hipGetErrorString(e));
        ret = DBEEL_ERR_HIP;
    } else if (derr) {
        set_err("corrupt entry/index record");
        ret = DBEEL_ERR_CORRUPT;
    } else {
        uint64_t last_kept = (n && (last_rec.src & RR_KEEP)) ? 1 : 0;
        uint64_t total_out =
            last_off + (last_kept ? last_rec.full_size : 0);
        uint64_t n_surv = (uint64_t)last_pos + last_kept;
        if (total_out)
            launch_copy(s, job->d_outindex, job->d_srcmap, job->d_winp0,
                        n_surv, total_out, job->d_outdata);
        e = hipStreamSynchronize(s);
        if (e == hipSuccess) e = hipGetLastError();
        if (e != hipSuccess) {
            set_err("scan copy failed: %s", hipGetErrorString(e));
            ret = DBEEL_ERR_HIP;
        } else {
            job->out_data_len = total_out;
            job->out_entries = n_surv;
            job->have_result = true;
            ret = dbeel_gpu_job_fetch(job, out);
        }
    }
    hipFree(d_kb);
    hipFree(d_ranges);
    dbeel_gpu_job_destroy(job);
    return ret;
}

extern "C" int dbeel_gpu_compact_timed(const dbeel_run_view* runs,
                                       size_t n_runs, int keep_tombstones,
                                       int device, dbeel_compact_result* out,
                                       dbeel_compact_timings* t) {
    if (!out) {
        set_err("out is null");
        return DBEEL_ERR_INVALID_ARG;
    }
    dbeel_gpu_job* job = nullptr;
    int rc = dbeel_gpu_job_create(runs, n_runs, device, &job);
    if (rc) return rc;
    rc = dbeel_gpu_job_run(job, keep_tombstones, nullptr, nullptr, t);
    if (!rc) rc = dbeel_gpu_job_fetch(job, out);
    dbeel_gpu_job_destroy(job);
    return rc;
}

extern "C" int dbeel_gpu_compact(const dbeel_run_view* runs, size_t n_runs,
                                 int keep_tombstones, int device,
                                 dbeel_compact_result* out) {
    return dbeel_gpu_compact_timed(runs, n_runs, keep_tombstones, device, out,
                                   nullptr);
}

/* ------------------------------------------------------------------ */
/* Sliced compaction: jobs whose inputs exceed HBM                    */
/*                                                                    */
/* The key space is cut at pivot keys into slices whose input fits    */
/* the device budget; each slice is an ordinary compaction job and    */
/* the outputs concatenate. Because slice boundaries are strict key   */
/* pivots, an equal-key group never splits, so winner/tombstone       */
/* decisions are slice-local and the concatenation is byte-identical  */
/* to one whole compaction (survivors verbatim in global key order,   */
/* offsets rebased). Requires densely-ordered runs (EntryWriter       */
/* layout — verified on-device).                                      */
/* ------------------------------------------------------------------ */

static int host_key_cmp(const uint8_t* a, uint64_t la, const uint8_t* b,
                        uint64_t lb) {
    uint64_t n = la < lb ? la : lb;
    int c = memcmp(a, b, n);
    if (c) return c;
    return la < lb ? -1 : (la > lb ? 1 : 0);
}

/* key view of entry i of a host run; NULL when the index record points
 * outside the run's data (corrupt input must not cause host OOB reads
 * during pivot selection — the device validation would catch it later,
 * but the slicer reads keys FIRST) */
static inline const uint8_t* host_entry_key(const dbeel_run_view* run,
                                            uint64_t i, uint64_t* klen) {
    const uint8_t* rec = run->index + i * 16;
    uint64_t off = ld_u64_host(rec);
    uint32_t key_size, full_size;
    memcpy(&key_size, rec + 8, 4);
    memcpy(&full_size, rec + 12, 4);
    if (key_size < 8 || full_size < 32 ||
        (uint64_t)key_size + 24 > full_size ||
        off + full_size > run->data_len)
        return NULL;
    *klen = key_size - 8;
    return run->data + off + 8;
}

extern "C" int dbeel_gpu_compact_sliced(const dbeel_run_view* runs,
                                        size_t n_runs, int keep_tombstones,
                                        int device,
                                        uint64_t max_resident_bytes,
                                        dbeel_compact_result* out) {
    g_err[0] = 0;
    if (!out) {
        set_err("out is null");
        return DBEEL_ERR_INVALID_ARG;
    }
    int rc = validate_runs(runs, n_runs);
    if (rc) return rc;
    if (device < 0) {
        set_err("device must be >= 0 (no CPU fallback)");
        return DBEEL_ERR_INVALID_ARG;
    }
    uint64_t total_input = 0, max_count = 0;
    size_t big_run = 0;
    for (size_t r = 0; r < n_runs; r++) {
        total_input += runs[r].data_len + runs[r].index_len;
        uint64_t c = runs[r].index_len / 16;
        if (c > max_count) {
            max_count = c;
            big_run = r;
        }
    }
    uint64_t budget = max_resident_bytes;
    if (!budget) {
        HIP_CHECK(hipSetDevice(device));
        size_t free_b = 0, total_b = 0;
        HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
        /* input slab + output data + pfx/aux/cr/rrec intermediates stay
         * under ~2.6x input for <= 64 runs; /3 leaves headroom */
        budget = (uint64_t)(free_b / 3);
    }
    uint64_t S = budget ? (total_input + budget - 1) / budget : 1;
    if (S <= 1 || max_count < 2 * S)
        return dbeel_gpu_compact(runs, n_runs, keep_tombstones, device, out);

    /* pivots: evenly spaced keys of the largest run */
    std::vector<std::pair<const uint8_t*, uint64_t>> pivots; /* key,len */
    for (uint64_t j = 1; j < S; j++) {
        uint64_t klen;
        const uint8_t* k =
            host_entry_key(&runs[big_run], j * max_count / S, &klen);
        if (!k) {
            set_err("corrupt index record in run %zu", big_run);
            return DBEEL_ERR_CORRUPT;
        }
        pivots.push_back({k, klen});
    }
    /* per run: slice boundary entry indices (first entry >= pivot) */
    std::vector<std::vector<uint64_t>> bounds(n_runs);
    for (size_t r = 0; r < n_runs; r++) {
        uint64_t n = runs[r].index_len / 16;
        bounds[r].push_back(0);
        for (auto& pv : pivots) {
            uint64_t lo = bounds[r].back(), hi = n;
            while (lo < hi) {
                uint64_t mid = (lo + hi) >> 1;
                uint64_t kl;
                const uint8_t* k = host_entry_key(&runs[r], mid, &kl);
                if (!k) {
                    set_err("corrupt index record in run %zu", r);
                    return DBEEL_ERR_CORRUPT;
                }
                if (host_key_cmp(k, kl, pv.first, pv.second) < 0)
                    lo = mid + 1;
                else
                    hi = mid;
            }
            bounds[r].push_back(lo);
        }
        bounds[r].push_back(n);
    }

    memset(out, 0, sizeof *out);
    std::vector<uint8_t> acc_data, acc_index;
    uint64_t written = 0;
    std::vector<uint8_t> idx_tmp;
    for (uint64_t sl = 0; sl < S; sl++) {
        std::vector<dbeel_run_view> views;
        std::vector<size_t> idx_tmp_off;
        idx_tmp.clear();
        uint64_t slice_entries = 0;
        for (size_t r = 0; r < n_runs; r++) {
            uint64_t lo = bounds[r][sl], hi = bounds[r][sl + 1];
            dbeel_run_view v{};
            if (hi > lo) {
                uint64_t off_lo = ld_u64_host(runs[r].index + lo * 16);
                uint64_t off_hi =
                    (hi < runs[r].index_len / 16)
                        ? ld_u64_host(runs[r].index + hi * 16)
                        : runs[r].data_len;
                v.data = runs[r].data + off_lo;
                v.data_len = off_hi - off_lo;
                /* rebased index records */
                size_t pos = idx_tmp.size();
                idx_tmp.resize(pos + (hi - lo) * 16);
                memcpy(idx_tmp.data() + pos, runs[r].index + lo * 16,
                       (hi - lo) * 16);
                for (uint64_t i = 0; i < hi - lo; i++) {
                    uint64_t o =
                        ld_u64_host(idx_tmp.data() + pos + i * 16) - off_lo;
                    memcpy(idx_tmp.data() + pos + i * 16, &o, 8);
                }
                idx_tmp_off.push_back(pos);
                v.index_len = (hi - lo) * 16;
                slice_entries += hi - lo;
            } else {
                v.data = runs[r].data; /* non-null, zero-length */
                v.data_len = 0;
                v.index = runs[r].index;
                v.index_len = 0;
                idx_tmp_off.push_back((size_t)-1);
            }
            views.push_back(v);
        }
        if (!slice_entries) continue;
        /* resolve idx pointers after idx_tmp stopped growing */
        for (size_t r = 0; r < n_runs; r++)
            if (idx_tmp_off[r] != (size_t)-1)
                views[r].index = idx_tmp.data() + idx_tmp_off[r];
        dbeel_compact_result sres{};
        rc = dbeel_gpu_compact(views.data(), n_runs, keep_tombstones,
                               device, &sres);
        if (rc) return rc;
        /* concatenate, rebasing index offsets */
        uint64_t base = acc_data.size();
        acc_data.insert(acc_data.end(), sres.data,
                        sres.data + sres.data_len);
        size_t ipos = acc_index.size();
        acc_index.insert(acc_index.end(), sres.index,
                         sres.index + sres.index_len);
        for (uint64_t i = 0; i < sres.index_len / 16; i++) {
            uint64_t o =
                ld_u64_host(acc_index.data() + ipos + i * 16) + base;
            memcpy(acc_index.data() + ipos + i * 16, &o, 8);
        }
        written += sres.entries_written;
        dbeel_gpu_result_free(&sres);
    }

    out->data = (uint8_t*)malloc(acc_data.size() ? acc_data.size() : 1);
    out->index = (uint8_t*)malloc(acc_index.size() ? acc_index.size() : 1);
    if (!out->data || !out->index) {
        free(out->data);
        free(out->index);
        memset(out, 0, sizeof *out);
        set_err("host alloc failed");
        return DBEEL_ERR_OOM;
    }
    memcpy(out->data, acc_data.data(), acc_data.size());
    memcpy(out->index, acc_index.data(), acc_index.size());
    out->data_len = acc_data.size();
    out->index_len = acc_index.size();
    out->entries_written = written;
    return DBEEL_OK;
}

extern "C" void dbeel_gpu_result_free(dbeel_compact_result* r) {
    if (!r) return;
    free(r->data);
    free(r->index);
    memset(r, 0, sizeof *r);
}
