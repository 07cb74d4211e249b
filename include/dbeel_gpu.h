/* dbeel_gpu.h — C ABI of the MI355X-native SSTable compaction engine.
 *
 * This is the drop-in seam for dbeel's `LSMTree::compact` hot path
 * (reference: src/storage_engine/lsm_tree.rs:950-1156). Everything from
 * lsm_tree.rs:974 down (read -> k-way merge/dedup -> write) is replaced by
 * `dbeel_gpu_compact`; the caller keeps the reference's surrounding side
 * effects (bloom file, .compact_action journal, renames, sstable-list swap —
 * lsm_tree.rs:1070-1153).
 *
 * Wire format (all integers little-endian, bincode fixint — reference
 * src/utils/bincode.rs:9-16):
 *   entry      = key_len:u64 | key bytes | data_len:u64 | data bytes |
 *                timestamp:i128                 (src/storage_engine/mod.rs:68-73,
 *                                               src/utils/timestamp_nanos.rs:6-11)
 *   index rec  = offset:u64 | key_size:u32 | full_size:u32   = 16 bytes
 *                (src/storage_engine/mod.rs:45-50, INDEX_ENTRY_SIZE mod.rs:33,
 *                 asserted lsm_tree.rs:408-413)
 *   key_size   = 8 + key_len;  full_size = 32 + key_len + data_len
 *
 * Merge semantics (lsm_tree.rs:1038-1066, CompactionItem ordering
 * lsm_tree.rs:52-71 + Entry::cmp mod.rs:75-81):
 *   total order: key bytes ascending (lexicographic), then timestamp
 *   ascending (signed i128), then run index ascending. Among entries with
 *   equal key, exactly one survives: the max by (timestamp, run index)
 *   ("last pop of an equal-key group wins"). A surviving entry whose data is
 *   empty (TOMBSTONE, mod.rs:14) is dropped unless keep_tombstones.
 *   Survivors are emitted verbatim (input bytes unchanged) in total order,
 *   densely packed; index offsets recomputed from 0
 *   (entry_writer.rs:71-98).
 *
 * Input invariants (guaranteed by dbeel's flush/compaction writers,
 * lsm_tree.rs:925-946): each run is sorted ascending by key with unique keys
 * within the run. Corrupt inputs (index size not a multiple of 16, entry
 * ranges out of bounds, full_size < 32 + key_len) return
 * DBEEL_ERR_CORRUPT_RECORD; the reference silently treats a failed decode as
 * end-of-stream (lsm_tree.rs:1014,1063 "if let Ok"), this engine errors
 * loudly instead (documented divergence).
 */
#ifndef DBEEL_GPU_H
#define DBEEL_GPU_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* One SSTable run, resident in host DRAM, caller-owned, read-only.
 * runs[] must be ordered by ascending sstable index — the (timestamp, run
 * index) tie-break depends on it (lsm_tree.rs:58-65). */
typedef struct {
    const uint8_t* data;      /* run data file bytes (concatenated entries) */
    size_t         data_len;
    const uint8_t* index;     /* run index file bytes (16-byte records)     */
    size_t         index_len;
} dbeel_run_view;

/* Engine-allocated outputs; free with dbeel_gpu_result_free. */
typedef struct {
    uint8_t* data;            /* output .data file bytes  */
    size_t   data_len;
    uint8_t* index;           /* output .index file bytes */
    size_t   index_len;
    uint64_t entries_written;
} dbeel_compact_result;

/* Optional per-call timing/throughput evidence (milliseconds, HIP events on
 * the job's stream). kernel_ms covers the whole device pipeline
 * (rank -> scan -> emit -> copy), excluding host<->device copies. */
typedef struct {
    double h2d_ms;
    double prep_ms;    /* k_prepare: validation + key-prefix/aux extract */
    double rank_ms;    /* k_corank + k_rankreduce: merge-path crossranks,
                          global ranks, winner flags                 */
    double scan_ms;    /* survivor size/position prefix sums        */
    double emit_ms;    /* output index build + survivor source map  */
    double copy_ms;    /* verbatim entry copy-out (incl. k_winmap)  */
    double kernel_ms;  /* prep+rank+scan+emit+copy                  */
    double d2h_ms;
} dbeel_compact_timings;

/* Error codes. Mirrors the reference error taxonomy at this seam
 * (src/error.rs; ItemTooLarge error.rs:60-61). */
enum {
    DBEEL_OK               = 0,
    DBEEL_ERR_INVALID_ARG  = 1,
    DBEEL_ERR_CORRUPT      = 2,  /* bincode-decode-failure analogue */
    DBEEL_ERR_ITEM_TOO_LARGE = 3,
    DBEEL_ERR_HIP          = 4,  /* device/runtime failure */
    DBEEL_ERR_NO_GPU       = 5,
    DBEEL_ERR_OOM          = 6,
    DBEEL_ERR_IO           = 7,  /* host filesystem failure (error.rs I/O
                                    analogue) — distinct from device errors */
};

/* Compact n_runs runs into one run. keep_tombstones: 0 = drop entries with
 * empty value data (lsm_tree.rs:1045-1046). device: HIP device ordinal
 * (>= 0). There is NO CPU fallback in this library: the CPU restatement
 * lives in liboracle.so (test infrastructure only) and device < 0 returns
 * DBEEL_ERR_INVALID_ARG. Returns 0 on success. Reentrant; concurrent calls
 * on distinct devices run independent jobs. */
int dbeel_gpu_compact(const dbeel_run_view* runs, size_t n_runs,
                      int keep_tombstones, int device,
                      dbeel_compact_result* out);

/* Same, and also reports timings (t may be NULL). */
int dbeel_gpu_compact_timed(const dbeel_run_view* runs, size_t n_runs,
                            int keep_tombstones, int device,
                            dbeel_compact_result* out,
                            dbeel_compact_timings* t);

void dbeel_gpu_result_free(dbeel_compact_result* r);

/* Thread-local message for the last error in this thread. */
const char* dbeel_gpu_last_error(void);

/* ---- Batched point lookup ----
 * GPU analogue of LSMTree::get over the sstables (lsm_tree.rs:674-723):
 * the reference scans sstables newest-INDEX-first and returns the first
 * key match (`sstables.iter().rev()`, lsm_tree.rs:692-696) — so the match
 * in the HIGHEST run index wins, regardless of timestamp. (This can
 * differ from the compaction winner rule when set_with_timestamp writes
 * an older timestamp into a newer sstable; compaction then reorders —
 * faithful to the reference read path.) run = -1 when absent;
 * is_tombstone = 1 distinguishes deleted from absent (the reference's
 * delete->get->KeyNotFound semantics). value_offset indexes into
 * runs[run].data. Bloom prefiltering stays host-side
 * (dbeel_bloom_contains). keys: concatenated blob + n+1 offsets. */
typedef struct {
    int32_t run;
    uint32_t is_tombstone;
    uint64_t value_offset;
    uint64_t value_len;
} dbeel_lookup_hit;

int dbeel_gpu_lookup(const dbeel_run_view* runs, size_t n_runs,
                     const uint8_t* keys, const uint64_t* key_offsets,
                     uint64_t n_keys, int device, dbeel_lookup_hit* out);

/* ---- Run encoder (the memtable-flush path) ----
 * Encodes an already-sorted (key, value, timestamp) stream into a run:
 * the same entry layout + 16-byte index records flush_memtable_to_disk /
 * EntryWriter produce (lsm_tree.rs:925-946, entry_writer.rs:71-98).
 * keys/values: concatenated byte blobs with n+1 exclusive offsets;
 * timestamps: n x 16-byte little-endian i128. Outputs are engine-allocated
 * (dbeel_gpu_result_free). Oversized entries (full_size > u32::MAX) return
 * DBEEL_ERR_ITEM_TOO_LARGE (error.rs:60-61). */
int dbeel_gpu_encode_run(uint64_t n_entries, const uint8_t* keys,
                         const uint64_t* key_offsets, const uint8_t* values,
                         const uint64_t* value_offsets,
                         const uint8_t* timestamps, int device,
                         dbeel_compact_result* out);

/* ---- Resident-job API (benchmarking / repeated compactions) ----
 * Uploads the runs to `device` once; each run executes the device pipeline
 * with inputs already resident in HBM (what BASELINE's MB/s is quoted on)
 * and leaves outputs on the device. */
typedef struct dbeel_gpu_job dbeel_gpu_job;

int dbeel_gpu_job_create(const dbeel_run_view* runs, size_t n_runs,
                         int device, dbeel_gpu_job** out_job);
/* Batched INDEPENDENT jobs in one launch set (one dbeel shard each —
 * BASELINE configs[3]'s many-jobs-per-GPU shape): runs[] holds every
 * job's runs back to back, runs_per_job[] their counts (each >= 1,
 * summing to n_runs). Jobs never interact: ranks, crossrank slots and
 * newest-wins flags are job-local, and a key present in two jobs
 * survives in both. Outputs land concatenated in job order;
 * dbeel_gpu_job_fetch_job slices job `job_idx`'s result back out with
 * index offsets rebased to its own run file. job_run/job_ingest/
 * job_fetch work on batched jobs unchanged (fetch returns the
 * concatenation). */
int dbeel_gpu_job_create_batched(const dbeel_run_view* runs, size_t n_runs,
                                 const uint32_t* runs_per_job,
                                 size_t n_jobs, int device,
                                 dbeel_gpu_job** out_job);
int dbeel_gpu_job_fetch_job(dbeel_gpu_job* job, size_t job_idx,
                            dbeel_compact_result* out);
/* Runs the pipeline; fills t (may be NULL) and the output sizes. */
int dbeel_gpu_job_run(dbeel_gpu_job* job, int keep_tombstones,
                      uint64_t* out_data_len, uint64_t* out_entries,
                      dbeel_compact_timings* t);
/* Copies the last run's outputs to host (engine-allocated). */
int dbeel_gpu_job_fetch(dbeel_gpu_job* job, dbeel_compact_result* out);
/* Builds the behavioral "DBLM" bloom over the last run's surviving keys
 * on the device (the Bloom::set-per-written-key step,
 * lsm_tree.rs:1049-1051; format in include/dbeel_lsm.h). Engine-allocated;
 * free with dbeel_gpu_bloom_free. */
int dbeel_gpu_job_bloom(dbeel_gpu_job* job, uint8_t** out_bytes,
                        uint64_t* out_len);
void dbeel_gpu_bloom_free(uint8_t* p);
void dbeel_gpu_job_destroy(dbeel_gpu_job* job);

/* ---- Streamed pinned ingest (north_star: "input SSTable runs are
 * pinned in host DRAM and streamed via hipMemcpyAsync with compute
 * overlap") ----
 *
 * dbeel_gpu_pin_host / unpin_host: page-lock a caller buffer
 * (hipHostRegister) so streamed copies run as true async DMA. A
 * production engine pins its run staging buffers once; pin before the
 * hot loop, not inside it.
 *
 * dbeel_gpu_job_ingest: re-uploads fresh run contents into an EXISTING
 * job's resident input slab, streaming each run's data in
 * entry-boundary-aligned chunks (default 64 MiB, env
 * DBEEL_STREAM_CHUNK_MB) on a dedicated copy stream while k_prepare
 * runs on already-arrived chunks on the compute stream — the
 * copy-engine/compute overlap the reference gets from its 16-deep
 * read-ahead DmaStreamReaders (lsm_tree.rs:974-993). Run shapes
 * (data_len/index_len per run) must match the job's. After a successful
 * ingest the job's prepare stage is already done: the next
 * dbeel_gpu_job_run skips it (the work happened, hidden behind the
 * PCIe transfer). stats (may be NULL) reports the wall ingest time and
 * how much prepare time was hidden. */
typedef struct {
    double ingest_ms;   /* wall: first copy enqueued -> all chunks +
                           prepare complete */
    double copy_ms;     /* pure H2D transfer time (copy stream span) */
    double prep_ms;     /* prepare kernel time, overlapped with copies */
    uint64_t bytes;     /* total bytes ingested (data + index) */
    uint64_t chunks;
} dbeel_ingest_stats;

int dbeel_gpu_pin_host(const void* ptr, size_t len);
int dbeel_gpu_unpin_host(const void* ptr);
int dbeel_gpu_job_ingest(dbeel_gpu_job* job, const dbeel_run_view* runs,
                         size_t n_runs, dbeel_ingest_stats* stats);

/* ---- Batched migration / iteration scan ----
 * GPU analogue of AsyncIter (lsm_tree.rs:141-282) feeding migration
 * (tasks/migration.rs:62-131): yields entries in the reference's yield
 * order — sstables ascending by index, entries in key order within each
 * (IterState walks index 0 upward, lsm_tree.rs:210-276) — with NO dedup
 * and NO tombstone filtering (the reference iterator yields every entry;
 * migration re-sends duplicates and resolves by timestamp at the
 * destination). Filters, combinable (entry kept iff it passes all):
 *   - key range [start_key, end_key), either bound NULL = unbounded;
 *   - murmur3_32(key, seed 0) hash ranges with wraparound semantics
 *     (hash_bytes shards.rs:99-101, murmur3 crate v0.5.2 restated;
 *     between_cmp tasks/migration.rs:54-60: start <= end means
 *     [start, end), end < start wraps to hash >= start || hash < end...
 *     precisely: kept iff hash < start || hash >= end when end < start).
 * Output = packed entries verbatim + rebuilt 16-B index records, like a
 * run file holding exactly the yielded entries. */
int dbeel_gpu_scan(const dbeel_run_view* runs, size_t n_runs,
                   const uint8_t* start_key, size_t start_key_len,
                   const uint8_t* end_key, size_t end_key_len,
                   const uint32_t* range_starts, const uint32_t* range_ends,
                   size_t n_ranges, int device, dbeel_compact_result* out);

/* ---- Sliced compaction (runs larger than HBM) ----
 * Splits the key space into slices that fit max_resident_bytes of input,
 * compacts each slice independently (equal keys never split: slice
 * boundaries are strict key pivots) and concatenates the outputs —
 * byte-identical to one whole-job compaction of the same runs.
 * max_resident_bytes = 0 picks a default from free device memory. */
int dbeel_gpu_compact_sliced(const dbeel_run_view* runs, size_t n_runs,
                             int keep_tombstones, int device,
                             uint64_t max_resident_bytes,
                             dbeel_compact_result* out);

#ifdef __cplusplus
}
#endif
#endif /* DBEEL_GPU_H */
