/* dbeel_lsm.h — file-level host layer over the compaction engine.
 *
 * Mirrors the file-system discipline of `LSMTree::compact`
 * (lsm_tree.rs:950-1156) and its crash-recovery replay
 * (lsm_tree.rs:424-438, run_compaction_action lsm_tree.rs:576-590), plus
 * the compaction trigger policy (tasks/compaction.rs:35-102), so that the
 * reference's artifact semantics (filenames `{index:020}.{ext}`
 * lsm_tree.rs:284-288, `.compact_*` staging files, the `.compact_action`
 * bincode journal with deletes-then-renames replay) keep working
 * unmodified above the GPU engine.
 *
 * Bloom: the reference's `.bloom` bytes are unpinnable (bloomfilter
 * v1.0.12 embeds random SipHash keys — SURVEY.md §8c); this layer writes
 * a behaviorally equivalent filter (zero false negatives, ~1% fp) in the
 * engine's own format ("DBLM" header), checked via dbeel_bloom_contains.
 */
#ifndef DBEEL_LSM_H
#define DBEEL_LSM_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Compact the runs at `indices` in `dir` into `output_index`, with the
 * reference's full artifact discipline:
 *   read {i:020}.data/.index -> GPU merge -> write
 *   {out:020}.compact_data/.compact_index (+ .compact_bloom when total
 *   input data > sstable_bloom_min_size, lsm_tree.rs:1026-1034) ->
 *   write {out:020}.compact_action journal (renames + deletes) ->
 *   perform renames -> delete inputs -> delete journal
 *   (lsm_tree.rs:1070-1153).
 * device: HIP ordinal. Returns 0 or a dbeel_gpu error code. */
int dbeel_lsm_compact(const char* dir, const uint64_t* indices,
                      size_t n_indices, uint64_t output_index,
                      int keep_tombstones, int device,
                      uint64_t sstable_bloom_min_size,
                      uint64_t* out_entries_written);

/* Replay any {n:020}.compact_action journals in dir: per journal, deletes
 * first (if present), then renames (if source present), then remove the
 * journal — idempotent compaction completion after a crash
 * (lsm_tree.rs:424-438, 576-590). Returns number of journals replayed in
 * *out_replayed (may be NULL). */
int dbeel_lsm_replay(const char* dir, uint32_t* out_replayed);

/* The compaction trigger policy of tasks/compaction.rs:35-102: discover
 * sstables ({i:020}.index), group by leading_zeros(entry_count), promote
 * groups whose combined count reaches a bigger size class, compact every
 * group with >= max(2, compaction_factor) members into the next odd
 * output index (+2 per group). Tombstones are dropped only when a group
 * covers EVERY live sstable: the reference's keep_tombstones = (i > 0)
 * over a HashMap enumeration (tasks/compaction.rs:82-92) is
 * nondeterministic and can resurrect deleted keys when a promoted group
 * of new runs enumerates first; this port implements the "only on the
 * final level" intent deterministically and safely. */
int dbeel_lsm_compact_tree(const char* dir, uint64_t compaction_factor,
                           int device, uint64_t sstable_bloom_min_size,
                           uint32_t* out_n_compactions);

/* Major compaction: merge EVERY live sstable in dir into one run at the
 * next odd output index, dropping tombstones (safe: nothing outside the
 * group can resurrect a deleted key). Bounds the tombstone buildup that
 * the conservative full-coverage rule in dbeel_lsm_compact_tree can
 * otherwise let accumulate (a steady-state tree may never form a
 * full-coverage group on its own — ADVICE r01); call periodically or
 * when space amplification matters. No-op (0 compactions) when fewer
 * than 2 sstables exist. */
int dbeel_lsm_major_compact(const char* dir, int device,
                            uint64_t sstable_bloom_min_size,
                            uint64_t* out_entries_written);

/* Membership test against an engine-format bloom file's bytes.
 * *out = 1 if possibly present, 0 if definitely absent. */
int dbeel_bloom_contains(const uint8_t* bloom_bytes, size_t bloom_len,
                         const uint8_t* key, size_t key_len, int* out);

#ifdef __cplusplus
}
#endif
#endif /* DBEEL_LSM_H */
