/* abi_demo.c — plain-C consumer of the engine's C ABI.
 *
 * Proves the drop-in boundary is host-language-free: this file compiles
 * with gcc (no hipcc, no C++), links only libdbeel_gpu.so, and performs a
 * compaction + a lookup + a scan through exactly the declarations of
 * include/dbeel_gpu.h — the same calls dbeel's Rust FFI stub
 *  would make through cgo/bindgen.
 *
 * Build (tests/test_abi.py does this on CPU; a GPU test runs it):
 *   gcc -O2 -I include tools/abi_demo.c -L dbeel_amd -ldbeel_gpu \
 *       -Wl,-rpath,'$ORIGIN/../dbeel_amd' -o abi_demo
 *
 * Exit codes: 0 ok, 1 ABI error, 2 wrong result.
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "dbeel_gpu.h"

/* Build one tiny run in the wire format (bincode fixint LE —
 * include/dbeel_gpu.h header comment). */
static size_t put_entry(uint8_t* d, uint8_t* idx, size_t* ecount,
                        size_t off, const char* key, const char* val,
                        long long ts) {
    size_t kl = strlen(key), vl = strlen(val);
    uint64_t kl64 = kl, vl64 = vl;
    uint8_t* p = d + off;
    memcpy(p, &kl64, 8);
    memcpy(p + 8, key, kl);
    memcpy(p + 8 + kl, &vl64, 8);
    memcpy(p + 16 + kl, val, vl);
    long long ts_lo = ts, ts_hi = ts < 0 ? -1 : 0;
    memcpy(p + 16 + kl + vl, &ts_lo, 8);
    memcpy(p + 24 + kl + vl, &ts_hi, 8);
    uint64_t o = off;
    uint32_t ks = (uint32_t)(8 + kl), fs = (uint32_t)(32 + kl + vl);
    uint8_t* r = idx + (*ecount) * 16;
    memcpy(r, &o, 8);
    memcpy(r + 8, &ks, 4);
    memcpy(r + 12, &fs, 4);
    (*ecount)++;
    return off + fs;
}

int main(void) {
    /* run 0: a=1 (ts 10), b=2 (ts 11); run 1: b=NEW (ts 20), c=3 (ts 21),
     * d tombstone (ts 22). Expected merge (drop tombstones):
     * a=1, b=NEW, c=3. */
    static uint8_t d0[4096], i0[256], d1[4096], i1[256];
    size_t n0 = 0, n1 = 0, o = 0;
    o = put_entry(d0, i0, &n0, o, "a", "1", 10);
    o = put_entry(d0, i0, &n0, o, "b", "2", 11);
    size_t len0 = o;
    o = 0;
    o = put_entry(d1, i1, &n1, o, "b", "NEW", 20);
    o = put_entry(d1, i1, &n1, o, "c", "3", 21);
    o = put_entry(d1, i1, &n1, o, "d", "", 22);
    size_t len1 = o;

    dbeel_run_view runs[2] = {
        {d0, len0, i0, n0 * 16},
        {d1, len1, i1, n1 * 16},
    };

    dbeel_compact_result res;
    int rc = dbeel_gpu_compact(runs, 2, /*keep_tombstones=*/0,
                               /*device=*/0, &res);
    if (rc != DBEEL_OK) {
        fprintf(stderr, "compact rc=%d: %s\n", rc, dbeel_gpu_last_error());
        return 1;
    }
    if (res.entries_written != 3) {
        fprintf(stderr, "expected 3 survivors, got %llu\n",
                (unsigned long long)res.entries_written);
        return 2;
    }
    /* survivor 1 must be key "b" with value "NEW" (newest wins) */
    uint64_t off1;
    uint32_t ks1, fs1;
    memcpy(&off1, res.index + 16, 8);
    memcpy(&ks1, res.index + 24, 4);
    memcpy(&fs1, res.index + 28, 4);
    const uint8_t* e = res.data + off1;
    if (ks1 != 9 || memcmp(e + 8, "b", 1) != 0 ||
        memcmp(e + 8 + 1 + 8, "NEW", 3) != 0) {
        fprintf(stderr, "survivor 1 is not b=NEW\n");
        dbeel_gpu_result_free(&res);
        return 2;
    }
    dbeel_gpu_result_free(&res);

    /* lookup: newest-index-first — "b" resolves from run 1 */
    uint8_t keys[2] = {'b', 'd'};
    uint64_t koff[3] = {0, 1, 2};
    dbeel_lookup_hit hits[2];
    rc = dbeel_gpu_lookup(runs, 2, keys, koff, 2, 0, hits);
    if (rc != DBEEL_OK) {
        fprintf(stderr, "lookup rc=%d: %s\n", rc, dbeel_gpu_last_error());
        return 1;
    }
    if (hits[0].run != 1 || hits[0].value_len != 3 ||
        hits[1].run != 1 || !hits[1].is_tombstone) {
        fprintf(stderr, "lookup results wrong\n");
        return 2;
    }

    /* scan: full iteration yields all 5 entries, runs ascending */
    rc = dbeel_gpu_scan(runs, 2, NULL, 0, NULL, 0, NULL, NULL, 0, 0,
                        &res);
    if (rc != DBEEL_OK) {
        fprintf(stderr, "scan rc=%d: %s\n", rc, dbeel_gpu_last_error());
        return 1;
    }
    if (res.entries_written != 5) {
        fprintf(stderr, "scan expected 5 entries, got %llu\n",
                (unsigned long long)res.entries_written);
        dbeel_gpu_result_free(&res);
        return 2;
    }
    dbeel_gpu_result_free(&res);

    printf("abi_demo OK: compact(3 survivors, b=NEW) + lookup + scan "
           "through the plain-C ABI\n");
    return 0;
}
