/* dbeel_lsm.cpp — file-level host layer (see include/dbeel_lsm.h).
 *
 * Pure host C++ (no device code). Restates, at the byte level:
 *   - filenames {index:020}.{ext}            lsm_tree.rs:284-288, mod.rs:21-30
 *   - the CompactionAction journal           lsm_tree.rs:73-77 (bincode
 *     fixint: Vec = u64 count, PathBuf = u64 len + utf8 bytes,
 *     utils/bincode.rs:9-16)
 *   - journal replay: deletes then renames   lsm_tree.rs:576-590
 *   - compact()'s artifact sequence          lsm_tree.rs:1070-1153
 *   - the trigger policy                     tasks/compaction.rs:35-102
 *   - bloom gating (input data > min size)   lsm_tree.rs:1026-1034
 */
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <string>
#include <vector>
#include <algorithm>

#include <dirent.h>
#include <sys/stat.h>
#include <unistd.h>

#include "../../include/dbeel_gpu.h"
#include "../../include/dbeel_lsm.h"

namespace {

std::string file_path(const std::string& dir, uint64_t index,
                      const char* ext) {
    char buf[64];
    snprintf(buf, sizeof buf, "%020llu.%s", (unsigned long long)index, ext);
    return dir + "/" + buf;
}

bool read_whole(const std::string& path, std::vector<uint8_t>& out) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) return false;
    fseek(f, 0, SEEK_END);
    long len = ftell(f);
    fseek(f, 0, SEEK_SET);
    out.resize(len < 0 ? 0 : (size_t)len);
    size_t rd = out.empty() ? 0 : fread(out.data(), 1, out.size(), f);
    fclose(f);
    return rd == out.size();
}

bool write_whole(const std::string& path, const uint8_t* data, size_t len) {
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) return false;
    size_t wr = len ? fwrite(data, 1, len, f) : 0;
    int rc = fclose(f);
    return wr == len && rc == 0;
}

bool exists(const std::string& p) {
    struct stat st;
    return stat(p.c_str(), &st) == 0;
}

/* files named ^(\d+)\.{ext}$ (create_file_path_regex, lsm_tree.rs:306-310) */
std::vector<uint64_t> scan_indices(const std::string& dir, const char* ext) {
    std::vector<uint64_t> out;
    DIR* d = opendir(dir.c_str());
    if (!d) return out;
    size_t elen = strlen(ext);
    while (dirent* e = readdir(d)) {
        const char* n = e->d_name;
        const char* dot = strchr(n, '.');
        if (!dot || dot == n) continue;
        if (strcmp(dot + 1, ext) != 0) continue;
        bool digits = true;
        for (const char* p = n; p < dot; p++)
            if (*p < '0' || *p > '9') digits = false;
        if (!digits) continue;
        out.push_back(strtoull(std::string(n, dot - n).c_str(), nullptr, 10));
        (void)elen;
    }
    closedir(d);
    std::sort(out.begin(), out.end());
    return out;
}

/* ---- bincode fixint LE helpers (utils/bincode.rs:9-16) ---- */
void put_u64(std::vector<uint8_t>& b, uint64_t v) {
    for (int i = 0; i < 8; i++) b.push_back((uint8_t)(v >> (8 * i)));
}
void put_str(std::vector<uint8_t>& b, const std::string& s) {
    put_u64(b, s.size());
    b.insert(b.end(), s.begin(), s.end());
}
bool get_u64(const uint8_t*& p, const uint8_t* end, uint64_t& v) {
    if (end - p < 8) return false;
    memcpy(&v, p, 8);
    p += 8;
    return true;
}
bool get_str(const uint8_t*& p, const uint8_t* end, std::string& s) {
    uint64_t n;
    if (!get_u64(p, end, n) || (uint64_t)(end - p) < n) return false;
    s.assign((const char*)p, n);
    p += n;
    return true;
}

/* ---- engine bloom: "DBLM" | ver | k | pad | n_bits | seed | bitmap ---- */
struct BloomHeader {
    char magic[4];
    uint32_t version;
    uint32_t k;
    uint32_t pad;
    uint64_t n_bits;
    uint64_t seed;
};
static_assert(sizeof(BloomHeader) == 32, "bloom header");

uint64_t fnv1a64(const uint8_t* p, size_t n, uint64_t seed) {
    uint64_t h = 1469598103934665603ull ^ seed;
    for (size_t i = 0; i < n; i++) {
        h ^= p[i];
        h *= 1099511628211ull;
    }
    /* final avalanche (splitmix64) */
    h ^= h >> 30;
    h *= 0xbf58476d1ce4e5b9ull;
    h ^= h >> 27;
    h *= 0x94d049bb133111ebull;
    h ^= h >> 31;
    return h;
}

} // namespace

extern "C" int dbeel_bloom_contains(const uint8_t* bloom_bytes,
                                    size_t bloom_len, const uint8_t* key,
                                    size_t key_len, int* out) {
    if (!bloom_bytes || !out || bloom_len < sizeof(BloomHeader))
        return DBEEL_ERR_INVALID_ARG;
    BloomHeader h;
    memcpy(&h, bloom_bytes, sizeof h);
    if (memcmp(h.magic, "DBLM", 4) != 0 || h.version != 1)
        return DBEEL_ERR_CORRUPT;
    if (bloom_len < sizeof h + (h.n_bits + 7) / 8) return DBEEL_ERR_CORRUPT;
    const uint8_t* bits = bloom_bytes + sizeof h;
    uint64_t h1 = fnv1a64(key, key_len, h.seed);
    uint64_t h2 = fnv1a64(key, key_len, h.seed ^ 0x9e3779b97f4a7c15ull) | 1;
    int present = 1;
    for (uint32_t j = 0; j < h.k; j++) {
        uint64_t bit = (h1 + j * h2) % h.n_bits;
        if (!(bits[bit / 8] & (1u << (bit % 8)))) {
            present = 0;
            break;
        }
    }
    *out = present;
    return DBEEL_OK;
}

extern "C" int dbeel_lsm_replay(const char* dir_c, uint32_t* out_replayed) {
    if (!dir_c) return DBEEL_ERR_INVALID_ARG;
    std::string dir(dir_c);
    uint32_t replayed = 0;
    for (uint64_t idx : scan_indices(dir, "compact_action")) {
        std::string jpath = file_path(dir, idx, "compact_action");
        std::vector<uint8_t> buf;
        if (!read_whole(jpath, buf)) continue;
        const uint8_t* p = buf.data();
        const uint8_t* end = p + buf.size();
        /* `while let Ok(action) = deserialize_from(...)`
         * (lsm_tree.rs:432-436): consume actions until parse fails */
        for (;;) {
            const uint8_t* save = p;
            uint64_t n_renames;
            std::vector<std::pair<std::string, std::string>> renames;
            std::vector<std::string> deletes;
            bool ok = get_u64(p, end, n_renames);
            for (uint64_t i = 0; ok && i < n_renames; i++) {
                std::string a, b;
                ok = get_str(p, end, a) && get_str(p, end, b);
                if (ok) renames.emplace_back(a, b);
            }
            uint64_t n_deletes = 0;
            ok = ok && get_u64(p, end, n_deletes);
            for (uint64_t i = 0; ok && i < n_deletes; i++) {
                std::string a;
                ok = get_str(p, end, a);
                if (ok) deletes.push_back(a);
            }
            if (!ok) {
                p = save;
                break;
            }
            /* run_compaction_action: deletes first, then renames
             * (lsm_tree.rs:576-590) */
            for (auto& d : deletes)
                if (exists(d)) unlink(d.c_str());
            for (auto& rn : renames)
                if (exists(rn.first))
                    if (rename(rn.first.c_str(), rn.second.c_str()) != 0)
                        return DBEEL_ERR_IO;
        }
        unlink(jpath.c_str());
        replayed++;
    }
    if (out_replayed) *out_replayed = replayed;
    return DBEEL_OK;
}

extern "C" int dbeel_lsm_compact(const char* dir_c, const uint64_t* indices,
                                 size_t n_indices, uint64_t output_index,
                                 int keep_tombstones, int device,
                                 uint64_t sstable_bloom_min_size,
                                 uint64_t* out_entries_written) {
    if (!dir_c || !indices || n_indices == 0) return DBEEL_ERR_INVALID_ARG;
    std::string dir(dir_c);

    /* read input runs (replaces the DmaStreamReader loop,
     * lsm_tree.rs:974-993) */
    std::vector<std::vector<uint8_t>> datas(n_indices), idxs(n_indices);
    std::vector<dbeel_run_view> views(n_indices);
    uint64_t total_input_data = 0;
    for (size_t i = 0; i < n_indices; i++) {
        if (!read_whole(file_path(dir, indices[i], "data"), datas[i]) ||
            !read_whole(file_path(dir, indices[i], "index"), idxs[i]))
            return DBEEL_ERR_IO;
        views[i] = {datas[i].data(), datas[i].size(), idxs[i].data(),
                    idxs[i].size()};
        total_input_data += datas[i].size();
    }

    dbeel_gpu_job* job = nullptr;
    int rc = dbeel_gpu_job_create(views.data(), n_indices, device, &job);
    if (rc != DBEEL_OK) return rc;
    rc = dbeel_gpu_job_run(job, keep_tombstones, nullptr, nullptr, nullptr);
    dbeel_compact_result res{};
    if (rc == DBEEL_OK) rc = dbeel_gpu_job_fetch(job, &res);

    /* bloom only when input data exceeds the threshold
     * (lsm_tree.rs:1026-1034; default 1 MiB, mod.rs:19) — built on the
     * device from the job's resident output */
    bool with_bloom = total_input_data > sstable_bloom_min_size;
    uint8_t* bloom = nullptr;
    uint64_t bloom_len = 0;
    if (rc == DBEEL_OK && with_bloom)
        rc = dbeel_gpu_job_bloom(job, &bloom, &bloom_len);
    dbeel_gpu_job_destroy(job);
    if (rc != DBEEL_OK) {
        dbeel_gpu_result_free(&res);
        dbeel_gpu_bloom_free(bloom);
        return rc;
    }

    std::string cd = file_path(dir, output_index, "compact_data");
    std::string ci = file_path(dir, output_index, "compact_index");
    std::string cb = file_path(dir, output_index, "compact_bloom");
    bool ok = write_whole(cd, res.data, res.data_len) &&
              write_whole(ci, res.index, res.index_len);
    if (ok && with_bloom) ok = write_whole(cb, bloom, bloom_len);
    uint64_t written = res.entries_written;
    dbeel_gpu_result_free(&res);
    dbeel_gpu_bloom_free(bloom);
    if (!ok) return DBEEL_ERR_IO;

    /* journal (CompactionAction, lsm_tree.rs:1090-1105): renames of all
     * three staging files, deletes of every input's data/index/bloom */
    std::vector<uint8_t> journal;
    put_u64(journal, 3); /* renames */
    put_str(journal, cd);
    put_str(journal, file_path(dir, output_index, "data"));
    put_str(journal, ci);
    put_str(journal, file_path(dir, output_index, "index"));
    put_str(journal, cb);
    put_str(journal, file_path(dir, output_index, "bloom"));
    put_u64(journal, n_indices * 3); /* deletes */
    std::vector<std::string> deletes;
    for (size_t i = 0; i < n_indices; i++) {
        deletes.push_back(file_path(dir, indices[i], "data"));
        deletes.push_back(file_path(dir, indices[i], "index"));
        deletes.push_back(file_path(dir, indices[i], "bloom"));
    }
    for (auto& d : deletes) put_str(journal, d);
    std::string jpath = file_path(dir, output_index, "compact_action");
    if (!write_whole(jpath, journal.data(), journal.size()))
        return DBEEL_ERR_IO;

    /* crash-injection hook for recovery tests (the flow_events analogue,
     * reference flow_events.rs:7-14): stop after the journal is durable —
     * dbeel_lsm_replay must complete the compaction idempotently */
    if (getenv("DBEEL_LSM_CRASH_AFTER_JOURNAL")) {
        if (out_entries_written) *out_entries_written = written;
        return DBEEL_OK;
    }

    /* renames (source may be absent — no bloom), then deletes, then the
     * journal itself (lsm_tree.rs:1107-1153) */
    struct {
        std::string from, to;
    } renames[3] = {{cd, file_path(dir, output_index, "data")},
                    {ci, file_path(dir, output_index, "index")},
                    {cb, file_path(dir, output_index, "bloom")}};
    for (auto& rn : renames)
        if (exists(rn.from))
            if (rename(rn.from.c_str(), rn.to.c_str()) != 0)
                return DBEEL_ERR_IO;
    for (auto& d : deletes)
        if (exists(d)) unlink(d.c_str());
    unlink(jpath.c_str());

    if (out_entries_written) *out_entries_written = written;
    return DBEEL_OK;
}

extern "C" int dbeel_lsm_major_compact(const char* dir_c, int device,
                                       uint64_t sstable_bloom_min_size,
                                       uint64_t* out_entries_written) {
    if (!dir_c) return DBEEL_ERR_INVALID_ARG;
    std::string dir(dir_c);
    std::vector<uint64_t> idxs = scan_indices(dir, "index");
    if (out_entries_written) *out_entries_written = 0;
    if (idxs.size() < 2) return DBEEL_OK;
    uint64_t out_index = 1;
    for (uint64_t i : idxs)
        if (i % 2 == 1 && i + 2 > out_index) out_index = i + 2;
    /* the group covers every live sstable -> dropping tombstones is safe */
    return dbeel_lsm_compact(dir_c, idxs.data(), idxs.size(), out_index,
                             /*keep_tombstones=*/0, device,
                             sstable_bloom_min_size, out_entries_written);
}

extern "C" int dbeel_lsm_compact_tree(const char* dir_c,
                                      uint64_t compaction_factor, int device,
                                      uint64_t sstable_bloom_min_size,
                                      uint32_t* out_n_compactions) {
    if (!dir_c) return DBEEL_ERR_INVALID_ARG;
    if (compaction_factor < 2) { /* MIN_COMPACTION_FACTOR,
                                    tasks/compaction.rs:13,105-107 */
        if (out_n_compactions) *out_n_compactions = 0;
        return DBEEL_OK;
    }
    std::string dir(dir_c);

    /* discover sstables + entry counts (sstable_indices_and_sizes;
     * size = index bytes / 16, lsm_tree.rs:450-453) */
    struct Tbl {
        uint64_t index, count;
    };
    std::vector<Tbl> tables;
    for (uint64_t idx : scan_indices(dir, "index")) {
        struct stat st;
        if (stat(file_path(dir, idx, "index").c_str(), &st) != 0) continue;
        tables.push_back({idx, (uint64_t)st.st_size / 16});
    }

    /* next odd output index (tasks/compaction.rs:38-43) */
    uint64_t out_index = 1;
    for (auto& t : tables)
        if (t.index % 2 == 1 && t.index + 2 > out_index)
            out_index = t.index + 2;

    auto lz = [](uint64_t v) {
        return v ? (uint64_t)__builtin_clzll(v) : 64ull;
    };

    /* group by leading_zeros(count), then promote groups whose combined
     * count reaches a bigger size class (tasks/compaction.rs:45-80).
     * Processing order: ascending leading_zeros = biggest first —
     * deterministic stand-in for the reference's HashMap order. */
    std::vector<std::pair<uint64_t, std::vector<uint64_t>>> groups; /* lz -> indices */
    for (auto& t : tables) {
        uint64_t o = lz(t.count);
        auto it = std::find_if(groups.begin(), groups.end(),
                               [&](auto& g) { return g.first == o; });
        if (it == groups.end())
            groups.push_back({o, {t.index}});
        else
            it->second.push_back(t.index);
    }
    /* descending size_order (= ascending size) like groups.sort in the
     * reference (tasks/compaction.rs:52) */
    std::sort(groups.begin(), groups.end(),
              [](auto& a, auto& b) { return a.first > b.first; });

    auto count_of = [&](uint64_t idx) {
        for (auto& t : tables)
            if (t.index == idx) return t.count;
        return (uint64_t)0;
    };

    std::vector<std::pair<uint64_t, std::vector<uint64_t>>> optimized;
    for (auto& [size_order, items0] : groups) {
        std::vector<uint64_t> items = items0;
        auto it = std::find_if(optimized.begin(), optimized.end(),
                               [&](auto& g) { return g.first == size_order; });
        if (it != optimized.end()) {
            items.insert(items.end(), it->second.begin(), it->second.end());
            optimized.erase(it);
        }
        uint64_t sum = 0;
        for (uint64_t i : items) sum += count_of(i);
        uint64_t est = lz(sum);
        uint64_t opt = est < size_order ? est : size_order;
        auto it2 = std::find_if(optimized.begin(), optimized.end(),
                                [&](auto& g) { return g.first == opt; });
        if (it2 == optimized.end())
            optimized.push_back({opt, items});
        else
            it2->second.insert(it2->second.end(), items.begin(),
                               items.end());
    }
    /* deterministic final order: biggest size class first */
    std::sort(optimized.begin(), optimized.end(),
              [](auto& a, auto& b) { return a.first < b.first; });

    /* Tombstone rule: the reference drops tombstones "only on the final
     * level" via keep_tombstones = (i > 0) over a HashMap enumeration
     * (tasks/compaction.rs:82-92) — nondeterministic, and a promoted
     * group of NEW runs can enumerate first and drop tombstones while
     * older runs still hold the deleted keys (data resurrection). This
     * port implements the comment's intent deterministically: drop
     * tombstones only when the group covers EVERY live sstable (nothing
     * outside it can resurrect). Documented divergence. */
    uint32_t done = 0;
    for (size_t i = 0; i < optimized.size(); i++) {
        auto& items = optimized[i].second;
        if (items.size() < 2 || items.size() < compaction_factor) continue;
        std::vector<uint64_t> idxs = items;
        std::sort(idxs.begin(), idxs.end()); /* ascending sstable index —
                                                the tie-break needs it */
        int keep = (items.size() == tables.size()) ? 0 : 1;
        int rc = dbeel_lsm_compact(dir_c, idxs.data(), idxs.size(),
                                   out_index, keep, device,
                                   sstable_bloom_min_size, nullptr);
        if (rc != DBEEL_OK) return rc;
        out_index += 2;
        done++;
    }
    if (out_n_compactions) *out_n_compactions = done;
    return DBEEL_OK;
}
