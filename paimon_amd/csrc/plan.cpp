// libpaimon_hip host side: session/plan lifecycle, planning
// (IntervalPartition restatement), file staging (footer parse, page tables,
// H2D of encoded column chunks), and the read_next device pipeline.
//
// Replaces, for this path, MergeFileSplitRead.createReader
// (paimon-core/.../operation/MergeFileSplitRead.java:242-270) +
// MergeTreeReaders (mergetree/MergeTreeReaders.java:40-102) +
// KeyValueFileReaderFactory (io/KeyValueFileReaderFactory.java:127-209).
// The KeyValue row layout contract (key cols | _SEQUENCE_NUMBER |
// _VALUE_KIND | value cols) follows KeyValueSerializer.java:34-99.

#include <hip/hip_runtime.h>

#include <algorithm>
#include <chrono>
#include <cstring>
#include <fstream>
#include <memory>
#include <queue>
#include <string>
#include <unordered_map>
#include <vector>

#include "../../include/paimon_hip.h"
#include "codec.h"
#include "zstd_core.h"
#include <map>
#include "common.h"
#include "json.h"
#include "kernels.h"
#include "orc_meta.h"
#include "parquet_meta.h"
#include "parquet_write.h"

#include <dlfcn.h>

namespace pmh {

std::string &last_error() {
    thread_local std::string e;
    return e;
}

#define HIP_TRY(expr)                                                     \
    do {                                                                  \
        hipError_t _e = (expr);                                           \
        if (_e != hipSuccess) {                                           \
            set_error("HIP error %s at %s:%d: %s", hipGetErrorString(_e), \
                      __FILE__, __LINE__, #expr);                         \
            return false;                                                 \
        }                                                                 \
    } while (0)

// ---------------------------------------------------------------- codecs

static bool zstd_decompress(const uint8_t *src, size_t src_n, uint8_t *dst,
                            size_t dst_n) {
    std::string err;
    if (!zstd_decompress_exact(src, src_n, dst, dst_n, err)) {
        set_error("%s", err.c_str());
        return false;
    }
    return true;
}



// ------------------------------------------------ on-GPU zstd page batch
//
// Decode all zstd pages of one column chunk on the GPU (k_zstd_pages, one
// wavefront per page frame) and copy the uncompressed image back to the
// host staging buffer. v1 keeps the host round trip so every downstream
// staging path (def-level peeks, dict pages, DELTA prescan) is unchanged;
// the PLAIN fast path staying device-resident is the follow-up. Returns
// false (WITHOUT set_error) to fall back to the host codec.
static bool gpu_zstd_enabled() {
    static int on = -1;
    if (on < 0) {
        const char *e = getenv("PMH_GPU_ZSTD");
        on = !(e && e[0] == '0');
        if (on) {
            int ndev = 0;
            if (hipGetDeviceCount(&ndev) != hipSuccess || ndev == 0) on = 0;
        }
    }
    return on;
}

struct ZstdBatchPage {
    int64_t src_off;  // into src_base
    int64_t src_len;
    int64_t dst_off;  // into the host image
    int64_t dst_len;
};

static bool gpu_zstd_batch_dev(
    const uint8_t *src_base, int64_t src_total,
    const std::vector<ZstdBatchPage> &pages, ::pmh_plan_t *plan,
    uint8_t **dev_out, int64_t out_total,
    const std::vector<std::pair<int64_t, int64_t>> &host_rngs,
    uint8_t *host_out);


static bool gpu_zstd_batch(const uint8_t *src_base, int64_t src_total,
                           const std::vector<ZstdBatchPage> &pages,
                           uint8_t *host_out, int64_t out_total) {
    if (pages.empty()) return true;
    int n = (int)pages.size();
    uint8_t *d_src = nullptr, *d_dst = nullptr, *d_scr = nullptr;
    ZstdJob *d_jobs = nullptr;
    int64_t *d_st = nullptr;
    std::vector<ZstdJob> jobs(n);
    for (int i = 0; i < n; i++) {
        jobs[i] = {(uint64_t)pages[i].src_off, (uint64_t)pages[i].dst_off,
                   (uint32_t)pages[i].src_len, (uint32_t)pages[i].dst_len};
    }
    bool ok = false;
    std::vector<int64_t> st(n);
    do {
        if (hipMalloc(&d_src, src_total) != hipSuccess) break;
        if (hipMalloc(&d_dst, out_total) != hipSuccess) break;
        if (hipMalloc(&d_scr, (size_t)n * PZ_SLOT) != hipSuccess) break;
        if (hipMalloc(&d_jobs, n * sizeof(ZstdJob)) != hipSuccess) break;
        if (hipMalloc(&d_st, n * 8) != hipSuccess) break;
        if (hipMemcpy(d_src, src_base, src_total, hipMemcpyHostToDevice) !=
            hipSuccess)
            break;
        if (hipMemcpy(d_jobs, jobs.data(), n * sizeof(ZstdJob),
                      hipMemcpyHostToDevice) != hipSuccess)
            break;
        if (pmh_launch_zstd_pages(d_src, d_jobs, n, d_dst, d_scr, d_st,
                                  nullptr) != hipSuccess)
            break;
        if (hipMemcpy(st.data(), d_st, n * 8, hipMemcpyDeviceToHost) !=
            hipSuccess)
            break;
        bool all = true;
        for (int i = 0; i < n; i++)
            if (st[i] != pages[i].dst_len) all = false;
        if (!all) break;
        if (hipMemcpy(host_out, d_dst, out_total, hipMemcpyDeviceToHost) !=
            hipSuccess)
            break;
        ok = true;
    } while (0);
    if (d_src) (void)hipFree(d_src);
    if (d_dst) (void)hipFree(d_dst);
    if (d_scr) (void)hipFree(d_scr);
    if (d_jobs) (void)hipFree(d_jobs);
    if (d_st) (void)hipFree(d_st);
    return ok;
}


// GPU batch page compression for the parquet write-back (k_zstd_compress).
// Returns false (no set_error) to fall back to the host codec.
// Opt-in (PMH_GPU_ZSTD_ENC=1): the v0 block compressor is spec-valid and
// wave-parallel, but the measured write-side A/B (profiles/ r02 zenc_ab)
// has host libzstd 4x faster AND ~2.5x smaller (huffman literals + a real
// parser beat predefined-FSE + greedy matching on typical column data) —
// so the HOST codec stays the write-side default. The debug entry and
// tests exercise the GPU path regardless.
bool pw_gpu_zstd_enc_enabled() {
    // read per call (cheap; tests toggle it within one process)
    const char *e = getenv("PMH_GPU_ZSTD_ENC");
    return e && e[0] == '1' && gpu_zstd_enabled();
}

bool pw_gpu_zstd_compress(const std::vector<std::string> &payloads,
                          std::vector<std::vector<uint8_t>> &outs) {
    if (!gpu_zstd_enabled() || payloads.empty()) return false;
    // one job per 128 KB BLOCK (pages can be multi-MB: block jobs keep
    // thousands of waves busy); the host stitches blocks into frames
    struct BlockRef {
        int page;
        int64_t dst_off;
    };
    std::vector<ZstdJob> jobs;
    std::vector<BlockRef> refs;
    int64_t src_total = 0, dst_total = 0;
    for (size_t i = 0; i < payloads.size(); i++) {
        int64_t sz = (int64_t)payloads[i].size();
        int64_t off = 0;
        while (off < sz) {
            int64_t bn = sz - off < PZ_BLOCK_MAX ? sz - off : PZ_BLOCK_MAX;
            int last = off + bn >= sz;
            int64_t cap = bn + (bn >> 8) + 256;
            ZstdJob j{(uint64_t)(src_total + off), (uint64_t)dst_total,
                      (uint32_t)bn,
                      (uint32_t)cap | (last ? 0x80000000u : 0u)};
            jobs.push_back(j);
            refs.push_back({(int)i, dst_total});
            dst_total += cap;
            off += bn;
        }
        src_total += sz;
    }
    int n = (int)jobs.size();
    if (n == 0) {  // all payloads empty: frames assemble host-side below
        n = 0;
    }
    std::vector<uint8_t> src_host(src_total ? src_total : 1);
    {
        int64_t o = 0;
        for (size_t i = 0; i < payloads.size(); i++) {
            memcpy(src_host.data() + o, payloads[i].data(),
                   payloads[i].size());
            o += (int64_t)payloads[i].size();
        }
    }
    uint8_t *d_src = nullptr, *d_dst = nullptr, *d_scr = nullptr;
    ZstdJob *d_jobs = nullptr;
    int64_t *d_st = nullptr;
    bool ok = false;
    std::vector<int64_t> st(n);
    do {
        if (hipMalloc(&d_src, src_host.size()) != hipSuccess) break;
        if (hipMalloc(&d_dst, dst_total ? dst_total : 1) != hipSuccess)
            break;
        if (hipMalloc(&d_scr, (size_t)n * sizeof(PzEnc)) != hipSuccess)
            break;
        if (hipMalloc(&d_jobs, n * sizeof(ZstdJob)) != hipSuccess) break;
        if (hipMalloc(&d_st, n * 8) != hipSuccess) break;
        if (hipMemcpy(d_src, src_host.data(), src_host.size(),
                      hipMemcpyHostToDevice) != hipSuccess)
            break;
        if (hipMemcpy(d_jobs, jobs.data(), n * sizeof(ZstdJob),
                      hipMemcpyHostToDevice) != hipSuccess)
            break;
        if (pmh_launch_zstd_compress(d_src, d_jobs, n, d_dst, d_scr, d_st,
                                     nullptr) != hipSuccess)
            break;
        if (hipMemcpy(st.data(), d_st, n * 8, hipMemcpyDeviceToHost) !=
            hipSuccess)
            break;
        bool all = true;
        for (int i = 0; i < n; i++)
            if (st[i] <= 0) all = false;
        if (!all) break;
        // stitch blocks into frames: header + concatenated block outputs
        std::vector<uint8_t> blob(dst_total ? dst_total : 1);
        if (dst_total &&
            hipMemcpy(blob.data(), d_dst, dst_total,
                      hipMemcpyDeviceToHost) != hipSuccess)
            break;
        outs.assign(payloads.size(), {});
        for (size_t i = 0; i < payloads.size(); i++) {
            uint8_t hdr[16];
            int hn = pz_frame_header(hdr, (int64_t)payloads[i].size());
            outs[i].insert(outs[i].end(), hdr, hdr + hn);
            if (payloads[i].empty()) {
                uint8_t raw0[3] = {1, 0, 0};  // empty raw last block
                outs[i].insert(outs[i].end(), raw0, raw0 + 3);
            }
        }
        for (int j = 0; j < n; j++) {
            auto &o = outs[refs[j].page];
            o.insert(o.end(), blob.data() + refs[j].dst_off,
                     blob.data() + refs[j].dst_off + st[j]);
        }
        ok = true;
    } while (0);
    if (d_src) (void)hipFree(d_src);
    if (d_dst) (void)hipFree(d_dst);
    if (d_scr) (void)hipFree(d_scr);
    if (d_jobs) (void)hipFree(d_jobs);
    if (d_st) (void)hipFree(d_st);
    return ok;
}

// ---------------------------------------------------- deletion vectors
//
// Paimon deletion vectors: per data file, a RoaringBitmap32
// of deleted row positions stored in a DV index file at (offset, length)
// (DeletionFile; deletionvectors/BitmapDeletionVector.java:98-112 wrapper =
// [i32 BE size][i32 BE magic 1581511376][roaring bytes][i32 BE crc];
// DeletionVector.read :101-118). The bitmap itself is the portable Roaring
// serialization (little-endian; cookie 12347 = no run containers, cookie
// low-16 12346 = with runs + bitset of run containers).
static bool parse_roaring32(const uint8_t *p, int64_t len,
                            std::vector<uint32_t> &out, std::string &err) {
    auto rd16 = [&](int64_t o) { return (uint32_t)p[o] | ((uint32_t)p[o + 1] << 8); };
    auto rd32 = [&](int64_t o) {
        return (uint32_t)p[o] | ((uint32_t)p[o + 1] << 8) |
               ((uint32_t)p[o + 2] << 16) | ((uint32_t)p[o + 3] << 24);
    };
    if (len < 8) { err = "roaring: short"; return false; }
    uint32_t cookie = rd32(0);
    int64_t o = 4;
    int32_t n_cont;
    bool has_run = false;
    std::vector<uint8_t> run_flags;
    if ((cookie & 0xFFFF) == 12346) {
        has_run = true;
        n_cont = (int32_t)(cookie >> 16) + 1;
        int64_t rb = (n_cont + 7) / 8;
        run_flags.assign(p + o, p + o + rb);
        o += rb;
    } else if (cookie == 12347) {
        n_cont = (int32_t)rd32(o);
        o += 4;
    } else {
        err = "roaring: bad cookie";
        return false;
    }
    std::vector<uint32_t> keys(n_cont), cards(n_cont);
    for (int i = 0; i < n_cont; i++) {
        keys[i] = rd16(o);
        cards[i] = rd16(o + 2) + 1;
        o += 4;
    }
    const bool has_offsets = !has_run || n_cont >= 4;
    if (has_offsets) o += 4 * (int64_t)n_cont;  // container offsets (unused)
    for (int i = 0; i < n_cont; i++) {
        const uint32_t hi = keys[i] << 16;
        const bool is_run =
            has_run && (run_flags[i / 8] >> (i % 8)) & 1;
        if (is_run) {
            uint32_t n_runs = rd16(o);
            o += 2;
            for (uint32_t r = 0; r < n_runs; r++) {
                uint32_t start = rd16(o), rl = rd16(o + 2);
                o += 4;
                for (uint32_t v = start; v <= start + rl; v++)
                    out.push_back(hi | v);
            }
        } else if (cards[i] > 4096) {  // bitmap container: 8 KB
            for (int w = 0; w < 1024; w++) {
                uint64_t word = 0;
                for (int b = 0; b < 8; b++)
                    word |= (uint64_t)p[o + w * 8 + b] << (8 * b);
                while (word) {
                    int bit = __builtin_ctzll(word);
                    out.push_back(hi | (uint32_t)(w * 64 + bit));
                    word &= word - 1;
                }
            }
            o += 8192;
        } else {  // array container
            for (uint32_t v = 0; v < cards[i]; v++)
                out.push_back(hi | rd16(o + 2 * v));
            o += 2 * (int64_t)cards[i];
        }
        if (o > len) { err = "roaring: overrun"; return false; }
    }
    return true;
}

// Read one file's DV from its index file slice; returns deleted positions.
static bool load_deletion_vector(const std::string &path, int64_t offset,
                                 int64_t length,
                                 std::vector<uint32_t> &out) {
    FILE *f = fopen(path.c_str(), "rb");
    if (!f) {
        set_error("cannot open deletion vector file %s", path.c_str());
        return false;
    }
    if (length <= 0) {  // whole file
        fseek(f, 0, SEEK_END);
        length = ftell(f) - offset;
    }
    std::vector<uint8_t> buf(length);
    fseek(f, offset, SEEK_SET);
    bool ok = fread(buf.data(), 1, length, f) == (size_t)length;
    fclose(f);
    if (!ok || length < 12) {
        set_error("deletion vector read failed (%s @%lld+%lld)",
                  path.c_str(), (long long)offset, (long long)length);
        return false;
    }
    auto be32 = [&](int64_t o) {
        return ((uint32_t)buf[o] << 24) | ((uint32_t)buf[o + 1] << 16) |
               ((uint32_t)buf[o + 2] << 8) | (uint32_t)buf[o + 3];
    };
    uint32_t size = be32(0);
    if (be32(4) != 1581511376u) {
        set_error("deletion vector magic mismatch in %s", path.c_str());
        return false;
    }
    if ((int64_t)size + 8 > length) {
        set_error("deletion vector truncated in %s", path.c_str());
        return false;
    }
    std::string err;
    if (!parse_roaring32(buf.data() + 8, size - 4, out, err)) {
        set_error("%s (%s)", err.c_str(), path.c_str());
        return false;
    }
    return true;
}

// ------------------------------------------- IntervalPartition (restated)

struct FileDesc {
    std::string dv_path;  // deletion vector (DeletionFile); empty = none
    int64_t dv_offset = 0;
    int64_t dv_length = 0;
    std::string path;
    int64_t row_count = 0;
    int64_t min_key = 0;
    int64_t max_key = 0;
    int level = 0;
    int input_index = 0;
};

// Restatement of IntervalPartition (mergetree/compact/IntervalPartition.java
// :40-126) for int64 keys: sort files by (minKey, maxKey); cut sections
// where minKey exceeds the running right bound; within a section, greedily
// pack files into the fewest sorted runs via a min-heap on each run's last
// maxKey.
static std::vector<std::vector<std::vector<FileDesc>>> interval_partition(
    std::vector<FileDesc> files, bool level_pure_runs = false) {
    std::sort(files.begin(), files.end(), [](const FileDesc &a, const FileDesc &b) {
        if (a.min_key != b.min_key) return a.min_key < b.min_key;
        return a.max_key < b.max_key;
    });
    std::vector<std::vector<std::vector<FileDesc>>> result;
    std::vector<FileDesc> section;
    int64_t bound = 0;
    bool has_bound = false;

    auto pack = [level_pure_runs](const std::vector<FileDesc> &metas) {
        // min-heap of runs keyed by last file's maxKey (IntervalPartition.java:93-125)
        auto cmp = [](const std::vector<FileDesc> &a, const std::vector<FileDesc> &b) {
            return a.back().max_key > b.back().max_key;  // min-heap
        };
        std::priority_queue<std::vector<FileDesc>, std::vector<std::vector<FileDesc>>,
                            decltype(cmp)>
            q(cmp);
        q.push({metas[0]});
        for (size_t i = 1; i < metas.size(); i++) {
            auto top = q.top();
            q.pop();
            if (metas[i].min_key > top.back().max_key &&
                (!level_pure_runs ||
                 metas[i].level == top.back().level)) {
                // level_pure_runs (changelog mode): records carry the level
                // of the FILE they were read from (KeyValue.setLevel per
                // file reader), so a run must not chain files of different
                // levels — start a new run instead (same merge semantics)
                top.push_back(metas[i]);
            } else {
                q.push({metas[i]});
            }
            q.push(top);
        }
        std::vector<std::vector<FileDesc>> runs;
        while (!q.empty()) {
            runs.push_back(q.top());
            q.pop();
        }
        return runs;
    };

    for (const auto &meta : files) {
        if (!section.empty() && meta.min_key > bound) {
            result.push_back(pack(section));
            section.clear();
            has_bound = false;
        }
        section.push_back(meta);
        if (!has_bound || meta.max_key > bound) {
            bound = meta.max_key;
            has_bound = true;
        }
    }
    if (!section.empty()) result.push_back(pack(section));
    return result;
}

// ------------------------------------------------------------- data model

struct ColSpec {
    std::string name;
    int dtype;  // pmh_dtype
    int out_esize;
    int stored_esize;  // parquet physical width (TINYINT stored as INT32)
    int precision = 0;  // DECIMAL(p,s) annotation on INT32/INT64
    int scale = 0;
};

// Plan-level GLOBAL dictionary for one string column: per-file parquet
// dictionaries remap into it at staging (id streams then decode to global
// ids on device and flow through merge/emit as int32), and the output
// batch exposes it — the same dictionary-vector shape the reference's
// columnar batches use (paimon-common heap vectors with setDictionary).
struct StrDict {
    std::vector<uint8_t> bytes;
    std::vector<int32_t> offsets{0};
    std::unordered_map<std::string, int32_t> index;
    int32_t add(const uint8_t *p, uint32_t len) {
        std::string s((const char *)p, len);
        auto it = index.find(s);
        if (it != index.end()) return it->second;
        int32_t id = (int32_t)offsets.size() - 1;
        bytes.insert(bytes.end(), p, p + len);
        offsets.push_back((int32_t)bytes.size());
        index.emplace(std::move(s), id);
        return id;
    }
};

struct DeviceBufs {
    std::vector<void *> bufs;
    void *alloc(size_t n) {
        void *p = nullptr;
        if (hipMalloc(&p, n ? n : 8) != hipSuccess) return nullptr;
        bufs.push_back(p);
        return p;
    }
    ~DeviceBufs() {
        for (void *p : bufs) (void)hipFree(p);
    }
};

// one (run, column), staged as ONE contiguous device column:
//  - PLAIN chunks: page value-payloads packed host-side, H2D straight into
//    `contig` at the chunk's row offset (the encoded PLAIN bytes ARE the
//    column values; packing is layout, not decode);
//  - dictionary chunks: the RLE id streams stay encoded in HBM; at read
//    time k_rle_decode + k_dict_gather materialize the chunk's row range of
//    `contig` (decode_ms, inside the timed region).
struct GatherTask {
    int64_t start = 0;  // run-row offset (or dense offset, to_dense)
    int64_t n = 0;
    void *dict_dev = nullptr;
    bool to_dense = false;  // nullable chunk: gather into the dense buffer,
                            // k_level_scatter positions the rows after
    bool in_place = false;  // ORC dictionary strings: the RLEv2 decode
                            // already wrote LOCAL ids at the target; remap
                            // them to global ids in place
};

struct RunCol {
    void *contig = nullptr;
    std::vector<DevPage> pages_host;
    DevPage *pages_dev = nullptr;
    int n_pages = 0;
    bool dict_encoded = false;  // any dictionary-encoded chunk
    std::vector<RleChunk> rle_host;
    RleChunk *rle_dev = nullptr;
    int32_t *ids_dev = nullptr;
    std::vector<GatherTask> gathers;
    int64_t n_rows = 0;
    // nullable columns with actual nulls: dense PLAIN values + def-level
    // streams stay encoded in HBM; k_level_scatter positions them at read
    // time (timed decode) and fills the byte-validity array.
    bool has_nulls = false;
    std::vector<uint8_t> dense_host;   // packed non-null values (staging)
    // (dense_start, nbytes) per dense_host append — PLAIN and dictionary
    // null-chunks can interleave, so host segments land at their own offsets
    std::vector<std::pair<int64_t, int64_t>> dense_segs;
    std::vector<uint8_t> levels_host;  // packed def-level streams (staging)
    std::vector<RleChunk> def_host;    // src = RELATIVE offset until upload
    int64_t dense_before = 0;          // running non-null count
    void *dense_dev = nullptr;
    RleChunk *def_dev = nullptr;
    uint8_t *valid_dev = nullptr;
    // ORC: RLEv2 / byte-RLE work chunks decode on the GPU at read time
    bool orc_encoded = false;
    std::vector<Rlev2Chunk> rlev2_host;
    Rlev2Chunk *rlev2_dev = nullptr;
    // parquet DELTA_BINARY_PACKED pages (decoded on GPU at read time)
    std::vector<DeltaChunk> delta_host;
    std::vector<DeltaStream> dstreams_host;
};

struct Run {
    int64_t length = 0;
    std::vector<RunCol> cols;
    uint8_t *tomb = nullptr;  // deletion-vector tombstones (1 = deleted)
    int level = 0;            // the SortedRun's level (files of one run
                              // share it; changelog top-level detection)
};

struct Section {
    std::vector<Run> runs;
    int64_t total_rows = 0;
    int64_t n_tiles = 0;
    // device-side descriptors
    DevCol *key_cols = nullptr;   // [k]
    DevCol *seq_cols = nullptr;   // [k]
    DevCol *kind_cols = nullptr;  // [k]
    DevCol *all_cols = nullptr;   // [k * n_cols] run-major
    int64_t *lens_dev = nullptr;
    int32_t *cuts = nullptr;
    uint32_t *winners = nullptr;
    int32_t *tile_counts = nullptr;
    int64_t *tile_offsets = nullptr;
    int64_t *total_dev = nullptr;
    uint16_t *group_start = nullptr;  // partial-update member offsets
    uint32_t *err_dev = nullptr;
    // fused merge+emit path (dedup / first-row): per-tile lookback words +
    // the tile ticket, zeroed per read
    uint64_t *status = nullptr;
    uint64_t *ticket = nullptr;
    uint32_t *dense_winners = nullptr;  // split mode (PMH_FSPLIT)
    // per-run tombstone byte arrays (deletion vectors; null entries = no
    // DV) + the device pointer table the merge kernels consume
    uint64_t *tombs_dev = nullptr;
    bool any_tomb = false;
    // packed per-row validity (PU/agg emit): one u64 per row per run,
    // bit c = column c non-null; built once per section by k_pack_valid
    std::vector<uint64_t *> row_masks;
    uint64_t **row_masks_dev = nullptr;  // [k] device array of the above
    // full-compaction changelog chain
    // hierarchical sections (> PMH_MAX_RUNS runs): the reference spills
    // to disk (MergeSorter.spillMergeSort); here the winner fold is
    // ASSOCIATIVE for deduplicate/first-row (max/min by (seq, isAdd)), so
    // run batches merge on-device into VIRTUAL runs first (keeping delete
    // winners), then the normal chain merges the virtual runs.
    bool hier = false;
    DevCol *rkeys = nullptr, *rseqs = nullptr, *rkinds = nullptr,
           *rall = nullptr;          // descriptors over the REAL runs
    int64_t *rlens = nullptr;
    uint64_t *rtombs = nullptr;
    std::vector<int> blo, bhi;       // batch run ranges
    std::vector<int64_t> brows, bntiles;
    std::vector<void **> bout_ptrs;      // [n_cols] device arrays per batch
    std::vector<uint8_t **> bout_valid;
    uint8_t *run_levels = nullptr;   // [k]
    uint64_t *cl_entries = nullptr;  // 2 slots per group, provisional
    uint64_t *cl_rows = nullptr;     // compacted rows (k_cl_finalize)
    int32_t *cl_counts = nullptr;
    int64_t *cl_offsets = nullptr;
    int64_t *cl_total_dev = nullptr;
    std::vector<int64_t *> ckeys;  // per-run composite keys (k_composite)
    // batched decode work (all run-columns in ONE launch each)
    Rlev2Chunk *rlev2_all = nullptr;
    int64_t n_rlev2 = 0;
    DeltaChunk *delta_all = nullptr;
    int64_t n_delta = 0;
    DeltaStream *dstreams_all = nullptr;
    int64_t n_dstreams = 0;
    int64_t *delta_sums = nullptr;   // per-chunk block sums
    int64_t *delta_bases = nullptr;  // per-chunk entering values
    RleChunk *def_all = nullptr;
    int64_t n_def = 0;
    bool any_dict = false;
    bool any_decode = false;  // dict or null-scatter work at read time
};

}  // namespace pmh

struct pmh_session_t {
    int device = -1;
};

struct pmh_plan_t {
    pmh_session_t *session = nullptr;
    hipStream_t stream = nullptr;
    hipStream_t stream_b = nullptr;  // split-mode value emission overlap
    pmh::DeviceBufs bufs;
    std::vector<pmh::ColSpec> cols;  // key..., seq, kind, value...
    int n_key_cols = 1;
    bool drop_delete = true;
    bool ignore_delete = false;
    bool host_output = false;
    bool pu = false;         // partial-update merge engine
    bool first_row = false;  // first-row merge engine
    bool fused = false;      // single-pass k_merge_emit (non-member-list
                             // engines; PMH_FUSED=0 falls back for A/B)
    bool fsplit = false;     // split value emission (PMH_FSPLIT=1 A/B)
    bool agg = false;        // aggregation merge engine (uses PU member lists)
    uint8_t *col_agg_dev = nullptr;  // per-column PMH_AGG_* codes
    // value filters (conjunction; applied to single-run sections only,
    // MergeFileSplitRead.java:227-239)
    std::vector<pmh::FilterTerm> filters;
    pmh::FilterTerm *filters_dev = nullptr;
    uint8_t *hier_dtype_dev = nullptr;  // batch-pass dtype map (int8/16 ->
                                        // int32 so virtual runs hold the
                                        // STORED width)
    // composite key (>1 key column): order-preserving packed comparand
    bool composite_key = false;
    uint64_t key_shifts = 0, key_bits = 0;  // 8 bits per sub-key
    // partial-update.remove-record-on-delete: a DELETE resets the row
    // (PartialUpdateMergeFunction.java:173-180); v1 accepts INSERT/DELETE
    // streams (UPDATE_BEFORE rejected)
    bool rrod = false;
    // sequence groups (fields.<seq>.sequence-group=members,
    // PartialUpdateMergeFunction.java:219-377): retracts become legal and
    // act on their groups
    bool seqg = false;
    bool agg_retract = false;  // every aggregator retract-capable
    // sequence.field (user-defined sequence comparator): compare listed
    // value columns before the sequence number
    int n_useq = 0;
    int16_t *useq_dev = nullptr;
    int n_seq_groups = 0;
    // changelog-producer = full-compaction (FullChangelogMergeFunction-
    // Wrapper): a second output stream of changelog rows per read_next
    bool changelog = false;
    bool cl_row_dedup = false;
    int max_level = 0;
    int64_t cl_rows_last = 0;
    std::vector<void *> out_dev2;
    void **out_ptrs2_dev = nullptr;
    std::vector<uint8_t *> out_valid2;
    uint8_t **out_valid2_dev = nullptr;
    std::vector<std::vector<uint8_t>> out_host2, out_valid_host2;
    std::vector<pmh_col> batch_cols2;
    uint8_t *col_group_dev = nullptr;
    int16_t *sg_fields_dev = nullptr;
    uint8_t *sg_nseq_dev = nullptr;
    std::vector<pmh::Section> sections;
    size_t cur_section = 0;
    int64_t rows_in_total = 0;
    int64_t encoded_bytes_total = 0;
    // output buffers (reused per section)
    std::vector<void *> out_dev;
    void **out_ptrs_dev = nullptr;
    uint8_t *col_dtype_dev = nullptr;
    std::vector<bool> col_nullable;        // any nulls staged for this col
    uint8_t *col_nullable_dev = nullptr;
    std::vector<uint8_t *> out_valid;      // per col; nullptr if never null
    uint8_t **out_valid_dev = nullptr;     // device array of the above
    std::vector<std::vector<uint8_t>> out_valid_host;
    std::vector<std::vector<uint8_t>> out_host;
    std::vector<pmh_col> batch_cols;
    std::vector<std::string> col_names;
    std::vector<std::unique_ptr<pmh::StrDict>> sdicts;  // per col; string only
    pmh_stats stats{};
    double h2d_ms = 0;
};

namespace pmh {

static int dtype_from_str(const std::string &s) {
    if (s == "int8" || s == "tinyint") return PMH_DT_INT8;
    if (s == "int16" || s == "smallint") return PMH_DT_INT16;
    if (s == "int32" || s == "int") return PMH_DT_INT32;
    if (s == "int64" || s == "bigint") return PMH_DT_INT64;
    if (s == "float" || s == "float32") return PMH_DT_FLOAT32;
    if (s == "double" || s == "float64") return PMH_DT_FLOAT64;
    if (s == "string" || s == "varchar" || s == "char") return PMH_DT_STRING;
    return -1;
}

// "decimal(p,s)": p <= 9 rides INT32, p <= 18 INT64 (unscaled values) —
// exactly the reference's physical mapping, ParquetSchemaConverter.java:
// 153-171 + is32BitDecimal/is64BitDecimal :334-341. Wider decimals (FLBA)
// are a later round.
static bool parse_decimal(const std::string &s, int *p, int *sc) {
    if (s.rfind("decimal(", 0) != 0 || s.back() != ')') return false;
    return sscanf(s.c_str(), "decimal(%d,%d)", p, sc) == 2;
}

static int dtype_out_esize(int dt) {
    switch (dt) {
    case PMH_DT_INT8: return 1;
    case PMH_DT_INT16: return 2;
    case PMH_DT_INT32: return 4;
    case PMH_DT_INT64: return 8;
    case PMH_DT_FLOAT32: return 4;
    case PMH_DT_FLOAT64: return 8;
    case PMH_DT_STRING: return 4;  // global dictionary ids
    }
    return 0;
}

static int dtype_stored_esize(int dt) {
    // parquet physical widths: INT8/16/32 stored as INT32
    switch (dt) {
    case PMH_DT_INT8:
    case PMH_DT_INT16:
    case PMH_DT_INT32: return 4;
    case PMH_DT_INT64: return 8;
    case PMH_DT_FLOAT32: return 4;
    case PMH_DT_FLOAT64: return 8;
    case PMH_DT_STRING: return 4;  // the RLE id streams decode to int32
    }
    return 0;
}

static inline int popcount8(uint8_t b) { return __builtin_popcount(b); }

static int expected_phys(int dtype) {
    switch (dtype) {
    case PMH_DT_INT8:
    case PMH_DT_INT16:
    case PMH_DT_INT32: return PHYS_INT32;
    case PMH_DT_INT64: return PHYS_INT64;
    case PMH_DT_FLOAT32: return PHYS_FLOAT;
    case PMH_DT_FLOAT64: return PHYS_DOUBLE;
    case PMH_DT_STRING: return PHYS_BYTE_ARRAY;
    }
    return -1;
}

// Walk a def-level stream (bit width 1): emit device work chunks with
// running dense offsets (aux) and report the number of non-null values.
// src offsets are RELATIVE to the (run,col) levels buffer (rel_base =
// offset of this stream within it); patched to device addresses at upload.
static bool prescan_def(const uint8_t *s, int64_t len, int64_t n,
                        int64_t out_row0, int64_t rel_base,
                        int64_t *dense_before, std::vector<RleChunk> &out) {
    int64_t p = 0, cnt = 0;
    // one wave per chunk in k_level_scatter: chunk size bounds the serial
    // 64-value passes per wave
    const int64_t MAX_CHUNK = 512;
    while (cnt < n) {
        if (p >= len) {
            set_error("def-level stream overrun");
            return false;
        }
        uint64_t header = 0;
        int shift = 0;
        for (;;) {
            uint8_t b = s[p++];
            header |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
        }
        if ((header & 1) == 0) {
            int64_t count = std::min<int64_t>((int64_t)(header >> 1), n - cnt);
            uint8_t v = s[p++] & 1;
            for (int64_t off = 0; off < count; off += MAX_CHUNK) {
                RleChunk c{};
                c.kind = 0;
                c.value = v;
                c.out_start = out_row0 + cnt + off;
                c.count = (int32_t)std::min<int64_t>(MAX_CHUNK, count - off);
                c.bit_width = 1;
                c.aux = *dense_before + (v ? off : 0);
                out.push_back(c);
            }
            if (v) *dense_before += count;
            cnt += count;
        } else {
            int64_t groups = (int64_t)(header >> 1);
            int64_t vals = std::min<int64_t>(groups * 8, n - cnt);
            int64_t voff = 0;
            while (voff < vals) {
                int64_t take = std::min<int64_t>(MAX_CHUNK, vals - voff);
                RleChunk c{};
                c.kind = 1;
                c.src = (uint64_t)(rel_base + p + voff / 8);
                c.out_start = out_row0 + cnt + voff;
                c.count = (int32_t)take;
                c.bit_width = 1;
                c.aux = *dense_before;
                out.push_back(c);
                // advance dense count by the set bits in this chunk
                for (int64_t i = 0; i < take; i += 8) {
                    uint8_t b = s[p + (voff + i) / 8];
                    int64_t rem = take - i;
                    uint8_t mask =
                        rem >= 8 ? 0xFF : (uint8_t)((1u << rem) - 1);
                    *dense_before += popcount8(b & mask);
                }
                voff += take;
            }
            p += groups;
            cnt += vals;
        }
    }
    return true;
}

// Walk an RLE/bit-packed def-level stream (bit width 1) and report whether
// any zero (null) occurs among the first n values.
static bool def_levels_have_nulls(const uint8_t *s, int64_t len, int64_t n) {
    int64_t p = 0, cnt = 0;
    while (cnt < n && p < len) {
        uint64_t header = 0;
        int shift = 0;
        for (;;) {
            if (p >= len) return true;  // malformed: treat as nulls
            uint8_t b = s[p++];
            header |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
        }
        if ((header & 1) == 0) {
            int64_t count = (int64_t)(header >> 1);
            if (p >= len) return true;
            uint8_t v = s[p++];
            if (count > n - cnt) count = n - cnt;
            if (v == 0 && count > 0) return true;
            cnt += count;
        } else {
            int64_t groups = (int64_t)(header >> 1);
            for (int64_t g = 0; g < groups && cnt < n; g++) {
                if (p >= len) return true;
                uint8_t v = s[p++];
                int64_t take = std::min<int64_t>(8, n - cnt);
                uint8_t mask = take >= 8 ? 0xFF : (uint8_t)((1u << take) - 1);
                if ((v & mask) != mask) return true;
                cnt += take;
            }
        }
    }
    return cnt < n ? true : false;
}

// Prescan an RLE stream into device work chunks (splitting long runs).
static bool prescan_rle(const uint8_t *s, int64_t len, int bit_width,
                        int64_t n, int64_t out_base, uint64_t dev_base,
                        int64_t host_off, std::vector<RleChunk> &out) {
    int64_t p = 0, cnt = 0;
    int byte_width = (bit_width + 7) / 8;
    const int64_t MAX_CHUNK = 16384;
    while (cnt < n) {
        if (p >= len) {
            set_error("RLE stream overrun");
            return false;
        }
        uint64_t header = 0;
        int shift = 0;
        for (;;) {
            uint8_t b = s[p++];
            header |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
        }
        if ((header & 1) == 0) {
            int64_t count = std::min<int64_t>((int64_t)(header >> 1), n - cnt);
            uint32_t v = 0;
            for (int i = 0; i < byte_width; i++) v |= (uint32_t)s[p + i] << (8 * i);
            p += byte_width;
            for (int64_t off = 0; off < count; off += MAX_CHUNK) {
                RleChunk c{};
                c.kind = 0;
                c.value = v;
                c.out_start = out_base + cnt + off;
                c.count = (int32_t)std::min<int64_t>(MAX_CHUNK, count - off);
                c.bit_width = bit_width;
                out.push_back(c);
            }
            cnt += count;
        } else {
            int64_t groups = (int64_t)(header >> 1);
            int64_t vals = std::min<int64_t>(groups * 8, n - cnt);
            // split at group boundaries
            for (int64_t voff = 0; voff < vals; voff += MAX_CHUNK) {
                RleChunk c{};
                c.kind = 1;
                c.src = dev_base + (uint64_t)(host_off + p) +
                        (uint64_t)((voff / 8) * bit_width);
                c.out_start = out_base + cnt + voff;
                c.count = (int32_t)std::min<int64_t>(MAX_CHUNK, vals - voff);
                c.bit_width = bit_width;
                out.push_back(c);
            }
            p += groups * bit_width;
            cnt += vals;
        }
    }
    return true;
}


// Prescan one DELTA_BINARY_PACKED page (VectorizedDeltaBinaryPackedReader
// .java / parquet-format Encodings.md): header <block_size><miniblocks_per_
// block><total_count><first zigzag>, then blocks of [min_delta zigzag]
// [miniblock widths][packed miniblocks]. Each PAGE is self-contained (its
// own first value), so it forms one DeltaStream; each block becomes one
// DeltaChunk. Miniblocks holding no real values carry a width byte but no
// data (their widths zero out in the packed u64 so device byte offsets
// skip nothing); partially-filled miniblocks are stored in full.
// Host DELTA_BINARY_PACKED decode of one complete stream (parquet
// delta-encoding spec; the GPU path decodes these on device — this host
// version serves DELTA_BYTE_ARRAY string staging, where values intern into
// the global dictionary anyway). Returns bytes consumed or -1.
static int64_t host_delta_i64(const uint8_t *pp, int64_t plen,
                              std::vector<int64_t> &out) {
    int64_t pos = 0;
    auto uleb = [&](uint64_t *v) -> bool {
        *v = 0;
        int sh = 0;
        for (;;) {
            if (pos >= plen) return false;
            uint8_t b = pp[pos++];
            *v |= (uint64_t)(b & 0x7f) << sh;
            if (!(b & 0x80)) return true;
            sh += 7;
        }
    };
    auto zz = [](uint64_t v) {
        return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
    };
    uint64_t bs, mpb, total, zfirst;
    if (!uleb(&bs) || !uleb(&mpb) || !uleb(&total) || !uleb(&zfirst))
        return -1;
    if (mpb == 0 || bs % mpb != 0) return -1;
    const int64_t vpm = (int64_t)(bs / mpb);
    out.clear();
    out.reserve(total);
    if (total == 0) return pos;
    int64_t cur = zz(zfirst);
    out.push_back(cur);
    int64_t remaining = (int64_t)total - 1;
    while (remaining > 0) {
        uint64_t zmd;
        if (!uleb(&zmd) || pos + (int64_t)mpb > plen) return -1;
        int64_t mind = zz(zmd);
        const uint8_t *widths = pp + pos;
        pos += mpb;
        for (uint64_t m = 0; m < mpb && remaining > 0; m++) {
            int w = widths[m];
            if (w > 64) return -1;
            int64_t take = remaining < vpm ? remaining : vpm;
            if (pos + (vpm * w + 7) / 8 > plen) return -1;
            for (int64_t j = 0; j < take; j++) {
                uint64_t d = 0;  // LSB-first bit-packed
                for (int b = 0; b < w; b++) {
                    int64_t bit = j * w + b;
                    d |= (uint64_t)((pp[pos + (bit >> 3)] >> (bit & 7)) & 1)
                         << b;
                }
                cur += mind + (int64_t)d;
                out.push_back(cur);
            }
            pos += (vpm * w + 7) / 8;  // full miniblock advances
            remaining -= take;
        }
    }
    return pos;
}

static bool prescan_delta(const uint8_t *pp, int64_t plen, int64_t n_values,
                          int64_t out_row0, uint64_t dev_base,
                          int64_t rel_off, uint64_t out_addr, int out_esize,
                          std::vector<DeltaChunk> &chunks,
                          std::vector<DeltaStream> &streams,
                          std::string &err) {
    int64_t pos = 0;
    auto uleb = [&](uint64_t *v) -> bool {
        *v = 0;
        int sh = 0;
        for (;;) {
            if (pos >= plen) return false;
            uint8_t b = pp[pos++];
            *v |= (uint64_t)(b & 0x7f) << sh;
            if (!(b & 0x80)) return true;
            sh += 7;
        }
    };
    auto zz = [](uint64_t v) {
        return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
    };
    uint64_t bs, mpb, total, zfirst;
    if (!uleb(&bs) || !uleb(&mpb) || !uleb(&total) || !uleb(&zfirst)) {
        err = "delta: truncated header";
        return false;
    }
    if (mpb == 0 || mpb > 8 || bs % mpb != 0) {
        err = "delta: unsupported miniblock layout (v1: <= 8 per block)";
        return false;
    }
    if ((int64_t)total != n_values) {
        err = "delta: header count disagrees with page";
        return false;
    }
    const int vpm = (int)(bs / mpb);
    if (n_values == 0) return true;
    DeltaStream st{};
    st.chunk_lo = (int64_t)chunks.size();
    st.first = zz(zfirst);
    st.out0 = out_row0;
    st.out_addr = out_addr;
    st.out_esize = out_esize;
    int64_t remaining = n_values - 1;
    int64_t out_i = out_row0 + 1;
    while (remaining > 0) {
        uint64_t zmd;
        if (!uleb(&zmd) || pos + (int64_t)mpb > plen) {
            err = "delta: truncated block";
            return false;
        }
        DeltaChunk ch{};
        ch.min_delta = zz(zmd);
        ch.vpm = (int16_t)vpm;
        ch.n_mini = (int16_t)mpb;
        ch.out_addr = out_addr;
        ch.out_start = out_i;
        ch.out_esize = out_esize;
        ch.count = (int32_t)(remaining < (int64_t)bs ? remaining
                                                     : (int64_t)bs);
        int64_t data = 0;
        for (uint64_t j = 0; j < mpb; j++) {
            const int w = pp[pos + j];
            if (w > 64) {
                err = "delta: miniblock width > 64";
                return false;
            }
            const int64_t real = remaining - (int64_t)j * vpm;
            if (real > 0) {
                ch.widths |= (uint64_t)w << (8 * j);
                data += (int64_t)vpm * w / 8;
            }
        }
        pos += mpb;
        ch.src = dev_base + (uint64_t)(rel_off + pos);
        if (pos + data > plen) {
            err = "delta: truncated miniblocks";
            return false;
        }
        pos += data;
        chunks.push_back(ch);
        out_i += ch.count;
        remaining -= ch.count;
    }
    st.chunk_hi = (int64_t)chunks.size();
    streams.push_back(st);
    return true;
}

struct StagedFile {
    std::vector<uint8_t> data;  // whole file (or decompressed payload view)
    bool orc = false;
    ParquetFileMeta meta;
    OrcFileMeta orc_meta;
};

static bool load_file(const std::string &path, StagedFile &sf) {
    std::ifstream f(path, std::ios::binary | std::ios::ate);
    if (!f) {
        set_error("cannot open %s", path.c_str());
        return false;
    }
    int64_t n = f.tellg();
    f.seekg(0);
    sf.data.resize(n);
    f.read((char *)sf.data.data(), n);
    if (is_orc_file(sf.data.data(), n)) {
        sf.orc = true;
        sf.orc_meta = parse_orc_meta(sf.data.data(), n);
        if (!sf.orc_meta.ok()) {
            set_error("%s: %s", path.c_str(), sf.orc_meta.error.c_str());
            return false;
        }
        return true;
    }
    sf.meta = parse_parquet_footer(sf.data.data(), n);
    if (!sf.meta.ok()) {
        set_error("%s: %s", path.c_str(), sf.meta.error.c_str());
        return false;
    }
    std::string err;
    for (auto &rg : sf.meta.row_groups)
        for (auto &cc : rg.columns)
            if (!scan_chunk_pages(sf.data.data(), n, cc, err)) {
                set_error("%s: %s", path.c_str(), err.c_str());
                return false;
            }
    return true;
}

// ------------------------------------------------- ORC staging (RLEv2)

// Walk an ORC RLEv2 stream (runs <= 512 values, byte-aligned) and emit one
// device work chunk per run. dev_base = device address of the uploaded
// stream bytes; dense0 = dense output offset of this stream's first value.
// Host RLEv2 decode (UNSIGNED interpretation) — ORC LENGTH / side streams
// that staging needs on the host (string id-ification). Restates the ORC
// v1 spec's RLEv2 (same arithmetic as k_rlev2;
// orc/impl/RunLengthIntegerReaderV2.java is the reference consumer).
static int rlev2_width(int enc) {
    static const int tbl[8] = {26, 28, 30, 32, 40, 48, 56, 64};
    if (enc < 24) return enc + 1;
    return tbl[enc - 24];
}

static bool host_rlev2_u(const uint8_t *s, int64_t len, int64_t n,
                         std::vector<uint64_t> &out) {
    out.clear();
    if (n > 0) out.reserve(n);
    int64_t p = 0;
    auto bits_be = [&](int64_t q, int64_t bitpos, int w) -> uint64_t {
        uint64_t v = 0;
        for (int i = 0; i < w; i++) {
            int64_t b = bitpos + i;
            v = (v << 1) | ((s[q + (b >> 3)] >> (7 - (b & 7))) & 1);
        }
        return v;
    };
    // n < 0: decode until the stream is exhausted (dictionary LENGTH
    // streams — the entry count comes from the stream itself)
    while (n < 0 ? p < len : (int64_t)out.size() < n) {
        if (p >= len) return false;
        uint8_t h = s[p];
        int mode = h >> 6;
        if (mode == 0) {  // SHORT_REPEAT
            int w = ((h >> 3) & 7) + 1;
            int cnt = (h & 7) + 3;
            if (p + 1 + w > len) return false;
            uint64_t v = 0;
            for (int i = 0; i < w; i++) v = (v << 8) | s[p + 1 + i];
            for (int i = 0; i < cnt; i++) out.push_back(v);
            p += 1 + w;
        } else if (mode == 1) {  // DIRECT
            int w = rlev2_width((h >> 1) & 31);
            if (p + 2 > len) return false;
            int cnt = (((int)(h & 1)) << 8 | s[p + 1]) + 1;
            int64_t nbytes = ((int64_t)cnt * w + 7) / 8;
            if (p + 2 + nbytes > len) return false;
            for (int i = 0; i < cnt; i++)
                out.push_back(bits_be(p + 2, (int64_t)i * w, w));
            p += 2 + nbytes;
        } else if (mode == 2) {  // PATCHED_BASE
            if (p + 4 > len) return false;
            int w = rlev2_width((h >> 1) & 31);
            int cnt = (((int)(h & 1)) << 8 | s[p + 1]) + 1;
            int bw = ((s[p + 2] >> 5) & 7) + 1;
            int pw = rlev2_width(s[p + 2] & 31);
            int pgw = ((s[p + 3] >> 5) & 7) + 1;
            int pll = s[p + 3] & 31;
            int64_t q = p + 4;
            if (q + bw > len) return false;
            uint64_t braw = 0;
            for (int i = 0; i < bw; i++) braw = (braw << 8) | s[q + i];
            int64_t base = (int64_t)braw;
            uint64_t sign = 1ull << (bw * 8 - 1);
            if (braw & sign) base = -(int64_t)(braw & (sign - 1));
            q += bw;
            int64_t dbytes = ((int64_t)cnt * w + 7) / 8;
            int pbits = pgw * 8 + pw;
            pbits = ((pbits + 7) / 8) * 8;
            int64_t pbytes = ((int64_t)pll * pbits + 7) / 8;
            if (q + dbytes + pbytes > len) return false;
            std::vector<uint64_t> vals(cnt);
            for (int i = 0; i < cnt; i++)
                vals[i] = bits_be(q, (int64_t)i * w, w);
            int64_t gap = 0;
            for (int i = 0; i < pll; i++) {
                uint64_t e = bits_be(q + dbytes, (int64_t)i * pbits, pbits);
                uint64_t patch =
                    e & ((pw >= 64) ? ~0ull : ((1ull << pw) - 1));
                gap += (int64_t)(e >> pw);
                if (gap >= cnt) return false;
                vals[gap] |= patch << w;
            }
            for (int i = 0; i < cnt; i++)
                out.push_back((uint64_t)(base + (int64_t)vals[i]));
            p = q + dbytes + pbytes;
        } else {  // DELTA
            int enc = (h >> 1) & 31;
            int w = enc == 0 ? 0 : rlev2_width(enc);
            if (p + 2 > len) return false;
            int cnt = (((int)(h & 1)) << 8 | s[p + 1]) + 1;
            int64_t q = p + 2;
            uint64_t base = 0;  // base: unsigned varint (UNSIGNED streams)
            int sh = 0;
            for (;;) {
                if (q >= len) return false;
                uint8_t b = s[q++];
                base |= (uint64_t)(b & 0x7f) << sh;
                if (!(b & 0x80)) break;
                sh += 7;
            }
            uint64_t zd = 0;  // delta0: signed varint (zigzag)
            sh = 0;
            for (;;) {
                if (q >= len) return false;
                uint8_t b = s[q++];
                zd |= (uint64_t)(b & 0x7f) << sh;
                if (!(b & 0x80)) break;
                sh += 7;
            }
            int64_t d0 = (int64_t)(zd >> 1) ^ -(int64_t)(zd & 1);
            out.push_back(base);
            int64_t cur = (int64_t)base;
            if (cnt > 1) {
                cur += d0;
                out.push_back((uint64_t)cur);
            }
            if (w == 0) {
                for (int i = 2; i < cnt; i++) {
                    cur += d0;
                    out.push_back((uint64_t)cur);
                }
            } else {
                int64_t nbytes = ((int64_t)(cnt - 2) * w + 7) / 8;
                if (q + nbytes > len) return false;
                for (int i = 2; i < cnt; i++) {
                    int64_t d = (int64_t)bits_be(q, (int64_t)(i - 2) * w, w);
                    cur += d0 < 0 ? -d : d;
                    out.push_back((uint64_t)cur);
                }
                q += nbytes;
            }
            p = q;
        }
    }
    return n < 0 || (int64_t)out.size() >= n;
}

static bool prescan_rlev2(const uint8_t *s, int64_t len, int64_t n_values,
                          int is_signed, int out_esize, int dense_target,
                          uint64_t dev_base, int64_t dense0,
                          std::vector<Rlev2Chunk> &out) {
    static const int fbs[32] = {1,  2,  3,  4,  5,  6,  7,  8,
                                9,  10, 11, 12, 13, 14, 15, 16,
                                17, 18, 19, 20, 21, 22, 23, 24,
                                26, 28, 30, 32, 40, 48, 56, 64};
    int64_t p = 0, cnt = 0;
    auto uvarint = [&](bool &ok) -> uint64_t {
        uint64_t v = 0;
        int shift = 0;
        for (;;) {
            if (p >= len) { ok = false; return 0; }
            uint8_t b = s[p++];
            v |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) return v;
            shift += 7;
        }
    };
    while (cnt < n_values) {
        if (p >= len) {
            set_error("RLEv2 stream overrun");
            return false;
        }
        uint8_t first = s[p++];
        int enc = (first >> 6) & 3;
        Rlev2Chunk c{};
        c.is_signed = (uint8_t)is_signed;
        c.out_esize = (uint8_t)out_esize;
        c.dense_target = (uint8_t)dense_target;
        c.out_start = dense0 + cnt;
        if (enc == 0) {  // SHORT_REPEAT: host decodes the literal
            int w = ((first >> 3) & 7) + 1;
            int rep = (first & 7) + 3;
            uint64_t raw = 0;
            for (int i = 0; i < w; i++) raw = (raw << 8) | s[p++];
            c.kind = 0;
            c.base = is_signed
                         ? ((int64_t)(raw >> 1) ^ -(int64_t)(raw & 1))
                         : (int64_t)raw;
            c.count = rep;
            cnt += rep;
        } else if (enc == 1) {  // DIRECT
            int width = fbs[(first >> 1) & 0x1f];
            int count = (int)(((first & 1) << 8) | s[p++]) + 1;
            c.kind = 1;
            c.width = (uint8_t)width;
            c.count = count;
            c.src = dev_base + p;
            p += ((int64_t)count * width + 7) / 8;
            cnt += count;
        } else if (enc == 2) {  // PATCHED_BASE
            int width = fbs[(first >> 1) & 0x1f];
            int count = (int)(((first & 1) << 8) | s[p++]) + 1;
            uint8_t third = s[p++], fourth = s[p++];
            int bw = ((third >> 5) & 7) + 1;
            int pw = fbs[third & 0x1f];
            int pgw = ((fourth >> 5) & 7) + 1;
            int pl = fourth & 0x1f;
            uint64_t braw = 0;
            for (int i = 0; i < bw; i++) braw = (braw << 8) | s[p++];
            uint64_t smask = 1ull << (bw * 8 - 1);
            c.kind = 2;
            c.width = (uint8_t)width;
            c.count = count;
            c.base = (braw & smask) ? -(int64_t)(braw & ~smask)
                                    : (int64_t)braw;
            c.src = dev_base + p;
            p += ((int64_t)count * width + 7) / 8;
            int cfb = 64;
            for (int i = 0; i < 32; i++)
                if (fbs[i] >= pw + pgw) { cfb = fbs[i]; break; }
            c.patch_src = dev_base + p;
            c.patch_pl = (uint16_t)pl;
            c.patch_pw = (uint8_t)pw;
            c.patch_pgw = (uint8_t)pgw;
            c.patch_cfb = (uint8_t)cfb;
            p += ((int64_t)pl * cfb + 7) / 8;
            cnt += count;
        } else {  // DELTA
            int wcode = (first >> 1) & 0x1f;
            int width = wcode == 0 ? 0 : fbs[wcode];
            int count = (int)(((first & 1) << 8) | s[p++]) + 1;
            bool ok = true;
            uint64_t braw = uvarint(ok);
            int64_t base = is_signed
                               ? ((int64_t)(braw >> 1) ^ -(int64_t)(braw & 1))
                               : (int64_t)braw;
            uint64_t draw = uvarint(ok);
            int64_t delta = (int64_t)(draw >> 1) ^ -(int64_t)(draw & 1);
            if (!ok) {
                set_error("RLEv2 delta varint overrun");
                return false;
            }
            c.kind = 3;
            c.width = (uint8_t)width;
            c.count = count;
            c.base = base;
            c.delta = delta;
            c.src = dev_base + p;
            if (width > 0 && count > 2)
                p += ((int64_t)(count - 2) * width + 7) / 8;
            cnt += count;
        }
        out.push_back(c);
        if (p > len) {
            set_error("RLEv2 stream overran its length");
            return false;
        }
    }
    return true;
}

// ORC byte-RLE (tinyint columns, e.g. _VALUE_KIND): header h >= 0 -> h+3
// copies of the next byte; h < 0 -> -h literal bytes.
static bool prescan_byterle(const uint8_t *s, int64_t len, int64_t n_values,
                            int out_esize, int dense_target,
                            uint64_t dev_base, int64_t dense0,
                            std::vector<Rlev2Chunk> &out) {
    int64_t p = 0, cnt = 0;
    while (cnt < n_values) {
        if (p >= len) {
            set_error("byte-RLE stream overrun");
            return false;
        }
        int8_t h = (int8_t)s[p++];
        Rlev2Chunk c{};
        c.out_esize = (uint8_t)out_esize;
        c.dense_target = (uint8_t)dense_target;
        c.out_start = dense0 + cnt;
        if (h >= 0) {
            c.kind = 4;
            c.base = (int8_t)s[p++];
            c.count = (int32_t)std::min<int64_t>(h + 3, n_values - cnt);
        } else {
            c.kind = 5;
            c.src = dev_base + p;
            c.count = (int32_t)std::min<int64_t>(-(int64_t)h, n_values - cnt);
            p += -(int64_t)h;
        }
        cnt += c.count;
        out.push_back(c);
    }
    return true;
}

static const uint8_t BITREV[256] = {
#define R2(n) n, n + 2 * 64, n + 1 * 64, n + 3 * 64
#define R4(n) R2(n), R2(n + 2 * 16), R2(n + 1 * 16), R2(n + 3 * 16)
#define R6(n) R4(n), R4(n + 2 * 4), R4(n + 1 * 4), R4(n + 3 * 4)
    R6(0), R6(2), R6(1), R6(3)
#undef R2
#undef R4
#undef R6
};

// ORC PRESENT stream (boolean RLE = byte-RLE over MSB-first bit-packed
// bytes) -> the parquet-style def-level RleChunks consumed by
// k_level_scatter (which reads LSB-first: bytes are bit-reversed into the
// (run,col) levels buffer). Tracks the dense (non-null) cursor.
static bool prescan_present(const uint8_t *s, int64_t len, int64_t n_rows,
                            int64_t out_row0, RunCol &rc) {
    int64_t p = 0, cnt = 0;
    while (cnt < n_rows) {
        if (p >= len) {
            set_error("PRESENT stream overrun");
            return false;
        }
        int8_t h = (int8_t)s[p++];
        if (h >= 0) {
            uint8_t v = s[p++];
            int64_t vals = std::min<int64_t>((int64_t)(h + 3) * 8,
                                             n_rows - cnt);
            if (v == 0x00 || v == 0xFF) {
                RleChunk c{};
                c.kind = 0;
                c.value = v ? 1 : 0;
                c.out_start = out_row0 + cnt;
                c.count = (int32_t)vals;
                c.bit_width = 1;
                c.aux = rc.dense_before;
                rc.def_host.push_back(c);
                if (v) rc.dense_before += vals;
            } else {
                RleChunk c{};
                c.kind = 1;
                c.src = (uint64_t)rc.levels_host.size();
                c.out_start = out_row0 + cnt;
                c.count = (int32_t)vals;
                c.bit_width = 1;
                c.aux = rc.dense_before;
                for (int i = 0; i < h + 3; i++)
                    rc.levels_host.push_back(BITREV[v]);
                rc.def_host.push_back(c);
                for (int64_t i = 0; i < vals; i += 8) {
                    int64_t rem = vals - i;
                    uint8_t mask = rem >= 8 ? 0xFF : (uint8_t)((1u << rem) - 1);
                    rc.dense_before += __builtin_popcount(BITREV[v] & mask);
                }
            }
            cnt += vals;
        } else {
            int lit = -(int)h;
            int64_t vals = std::min<int64_t>((int64_t)lit * 8, n_rows - cnt);
            RleChunk c{};
            c.kind = 1;
            c.src = (uint64_t)rc.levels_host.size();
            c.out_start = out_row0 + cnt;
            c.count = (int32_t)vals;
            c.bit_width = 1;
            c.aux = rc.dense_before;
            for (int i = 0; i < lit; i++)
                rc.levels_host.push_back(BITREV[s[p + i]]);
            rc.def_host.push_back(c);
            for (int64_t i = 0; i < vals; i++) {
                uint8_t b = BITREV[s[p + (i >> 3)]];
                rc.dense_before += (b >> (i & 7)) & 1;
            }
            p += lit;
            cnt += vals;
        }
    }
    return true;
}

// Stage one ORC file's stripes for the required columns.
static bool stage_orc_file(pmh_plan_t *plan, const FileDesc &fd,
                           const StagedFile &sf, Run &run, int64_t row_base) {
    const auto &cols = plan->cols;
    const OrcFileMeta &om = sf.orc_meta;
    std::vector<int> col_id(cols.size(), -1);
    for (size_t c = 0; c < cols.size(); c++) {
        for (size_t i = 0; i < om.column_names.size(); i++)
            if (om.column_names[i] == cols[c].name) col_id[c] = (int)i + 1;
        if (col_id[c] < 0) {
            set_error("%s: column %s not found", fd.path.c_str(),
                      cols[c].name.c_str());
            return false;
        }
    }
    int64_t stripe_row = 0;
    for (const auto &st : om.stripes) {
        for (size_t c = 0; c < cols.size(); c++) {
            RunCol &rc = run.cols[c];
            rc.orc_encoded = true;
            int cid = col_id[c];
            int ckind = om.column_kinds[cid - 1];
            if (cols[c].dtype == PMH_DT_STRING && ckind != ORC_STRING) {
                set_error("%s col %s: declared string but ORC kind is %d",
                          fd.path.c_str(), cols[c].name.c_str(), ckind);
                return false;
            }
            const int stored = cols[c].stored_esize;
            const OrcStream *data = orc_find_stream(st, cid, ORC_STREAM_DATA);
            const OrcStream *present =
                orc_find_stream(st, cid, ORC_STREAM_PRESENT);
            if (!data) {
                set_error("%s col %s: DATA stream missing", fd.path.c_str(),
                          cols[c].name.c_str());
                return false;
            }
            const int cenc =
                (int)st.encodings.size() > cid ? st.encodings[cid] : 0;
            if (ckind == ORC_STRING
                    ? (cenc != 2 && cenc != 3)
                    : (cenc != 0 && cenc != 2)) {
                set_error("%s col %s: ORC column encoding %d not supported "
                          "(ints: DIRECT/DIRECT_V2; strings: DIRECT_V2 | "
                          "DICTIONARY_V2)",
                          fd.path.c_str(), cols[c].name.c_str(), cenc);
                return false;
            }
            // compressed streams decompress on the host at staging, like
            // the parquet zstd path (on-GPU codecs: §8f); the RLEv2/byte-RLE
            // bytes then upload verbatim
            const uint8_t *data_ptr = sf.data.data() + data->offset;
            int64_t data_len = data->length;
            const uint8_t *pres_ptr =
                present ? sf.data.data() + present->offset : nullptr;
            int64_t pres_len = present ? present->length : 0;
            std::vector<uint8_t> data_dec, pres_dec;
            if (om.compression != 0) {
                std::string cerr;
                if (!orc_decompress(data_ptr, data_len, om.compression,
                                    om.compression_block_size, data_dec,
                                    cerr)) {
                    set_error("%s col %s: %s", fd.path.c_str(),
                              cols[c].name.c_str(), cerr.c_str());
                    return false;
                }
                data_ptr = data_dec.data();
                data_len = (int64_t)data_dec.size();
                if (present) {
                    if (!orc_decompress(pres_ptr, pres_len, om.compression,
                                        om.compression_block_size, pres_dec,
                                        cerr)) {
                        set_error("%s col %s PRESENT: %s", fd.path.c_str(),
                                  cols[c].name.c_str(), cerr.c_str());
                        return false;
                    }
                    pres_ptr = pres_dec.data();
                    pres_len = (int64_t)pres_dec.size();
                }
            }
            if (ckind == ORC_FLOAT || ckind == ORC_DOUBLE) {
                // FLOAT/DOUBLE DATA streams are raw IEEE754 LE values — the
                // stream IS the decoded representation, so staging is a copy
                // (layout, not decode), exactly like parquet PLAIN packing.
                // With PRESENT, the stream holds only the non-null values
                // and feeds the dense buffer the level scatter consumes.
                int64_t row0 = row_base + stripe_row;
                if (present) {
                    rc.has_nulls = true;
                    int64_t before = rc.dense_before;
                    if (!prescan_present(pres_ptr, pres_len, st.num_rows,
                                         row0, rc))
                        return false;
                    int64_t n_dense = rc.dense_before - before;
                    if (data_len < n_dense * stored) {
                        set_error("%s col %s: FLOAT/DOUBLE stream short",
                                  fd.path.c_str(), cols[c].name.c_str());
                        return false;
                    }
                    rc.dense_host.insert(rc.dense_host.end(), data_ptr,
                                         data_ptr + n_dense * stored);
                    rc.dense_segs.emplace_back(before, n_dense * stored);
                } else {
                    if (data_len < st.num_rows * stored) {
                        set_error("%s col %s: FLOAT/DOUBLE stream short",
                                  fd.path.c_str(), cols[c].name.c_str());
                        return false;
                    }
                    if (hipMemcpy((uint8_t *)rc.contig + row0 * stored,
                                  data_ptr, st.num_rows * stored,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                }
                plan->encoded_bytes_total += data_len;
                continue;
            }
            if (ckind == ORC_STRING) {
                // ORC strings (WriterImpl StringTreeWriter):
                //  - DICTIONARY_V2: DICTIONARY_DATA bytes + LENGTH (RLEv2
                //    u) describe the stripe dictionary; DATA = RLEv2 u
                //    local ids -> decode on GPU, then remap IN PLACE to
                //    the column's plan-level GLOBAL dictionary;
                //  - DIRECT_V2: LENGTH + DATA bytes -> each value interns
                //    into the global dictionary at staging (host) and the
                //    int32 ids upload directly (ids ARE the decoded form).
                if (cols[c].dtype != PMH_DT_STRING) {
                    set_error("%s col %s: ORC STRING column read as "
                              "non-string type", fd.path.c_str(),
                              cols[c].name.c_str());
                    return false;
                }
                const OrcStream *lenst =
                    orc_find_stream(st, cid, ORC_STREAM_LENGTH);
                if (!lenst) {
                    set_error("%s col %s: LENGTH stream missing",
                              fd.path.c_str(), cols[c].name.c_str());
                    return false;
                }
                const uint8_t *len_ptr = sf.data.data() + lenst->offset;
                int64_t len_len = lenst->length;
                std::vector<uint8_t> len_dec;
                if (om.compression != 0) {
                    std::string cerr;
                    if (!orc_decompress(len_ptr, len_len, om.compression,
                                        om.compression_block_size, len_dec,
                                        cerr)) {
                        set_error("%s col %s LENGTH: %s", fd.path.c_str(),
                                  cols[c].name.c_str(), cerr.c_str());
                        return false;
                    }
                    len_ptr = len_dec.data();
                    len_len = (int64_t)len_dec.size();
                }
                if (!plan->sdicts[c])
                    plan->sdicts[c].reset(new StrDict());
                StrDict *sd = plan->sdicts[c].get();
                int64_t row0 = row_base + stripe_row;
                int64_t n_dense = st.num_rows;
                int64_t dense0 = row0;
                int dense_target = 0;
                if (present) {
                    rc.has_nulls = true;
                    int64_t before = rc.dense_before;
                    if (!prescan_present(pres_ptr, pres_len, st.num_rows,
                                         row0, rc))
                        return false;
                    n_dense = rc.dense_before - before;
                    dense0 = before;
                    dense_target = 1;
                }
                if (cenc == 3) {  // DICTIONARY_V2
                    const OrcStream *dct =
                        orc_find_stream(st, cid, ORC_STREAM_DICTIONARY);
                    if (!dct) {
                        set_error("%s col %s: DICTIONARY_DATA missing",
                                  fd.path.c_str(), cols[c].name.c_str());
                        return false;
                    }
                    const uint8_t *dp = sf.data.data() + dct->offset;
                    int64_t dl = dct->length;
                    std::vector<uint8_t> dict_dec;
                    if (om.compression != 0) {
                        std::string cerr;
                        if (!orc_decompress(dp, dl, om.compression,
                                            om.compression_block_size,
                                            dict_dec, cerr)) {
                            set_error("%s col %s DICTIONARY: %s",
                                      fd.path.c_str(), cols[c].name.c_str(),
                                      cerr.c_str());
                            return false;
                        }
                        dp = dict_dec.data();
                        dl = (int64_t)dict_dec.size();
                    }
                    std::vector<uint64_t> lens;
                    if (!host_rlev2_u(len_ptr, len_len, -1, lens)) {
                        set_error("%s col %s: LENGTH stream decode failed",
                                  fd.path.c_str(), cols[c].name.c_str());
                        return false;
                    }
                    std::vector<int32_t> remap(lens.size());
                    int64_t off = 0;
                    for (size_t i = 0; i < lens.size(); i++) {
                        if (off + (int64_t)lens[i] > dl) {
                            set_error("%s col %s: dictionary bytes short",
                                      fd.path.c_str(), cols[c].name.c_str());
                            return false;
                        }
                        remap[i] = sd->add(dp + off, (uint32_t)lens[i]);
                        off += (int64_t)lens[i];
                    }
                    void *rdev = plan->bufs.alloc(
                        (lens.empty() ? 1 : lens.size()) * 4);
                    if (!rdev) return false;
                    if (!lens.empty() &&
                        hipMemcpy(rdev, remap.data(), lens.size() * 4,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D ORC string remap failed");
                        return false;
                    }
                    void *dev = plan->bufs.alloc(data_len + 16);
                    if (!dev) return false;
                    if (hipMemcpy(dev, data_ptr, data_len,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                    plan->encoded_bytes_total += data_len + len_len + dl;
                    if (!prescan_rlev2(data_ptr, data_len, n_dense, 0, 4,
                                       dense_target, (uint64_t)dev, dense0,
                                       rc.rlev2_host))
                        return false;
                    GatherTask gt;
                    gt.start = dense0;
                    gt.n = n_dense;
                    gt.dict_dev = rdev;
                    gt.to_dense = dense_target != 0;
                    gt.in_place = true;
                    rc.gathers.push_back(gt);
                } else {  // DIRECT_V2: host id-ification
                    std::vector<uint64_t> lens;
                    if (!host_rlev2_u(len_ptr, len_len, n_dense, lens)) {
                        set_error("%s col %s: LENGTH stream decode failed",
                                  fd.path.c_str(), cols[c].name.c_str());
                        return false;
                    }
                    std::vector<int32_t> ids(n_dense);
                    int64_t off = 0;
                    for (int64_t i = 0; i < n_dense; i++) {
                        if (off + (int64_t)lens[i] > data_len) {
                            set_error("%s col %s: string bytes short",
                                      fd.path.c_str(), cols[c].name.c_str());
                            return false;
                        }
                        ids[i] = sd->add(data_ptr + off,
                                         (uint32_t)lens[i]);
                        off += (int64_t)lens[i];
                    }
                    plan->encoded_bytes_total += data_len + len_len;
                    if (present) {
                        const uint8_t *ib = (const uint8_t *)ids.data();
                        rc.dense_host.insert(rc.dense_host.end(), ib,
                                             ib + n_dense * 4);
                        rc.dense_segs.emplace_back(dense0, n_dense * 4);
                    } else if (n_dense &&
                               hipMemcpy((uint8_t *)rc.contig + row0 * 4,
                                         ids.data(), n_dense * 4,
                                         hipMemcpyHostToDevice) !=
                                   hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                }
                continue;
            }
            // upload the encoded DATA stream (+16 B pad: the RLEv2
            // bit reader uses an aligned 16-byte window)
            void *dev = plan->bufs.alloc(data_len + 16);
            if (!dev) return false;
            if (hipMemcpy(dev, data_ptr, data_len, hipMemcpyHostToDevice) !=
                hipSuccess) {
                set_error("H2D failed");
                return false;
            }
            plan->encoded_bytes_total += data_len;
            int64_t row0 = row_base + stripe_row;
            int64_t n_dense = st.num_rows;
            int64_t dense0 = row0;
            int dense_target = 0;
            if (present) {
                rc.has_nulls = true;
                int64_t before = rc.dense_before;
                if (!prescan_present(pres_ptr, pres_len, st.num_rows, row0,
                                     rc))
                    return false;
                n_dense = rc.dense_before - before;
                dense0 = before;
                dense_target = 1;
            }
            bool ok;
            if (ckind == ORC_BYTE) {
                ok = prescan_byterle(data_ptr, data_len, n_dense, stored,
                                     dense_target, (uint64_t)dev, dense0,
                                     rc.rlev2_host);
            } else if (ckind == ORC_SHORT || ckind == ORC_INT ||
                       ckind == ORC_LONG) {
                ok = prescan_rlev2(data_ptr, data_len, n_dense, 1, stored,
                                   dense_target, (uint64_t)dev, dense0,
                                   rc.rlev2_host);
            } else {
                set_error("%s col %s: ORC type kind %d not supported yet",
                          fd.path.c_str(), cols[c].name.c_str(), ckind);
                return false;
            }
            if (!ok) return false;
        }
        stripe_row += st.num_rows;
    }
    if (stripe_row != fd.row_count) {
        set_error("%s: rowCount %lld != file rows %lld", fd.path.c_str(),
                  (long long)fd.row_count, (long long)stripe_row);
        return false;
    }
    return true;
}

// Stage one run (list of files) for the required columns onto the device.
// Upload one chunk's dictionary for k_dict_gather. Numeric columns upload
// the raw fixed-width values; STRING columns remap the chunk's byte-array
// entries into the column's plan-level GLOBAL dictionary and upload the
// int32 remap table instead (ids then decode straight to global ids).
static bool upload_chunk_dict(pmh_plan_t *plan, int c,
                              const uint8_t *dict_host, int64_t dict_count,
                              int stored, void **out) {
    if (plan->cols[c].dtype == PMH_DT_STRING) {
        if (!plan->sdicts[c]) plan->sdicts[c].reset(new StrDict());
        StrDict *sd = plan->sdicts[c].get();
        std::vector<int32_t> remap((size_t)dict_count);
        const uint8_t *p = dict_host;
        for (int64_t i = 0; i < dict_count; i++) {
            uint32_t len;
            memcpy(&len, p, 4);
            p += 4;
            remap[i] = sd->add(p, len);
            p += len;
        }
        void *dev = plan->bufs.alloc((dict_count ? dict_count : 1) * 4);
        if (!dev) return false;
        if (dict_count &&
            hipMemcpy(dev, remap.data(), dict_count * 4,
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_error("H2D string remap failed");
            return false;
        }
        *out = dev;
        return true;
    }
    void *dev = plan->bufs.alloc(dict_count * stored);
    if (!dev) return false;
    if (hipMemcpy(dev, dict_host, dict_count * stored,
                  hipMemcpyHostToDevice) != hipSuccess) {
        set_error("H2D dict failed");
        return false;
    }
    *out = dev;
    return true;
}

static bool stage_run(pmh_plan_t *plan, const std::vector<FileDesc> &files,
                      Run &run) {
    const auto &cols = plan->cols;
    run.cols.resize(cols.size());
    int64_t total_rows = 0;
    for (const auto &fd : files) total_rows += fd.row_count;
    run.level = files.empty() ? 0 : files[0].level;
    if (total_rows >= ((int64_t)1 << PMH_ROW_BITS)) {
        // a run is the CONCATENATION of non-overlapping files; the winner
        // packing (run | row) indexes into the whole run, so the cap applies
        // to the run total, not per file (a >=2^row-bits run would silently
        // overflow into the run bits and merge wrong)
        set_error("run totals %lld rows >= 2^%d per-run limit (%zu files)",
                  (long long)total_rows, PMH_ROW_BITS, files.size());
        return false;
    }
    for (size_t c = 0; c < cols.size(); c++) {
        RunCol &rc = run.cols[c];
        rc.contig = plan->bufs.alloc(total_rows * cols[c].stored_esize);
        if (!rc.contig) {
            set_error("hipMalloc failed (%lld rows col %s)",
                      (long long)total_rows, cols[c].name.c_str());
            return false;
        }
        plan->encoded_bytes_total += total_rows * cols[c].stored_esize;
    }
    std::vector<uint8_t> tomb_host;
    {
        int64_t rb = 0;
        for (const auto &fd : files) {
            if (!fd.dv_path.empty()) {
                // ApplyDeletionVectorReader semantics (io/
                // KeyValueFileReaderFactory.java:139-143): deleted positions
                // never reach the merge — staged as per-run tombstones the
                // merge kernels treat as nonexistent rows
                std::vector<uint32_t> pos;
                if (!load_deletion_vector(fd.dv_path, fd.dv_offset,
                                          fd.dv_length, pos))
                    return false;
                if (tomb_host.empty()) tomb_host.assign(total_rows, 0);
                for (uint32_t pp : pos) {
                    if ((int64_t)pp < fd.row_count)
                        tomb_host[rb + pp] = 1;
                }
            }
            rb += fd.row_count;
        }
    }
    if (!tomb_host.empty()) {
        run.tomb = (uint8_t *)plan->bufs.alloc(total_rows);
        if (!run.tomb ||
            hipMemcpy(run.tomb, tomb_host.data(), total_rows,
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_error("H2D tombstones failed");
            return false;
        }
    }
    int64_t row_base = 0;
    for (const auto &fd : files) {
        StagedFile sf;
        if (!load_file(fd.path, sf)) return false;
        if (sf.orc) {
            if (!stage_orc_file(plan, fd, sf, run, row_base)) return false;
            row_base += fd.row_count;
            continue;
        }
        std::vector<int> leaf(cols.size(), -1);
        for (size_t c = 0; c < cols.size(); c++) {
            for (size_t i = 0; i < sf.meta.schema_names.size(); i++)
                if (sf.meta.schema_names[i] == cols[c].name) leaf[c] = (int)i;
            if (leaf[c] < 0) {
                set_error("%s: column %s not found", fd.path.c_str(),
                          cols[c].name.c_str());
                return false;
            }
            if ((int)sf.meta.phys_types.size() > leaf[c] &&
                sf.meta.phys_types[leaf[c]] != expected_phys(cols[c].dtype)) {
                set_error("%s: column %s physical type %d does not match "
                          "declared type (expect %d; decimals ride "
                          "INT32/INT64 per ParquetSchemaConverter.java:"
                          "153-171, strings BYTE_ARRAY)",
                          fd.path.c_str(), cols[c].name.c_str(),
                          sf.meta.phys_types[leaf[c]],
                          expected_phys(cols[c].dtype));
                return false;
            }
        }
        // file-level GPU zstd pre-pass: decode EVERY
        // selected zstd chunk's pages in ONE k_zstd_pages batch — thousands
        // of wavefronts in flight hide the per-page serial decode latency
        // that per-chunk batches cannot. Failure falls back to the host
        // codec per chunk.
        std::vector<uint8_t> file_unc;
        std::map<const void *, int64_t> zchunk_base;
        // chunks whose decoded image can stay DEVICE-RESIDENT (simple
        // PLAIN, no def levels, no dictionary page): their uncompressed
        // bytes never cross PCIe — the PLAIN staging copies D2D
        std::map<const void *, uint8_t *> zchunk_dev;
        if (gpu_zstd_enabled()) {
            std::vector<ZstdBatchPage> zp;
            std::vector<std::pair<int64_t, int64_t>> host_rngs;
            int64_t lo = INT64_MAX, hi = 0, out = 0;
            // chunk classes: HOST (dict pages / non-PLAIN / strings:
            // full D2H), PREFIX (PLAIN under a nullable schema: D2H only
            // each page's def-level prefix; values stay on device unless
            // the prefix shows REAL nulls), DEV (required PLAIN: nothing
            // copies back)
            struct PfxChunk {
                const void *cc;
                int64_t base;
                int leafc;
            };
            std::vector<PfxChunk> pfx_chunks;
            const int64_t PFX = 16384;
            for (auto &rg : sf.meta.row_groups)
                for (size_t c = 0; c < cols.size(); c++) {
                    auto &cc = rg.columns[leaf[c]];
                    if (cc.codec != CODEC_ZSTD || cc.pages.empty()) continue;
                    if (zchunk_base.count(&cc)) continue;
                    zchunk_base[&cc] = out;
                    bool needs_host = cols[c].dtype == PMH_DT_STRING;
                    const bool has_def =
                        sf.meta.max_def_levels[leaf[c]] > 0;
                    int64_t c0 = out;
                    for (auto &pg : cc.pages) {
                        if (pg.page_type == 2 ||
                            (pg.page_type == 0 &&
                             pg.encoding != ENC_PLAIN))
                            needs_host = true;
                        if (pg.data_off < lo) lo = pg.data_off;
                        if (pg.data_off + pg.compressed_size > hi)
                            hi = pg.data_off + pg.compressed_size;
                        zp.push_back({pg.data_off, pg.compressed_size, out,
                                      pg.uncompressed_size});
                        out += pg.uncompressed_size;
                    }
                    if (needs_host)
                        host_rngs.push_back({c0, out});
                    else if (has_def)
                        pfx_chunks.push_back({&cc, c0, leaf[c]});
                    else
                        zchunk_dev[&cc] = nullptr;  // patched below
                }
            if (!zp.empty()) {
                for (auto &p : zp) p.src_off -= lo;
                file_unc.resize(out);
                uint8_t *dev_blob = nullptr;
                // page-prefix ranges for PREFIX chunks (def levels live
                // at page start)
                for (const auto &pc : pfx_chunks) {
                    const auto &cc =
                        *(const ColumnChunkMeta *)pc.cc;
                    int64_t o = pc.base;
                    for (auto &pg : cc.pages) {
                        int64_t take = pg.uncompressed_size < PFX
                                           ? pg.uncompressed_size
                                           : PFX;
                        host_rngs.push_back({o, o + take});
                        o += pg.uncompressed_size;
                    }
                }
                if (gpu_zstd_batch_dev(sf.data.data() + lo, hi - lo, zp,
                                       plan, &dev_blob, out, host_rngs,
                                       file_unc.data())) {
                    plan->stats.gpu_zstd_pages += (int64_t)zp.size();
                    // PREFIX chunks: stay on device unless a page's def
                    // stream shows real nulls (or overflows the prefix) —
                    // then the chunk needs its full host image
                    for (const auto &pc : pfx_chunks) {
                        const auto &cc =
                            *(const ColumnChunkMeta *)pc.cc;
                        bool full = false;
                        int64_t o = pc.base, cend = pc.base;
                        for (auto &pg : cc.pages)
                            cend += pg.uncompressed_size;
                        for (auto &pg : cc.pages) {
                            if (pg.page_type == 0) {
                                uint32_t dl;
                                memcpy(&dl, file_unc.data() + o, 4);
                                if (4 + (int64_t)dl > PFX ||
                                    def_levels_have_nulls(
                                        file_unc.data() + o + 4, dl,
                                        pg.num_values))
                                    full = true;
                            }
                            o += pg.uncompressed_size;
                            if (full) break;
                        }
                        if (full) {
                            if (hipMemcpy(file_unc.data() + pc.base,
                                          dev_blob + pc.base,
                                          cend - pc.base,
                                          hipMemcpyDeviceToHost) !=
                                hipSuccess) {
                                set_error("zstd prefix D2H failed");
                                return false;
                            }
                        } else {
                            zchunk_dev[pc.cc] = nullptr;
                        }
                    }
                    for (auto &kv : zchunk_dev)
                        kv.second = dev_blob + zchunk_base[kv.first];
                    if (getenv("PMH_DEBUG_ZSTD"))
                        fprintf(stderr,
                                "[zstd] file chunks=%zu dev-resident=%zu "
                                "host-rngs=%zu\n",
                                zchunk_base.size(), zchunk_dev.size(),
                                host_rngs.size());
                } else {
                    file_unc.clear();
                    zchunk_base.clear();
                    zchunk_dev.clear();
                }
            }
        }
        int64_t rg_row = 0;
        for (auto &rg : sf.meta.row_groups) {
            for (size_t c = 0; c < cols.size(); c++) {
                auto &cc = rg.columns[leaf[c]];
                RunCol &rc = run.cols[c];
                const int stored = cols[c].stored_esize;
                const int max_def = sf.meta.max_def_levels[leaf[c]];
                const int64_t chunk_row0 = row_base + rg_row;
                // uncompressed payload image of this chunk
                int64_t chunk_start = cc.dictionary_page_offset
                                          ? cc.dictionary_page_offset
                                          : cc.data_page_offset;
                std::vector<uint8_t> packed;  // compressed codecs
                std::vector<int64_t> ppo(cc.pages.size());
                const uint8_t *payload_base = sf.data.data() + chunk_start;
                const uint8_t *zdev_base = nullptr;
                auto zit = zchunk_base.find((const void *)&cc);
                if (zit != zchunk_base.end()) {
                    // pages already decoded by the file-level k_zstd_pages
                    // batch; the chunk's image starts at zit->second.
                    // Device-resident chunks (zchunk_dev) never copied
                    // back — the PLAIN staging below goes D2D.
                    int64_t o = 0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        ppo[pi] = o;
                        o += cc.pages[pi].uncompressed_size;
                    }
                    payload_base = file_unc.data() + zit->second;
                    auto zdv = zchunk_dev.find((const void *)&cc);
                    if (zdv != zchunk_dev.end()) zdev_base = zdv->second;
                } else if (cc.codec == CODEC_ZSTD || cc.codec == CODEC_GZIP ||
                    cc.codec == CODEC_SNAPPY) {
                    int64_t total_unc = 0;
                    for (auto &pg : cc.pages) total_unc += pg.uncompressed_size;
                    packed.resize(total_unc);
                    int64_t off = 0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        std::string cerr;
                        size_t got = 0;
                        bool ok;
                        if (cc.codec == CODEC_ZSTD)
                            ok = zstd_decompress(
                                sf.data.data() + pg.data_off,
                                pg.compressed_size, packed.data() + off,
                                pg.uncompressed_size);
                        else if (cc.codec == CODEC_GZIP)
                            ok = gzip_decompress_exact(
                                sf.data.data() + pg.data_off,
                                pg.compressed_size, packed.data() + off,
                                pg.uncompressed_size, cerr);
                        else
                            ok = snappy_decompress(
                                     sf.data.data() + pg.data_off,
                                     pg.compressed_size, packed.data() + off,
                                     pg.uncompressed_size, got, cerr) &&
                                 got == (size_t)pg.uncompressed_size;
                        if (!ok) {
                            if (!cerr.empty()) set_error("%s", cerr.c_str());
                            else if (cc.codec == CODEC_SNAPPY)
                                set_error("snappy page size mismatch");
                            return false;
                        }
                        ppo[pi] = off;
                        off += pg.uncompressed_size;
                    }
                    payload_base = packed.data();
                } else if (cc.codec == CODEC_UNCOMPRESSED) {
                    for (size_t pi = 0; pi < cc.pages.size(); pi++)
                        ppo[pi] = cc.pages[pi].data_off - chunk_start;
                } else {
                    set_error("%s: unsupported codec %d", fd.path.c_str(),
                              cc.codec);
                    return false;
                }
                // classify data pages
                bool has_plain = false, has_dict = false, has_delta = false,
                     has_dba = false;
                const uint8_t *dict_host = nullptr;
                int64_t dict_count = 0;
                for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                    auto &pg = cc.pages[pi];
                    if (pg.page_type == 2) {
                        dict_host = payload_base + ppo[pi];
                        dict_count = pg.num_values;
                    } else if (pg.encoding == ENC_PLAIN) {
                        has_plain = true;
                    } else if (pg.encoding == ENC_RLE_DICTIONARY ||
                               pg.encoding == ENC_PLAIN_DICTIONARY) {
                        has_dict = true;
                    } else if (pg.encoding == ENC_DELTA_BINARY_PACKED) {
                        has_delta = true;
                    } else if ((pg.encoding == ENC_DELTA_BYTE_ARRAY ||
                                pg.encoding ==
                                    ENC_DELTA_LENGTH_BYTE_ARRAY) &&
                               cols[c].dtype == PMH_DT_STRING) {
                        // DELTA_LENGTH_BYTE_ARRAY = DELTA_BYTE_ARRAY with
                        // no prefix stream (lengths + bytes)
                        has_dba = true;
                    } else {
                        set_error("%s: unsupported encoding %d",
                                  fd.path.c_str(), pg.encoding);
                        return false;
                    }
                }
                if (has_plain && has_dict) {
                    set_error("%s col %s: mixed PLAIN/dictionary pages in one "
                              "chunk not supported yet",
                              fd.path.c_str(), cols[c].name.c_str());
                    return false;
                }
                if ((has_delta || has_dba) &&
                    (has_plain || has_dict || (has_delta && has_dba))) {
                    set_error("%s col %s: mixed DELTA/other pages in one "
                              "chunk not supported", fd.path.c_str(),
                              cols[c].name.c_str());
                    return false;
                }
                if (cols[c].dtype == PMH_DT_STRING && has_plain) {
                    set_error("%s col %s: PLAIN byte-array pages are not on "
                              "the GPU path yet — string columns must be "
                              "dictionary-encoded (the parquet writer "
                              "default; C5's dictionary strings are)",
                              fd.path.c_str(), cols[c].name.c_str());
                    return false;
                }
                // per-page value offsets past def levels (+ null detection)
                std::vector<int64_t> vpos(cc.pages.size(), 0);
                bool chunk_nulls = false;
                for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                    auto &pg = cc.pages[pi];
                    if (pg.page_type != 0) continue;
                    const uint8_t *pp = payload_base + ppo[pi];
                    int64_t pos = 0;
                    if (max_def > 0) {
                        uint32_t dl_len;
                        memcpy(&dl_len, pp, 4);
                        chunk_nulls |= def_levels_have_nulls(pp + 4, dl_len,
                                                             pg.num_values);
                        pos = 4 + dl_len;
                    }
                    vpos[pi] = pos;
                }
                if (has_dba) {
                    // DELTA_BYTE_ARRAY strings (VectorizedDeltaByteArray-
                    // Reader.java): per page, prefix lengths (DELTA) +
                    // suffix lengths (DELTA) + suffix bytes; values share
                    // prefixes with their predecessor. They intern into
                    // the plan-level GLOBAL dictionary at staging (the ids
                    // are the device representation, like ORC DIRECT_V2).
                    if (!plan->sdicts[c])
                        plan->sdicts[c].reset(new StrDict());
                    StrDict *sd = plan->sdicts[c].get();
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        const uint8_t *pp = payload_base + ppo[pi];
                        int64_t plen2 = (cc.codec != CODEC_UNCOMPRESSED
                                             ? pg.uncompressed_size
                                             : pg.compressed_size);
                        int64_t row0 = chunk_row0 + pg.first_row;
                        int64_t n_dense = pg.num_values;
                        int64_t dense0 = row0;
                        bool to_dense = false;
                        if (max_def > 0) {
                            uint32_t dl_len;
                            memcpy(&dl_len, pp, 4);
                            if (chunk_nulls) {
                                rc.has_nulls = true;
                                int64_t rel =
                                    (int64_t)rc.levels_host.size();
                                rc.levels_host.insert(rc.levels_host.end(),
                                                      pp + 4,
                                                      pp + 4 + dl_len);
                                int64_t before = rc.dense_before;
                                if (!prescan_def(pp + 4, dl_len,
                                                 pg.num_values, row0, rel,
                                                 &rc.dense_before,
                                                 rc.def_host))
                                    return false;
                                n_dense = rc.dense_before - before;
                                dense0 = before;
                                to_dense = true;
                            }
                        }
                        int64_t pos = vpos[pi];
                        std::vector<int64_t> pre, suf;
                        const bool dlba =
                            pg.encoding == ENC_DELTA_LENGTH_BYTE_ARRAY;
                        int64_t used;
                        if (dlba) {
                            pre.assign((size_t)n_dense, 0);  // no prefixes
                        } else {
                            used = host_delta_i64(pp + pos, plen2 - pos,
                                                  pre);
                            if (used < 0 ||
                                (int64_t)pre.size() < n_dense) {
                                set_error("%s col %s: DELTA_BYTE_ARRAY "
                                          "prefix stream bad",
                                          fd.path.c_str(),
                                          cols[c].name.c_str());
                                return false;
                            }
                            pos += used;
                        }
                        used = host_delta_i64(pp + pos, plen2 - pos, suf);
                        if (used < 0 ||
                            (int64_t)suf.size() < n_dense) {
                            set_error("%s col %s: DELTA_BYTE_ARRAY suffix "
                                      "stream bad", fd.path.c_str(),
                                      cols[c].name.c_str());
                            return false;
                        }
                        pos += used;
                        std::string prev;
                        std::vector<int32_t> ids(n_dense);
                        for (int64_t i = 0; i < n_dense; i++) {
                            int64_t pl = pre[i], sl = suf[i];
                            if (pl < 0 || sl < 0 ||
                                pl > (int64_t)prev.size() ||
                                pos + sl > plen2) {
                                set_error("%s col %s: DELTA_BYTE_ARRAY "
                                          "lengths bad", fd.path.c_str(),
                                          cols[c].name.c_str());
                                return false;
                            }
                            std::string v = prev.substr(0, pl);
                            v.append((const char *)pp + pos, (size_t)sl);
                            pos += sl;
                            ids[i] = sd->add((const uint8_t *)v.data(),
                                             (uint32_t)v.size());
                            prev = std::move(v);
                        }
                        plan->encoded_bytes_total += plen2;
                        if (to_dense) {
                            const uint8_t *ib = (const uint8_t *)ids.data();
                            rc.dense_host.insert(rc.dense_host.end(), ib,
                                                 ib + n_dense * 4);
                            rc.dense_segs.emplace_back(dense0, n_dense * 4);
                        } else if (n_dense &&
                                   hipMemcpy((uint8_t *)rc.contig +
                                                 row0 * 4,
                                             ids.data(), n_dense * 4,
                                             hipMemcpyHostToDevice) !=
                                       hipSuccess) {
                            set_error("H2D failed");
                            return false;
                        }
                    }
                } else if (has_delta) {
                    if (cols[c].dtype == PMH_DT_STRING) {
                        set_error("%s col %s: DELTA byte arrays are a later "
                                  "round", fd.path.c_str(),
                                  cols[c].name.c_str());
                        return false;
                    }
                    // upload the encoded payload image (+16 B pad for the
                    // device bit-window loads); blocks decode on GPU at
                    // read time (k_delta_sum/scan/emit)
                    int64_t payload_len = 0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++)
                        payload_len = std::max(
                            payload_len,
                            ppo[pi] + (cc.codec != CODEC_UNCOMPRESSED
                                           ? cc.pages[pi].uncompressed_size
                                           : cc.pages[pi].compressed_size));
                    void *dev = plan->bufs.alloc(payload_len + 16);
                    if (!dev) return false;
                    if (hipMemcpy(dev, payload_base, payload_len,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                    plan->encoded_bytes_total += payload_len;
                    if (chunk_nulls) rc.has_nulls = true;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        const uint8_t *pp = payload_base + ppo[pi];
                        int64_t pos = vpos[pi];
                        int64_t plen = (cc.codec != CODEC_UNCOMPRESSED
                                            ? pg.uncompressed_size
                                            : pg.compressed_size);
                        std::string derr;
                        if (chunk_nulls && max_def > 0) {
                            // nullable DELTA: the stream encodes only the
                            // NON-NULL values — decode them to the dense
                            // buffer (out_addr 0 patched to dense_dev at
                            // finalize) and let k_level_scatter position
                            // them like dense PLAIN/dictionary values
                            uint32_t dl_len;
                            memcpy(&dl_len, pp, 4);
                            int64_t rel = (int64_t)rc.levels_host.size();
                            rc.levels_host.insert(rc.levels_host.end(),
                                                  pp + 4, pp + 4 + dl_len);
                            int64_t before = rc.dense_before;
                            if (!prescan_def(pp + 4, dl_len, pg.num_values,
                                             chunk_row0 + pg.first_row, rel,
                                             &rc.dense_before, rc.def_host))
                                return false;
                            int64_t nvalid = rc.dense_before - before;
                            if (!prescan_delta(pp + pos, plen - pos, nvalid,
                                               before, (uint64_t)dev,
                                               ppo[pi] + pos, 0ull, stored,
                                               rc.delta_host,
                                               rc.dstreams_host, derr)) {
                                set_error("%s col %s: %s", fd.path.c_str(),
                                          cols[c].name.c_str(),
                                          derr.c_str());
                                return false;
                            }
                            continue;
                        }
                        if (!prescan_delta(pp + pos, plen - pos,
                                           pg.num_values,
                                           chunk_row0 + pg.first_row,
                                           (uint64_t)dev, ppo[pi] + pos,
                                           (uint64_t)rc.contig, stored,
                                           rc.delta_host, rc.dstreams_host,
                                           derr)) {
                            set_error("%s col %s: %s", fd.path.c_str(),
                                      cols[c].name.c_str(), derr.c_str());
                            return false;
                        }
                    }
                } else if (chunk_nulls && has_dict) {
                    // dictionary chunk with nulls: def-level streams stay
                    // encoded; the id stream decodes to DENSE positions on
                    // the GPU (k_rle_decode), k_dict_gather fills the dense
                    // buffer, and k_level_scatter positions the rows
                    rc.has_nulls = true;
                    if (!dict_host) {
                        set_error("%s: dictionary page missing",
                                  fd.path.c_str());
                        return false;
                    }
                    rc.dict_encoded = true;
                    int64_t payload_len = 0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++)
                        payload_len = std::max(
                            payload_len,
                            ppo[pi] + (cc.codec != CODEC_UNCOMPRESSED
                                           ? cc.pages[pi].uncompressed_size
                                           : cc.pages[pi].compressed_size));
                    void *dev = plan->bufs.alloc(payload_len);
                    if (!dev) return false;
                    if (hipMemcpy(dev, payload_base, payload_len,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                    plan->encoded_bytes_total += payload_len;
                    int64_t dense_start = rc.dense_before;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        const uint8_t *pp = payload_base + ppo[pi];
                        uint32_t dl_len;
                        memcpy(&dl_len, pp, 4);
                        int64_t rel = (int64_t)rc.levels_host.size();
                        rc.levels_host.insert(rc.levels_host.end(), pp + 4,
                                              pp + 4 + dl_len);
                        int64_t before = rc.dense_before;
                        if (!prescan_def(pp + 4, dl_len, pg.num_values,
                                         chunk_row0 + pg.first_row, rel,
                                         &rc.dense_before, rc.def_host))
                            return false;
                        int64_t nvalid = rc.dense_before - before;
                        int64_t pos = vpos[pi];
                        int bw = pp[pos];
                        int64_t plen = (cc.codec != CODEC_UNCOMPRESSED
                                            ? pg.uncompressed_size
                                            : pg.compressed_size);
                        if (!prescan_rle(pp + pos + 1, plen - pos - 1, bw,
                                         nvalid, before, (uint64_t)dev,
                                         ppo[pi] + pos + 1, rc.rle_host))
                            return false;
                    }
                    void *dict_dev = nullptr;
                    if (!upload_chunk_dict(plan, (int)c, dict_host,
                                           dict_count, stored, &dict_dev))
                        return false;
                    GatherTask gt;
                    gt.start = dense_start;
                    gt.n = rc.dense_before - dense_start;
                    gt.dict_dev = dict_dev;
                    gt.to_dense = true;
                    rc.gathers.push_back(gt);
                } else if (chunk_nulls) {
                    // dense PLAIN values + def-level streams stay encoded;
                    // k_level_scatter positions them at read time
                    rc.has_nulls = true;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        const uint8_t *pp = payload_base + ppo[pi];
                        uint32_t dl_len;
                        memcpy(&dl_len, pp, 4);
                        int64_t rel = (int64_t)rc.levels_host.size();
                        rc.levels_host.insert(rc.levels_host.end(), pp + 4,
                                              pp + 4 + dl_len);
                        int64_t before = rc.dense_before;
                        if (!prescan_def(pp + 4, dl_len, pg.num_values,
                                         chunk_row0 + pg.first_row, rel,
                                         &rc.dense_before, rc.def_host))
                            return false;
                        int64_t nvalid = rc.dense_before - before;
                        rc.dense_host.insert(
                            rc.dense_host.end(), pp + vpos[pi],
                            pp + vpos[pi] + nvalid * stored);
                        rc.dense_segs.emplace_back(before,
                                                   nvalid * stored);
                    }
                } else if ((has_plain || (!has_dict && cc.num_values > 0))
                           && zdev_base) {
                    // PLAIN pages already decoded on DEVICE: stage D2D
                    int64_t row_at = chunk_row0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        int64_t nbytes = (int64_t)pg.num_values * stored;
                        if (nbytes &&
                            hipMemcpy((uint8_t *)rc.contig +
                                          row_at * stored,
                                      zdev_base + ppo[pi] + vpos[pi],
                                      nbytes,
                                      hipMemcpyDeviceToDevice) !=
                                hipSuccess) {
                            set_error("D2D staging failed");
                            return false;
                        }
                        row_at += pg.num_values;
                    }
                } else if (has_plain || (!has_dict && cc.num_values > 0)) {
                    // pack PLAIN value payloads and copy into contig
                    std::vector<uint8_t> pack(cc.num_values * stored);
                    int64_t off = 0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        int64_t nbytes = (int64_t)pg.num_values * stored;
                        memcpy(pack.data() + off,
                               payload_base + ppo[pi] + vpos[pi], nbytes);
                        off += nbytes;
                    }
                    if (off != (int64_t)pack.size()) {
                        set_error("%s: page value counts disagree with chunk",
                                  fd.path.c_str());
                        return false;
                    }
                    if (hipMemcpy((uint8_t *)rc.contig + chunk_row0 * stored,
                                  pack.data(), off,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                } else if (has_dict) {
                    if (!dict_host) {
                        set_error("%s: dictionary page missing",
                                  fd.path.c_str());
                        return false;
                    }
                    rc.dict_encoded = true;
                    // upload the id-stream payload image
                    int64_t payload_len = 0;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++)
                        payload_len =
                            std::max(payload_len,
                                     ppo[pi] + (cc.codec != CODEC_UNCOMPRESSED
                                                    ? cc.pages[pi].uncompressed_size
                                                    : cc.pages[pi].compressed_size));
                    void *dev = plan->bufs.alloc(payload_len);
                    if (!dev) return false;
                    if (hipMemcpy(dev, payload_base, payload_len,
                                  hipMemcpyHostToDevice) != hipSuccess) {
                        set_error("H2D failed");
                        return false;
                    }
                    plan->encoded_bytes_total += payload_len;
                    for (size_t pi = 0; pi < cc.pages.size(); pi++) {
                        auto &pg = cc.pages[pi];
                        if (pg.page_type != 0) continue;
                        const uint8_t *pp = payload_base + ppo[pi];
                        int64_t pos = vpos[pi];
                        int bw = pp[pos];
                        int64_t plen = (cc.codec != CODEC_UNCOMPRESSED
                                            ? pg.uncompressed_size
                                            : pg.compressed_size);
                        if (!prescan_rle(pp + pos + 1, plen - pos - 1, bw,
                                         pg.num_values,
                                         chunk_row0 + pg.first_row,
                                         (uint64_t)dev, ppo[pi] + pos + 1,
                                         rc.rle_host))
                            return false;
                    }
                    void *dict_dev = nullptr;
                    if (!upload_chunk_dict(plan, (int)c, dict_host,
                                           dict_count, stored, &dict_dev))
                        return false;
                    GatherTask gt;
                    gt.start = chunk_row0;
                    gt.n = cc.num_values;
                    gt.dict_dev = dict_dev;
                    rc.gathers.push_back(gt);
                }
            }
            rg_row += rg.num_rows;
        }
        if (rg_row != fd.row_count) {
            set_error("%s: rowCount %lld != file rows %lld", fd.path.c_str(),
                      (long long)fd.row_count, (long long)rg_row);
            return false;
        }
        row_base += rg_row;
    }
    run.length = row_base;
    for (size_t c = 0; c < run.cols.size(); c++) {
        RunCol &rc = run.cols[c];
        rc.n_rows = run.length;
        if (rc.dict_encoded) {
            rc.ids_dev = (int32_t *)plan->bufs.alloc(run.length * 4);
            rc.rle_dev = (RleChunk *)plan->bufs.alloc(rc.rle_host.size() *
                                                      sizeof(RleChunk));
            if (!rc.ids_dev || !rc.rle_dev) return false;
            if (hipMemcpy(rc.rle_dev, rc.rle_host.data(),
                          rc.rle_host.size() * sizeof(RleChunk),
                          hipMemcpyHostToDevice) != hipSuccess)
                return false;
        }
        if (rc.has_nulls) {
            // parquet PLAIN: dense values packed host-side (dense_host);
            // parquet dictionary / ORC: dense values are produced on-device
            // (k_dict_gather / k_rlev2) — size by the global dense count
            size_t dense_bytes =
                (size_t)rc.dense_before * plan->cols[c].stored_esize;
            rc.valid_dev = (uint8_t *)plan->bufs.alloc(run.length);
            rc.dense_dev = plan->bufs.alloc(dense_bytes);
            void *levels_dev = plan->bufs.alloc(rc.levels_host.size());
            if (!rc.valid_dev || !rc.dense_dev || !levels_dev)
                return false;
            // rows in non-null chunks of this column keep validity 1
            if (hipMemset(rc.valid_dev, 1, run.length) != hipSuccess)
                return false;
            size_t ho = 0;
            for (const auto &sg : rc.dense_segs) {
                if (sg.second > 0 &&
                    hipMemcpy((uint8_t *)rc.dense_dev +
                                  sg.first * plan->cols[c].stored_esize,
                              rc.dense_host.data() + ho, sg.second,
                              hipMemcpyHostToDevice) != hipSuccess)
                    return false;
                ho += sg.second;
            }
            if (hipMemcpy(levels_dev, rc.levels_host.data(),
                          rc.levels_host.size(),
                          hipMemcpyHostToDevice) != hipSuccess)
                return false;
            for (auto &dc : rc.def_host) {
                if (dc.kind == 1) dc.src += (uint64_t)levels_dev;
                dc.out_addr = (uint64_t)rc.contig;
                dc.valid_addr = (uint64_t)rc.valid_dev;
                dc.dense_addr = (uint64_t)rc.dense_dev;
                dc.esize = plan->cols[c].stored_esize;
            }
            // nullable DELTA pages decode into the dense buffer
            // (k_delta_emit runs before k_level_scatter)
            for (auto &ch : rc.delta_host)
                if (!ch.out_addr) ch.out_addr = (uint64_t)rc.dense_dev;
            for (auto &ds : rc.dstreams_host)
                if (!ds.out_addr) ds.out_addr = (uint64_t)rc.dense_dev;
            plan->encoded_bytes_total += rc.levels_host.size();
            rc.dense_host.clear();
            rc.dense_host.shrink_to_fit();
            rc.levels_host.clear();
            rc.levels_host.shrink_to_fit();
        }
        for (auto &vc : rc.rlev2_host)
            vc.out_addr =
                vc.dense_target ? (uint64_t)rc.dense_dev : (uint64_t)rc.contig;
        DevPage dp{(uint64_t)rc.contig, 0};
        rc.pages_host.assign(1, dp);
        rc.n_pages = 1;
        rc.pages_dev = (DevPage *)plan->bufs.alloc(sizeof(DevPage));
        if (!rc.pages_dev) return false;
        if (hipMemcpy(rc.pages_dev, rc.pages_host.data(), sizeof(DevPage),
                      hipMemcpyHostToDevice) != hipSuccess)
            return false;
    }
    return true;
}

// Like gpu_zstd_batch, but the decoded blob STAYS on device (allocated
// from the plan arena; *dev_out) and only `host_rngs` (chunks whose
// staging needs host peeks) copy back. Simple PLAIN chunks then stage
// with D2D copies — no PCIe round trip for their bytes.
static bool gpu_zstd_batch_dev(
    const uint8_t *src_base, int64_t src_total,
    const std::vector<ZstdBatchPage> &pages, pmh_plan_t *plan,
    uint8_t **dev_out, int64_t out_total,
    const std::vector<std::pair<int64_t, int64_t>> &host_rngs,
    uint8_t *host_out) {
    if (pages.empty()) return true;
    int n = (int)pages.size();
    uint8_t *d_src = nullptr, *d_scr = nullptr;
    ZstdJob *d_jobs = nullptr;
    int64_t *d_st = nullptr;
    uint8_t *d_dst = (uint8_t *)plan->bufs.alloc(out_total ? out_total : 1);
    if (!d_dst) return false;
    std::vector<ZstdJob> jobs(n);
    for (int i = 0; i < n; i++)
        jobs[i] = {(uint64_t)pages[i].src_off, (uint64_t)pages[i].dst_off,
                   (uint32_t)pages[i].src_len, (uint32_t)pages[i].dst_len};
    bool ok = false;
    std::vector<int64_t> st(n);
    do {
        if (hipMalloc(&d_src, src_total) != hipSuccess) break;
        if (hipMalloc(&d_scr, (size_t)n * PZ_SLOT) != hipSuccess) break;
        if (hipMalloc(&d_jobs, n * sizeof(ZstdJob)) != hipSuccess) break;
        if (hipMalloc(&d_st, n * 8) != hipSuccess) break;
        if (hipMemcpy(d_src, src_base, src_total, hipMemcpyHostToDevice) !=
            hipSuccess)
            break;
        if (hipMemcpy(d_jobs, jobs.data(), n * sizeof(ZstdJob),
                      hipMemcpyHostToDevice) != hipSuccess)
            break;
        if (pmh_launch_zstd_pages(d_src, d_jobs, n, d_dst, d_scr, d_st,
                                  nullptr) != hipSuccess)
            break;
        if (hipMemcpy(st.data(), d_st, n * 8, hipMemcpyDeviceToHost) !=
            hipSuccess)
            break;
        bool all = true;
        for (int i = 0; i < n; i++)
            if (st[i] != pages[i].dst_len) all = false;
        if (!all) break;
        ok = true;
        for (auto &r : host_rngs) {
            if (r.second > r.first &&
                hipMemcpy(host_out + r.first, d_dst + r.first,
                          r.second - r.first,
                          hipMemcpyDeviceToHost) != hipSuccess) {
                ok = false;
                break;
            }
        }
    } while (0);
    if (d_src) (void)hipFree(d_src);
    if (d_scr) (void)hipFree(d_scr);
    if (d_jobs) (void)hipFree(d_jobs);
    if (d_st) (void)hipFree(d_st);
    if (ok) *dev_out = d_dst;
    return ok;
}

// Hierarchical section setup: group real runs into batches (<= PMH_MAX_RUNS
// runs AND < 2^PMH_ROW_BITS input rows each, so every virtual run respects
// the winner row packing), allocate the virtual runs' columns, and swap the
// section's chain descriptors to point at them. The batch pass runs in
// pmh_read_next (it needs decoded columns).
static bool build_hier_section(pmh_plan_t *plan, Section &sec);

static bool build_hier_section(pmh_plan_t *plan, Section &sec) {
    const int n_real = (int)sec.runs.size();
    const int n_cols = (int)plan->cols.size();
    // keep the real-run descriptors for the batch pass
    sec.rkeys = sec.key_cols;
    sec.rseqs = sec.seq_cols;
    sec.rkinds = sec.kind_cols;
    sec.rall = sec.all_cols;
    sec.rlens = sec.lens_dev;
    sec.rtombs = sec.tombs_dev;
    sec.tombs_dev = nullptr;  // tombstones are consumed by the batch pass
    // batches: <= PMH_MAX_RUNS runs and < 2^PMH_ROW_BITS input rows each
    {
        int lo = 0;
        while (lo < n_real) {
            int hi = lo;
            int64_t rows = 0;
            while (hi < n_real && hi - lo < PMH_MAX_RUNS &&
                   rows + sec.runs[hi].length < ((int64_t)1 << PMH_ROW_BITS))
                rows += sec.runs[hi++].length;
            if (hi == lo) {
                set_error("run of %lld rows cannot batch under the 2^%d "
                          "winner row limit",
                          (long long)sec.runs[lo].length, PMH_ROW_BITS);
                return false;
            }
            sec.blo.push_back(lo);
            sec.bhi.push_back(hi);
            sec.brows.push_back(rows);
            sec.bntiles.push_back((rows + PMH_TILE_ROWS - 1) / PMH_TILE_ROWS);
            lo = hi;
        }
        if ((int)sec.blo.size() > PMH_MAX_RUNS) {
            set_error("section needs %zu virtual runs > %d (two-level "
                      "hierarchy is a later round)",
                      sec.blo.size(), PMH_MAX_RUNS);
            return false;
        }
    }
    const int nb = (int)sec.blo.size();
    // batch-pass dtype map: int8/int16 emit as int32 so the virtual runs
    // hold the STORED width the second merge/emit pass reads
    if (!plan->hier_dtype_dev) {
        std::vector<uint8_t> hd(n_cols);
        for (int c = 0; c < n_cols; c++) {
            uint8_t d = (uint8_t)plan->cols[c].dtype;
            hd[c] = (d == 1 || d == 2) ? 3 : d;
        }
        plan->hier_dtype_dev = (uint8_t *)plan->bufs.alloc(n_cols);
        if (!plan->hier_dtype_dev ||
            hipMemcpy(plan->hier_dtype_dev, hd.data(), n_cols,
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_error("H2D hier dtype map failed");
            return false;
        }
    }
    // virtual runs: one per batch; columns at stored width
    std::vector<DevCol> vkeys(nb), vseqs(nb), vkinds(nb), vall(nb * n_cols);
    int seq_idx = plan->n_key_cols;
    int kind_idx = plan->n_key_cols + 1;
    for (int b = 0; b < nb; b++) {
        std::vector<void *> outs(n_cols);
        std::vector<uint8_t *> valids(n_cols, nullptr);
        for (int c = 0; c < n_cols; c++) {
            int es = plan->cols[c].stored_esize;
            outs[c] = plan->bufs.alloc((size_t)sec.brows[b] * es);
            if (!outs[c]) return false;
            // nullability decided per plan at create end; allocate byte
            // validity for every value column that CAN be null in any run
            bool nullable = false;
            for (auto &run : sec.runs)
                if (run.cols[c].has_nulls) nullable = true;
            if (nullable) {
                valids[c] = (uint8_t *)plan->bufs.alloc(sec.brows[b]);
                if (!valids[c]) return false;
            }
            DevCol dc{(uint64_t)outs[c], (uint64_t)valids[c], nullptr, 1,
                      es};
            vall[b * n_cols + c] = dc;
            if (c == 0) vkeys[b] = dc;
            if (c == seq_idx) vseqs[b] = dc;
            if (c == kind_idx) vkinds[b] = dc;
        }
        void **op = (void **)plan->bufs.alloc(n_cols * sizeof(void *));
        uint8_t **vp =
            (uint8_t **)plan->bufs.alloc(n_cols * sizeof(void *));
        if (!op || !vp ||
            hipMemcpy(op, outs.data(), n_cols * sizeof(void *),
                      hipMemcpyHostToDevice) != hipSuccess ||
            hipMemcpy(vp, valids.data(), n_cols * sizeof(void *),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_error("H2D hier batch outputs failed");
            return false;
        }
        sec.bout_ptrs.push_back(op);
        sec.bout_valid.push_back(vp);
    }
    auto up = [&](const void *host, size_t n) -> void * {
        void *d = plan->bufs.alloc(n);
        if (d && host &&
            hipMemcpy(d, host, n, hipMemcpyHostToDevice) != hipSuccess)
            return nullptr;
        return d;
    };
    sec.key_cols = (DevCol *)up(vkeys.data(), nb * sizeof(DevCol));
    sec.seq_cols = (DevCol *)up(vseqs.data(), nb * sizeof(DevCol));
    sec.kind_cols = (DevCol *)up(vkinds.data(), nb * sizeof(DevCol));
    sec.all_cols = (DevCol *)up(vall.data(), vall.size() * sizeof(DevCol));
    sec.lens_dev = (int64_t *)plan->bufs.alloc(nb * 8);
    if (!sec.key_cols || !sec.seq_cols || !sec.kind_cols || !sec.all_cols ||
        !sec.lens_dev) {
        set_error("hier descriptor allocation failed");
        return false;
    }
    return true;
}

static bool build_section_descriptors(pmh_plan_t *plan, Section &sec) {
    int k = (int)sec.runs.size();
    int n_cols = (int)plan->cols.size();
    std::vector<DevCol> keyv(k), seqv(k), kindv(k), allv(k * n_cols);
    std::vector<int64_t> lens(k);
    int seq_idx = plan->n_key_cols;
    int kind_idx = plan->n_key_cols + 1;
    if (plan->composite_key) sec.ckeys.assign(k, nullptr);
    for (int r = 0; r < k; r++) {
        Run &run = sec.runs[r];
        lens[r] = run.length;
        sec.total_rows += run.length;
        for (int c = 0; c < n_cols; c++) {
            DevCol dc{(uint64_t)run.cols[c].contig,
                      (uint64_t)run.cols[c].valid_dev,
                      run.cols[c].pages_dev, run.cols[c].n_pages,
                      plan->cols[c].stored_esize};
            allv[r * n_cols + c] = dc;
            if (c == 0) keyv[r] = dc;  // single-column key: the column itself
            if (c == seq_idx) seqv[r] = dc;
            if (c == kind_idx) kindv[r] = dc;
        }
        if (plan->composite_key) {
            // the partition/merge comparand is the packed composite key,
            // built per pass by k_composite
            int64_t n = run.length > 0 ? run.length : 1;
            sec.ckeys[r] = (int64_t *)plan->bufs.alloc(n * 8);
            if (!sec.ckeys[r]) return false;
            keyv[r] = DevCol{(uint64_t)sec.ckeys[r], 0, nullptr, 1, 8};
        }
    }
    sec.n_tiles = (sec.total_rows + PMH_TILE_ROWS - 1) / PMH_TILE_ROWS;
    if (sec.n_tiles == 0) sec.n_tiles = 1;
    auto up = [&](const void *host, size_t n) -> void * {
        void *d = plan->bufs.alloc(n);
        if (d && host) (void)hipMemcpy(d, host, n, hipMemcpyHostToDevice);
        return d;
    };
    sec.key_cols = (DevCol *)up(keyv.data(), k * sizeof(DevCol));
    sec.seq_cols = (DevCol *)up(seqv.data(), k * sizeof(DevCol));
    sec.kind_cols = (DevCol *)up(kindv.data(), k * sizeof(DevCol));
    sec.all_cols = (DevCol *)up(allv.data(), allv.size() * sizeof(DevCol));
    sec.lens_dev = (int64_t *)up(lens.data(), k * sizeof(int64_t));
    sec.cuts = (int32_t *)plan->bufs.alloc((sec.n_tiles + 1) * k * 4);
    if (plan->fused) {
        // single-pass path: no winners round-trip, no scan arrays — just
        // the per-tile lookback words and the tile ticket
        sec.status = (uint64_t *)plan->bufs.alloc(sec.n_tiles * 8);
        sec.ticket = (uint64_t *)plan->bufs.alloc(8);
        if (!sec.status || !sec.ticket) return false;
        if (plan->fsplit) {
            sec.dense_winners =
                (uint32_t *)plan->bufs.alloc(sec.total_rows * 4);
            if (!sec.dense_winners) return false;
        }
    } else {
        sec.winners = (uint32_t *)plan->bufs.alloc(
            sec.n_tiles * (PMH_TILE_ROWS + PMH_MAX_RUNS) * 4);
        sec.tile_counts = (int32_t *)plan->bufs.alloc(sec.n_tiles * 4);
        sec.tile_offsets = (int64_t *)plan->bufs.alloc(sec.n_tiles * 8);
        if (!sec.winners || !sec.tile_counts || !sec.tile_offsets)
            return false;
    }
    if (plan->changelog) {
        std::vector<uint8_t> lv(k);
        for (int r = 0; r < k; r++)
            lv[r] = (uint8_t)sec.runs[r].level;
        sec.run_levels = (uint8_t *)up(lv.data(), k);
        int64_t slots2 = sec.n_tiles * 2 * (PMH_TILE_ROWS + PMH_MAX_RUNS);
        sec.cl_entries = (uint64_t *)plan->bufs.alloc(slots2 * 8);
        sec.cl_rows = (uint64_t *)plan->bufs.alloc(slots2 * 8);
        sec.cl_counts = (int32_t *)plan->bufs.alloc(sec.n_tiles * 4);
        sec.cl_offsets = (int64_t *)plan->bufs.alloc(sec.n_tiles * 8);
        sec.cl_total_dev = (int64_t *)plan->bufs.alloc(8);
        if (!sec.run_levels || !sec.cl_entries || !sec.cl_rows ||
            !sec.cl_counts || !sec.cl_offsets || !sec.cl_total_dev)
            return false;
    }
    sec.total_dev = (int64_t *)plan->bufs.alloc(8);
    sec.err_dev = (uint32_t *)plan->bufs.alloc(4);
    if (sec.err_dev) (void)hipMemset(sec.err_dev, 0, 4);
    {
        std::vector<uint64_t> th(k, 0);
        for (int r = 0; r < k; r++) {
            th[r] = (uint64_t)sec.runs[r].tomb;
            sec.any_tomb |= sec.runs[r].tomb != nullptr;
        }
        if (sec.any_tomb) {
            sec.tombs_dev = (uint64_t *)up(th.data(), k * 8);
            if (!sec.tombs_dev) return false;
        }
    }
    if (plan->pu) {
        sec.group_start = (uint16_t *)plan->bufs.alloc(
            sec.n_tiles * (PMH_TILE_ROWS + 1) * 2);
        if (!sec.group_start) return false;
        bool any_valid = false;
        for (auto &run : sec.runs)
            for (auto &rc : run.cols)
                if (rc.valid_dev) any_valid = true;
        if (any_valid && n_cols <= 64) {
            sec.row_masks.assign(k, nullptr);
            for (int r = 0; r < k; r++) {
                int64_t n = sec.runs[r].length > 0 ? sec.runs[r].length : 1;
                sec.row_masks[r] = (uint64_t *)plan->bufs.alloc(n * 8);
                if (!sec.row_masks[r]) return false;
            }
            sec.row_masks_dev =
                (uint64_t **)up(sec.row_masks.data(), k * sizeof(void *));
            if (!sec.row_masks_dev) return false;
        }
    }
    std::vector<Rlev2Chunk> all_v;
    std::vector<RleChunk> all_d;
    std::vector<DeltaChunk> all_dc;
    std::vector<DeltaStream> all_ds;
    for (auto &run : sec.runs)
        for (auto &rc : run.cols) {
            sec.any_dict |= rc.dict_encoded;
            sec.any_decode |= rc.dict_encoded || rc.has_nulls ||
                              !rc.rlev2_host.empty() ||
                              !rc.delta_host.empty() ||
                              !rc.gathers.empty();
            all_v.insert(all_v.end(), rc.rlev2_host.begin(),
                         rc.rlev2_host.end());
            all_d.insert(all_d.end(), rc.def_host.begin(), rc.def_host.end());
            const int64_t cbase = (int64_t)all_dc.size();
            all_dc.insert(all_dc.end(), rc.delta_host.begin(),
                          rc.delta_host.end());
            for (DeltaStream st : rc.dstreams_host) {
                st.chunk_lo += cbase;  // indices into the BATCHED arrays
                st.chunk_hi += cbase;
                all_ds.push_back(st);
            }
        }
    sec.n_rlev2 = (int64_t)all_v.size();
    sec.n_def = (int64_t)all_d.size();
    if (sec.n_rlev2) {
        sec.rlev2_all =
            (Rlev2Chunk *)up(all_v.data(), all_v.size() * sizeof(Rlev2Chunk));
        if (!sec.rlev2_all) return false;
    }
    if (sec.n_def) {
        sec.def_all =
            (RleChunk *)up(all_d.data(), all_d.size() * sizeof(RleChunk));
        if (!sec.def_all) return false;
    }
    sec.n_delta = (int64_t)all_dc.size();
    sec.n_dstreams = (int64_t)all_ds.size();
    if (sec.n_delta) {
        sec.delta_all = (DeltaChunk *)up(all_dc.data(),
                                         all_dc.size() * sizeof(DeltaChunk));
        sec.dstreams_all = (DeltaStream *)up(
            all_ds.data(), all_ds.size() * sizeof(DeltaStream));
        sec.delta_sums = (int64_t *)plan->bufs.alloc(sec.n_delta * 8);
        sec.delta_bases = (int64_t *)plan->bufs.alloc(sec.n_delta * 8);
        if (!sec.delta_all || !sec.dstreams_all || !sec.delta_sums ||
            !sec.delta_bases)
            return false;
    }
    return sec.key_cols && sec.seq_cols && sec.kind_cols && sec.all_cols &&
           sec.lens_dev && sec.cuts && sec.total_dev && sec.err_dev;
}

}  // namespace pmh

// ==================================================================== C ABI

using namespace pmh;

extern "C" {

const char *pmh_last_error(void) { return last_error().c_str(); }

pmh_session_t *pmh_open_session(int device) {
    auto *s = new pmh_session_t();
    s->device = device;
    if (device >= 0) {
        if (hipSetDevice(device) != hipSuccess) {
            set_error("hipSetDevice(%d) failed", device);
            delete s;
            return nullptr;
        }
    }
    return s;
}

void pmh_close_session(pmh_session_t *s) { delete s; }

pmh_plan_t *pmh_plan_create(pmh_session_t *s, const char *plan_json) {
    if (!s) {
        set_error("null session");
        return nullptr;
    }
    if (s->device < 0) {
        set_error("pmh_plan_create requires a GPU session (device >= 0)");
        return nullptr;
    }
    (void)hipSetDevice(s->device);
    std::unique_ptr<pmh_plan_t> plan(new pmh_plan_t());
    plan->session = s;
    try {
        Json j = JsonParser(plan_json).parse();
        // columns: key cols, then seq, kind, then value cols
        auto add_col = [&](const Json &cj) -> bool {
            ColSpec cs;
            cs.name = cj["name"].as_str();
            const std::string ts = cj["type"].as_str();
            int dp = 0, ds = 0;
            if (parse_decimal(ts, &dp, &ds)) {
                if (dp <= 0 || dp > 18 || ds < 0 || ds > dp) {
                    set_error("decimal(%d,%d): v1 supports precision 1..18 "
                              "(INT32/INT64 physical per the reference; "
                              "FLBA decimals are a later round)", dp, ds);
                    return false;
                }
                cs.dtype = dp <= 9 ? PMH_DT_INT32 : PMH_DT_INT64;
                cs.precision = dp;
                cs.scale = ds;
            } else {
                cs.dtype = dtype_from_str(ts);
            }
            if (cs.dtype < 0 || cs.name.empty()) {
                set_error("bad column spec");
                return false;
            }
            cs.out_esize = dtype_out_esize(cs.dtype);
            cs.stored_esize = dtype_stored_esize(cs.dtype);
            plan->cols.push_back(cs);
            return true;
        };
        const Json &kc = j["key_cols"];
        if (kc.arr.empty() || kc.arr.size() > 8) {
            set_error("v1 supports 1..8 integer key columns (got %zu)",
                      kc.arr.size());
            return nullptr;
        }
        int key_total_bits = 0;
        {
            int shift = 64;
            for (size_t i = 0; i < kc.arr.size(); i++) {
                int dt = dtype_from_str(kc.arr[i]["type"].as_str());
                if (dt < PMH_DT_INT8 || dt > PMH_DT_INT64) {
                    set_error("key column %zu: integer types only "
                              "(TINYINT..BIGINT) in v1",
                              i);
                    return nullptr;
                }
                int bits = 8 << (dt - PMH_DT_INT8);
                key_total_bits += bits;
                shift -= bits;
                if (key_total_bits > 64) {
                    set_error("composite key widths sum to %d bits > 64 "
                              "(the order-preserving packed comparand is one "
                              "int64; wider keys are a later round)",
                              key_total_bits);
                    return nullptr;
                }
                plan->key_shifts |= (uint64_t)(uint8_t)shift << (8 * i);
                plan->key_bits |= (uint64_t)(uint8_t)bits << (8 * i);
            }
        }
        plan->n_key_cols = (int)kc.arr.size();
        plan->composite_key = plan->n_key_cols > 1;
        for (const auto &c : kc.arr)
            if (!add_col(c)) return nullptr;
        {  // _SEQUENCE_NUMBER, _VALUE_KIND (SpecialFields.java:79-83)
            ColSpec seq{"_SEQUENCE_NUMBER", PMH_DT_INT64, 8, 8};
            ColSpec kind{"_VALUE_KIND", PMH_DT_INT8, 1, 4};
            plan->cols.push_back(seq);
            plan->cols.push_back(kind);
        }
        for (const auto &c : j["value_cols"].arr)
            if (!add_col(c)) return nullptr;
        plan->sdicts.resize(plan->cols.size());  // before staging touches it
        std::string engine = j["merge_engine"].as_str("deduplicate");
        if (engine == "partial-update") {
            plan->pu = true;  // INSERT-only unless remove-record-on-delete
                              // or sequence groups
                              // (PartialUpdateMergeFunction rejects retracts
                              // by default, :170-186)
            plan->rrod = j["remove_record_on_delete"].as_bool(false);
            const Json &sgs = j["sequence_groups"];
            if (!sgs.arr.empty()) {
                plan->seqg = true;
                if (sgs.arr.size() > 16) {
                    set_error("at most 16 sequence groups in v1");
                    return nullptr;
                }
                const int first_val = plan->n_key_cols + 2;
                const int n_cols = (int)plan->cols.size();
                std::vector<uint8_t> cg(n_cols, 0xff);
                std::vector<int16_t> sf(sgs.arr.size() * 4, -1);
                std::vector<uint8_t> ns(sgs.arr.size(), 0);
                auto col_idx = [&](const std::string &nm) -> int {
                    for (int c = first_val; c < n_cols; c++)
                        if (plan->cols[c].name == nm) return c;
                    return -1;
                };
                for (size_t g = 0; g < sgs.arr.size(); g++) {
                    const Json &sg = sgs.arr[g];
                    const auto &sfj = sg["sequence_fields"].arr;
                    if (sfj.empty() || sfj.size() > 4) {
                        set_error("sequence group %zu: 1..4 sequence "
                                  "fields in v1", g);
                        return nullptr;
                    }
                    for (size_t i2 = 0; i2 < sfj.size(); i2++) {
                        int c = col_idx(sfj[i2].as_str());
                        if (c < 0 || plan->cols[c].dtype > PMH_DT_INT64) {
                            set_error("sequence field '%s': integer value "
                                      "columns only in v1",
                                      sfj[i2].as_str().c_str());
                            return nullptr;
                        }
                        if (cg[c] != 0xff) {
                            set_error("column '%s' is in two sequence "
                                      "groups", sfj[i2].as_str().c_str());
                            return nullptr;
                        }
                        cg[c] = (uint8_t)g;
                        sf[g * 4 + i2] = (int16_t)c;
                    }
                    ns[g] = (uint8_t)sfj.size();
                    for (const auto &mf : sg["group_fields"].arr) {
                        int c = col_idx(mf.as_str());
                        if (c < 0) {
                            set_error("sequence-group member '%s' is not a "
                                      "value column", mf.as_str().c_str());
                            return nullptr;
                        }
                        if (cg[c] != 0xff) {
                            set_error("column '%s' is in two sequence "
                                      "groups", mf.as_str().c_str());
                            return nullptr;
                        }
                        cg[c] = (uint8_t)g;
                    }
                }
                plan->n_seq_groups = (int)sgs.arr.size();
                plan->col_group_dev = (uint8_t *)plan->bufs.alloc(n_cols);
                plan->sg_fields_dev =
                    (int16_t *)plan->bufs.alloc(sf.size() * 2);
                plan->sg_nseq_dev = (uint8_t *)plan->bufs.alloc(ns.size());
                if (!plan->col_group_dev || !plan->sg_fields_dev ||
                    !plan->sg_nseq_dev ||
                    hipMemcpy(plan->col_group_dev, cg.data(), n_cols,
                              hipMemcpyHostToDevice) != hipSuccess ||
                    hipMemcpy(plan->sg_fields_dev, sf.data(), sf.size() * 2,
                              hipMemcpyHostToDevice) != hipSuccess ||
                    hipMemcpy(plan->sg_nseq_dev, ns.data(), ns.size(),
                              hipMemcpyHostToDevice) != hipSuccess) {
                    set_error("H2D of sequence-group tables failed");
                    return nullptr;
                }
            }
        } else if (engine == "first-row") {
            plan->first_row = true;  // FirstRowMergeFunction.java:32-77
        } else if (engine == "aggregation") {
            // AggregateMergeFunction.java:50-125 — per-field aggregators
            // over the same grouped stream as partial-update
            plan->pu = true;
            plan->agg = true;
            plan->rrod = j["remove_record_on_delete"].as_bool(false);
        } else if (engine != "deduplicate") {
            set_error("merge engine '%s' not on the GPU path (deduplicate | "
                      "partial-update | aggregation | first-row)",
                      engine.c_str());
            return nullptr;
        }
        // sort-engine (CoreOptions.java:733-736, default LOSER_TREE):
        // both engines share one output contract (SortMergeReader.java:41-57
        // — identical merged stream), so either value maps to the same GPU
        // merge; accepted for drop-in config compatibility.
        {
            std::string se = j["sort_engine"].as_str("loser-tree");
            if (se != "loser-tree" && se != "min-heap") {
                set_error("unknown sort-engine '%s'", se.c_str());
                return nullptr;
            }
        }
        {
            const Json &uf = j["sequence_fields"];
            if (!uf.arr.empty()) {
                if (plan->pu) {
                    set_error("sequence.field with partial-update/"
                              "aggregation engines is a later round (v1: "
                              "deduplicate / first-row)");
                    return nullptr;
                }
                if (uf.arr.size() > 4) {
                    set_error("at most 4 sequence fields in v1");
                    return nullptr;
                }
                const int first_val = plan->n_key_cols + 2;
                const int n_cols = (int)plan->cols.size();
                std::vector<int16_t> uc;
                for (const auto &fj : uf.arr) {
                    int idx = -1;
                    for (int c = first_val; c < n_cols; c++)
                        if (plan->cols[c].name == fj.as_str()) idx = c;
                    if (idx < 0 ||
                        plan->cols[idx].dtype > PMH_DT_INT64) {
                        set_error("sequence field '%s': integer value "
                                  "columns only in v1",
                                  fj.as_str().c_str());
                        return nullptr;
                    }
                    uc.push_back((int16_t)idx);
                }
                plan->n_useq = (int)uc.size();
                plan->useq_dev =
                    (int16_t *)plan->bufs.alloc(uc.size() * 2);
                if (!plan->useq_dev ||
                    hipMemcpy(plan->useq_dev, uc.data(), uc.size() * 2,
                              hipMemcpyHostToDevice) != hipSuccess) {
                    set_error("H2D of sequence fields failed");
                    return nullptr;
                }
            }
        }
        plan->drop_delete = j["drop_delete"].as_bool(true);
        plan->ignore_delete = j["ignore_delete"].as_bool(false);
        {
            std::string cp = j["changelog_producer"].as_str("none");
            if (cp == "full-compaction") {
                plan->changelog = true;
                plan->cl_row_dedup =
                    j["changelog_row_deduplicate"].as_bool(false);
                plan->max_level = (int)j["max_level"].as_i64(-1);
                if (plan->max_level < 0) {
                    set_error("changelog_producer=full-compaction needs "
                              "max_level (the table's num-levels - 1; "
                              "FullChangelogMergeFunctionWrapper maxLevel)");
                    return nullptr;
                }
                if (plan->pu || plan->agg) {
                    set_error("changelog_producer=full-compaction is "
                              "supported for deduplicate/first-row in v1 "
                              "(partial-update/aggregation changelog is a "
                              "later round)");
                    return nullptr;
                }
            } else if (cp != "none") {
                set_error("changelog_producer '%s' not supported (none | "
                          "full-compaction; lookup changelog is a later "
                          "round)", cp.c_str());
                return nullptr;
            }
        }
        {
            // Same-box A/B (DESIGN.md §7 experiment log): the classic
            // partition->merge->scan->emit chain with the two-level
            // partition is the fastest winner-engine configuration
            // (C2 6.4ms vs 7.1ms split-fused; 16x20M 31.6 vs 36.4ms), so
            // the chain is the product default. The fused single-pass path
            // stays as the path for user-defined sequence fields (its
            // comparator lives there) and behind PMH_FUSED=1 for A/B.
            const char *pf = getenv("PMH_FUSED");
            bool want_fused = pf ? pf[0] != '0' : (plan->n_useq > 0);
            plan->fused = !plan->pu && want_fused;
            if (plan->changelog && plan->fused) {
                if (plan->n_useq > 0 || (pf && pf[0] != '0')) {
                    set_error("changelog_producer=full-compaction runs on "
                              "the classic merge chain (no sequence.field / "
                              "PMH_FUSED=1 combination in v1)");
                    return nullptr;
                }
                plan->fused = false;
            }
            if (plan->n_useq > 0 && !plan->pu && !plan->fused) {
                set_error("sequence.field needs the fused merge path "
                          "(unset PMH_FUSED=0)");
                return nullptr;
            }
            // within the fused path, split value emission (merge_emit +
            // emit_dense) is the faster variant; PMH_FSPLIT=0 keeps
            // in-kernel emission for A/B
            const char *fs = getenv("PMH_FSPLIT");
            plan->fsplit = plan->fused && !(fs && fs[0] == '0');
        }
        if (plan->rrod && plan->ignore_delete) {
            set_error("remove-record-on-delete cannot be used with "
                      "ignore-delete (PartialUpdateMergeFunction.java:"
                      "491-495)");
            return nullptr;
        }
        plan->host_output = j["output"].as_str("device") == "host";
        {
            // "filters": [{"field": "...", "op": "eq|ne|lt|le|gt|ge|
            // is_null|is_not_null", "literal": x}] — a conjunction; the
            // reference pushes VALUE filters only into single-run
            // sections (overlapping runs would lose newer records,
            // MergeFileSplitRead.java:227-239). Key filters prune by file
            // stats before planning (caller side, as DataSplit already
            // carries the pruned file list).
            const Json &fj = j["filters"];
            for (const auto &f : fj.arr) {
                std::string field = f["field"].as_str("");
                std::string op = f["op"].as_str("");
                int idx = -1;
                for (size_t c = 0; c < plan->cols.size(); c++)
                    if (plan->cols[c].name == field) idx = (int)c;
                if (idx < 0) {
                    set_error("filters: unknown field '%s'", field.c_str());
                    return nullptr;
                }
                static const char *ops[] = {"eq", "ne", "lt", "le",
                                            "gt", "ge", "is_null",
                                            "is_not_null"};
                int opc = -1;
                for (int o = 0; o < 8; o++)
                    if (op == ops[o]) opc = o;
                if (opc < 0) {
                    set_error("filters: unknown op '%s'", op.c_str());
                    return nullptr;
                }
                FilterTerm ft{};
                ft.col = idx;
                ft.op = opc;
                int dt = plan->cols[idx].dtype;
                if (dt == PMH_DT_STRING && opc < 6 && opc > 1) {
                    set_error("filters: order comparisons on string "
                              "columns are not supported (dictionary ids "
                              "are unordered); eq/ne/is_null only");
                    return nullptr;
                }
                ft.is_fp = (dt == PMH_DT_FLOAT32 || dt == PMH_DT_FLOAT64);
                if (opc < 6) {
                    if (ft.is_fp)
                        ft.dlit = f["literal"].type == Json::NUM
                                      ? f["literal"].num
                                      : 0.0;
                    else ft.ilit = f["literal"].as_i64(0);
                }
                plan->filters.push_back(ft);
            }
            if (!plan->filters.empty()) {
                plan->filters_dev = (FilterTerm *)plan->bufs.alloc(
                    plan->filters.size() * sizeof(FilterTerm));
                if (!plan->filters_dev ||
                    hipMemcpy(plan->filters_dev, plan->filters.data(),
                              plan->filters.size() * sizeof(FilterTerm),
                              hipMemcpyHostToDevice) != hipSuccess) {
                    set_error("H2D filters failed");
                    return nullptr;
                }
            }
        }

        std::vector<FileDesc> files;
        int idx = 0;
        for (const auto &fj : j["files"].arr) {
            FileDesc fd;
            fd.path = fj["path"].as_str();
            fd.row_count = fj["rowCount"].as_i64();
            fd.min_key = fj["minKey"].as_i64();
            fd.max_key = fj["maxKey"].as_i64();
            fd.level = (int)fj["level"].as_i64();
            fd.input_index = idx++;
            const Json &dv = fj["deletionVector"];
            if (!dv.obj.empty()) {
                fd.dv_path = dv["file"].as_str();
                fd.dv_offset = dv["offset"].as_i64();
                fd.dv_length = dv["length"].as_i64();
            }
            if (fd.row_count >= ((int64_t)1 << PMH_ROW_BITS)) {
                set_error("file %s exceeds 2^%d rows-per-run limit",
                          fd.path.c_str(), PMH_ROW_BITS);
                return nullptr;
            }
            files.push_back(fd);
        }
        if (files.empty()) {
            set_error("no files in plan");
            return nullptr;
        }

        if (hipStreamCreate(&plan->stream) != hipSuccess) {
            set_error("hipStreamCreate failed");
            return nullptr;
        }
        if (plan->fsplit &&
            hipStreamCreate(&plan->stream_b) != hipSuccess) {
            set_error("hipStreamCreate (b) failed");
            return nullptr;
        }

        auto sections = interval_partition(files, plan->changelog);
        auto t0 = std::chrono::steady_clock::now();
        for (auto &sec_files : sections) {
            Section sec;
            if ((int)sec_files.size() > PMH_MAX_RUNS) {
                if (plan->pu || plan->agg || plan->ignore_delete ||
                    plan->changelog || plan->fused || plan->composite_key) {
                    set_error(
                        "section has %zu runs > %d: the hierarchical "
                        "winner merge serves deduplicate/first-row without "
                        "ignore-delete / changelog / sequence-fields / "
                        "composite keys in v1 (the reference spills to "
                        "disk here, MergeSorter.java:112-125)",
                        sec_files.size(), PMH_MAX_RUNS);
                    return nullptr;
                }
                sec.hier = true;
            }
            sec.runs.resize(sec_files.size());
            int64_t run_rows = 0;
            for (size_t r = 0; r < sec_files.size(); r++) {
                if (!stage_run(plan.get(), sec_files[r], sec.runs[r]))
                    return nullptr;
                run_rows += sec.runs[r].length;
            }
            if (!plan->filters.empty() && sec.runs.size() == 1 &&
                !sec.runs[0].tomb && sec.runs[0].length > 0) {
                // single-run section: value filters apply via the
                // tombstone mechanism (k_filter marks failing rows)
                sec.runs[0].tomb =
                    (uint8_t *)plan->bufs.alloc(sec.runs[0].length);
                if (!sec.runs[0].tomb ||
                    hipMemset(sec.runs[0].tomb, 0,
                              sec.runs[0].length) != hipSuccess) {
                    set_error("filter tombstone allocation failed");
                    return nullptr;
                }
            }
            if (!build_section_descriptors(plan.get(), sec)) {
                set_error("device allocation failed for section");
                return nullptr;
            }
            if (sec.hier && !build_hier_section(plan.get(), sec))
                return nullptr;
            plan->rows_in_total += run_rows;
            plan->sections.push_back(std::move(sec));
        }
        auto t1 = std::chrono::steady_clock::now();
        plan->h2d_ms =
            std::chrono::duration<double, std::milli>(t1 - t0).count();

        // output buffers: sized for the largest section
        int64_t max_rows = 0;
        for (auto &sec : plan->sections)
            max_rows = std::max(max_rows, sec.total_rows);
        int n_cols = (int)plan->cols.size();
        std::vector<void *> outs(n_cols);
        std::vector<uint8_t> dts(n_cols);
        for (int c = 0; c < n_cols; c++) {
            outs[c] = plan->bufs.alloc(max_rows * plan->cols[c].out_esize);
            if (!outs[c]) {
                set_error("output allocation failed");
                return nullptr;
            }
            dts[c] = (uint8_t)plan->cols[c].dtype;
        }
        plan->out_dev = outs;
        plan->out_ptrs_dev = (void **)plan->bufs.alloc(n_cols * sizeof(void *));
        plan->col_dtype_dev = (uint8_t *)plan->bufs.alloc(n_cols);
        if (hipMemcpy(plan->out_ptrs_dev, outs.data(),
                      n_cols * sizeof(void *),
                      hipMemcpyHostToDevice) != hipSuccess ||
            hipMemcpy(plan->col_dtype_dev, dts.data(), n_cols,
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_error("H2D of output descriptors failed");
            return nullptr;
        }
        // per-column nullability (any staged nulls in any section/run) +
        // output validity buffers
        plan->col_nullable.assign(n_cols, false);
        for (auto &sec : plan->sections)
            for (auto &run : sec.runs)
                for (int c = 0; c < n_cols; c++)
                    if (run.cols[c].has_nulls) plan->col_nullable[c] = true;
        plan->out_valid.assign(n_cols, nullptr);
        std::vector<uint8_t> nul(n_cols, 0);
        for (int c = 0; c < n_cols; c++) {
            nul[c] = plan->col_nullable[c] ? 1 : 0;
            if (plan->col_nullable[c]) {
                plan->out_valid[c] = (uint8_t *)plan->bufs.alloc(max_rows);
                if (!plan->out_valid[c]) {
                    set_error("validity allocation failed");
                    return nullptr;
                }
            }
        }
        plan->col_nullable_dev = (uint8_t *)plan->bufs.alloc(n_cols);
        plan->out_valid_dev =
            (uint8_t **)plan->bufs.alloc(n_cols * sizeof(void *));
        if (hipMemcpy(plan->col_nullable_dev, nul.data(), n_cols,
                      hipMemcpyHostToDevice) != hipSuccess ||
            hipMemcpy(plan->out_valid_dev, plan->out_valid.data(),
                      n_cols * sizeof(void *),
                      hipMemcpyHostToDevice) != hipSuccess) {
            set_error("H2D of validity descriptors failed");
            return nullptr;
        }
        if (plan->changelog) {
            // changelog outputs: up to 2 rows per key group <= 2 * inputs
            int64_t cap2 = 2 * max_rows;
            plan->out_dev2.resize(n_cols);
            plan->out_valid2.assign(n_cols, nullptr);
            for (int c = 0; c < n_cols; c++) {
                plan->out_dev2[c] =
                    plan->bufs.alloc(cap2 * plan->cols[c].out_esize);
                if (!plan->out_dev2[c]) {
                    set_error("changelog output allocation failed");
                    return nullptr;
                }
                if (plan->col_nullable[c]) {
                    plan->out_valid2[c] =
                        (uint8_t *)plan->bufs.alloc(cap2);
                    if (!plan->out_valid2[c]) {
                        set_error("changelog validity allocation failed");
                        return nullptr;
                    }
                }
            }
            plan->out_ptrs2_dev =
                (void **)plan->bufs.alloc(n_cols * sizeof(void *));
            plan->out_valid2_dev =
                (uint8_t **)plan->bufs.alloc(n_cols * sizeof(void *));
            if (!plan->out_ptrs2_dev || !plan->out_valid2_dev ||
                hipMemcpy(plan->out_ptrs2_dev, plan->out_dev2.data(),
                          n_cols * sizeof(void *),
                          hipMemcpyHostToDevice) != hipSuccess ||
                hipMemcpy(plan->out_valid2_dev, plan->out_valid2.data(),
                          n_cols * sizeof(void *),
                          hipMemcpyHostToDevice) != hipSuccess) {
                set_error("H2D of changelog descriptors failed");
                return nullptr;
            }
        }
        for (auto &cs : plan->cols) plan->col_names.push_back(cs.name);
        if (plan->agg) {
            // "aggregations": {"col": "sum", ...}; unnamed value columns get
            // last_non_null_value (AggregateMergeFunction.java:197-203);
            // "first_not_null_value" is the legacy alias
            // (FieldFirstNonNullValueAggLegacyFactory.java:29)
            std::vector<uint8_t> ca(n_cols, PMH_AGG_LAST_NON_NULL);
            auto code_of = [](const std::string &s) -> int {
                if (s == "last_non_null_value") return PMH_AGG_LAST_NON_NULL;
                if (s == "last_value") return PMH_AGG_LAST_VALUE;
                if (s == "first_value") return PMH_AGG_FIRST_VALUE;
                if (s == "first_non_null_value" ||
                    s == "first_not_null_value")
                    return PMH_AGG_FIRST_NON_NULL;
                if (s == "sum") return PMH_AGG_SUM;
                if (s == "max") return PMH_AGG_MAX;
                if (s == "min") return PMH_AGG_MIN;
                if (s == "primary_key") return PMH_AGG_PRIMARY_KEY;
                return -1;
            };
            const int first_val = plan->n_key_cols + 2;
            for (const auto &kv : j["aggregations"].obj) {
                int idx = -1;
                for (int c = first_val; c < n_cols; c++)
                    if (plan->cols[c].name == kv.first) idx = c;
                if (idx < 0) {
                    set_error("aggregations: '%s' is not a value column",
                              kv.first.c_str());
                    return nullptr;
                }
                int code = code_of(kv.second.as_str());
                if (code < 0) {
                    set_error("aggregate function '%s' not on the GPU path "
                              "(v1: sum, max, min, last_value, first_value, "
                              "last_non_null_value, first_non_null_value)",
                              kv.second.as_str().c_str());
                    return nullptr;
                }
                if (plan->cols[idx].dtype == PMH_DT_STRING &&
                    (code == PMH_AGG_SUM || code == PMH_AGG_MAX ||
                     code == PMH_AGG_MIN)) {
                    set_error("aggregate '%s' on a string column is not on "
                              "the GPU path (ids are not ordered by value)",
                              kv.second.as_str().c_str());
                    return nullptr;
                }
                if (plan->rrod && (code == PMH_AGG_FIRST_VALUE ||
                                   code == PMH_AGG_FIRST_NON_NULL)) {
                    set_error("aggregate function '%s' is not supported "
                              "with remove-record-on-delete in v1 (its "
                              "initialized-state does not reset on DELETE "
                              "in the reference; later round)",
                              kv.second.as_str().c_str());
                    return nullptr;
                }
                ca[idx] = (uint8_t)code;
            }
            // fields.<f>.ignore-retract = true (FieldIgnoreRetractAgg):
            // retract records leave that field's accumulator untouched
            for (const auto &irj : j["ignore_retract"].arr) {
                int idx = -1;
                for (int c = first_val; c < n_cols; c++)
                    if (plan->cols[c].name == irj.as_str()) idx = c;
                if (idx < 0) {
                    set_error("ignore_retract: '%s' is not a value column",
                              irj.as_str().c_str());
                    return nullptr;
                }
                ca[idx] |= PMH_AGG_IGNORE_RETRACT;
            }
            // retract records are accepted iff EVERY value column's
            // aggregator supports retraction (FieldSumAgg.retract,
            // FieldPrimaryKeyAgg) or ignores it (ignore-retract wrapper) —
            // otherwise the reference throws on the first retract
            // (FieldAggregator.retract :47-53) and so do we, at merge time
            {
                bool all_rt = true;
                for (int c = first_val; c < n_cols; c++) {
                    uint8_t a = ca[c];
                    if (a & PMH_AGG_IGNORE_RETRACT) continue;
                    if (a == PMH_AGG_SUM || a == PMH_AGG_PRIMARY_KEY)
                        continue;
                    all_rt = false;
                }
                plan->agg_retract = all_rt && !plan->rrod;
            }
            plan->col_agg_dev = (uint8_t *)plan->bufs.alloc(n_cols);
            if (!plan->col_agg_dev ||
                hipMemcpy(plan->col_agg_dev, ca.data(), n_cols,
                          hipMemcpyHostToDevice) != hipSuccess) {
                set_error("H2D of aggregator codes failed");
                return nullptr;
            }
        }
    } catch (const std::exception &e) {
        set_error("plan parse: %s", e.what());
        return nullptr;
    }
    plan->stats.h2d_ms = plan->h2d_ms;
    return plan.release();
}

int64_t pmh_read_next(pmh_plan_t *p, pmh_batch *out) {
    if (!p) {
        set_error("null plan");
        return -1;
    }
    if (p->cur_section >= p->sections.size()) return 0;
    (void)hipSetDevice(p->session->device);
    Section &sec = p->sections[p->cur_section++];
    hipStream_t st = p->stream;
    int k = (int)sec.runs.size();
    int n_cols = (int)p->cols.size();

    hipEvent_t ev[6];
    for (auto &e : ev) (void)hipEventCreate(&e);
    // ev: 0 start, 1 decode done, 2 partition done, 3 merge done,
    //     4 scan done, 5 emit done

    auto fail = [&](const char *what, hipError_t e) -> int64_t {
        set_error("%s: %s", what, hipGetErrorString(e));
        for (auto &evv : ev) (void)hipEventDestroy(evv);
        return -1;
    };

    (void)hipEventRecord(ev[0], st);
    // decode: batched RLEv2/byte-RLE + PRESENT/def-level scatter (one
    // launch each across every run-column — the per-run-column launches
    // serialized ~350 small kernels per step on C3), plus parquet
    // dictionary materialization per chunk
    if (sec.any_decode) {
        if (sec.n_rlev2) {
            hipError_t e = pmh_launch_rlev2(sec.rlev2_all, sec.n_rlev2, st);
            if (e != hipSuccess) return fail("rlev2", e);
        }
        if (sec.n_delta) {
            hipError_t e = pmh_launch_delta_sum(sec.delta_all, sec.n_delta,
                                                sec.delta_sums, st);
            if (e == hipSuccess)
                e = pmh_launch_delta_scan(sec.dstreams_all, sec.n_dstreams,
                                          sec.delta_sums, sec.delta_bases,
                                          st);
            if (e == hipSuccess)
                e = pmh_launch_delta_emit(sec.delta_all, sec.n_delta,
                                          sec.delta_bases, st);
            if (e != hipSuccess) return fail("delta", e);
        }
        for (auto &run : sec.runs) {
            for (size_t c = 0; c < run.cols.size(); c++) {
                RunCol &rc = run.cols[c];
                int es = p->cols[c].stored_esize;
                if (!rc.dict_encoded && rc.gathers.empty()) continue;
                if (rc.dict_encoded) {
                    hipError_t e = pmh_launch_rle_decode(
                        rc.rle_dev, (int64_t)rc.rle_host.size(), rc.ids_dev,
                        st);
                    if (e != hipSuccess) return fail("rle_decode", e);
                }
                for (const GatherTask &gt : rc.gathers) {
                    if (!gt.n) continue;
                    uint8_t *dst = gt.to_dense
                                       ? (uint8_t *)rc.dense_dev
                                       : (uint8_t *)rc.contig;
                    // in_place: ORC dictionary-string LOCAL ids were
                    // RLEv2-decoded at the target already; remap to global
                    const int32_t *ids =
                        gt.in_place
                            ? (const int32_t *)(dst + gt.start * es)
                            : rc.ids_dev + gt.start;
                    hipError_t e = pmh_launch_dict_gather(
                        ids, gt.dict_dev, gt.n, dst + gt.start * es, es,
                        st);
                    if (e != hipSuccess) return fail("dict_gather", e);
                }
            }
        }
        if (sec.n_def) {
            hipError_t e =
                pmh_launch_level_scatter(sec.def_all, sec.n_def, st);
            if (e != hipSuccess) return fail("level_scatter", e);
        }
    }
    if (p->composite_key) {
        // rebuilt per pass (decode-derived, like the validity masks)
        for (int r = 0; r < k; r++) {
            if (sec.runs[r].length <= 0) continue;
            hipError_t ce = pmh_launch_composite(
                sec.all_cols + (size_t)r * n_cols, p->n_key_cols,
                p->key_shifts, p->key_bits, sec.runs[r].length, sec.ckeys[r],
                st);
            if (ce != hipSuccess) return fail("composite", ce);
        }
    }
    if (sec.row_masks_dev) {
        // rebuilt every pass: masks derive from the per-step level decode,
        // so caching them across steps would skip timed work
        for (int r = 0; r < k; r++) {
            if (sec.runs[r].length <= 0) continue;
            hipError_t pe = pmh_launch_pack_valid(
                sec.all_cols + (size_t)r * n_cols, n_cols,
                sec.runs[r].length, sec.row_masks[r], st);
            if (pe != hipSuccess) return fail("pack_valid", pe);
        }
    }
    (void)hipEventRecord(ev[1], st);
    if (p->filters_dev && k == 1 && !sec.hier && sec.runs[0].tomb &&
        sec.runs[0].length > 0) {
        // value-filter pushdown: single-run sections only (overlapping
        // sections would lose newer records — MergeFileSplitRead.java:
        // 227-239); idempotent across repeats (OR into the tombstones)
        hipError_t fe = pmh_launch_filter(
            sec.all_cols, n_cols, p->filters_dev, (int)p->filters.size(),
            sec.runs[0].length, sec.runs[0].tomb, st);
        if (fe != hipSuccess) return fail("filter", fe);
    }
    int64_t n_tiles_eff = sec.n_tiles;
    int64_t rows_eff = sec.total_rows;
    if (sec.hier) {
        // batch pass: merge run batches into the virtual runs
        // (winner-of-winners — the dedup/first-row fold is associative;
        // deletes are KEPT here and drop only in the final chain)
        const int nb = (int)sec.blo.size();
        k = nb;
        const int bflags = p->first_row ? 8 : 0;  // no drop, no ignore
        for (int b = 0; b < nb; b++) {
            int lo = sec.blo[b];
            int kb = sec.bhi[b] - lo;
            int64_t btiles = sec.bntiles[b];
            hipError_t hb = pmh_launch_partition(
                sec.rkeys + lo, sec.rlens + lo, kb, PMH_TILE_ROWS,
                btiles + 1, sec.brows[b], sec.cuts, st);
            if (hb != hipSuccess) return fail("hier partition", hb);
            hb = pmh_launch_merge_tiles(
                sec.rkeys + lo, sec.rseqs + lo, sec.rkinds + lo,
                sec.rlens + lo, kb, sec.cuts, btiles, PMH_TILE_ROWS, bflags,
                sec.rtombs ? sec.rtombs + lo : nullptr, sec.winners,
                sec.tile_counts, sec.group_start, sec.err_dev, nullptr, 0,
                nullptr, nullptr, st);
            if (hb != hipSuccess) return fail("hier merge", hb);
            hb = pmh_launch_scan_tiles(sec.tile_counts, btiles,
                                       sec.tile_offsets, sec.lens_dev + b,
                                       st);
            if (hb != hipSuccess) return fail("hier scan", hb);
            hb = pmh_launch_emit(sec.rall + (size_t)lo * n_cols,
                                 p->hier_dtype_dev, p->col_nullable_dev,
                                 n_cols, kb, sec.winners, sec.tile_counts,
                                 sec.tile_offsets, btiles, PMH_TILE_ROWS,
                                 sec.lens_dev + b, sec.bout_ptrs[b],
                                 sec.bout_valid[b], st);
            if (hb != hipSuccess) return fail("hier emit", hb);
        }
        std::vector<int64_t> vl(nb);
        if (hipMemcpy(vl.data(), sec.lens_dev, nb * 8,
                      hipMemcpyDeviceToHost) != hipSuccess)
            return fail("hier lens D2H", hipErrorUnknown);
        rows_eff = 0;
        for (int b = 0; b < nb; b++) rows_eff += vl[b];
        n_tiles_eff = (rows_eff + PMH_TILE_ROWS - 1) / PMH_TILE_ROWS;
        if (n_tiles_eff == 0) n_tiles_eff = 1;
    }
    hipError_t e = pmh_launch_partition(sec.key_cols, sec.lens_dev, k,
                                        PMH_TILE_ROWS, n_tiles_eff + 1,
                                        rows_eff, sec.cuts, st);
    if (e != hipSuccess) return fail("partition", e);
    (void)hipEventRecord(ev[2], st);
    int flags = (p->drop_delete ? 1 : 0) | (p->ignore_delete ? 2 : 0) |
                (p->pu ? 4 : 0) | (p->first_row ? 8 : 0) |
                (p->rrod ? 16 : 0) | (p->seqg ? 32 : 0) |
                (p->agg_retract ? 64 : 0);
    if (const char *ab = getenv("PMH_ABLATE"))  // profiling-only phase knob
        flags |= (atoi(ab) & 0xf) << 8;
    if (const char *ab = getenv("PMH_FABL"))  // fused-kernel phase knob
        flags |= (atoi(ab) & 0x3) << 12;      // (profiling only)
    if (const char *fs = getenv("PMH_FSTAGE"))  // A/B: LDS-staged emission
        flags |= (atoi(fs) & 1) << 14;          // instead of direct gather
    if (p->fused) {
        // single-pass merge + emit: zero the lookback words + ticket, then
        // one kernel does merge, offsets and emission (scan/emit launches
        // and the winners round-trip disappear)
        e = hipMemsetAsync(sec.status, 0, n_tiles_eff * 8, st);
        if (e != hipSuccess) return fail("status memset", e);
        e = hipMemsetAsync(sec.ticket, 0, 8, st);
        if (e != hipSuccess) return fail("ticket memset", e);
        int key_col = p->composite_key ? -1 : 0;
        {
            // split mode: chunk the tile space and pipeline — kernel A
            // (merge + key/seq/kind + dense winners) for chunk i+1 runs on
            // the plan stream while kernel B (value gather) for chunk i
            // runs on stream_b, gated by a per-chunk event. Non-split: one
            // A launch does everything.
            // measured: A/B stream overlap is a wash — B's full-occupancy
            // gathers stretch A by about what the B tail saves (ch=1
            // 7.19 ms vs ch=8 7.57 on the same box) — so the default is
            // ONE chunk (A then B); PMH_FCHUNKS re-enables the pipeline
            int64_t want = 1;
            if (const char *fc = getenv("PMH_FCHUNKS"))
                want = sec.dense_winners ? atoll(fc) : 1;
            const int64_t n_chunks =
                want < 1 ? 1 : (want > 8 ? 8 : want);
            const int64_t per =
                (n_tiles_eff + n_chunks - 1) / n_chunks;
            hipEvent_t cev[8];
            for (int64_t c = 0; c < n_chunks; c++)
                (void)hipEventCreateWithFlags(&cev[c],
                                              hipEventDisableTiming);
            for (int64_t c = 0; c < n_chunks; c++) {
                int64_t t0 = c * per;
                int64_t t1 = t0 + per < n_tiles_eff ? t0 + per : n_tiles_eff;
                if (t0 >= t1) { (void)hipEventRecord(cev[c], st); continue; }
                if (c > 0) {  // ticket restarts per chunk
                    e = hipMemsetAsync(sec.ticket, 0, 8, st);
                    if (e != hipSuccess) return fail("ticket memset", e);
                }
                e = pmh_launch_merge_emit(
                    sec.key_cols, sec.seq_cols, sec.kind_cols, sec.lens_dev,
                    k, sec.cuts, t0, t1, n_tiles_eff, PMH_TILE_ROWS, flags,
                    sec.tombs_dev,
                    sec.all_cols, p->col_dtype_dev, p->col_nullable_dev,
                    n_cols, key_col, p->n_key_cols, p->n_key_cols + 1,
                    p->useq_dev, p->n_useq,
                    sec.status, sec.ticket, sec.total_dev, sec.dense_winners,
                    p->out_ptrs_dev, p->out_valid_dev, sec.err_dev, st);
                if (e != hipSuccess) return fail("merge_emit", e);
                (void)hipEventRecord(cev[c], st);
            }
            (void)hipEventRecord(ev[3], st);
            (void)hipEventRecord(ev[4], st);
            if (sec.dense_winners) {
                hipStream_t sb = p->stream_b;
                for (int64_t c = 0; c < n_chunks; c++) {
                    int64_t t0 = c * per;
                    int64_t t1 =
                        t0 + per < n_tiles_eff ? t0 + per : n_tiles_eff;
                    if (t0 >= t1) continue;
                    (void)hipStreamWaitEvent(sb, cev[c], 0);
                    e = pmh_launch_emit_dense(
                        sec.all_cols, p->col_dtype_dev, p->col_nullable_dev,
                        n_cols, key_col, p->n_key_cols, p->n_key_cols + 1,
                        sec.dense_winners, t0, t1, sec.status,
                        p->out_ptrs_dev, p->out_valid_dev, sb);
                    if (e != hipSuccess) return fail("emit_dense", e);
                }
                // rejoin: the plan stream waits for the last B chunk
                hipEvent_t done;
                (void)hipEventCreateWithFlags(&done, hipEventDisableTiming);
                (void)hipEventRecord(done, sb);
                (void)hipStreamWaitEvent(st, done, 0);
                (void)hipEventDestroy(done);
            }
            (void)hipEventRecord(ev[5], st);
            for (int64_t c = 0; c < n_chunks; c++)
                (void)hipEventDestroy(cev[c]);
        }
        goto collect;
    }
    e = pmh_launch_merge_tiles(sec.key_cols, sec.seq_cols, sec.kind_cols,
                               sec.lens_dev, k, sec.cuts, n_tiles_eff,
                               PMH_TILE_ROWS,
                               flags | (p->cl_row_dedup ? 128 : 0),
                               sec.tombs_dev, sec.winners,
                               sec.tile_counts, sec.group_start, sec.err_dev,
                               sec.run_levels, p->max_level, sec.cl_entries,
                               sec.cl_counts, st);
    if (e != hipSuccess) return fail("merge_tiles", e);
    (void)hipEventRecord(ev[3], st);
    e = pmh_launch_scan_tiles(sec.tile_counts, n_tiles_eff, sec.tile_offsets,
                              sec.total_dev, st);
    if (e != hipSuccess) return fail("scan_tiles", e);
    (void)hipEventRecord(ev[4], st);
    if (p->agg) {
        e = pmh_launch_emit_agg(
            sec.all_cols, p->col_dtype_dev, p->col_nullable_dev,
            p->col_agg_dev, n_cols, k, p->n_key_cols, p->n_key_cols + 1,
            flags, sec.winners, sec.group_start, sec.tile_offsets,
            n_tiles_eff, PMH_TILE_ROWS, sec.total_dev, sec.row_masks_dev,
            p->out_ptrs_dev, p->out_valid_dev, st);
    } else if (p->pu && p->seqg) {
        e = pmh_launch_emit_pu_sg(
            sec.all_cols, p->col_dtype_dev, p->col_nullable_dev, n_cols, k,
            p->n_key_cols, p->n_key_cols + 1, flags, p->col_group_dev,
            p->sg_fields_dev, p->sg_nseq_dev, p->n_seq_groups, sec.winners,
            sec.group_start, sec.tile_offsets, n_tiles_eff, PMH_TILE_ROWS,
            sec.total_dev, sec.row_masks_dev, p->out_ptrs_dev,
            p->out_valid_dev, st);
    } else if (p->pu) {
        e = pmh_launch_emit_pu(
            sec.all_cols, p->col_dtype_dev, p->col_nullable_dev, n_cols, k,
            p->n_key_cols, p->n_key_cols + 1, flags, sec.winners,
            sec.group_start, sec.tile_offsets, n_tiles_eff, PMH_TILE_ROWS,
            sec.total_dev, sec.row_masks_dev, p->out_ptrs_dev,
            p->out_valid_dev, st);
    } else {
        e = pmh_launch_emit(sec.all_cols, p->col_dtype_dev,
                            p->col_nullable_dev, n_cols, k, sec.winners,
                            sec.tile_counts, sec.tile_offsets, n_tiles_eff,
                            PMH_TILE_ROWS, sec.total_dev, p->out_ptrs_dev,
                            p->out_valid_dev, st);
    }
    if (e != hipSuccess) return fail("emit", e);
    if (p->changelog) {
        e = pmh_launch_cl_finalize(sec.all_cols, p->col_dtype_dev, n_cols,
                                   p->n_key_cols + 2, k, sec.cl_entries,
                                   sec.cl_rows, sec.cl_counts, n_tiles_eff,
                                   PMH_TILE_ROWS, st);
        if (e != hipSuccess) return fail("cl_finalize", e);
        e = pmh_launch_scan_tiles(sec.cl_counts, n_tiles_eff, sec.cl_offsets,
                                  sec.cl_total_dev, st);
        if (e != hipSuccess) return fail("cl_scan", e);
        e = pmh_launch_cl_emit(sec.all_cols, p->col_dtype_dev,
                               p->col_nullable_dev, n_cols,
                               p->n_key_cols + 1, sec.cl_rows, sec.cl_counts,
                               sec.cl_offsets, n_tiles_eff, PMH_TILE_ROWS,
                               p->out_ptrs2_dev, p->out_valid2_dev, st);
        if (e != hipSuccess) return fail("cl_emit", e);
    }
    (void)hipEventRecord(ev[5], st);

collect:
    int64_t total = 0;
    int64_t cl_total = 0;
    uint32_t err_word = 0;
    e = hipMemcpyAsync(&total, sec.total_dev, 8, hipMemcpyDeviceToHost, st);
    if (e != hipSuccess) return fail("total D2H", e);
    if (p->changelog && sec.cl_total_dev) {
        e = hipMemcpyAsync(&cl_total, sec.cl_total_dev, 8,
                           hipMemcpyDeviceToHost, st);
        if (e != hipSuccess) return fail("cl total D2H", e);
    }
    e = hipMemcpyAsync(&err_word, sec.err_dev, 4, hipMemcpyDeviceToHost, st);
    if (e != hipSuccess) return fail("err D2H", e);
    e = hipStreamSynchronize(st);
    if (e != hipSuccess) return fail("stream sync", e);
    if (err_word & 1) {
        if (p->rrod)
            set_error("remove-record-on-delete accepts INSERT/DELETE "
                      "streams in v1; UPDATE_BEFORE records are not "
                      "supported yet");
        else
            set_error("%s with retract records is not supported "
                      "(the reference default also rejects them, "
                      "PartialUpdateMergeFunction.java:170-186); configure "
                      "'partial-update.remove-record-on-delete', or wait "
                      "for the sequence-group/retract-aggregator rounds",
                      p->agg ? "aggregation" : "partial-update");
        return -1;
    }
    if (err_word & 2) {
        set_error("first-row merge engine cannot accept DELETE/UPDATE_BEFORE "
                  "records; configure 'ignore-delete' to skip them "
                  "(FirstRowMergeFunction.java:49-59)");
        return -1;
    }
    if (err_word & 4) {
        set_error("internal: fused merge lookback timed out waiting for a "
                  "predecessor tile");
        return -1;
    }
    if (err_word & 8) {
        set_error("Top level key-value already exists (two runs at "
                  "max_level hold the same key; "
                  "FullChangelogMergeFunctionWrapper.java:76-78 checkState)");
        return -1;
    }

    float ms;
    (void)hipEventElapsedTime(&ms, ev[0], ev[1]);
    p->stats.decode_ms += ms;
    (void)hipEventElapsedTime(&ms, ev[1], ev[2]);
    p->stats.partition_ms += ms;
    (void)hipEventElapsedTime(&ms, ev[2], ev[3]);
    p->stats.merge_ms += ms;
    (void)hipEventElapsedTime(&ms, ev[3], ev[4]);
    p->stats.scan_ms += ms;
    (void)hipEventElapsedTime(&ms, ev[4], ev[5]);
    p->stats.emit_ms += ms;
    (void)hipEventElapsedTime(&ms, ev[0], ev[5]);
    p->stats.total_device_ms += ms;
    for (auto &evv : ev) (void)hipEventDestroy(evv);

    p->stats.rows_in += sec.total_rows;
    p->stats.rows_out += total;

    // assemble batch
    p->batch_cols.resize(n_cols);
    if (p->host_output) {
        p->out_host.resize(n_cols);
        for (int c = 0; c < n_cols; c++) {
            p->out_host[c].resize(total * p->cols[c].out_esize);
            if (total > 0 &&
                hipMemcpy(p->out_host[c].data(), p->out_dev[c],
                          total * p->cols[c].out_esize,
                          hipMemcpyDeviceToHost) != hipSuccess) {
                set_error("D2H output failed");
                return -1;
            }
        }
    }
    if (p->host_output) {
        p->out_valid_host.resize(n_cols);
        for (int c = 0; c < n_cols; c++) {
            if (!p->out_valid[c]) continue;
            p->out_valid_host[c].resize(total);
            if (total > 0 &&
                hipMemcpy(p->out_valid_host[c].data(), p->out_valid[c], total,
                          hipMemcpyDeviceToHost) != hipSuccess) {
                set_error("D2H validity failed");
                return -1;
            }
        }
    }
    for (int c = 0; c < n_cols; c++) {
        pmh_col &pc = p->batch_cols[c];
        pc.name = p->col_names[c].c_str();
        pc.dtype = p->cols[c].dtype;
        pc.data = p->host_output ? (const void *)p->out_host[c].data()
                                 : (const void *)p->out_dev[c];
        pc.valid = nullptr;
        if (p->out_valid[c])
            pc.valid = p->host_output
                           ? (const uint8_t *)p->out_valid_host[c].data()
                           : (const uint8_t *)p->out_valid[c];
        pc.dict_data = nullptr;
        pc.dict_offsets = nullptr;
        pc.dict_len = 0;
        pc.precision = p->cols[c].precision;
        pc.scale = p->cols[c].scale;
        if (p->cols[c].dtype == PMH_DT_STRING && p->sdicts[c]) {
            StrDict *sd = p->sdicts[c].get();
            pc.dict_data = sd->bytes.data();
            pc.dict_offsets = sd->offsets.data();
            pc.dict_len = (int32_t)sd->offsets.size() - 1;
        }
    }
    if (p->changelog) {
        p->cl_rows_last = cl_total;
        p->batch_cols2.resize(n_cols);
        if (p->host_output) {
            p->out_host2.resize(n_cols);
            p->out_valid_host2.resize(n_cols);
            for (int c = 0; c < n_cols; c++) {
                p->out_host2[c].resize(cl_total * p->cols[c].out_esize);
                if (cl_total > 0 &&
                    hipMemcpy(p->out_host2[c].data(), p->out_dev2[c],
                              cl_total * p->cols[c].out_esize,
                              hipMemcpyDeviceToHost) != hipSuccess) {
                    set_error("D2H changelog failed");
                    return -1;
                }
                if (p->out_valid2[c]) {
                    p->out_valid_host2[c].resize(cl_total);
                    if (cl_total > 0 &&
                        hipMemcpy(p->out_valid_host2[c].data(),
                                  p->out_valid2[c], cl_total,
                                  hipMemcpyDeviceToHost) != hipSuccess) {
                        set_error("D2H changelog validity failed");
                        return -1;
                    }
                }
            }
        }
        for (int c = 0; c < n_cols; c++) {
            pmh_col pc = p->batch_cols[c];  // share names/dict/decimal info
            pc.data = p->host_output ? (const void *)p->out_host2[c].data()
                                     : (const void *)p->out_dev2[c];
            pc.valid = nullptr;
            if (p->out_valid2[c])
                pc.valid =
                    p->host_output
                        ? (const uint8_t *)p->out_valid_host2[c].data()
                        : (const uint8_t *)p->out_valid2[c];
            p->batch_cols2[c] = pc;
        }
    }
    if (out) {
        out->n_rows = total;
        out->n_cols = n_cols;
        out->device = p->host_output ? -1 : p->session->device;
        out->cols = p->batch_cols.data();
    }
    return total;
}

// Changelog batch of the LAST pmh_read_next call (changelog_producer =
// full-compaction). Valid until the next read_next, like the main batch.
int64_t pmh_changelog_next(pmh_plan_t *p, pmh_batch *out) {
    if (!p) return -1;
    if (!p->changelog) {
        set_error("plan has no changelog_producer configured");
        return -1;
    }
    if (out) {
        out->n_rows = p->cl_rows_last;
        out->n_cols = (int32_t)p->cols.size();
        out->device = p->host_output ? -1 : p->session->device;
        out->cols = p->batch_cols2.data();
    }
    return p->cl_rows_last;
}

int pmh_plan_reset(pmh_plan_t *p) {
    if (!p) return -1;
    p->cur_section = 0;
    return 0;
}

int pmh_plan_close(pmh_plan_t *p) {
    if (!p) return 0;
    if (p->stream) (void)hipStreamDestroy(p->stream);
    if (p->stream_b) (void)hipStreamDestroy(p->stream_b);
    delete p;
    return 0;
}

int pmh_stats_get(pmh_plan_t *p, pmh_stats *out) {
    if (p) p->stats.path_mode = !p->fused ? 0 : (p->fsplit ? 2 : 1);
    if (!p || !out) return -1;
    p->stats.hbm_bytes_algo = 0;  // filled by bench from encoded+output sizes
    *out = p->stats;
    out->hbm_bytes_algo = p->encoded_bytes_total;
    return 0;
}

void pmh_free_string(char *s) { free(s); }

int64_t pmh_debug_parse_dv(const char *path, int64_t offset, int64_t length,
                           int64_t *out, int64_t cap) {
    std::vector<uint32_t> pos;
    if (!load_deletion_vector(path, offset, length, pos)) return -1;
    int64_t n = (int64_t)pos.size() < cap ? (int64_t)pos.size() : cap;
    for (int64_t i = 0; i < n; i++) out[i] = pos[i];
    return (int64_t)pos.size();
}

// CPU-side entry to the scalar zstd restatement (zstd_core.h): decode one
// frame; the CPU tests fuzz this against libzstd before the GPU kernel
// (which shares the same core) ever runs.
int64_t pmh_debug_zstd_cpu(const void *src, int64_t n, void *dst,
                           int64_t cap) {
    std::vector<uint8_t> lit(PZ_BLOCK_MAX);
    std::vector<PzCtx> cx(1);
    int64_t r = pz_decode_frame((const uint8_t *)src, n, (uint8_t *)dst, cap,
                                lit.data(), cx.data());
    if (r < 0) set_error("pz_decode_frame: error %lld", (long long)r);
    return r;
}

// GPU round trip of the zstd page kernel: decode one frame whose
// decompressed size the caller knows (parity tests vs libzstd on the box).
int64_t pmh_debug_zstd_gpu(const void *src, int64_t n, void *dst,
                           int64_t expected) {
    std::vector<ZstdBatchPage> zp(1);
    zp[0] = {0, n, 0, expected};
    if (!gpu_zstd_batch((const uint8_t *)src, n, zp, (uint8_t *)dst,
                        expected)) {
        set_error("gpu zstd decode failed (batch of 1)");
        return -1;
    }
    return expected;
}

// CPU-side entry to the from-scratch zstd ENCODER (zstd_core.h): compress
// one frame; tests round-trip it through libzstd/pyarrow to prove the
// frames are spec-valid (the GPU kernel shares the same core).
int64_t pmh_debug_zstd_enc_cpu(const void *src, int64_t n, void *dst,
                               int64_t cap) {
    std::vector<PzEnc> e(1);
    std::vector<int32_t> ht(1 << PZ_ENC_HLOG);
    e[0].htab = ht.data();
    int64_t r = pz_encode_frame((const uint8_t *)src, n, (uint8_t *)dst,
                                cap, e.data());
    if (r < 0) set_error("pz_encode_frame: error %lld", (long long)r);
    return r;
}

// GPU round trip of the zstd COMPRESS kernel (batch of 1).
int64_t pmh_debug_zstd_enc_gpu(const void *src, int64_t n, void *dst,
                               int64_t cap) {
    std::vector<std::string> in(1);
    in[0].assign((const char *)src, (size_t)n);
    std::vector<std::vector<uint8_t>> outs;
    if (!pw_gpu_zstd_compress(in, outs)) {
        set_error("gpu zstd compress failed (batch of 1)");
        return -1;
    }
    if ((int64_t)outs[0].size() > cap) return PZ_ERR_DST_SMALL;
    memcpy(dst, outs[0].data(), outs[0].size());
    return (int64_t)outs[0].size();
}

int64_t pmh_debug_snappy(const void *src, int64_t n, void *dst, int64_t cap) {
    size_t got = 0;
    std::string err;
    if (!snappy_decompress((const uint8_t *)src, (size_t)n, (uint8_t *)dst,
                           (size_t)cap, got, err)) {
        set_error("%s", err.c_str());
        return -1;
    }
    return (int64_t)got;
}

int pmh_write_parquet(const pmh_col *cols, int32_t n_cols, int64_t n_rows,
                      const char *path, int64_t row_group_rows,
                      int64_t page_rows, const char *compression) {
    int codec = CODEC_UNCOMPRESSED;
    if (compression && *compression) {
        std::string cs(compression);
        if (cs == "zstd" || cs == "ZSTD") codec = CODEC_ZSTD;
        else if (cs != "NONE" && cs != "none" && cs != "UNCOMPRESSED") {
            set_error("write compression '%s' not supported (NONE | zstd)",
                      compression);
            return -1;
        }
    }
    if (!cols || n_cols <= 0 || !path) {
        set_error("pmh_write_parquet: bad arguments");
        return -1;
    }
    std::vector<PwCol> pc(n_cols);
    for (int32_t i = 0; i < n_cols; i++) {
        pc[i].name = cols[i].name ? cols[i].name : "";
        pc[i].dtype = cols[i].dtype;
        pc[i].data = cols[i].data;
        pc[i].valid = cols[i].valid;
        pc[i].dict_data = (const uint8_t *)cols[i].dict_data;
        pc[i].dict_offsets = cols[i].dict_offsets;
        pc[i].dict_len = cols[i].dict_len;
        pc[i].precision = cols[i].precision;
        pc[i].scale = cols[i].scale;
        if (pc[i].name.empty() || (!pc[i].data && n_rows > 0)) {
            set_error("pmh_write_parquet: column %d missing name/data", i);
            return -1;
        }
    }
    std::string err;
    if (!write_parquet(pc, n_rows, path, row_group_rows, page_rows, codec,
                       err)) {
        set_error("%s", err.c_str());
        return -1;
    }
    return 0;
}

char *pmh_debug_footer_json(const char *path) {
    StagedFile sf;
    if (!load_file(path, sf)) return nullptr;
    std::string out = "{\"num_rows\": " + std::to_string(sf.meta.num_rows) +
                      ", \"columns\": [";
    for (size_t i = 0; i < sf.meta.schema_names.size(); i++) {
        if (i) out += ",";
        out += "{\"name\": \"" + json_escape(sf.meta.schema_names[i]) +
               "\", \"max_def\": " + std::to_string(sf.meta.max_def_levels[i]) +
               ", \"phys\": " + std::to_string(sf.meta.phys_types[i]) + "}";
    }
    out += "], \"row_groups\": [";
    for (size_t g = 0; g < sf.meta.row_groups.size(); g++) {
        auto &rg = sf.meta.row_groups[g];
        if (g) out += ",";
        out += "{\"num_rows\": " + std::to_string(rg.num_rows) + ", \"chunks\": [";
        for (size_t c = 0; c < rg.columns.size(); c++) {
            auto &cc = rg.columns[c];
            if (c) out += ",";
            int64_t n_data_pages = 0;
            for (auto &pg : cc.pages)
                if (pg.page_type == 0) n_data_pages++;
            out += "{\"name\": \"" + json_escape(cc.name) +
                   "\", \"num_values\": " + std::to_string(cc.num_values) +
                   ", \"codec\": " + std::to_string(cc.codec) +
                   ", \"data_page_offset\": " +
                   std::to_string(cc.data_page_offset) +
                   ", \"dict_page_offset\": " +
                   std::to_string(cc.dictionary_page_offset) +
                   ", \"n_data_pages\": " + std::to_string(n_data_pages) + "}";
        }
        out += "]}";
    }
    out += "]}";
    return strdup(out.c_str());
}

int pmh_debug_interval_partition(int n, const int64_t *min_keys,
                                 const int64_t *max_keys, int32_t *out_section,
                                 int32_t *out_run) {
    std::vector<FileDesc> files(n);
    for (int i = 0; i < n; i++) {
        files[i].min_key = min_keys[i];
        files[i].max_key = max_keys[i];
        files[i].input_index = i;
    }
    auto sections = interval_partition(files);
    for (size_t s = 0; s < sections.size(); s++)
        for (size_t r = 0; r < sections[s].size(); r++)
            for (const auto &fd : sections[s][r]) {
                out_section[fd.input_index] = (int32_t)s;
                out_run[fd.input_index] = (int32_t)r;
            }
    return (int)sections.size();
}

}  // extern "C"
