// Host-side Parquet v1 writer: PLAIN encoding, uncompressed, thrift-compact
// footer/page headers — the write-back half of the compaction surface.
//
// Replaces, for this path, the reference's format writer stack
// (paimon-format/src/main/java/org/apache/paimon/format/parquet/
// ParquetWriterFactory.java + the vendored parquet-mr writer) as used by
// KeyValueDataFileWriter (io/KeyValueDataFileWriter.java:121-170) under
// CompactRewriter. v1 matrix: INT8/16 (stored INT32), INT32, INT64, FLOAT,
// DOUBLE; REQUIRED or OPTIONAL (byte validity -> RLE/bit-packed def levels);
// one PLAIN data page per `page_rows`; UNCOMPRESSED or ZSTD pages (host
// compress; no dictionary — GPU-side encode is roadmap §8f.1). Readable by
// parquet-mr/pyarrow and by this library's own reader (tests pin both).

#include "parquet_write.h"

#include <cstdio>
#include <cstring>

#include "codec.h"
#include "parquet_meta.h"

namespace pmh {

namespace {

// ---- thrift compact protocol writer (parquet.thrift field ids) ----
struct TC {
    std::string out;
    void byte(uint8_t b) { out.push_back((char)b); }
    void uvarint(uint64_t v) {
        while (v >= 0x80) {
            byte((uint8_t)(v | 0x80));
            v >>= 7;
        }
        byte((uint8_t)v);
    }
    static uint64_t zz(int64_t v) {
        return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63);
    }
    // field header; caller tracks the previous field id per struct level
    void field(int &last, int id, int type) {
        int delta = id - last;
        if (delta >= 1 && delta <= 15) {
            byte((uint8_t)((delta << 4) | type));
        } else {
            byte((uint8_t)type);
            uvarint(zz(id));
        }
        last = id;
    }
    void stop() { byte(0); }
    void i32(int &last, int id, int32_t v) {
        field(last, id, 5);
        uvarint(zz(v));
    }
    void i64(int &last, int id, int64_t v) {
        field(last, id, 6);
        uvarint(zz(v));
    }
    void str(int &last, int id, const std::string &s) {
        field(last, id, 8);
        uvarint(s.size());
        out.append(s);
    }
    void list_begin(int &last, int id, int elem_type, size_t n) {
        field(last, id, 9);
        if (n < 15) byte((uint8_t)((n << 4) | elem_type));
        else {
            byte((uint8_t)(0xF0 | elem_type));
            uvarint(n);
        }
    }
};

int phys_of(int dtype) {
    switch (dtype) {
    case 1:
    case 2:
    case 3: return PHYS_INT32;
    case 4: return PHYS_INT64;
    case 5: return PHYS_FLOAT;
    case 6: return PHYS_DOUBLE;
    case 7: return PHYS_BYTE_ARRAY;  // dictionary string
    }
    return -1;
}

int converted_of(const PwCol &c) {  // ConvertedType; -1 = none
    if (c.precision > 0) return 5;  // DECIMAL (INT32/INT64 physical, as
                                    // ParquetSchemaConverter.java:153-171)
    switch (c.dtype) {
    case 1: return 15;  // INT_8
    case 2: return 16;  // INT_16
    case 7: return 0;   // UTF8
    }
    return -1;
}

// RLE/bit-packed hybrid id stream for RLE_DICTIONARY pages: a leading
// bit-width byte, then bit-packed groups of 8 (the format
// VectorizedRleValuesReader.java:977-1018 consumes; nulls skipped)
void encode_dict_ids(const PwCol &c, int64_t s, int64_t e,
                     std::string &out) {
    int bw = 1;
    while ((1 << bw) < c.dict_len) bw++;
    std::vector<int32_t> ids;
    ids.reserve(e - s);
    const int32_t *d = (const int32_t *)c.data;
    for (int64_t i = s; i < e; i++) {
        if (c.valid && !c.valid[i]) continue;
        ids.push_back(d[i]);
    }
    out.push_back((char)bw);
    const int64_t ngroups = ((int64_t)ids.size() + 7) / 8;
    TC t;
    t.uvarint(((uint64_t)ngroups << 1) | 1);
    out.append(t.out);
    // streaming bit accumulator: a group is 8 values x bw bits = bw bytes,
    // LSB-first (Packer.LITTLE_ENDIAN); bw can exceed 8 (large dicts)
    uint64_t acc = 0;
    int nbits = 0;
    for (int64_t i = 0; i < ngroups * 8; i++) {
        uint64_t v = i < (int64_t)ids.size() ? (uint64_t)ids[i] : 0;
        acc |= v << nbits;
        nbits += bw;
        while (nbits >= 8) {
            out.push_back((char)(acc & 0xFF));
            acc >>= 8;
            nbits -= 8;
        }
    }
}

void dict_page_header(int64_t n_vals, int32_t unc, int32_t comp,
                      std::string &out) {
    TC t;
    int l0 = 0;
    t.i32(l0, 1, 2);     // type = DICTIONARY_PAGE
    t.i32(l0, 2, unc);
    t.i32(l0, 3, comp);
    t.field(l0, 7, 12);  // dictionary_page_header
    {
        int l1 = 0;
        t.i32(l1, 1, (int32_t)n_vals);
        t.i32(l1, 2, ENC_PLAIN);
        t.stop();
    }
    t.stop();
    out.append(t.out);
}

// RLE/bit-packed hybrid def levels (bit width 1), with the 4-byte LE length
// prefix of v1 data pages. All-valid pages use one RLE run; mixed pages use
// one bit-packed literal (LSB-first, zero-padded).
void encode_def_levels(const uint8_t *valid, int64_t n, std::string &out) {
    bool all = true;
    for (int64_t i = 0; i < n && all; i++) all = valid[i] != 0;
    std::string body;
    if (all) {
        TC t;
        t.uvarint((uint64_t)n << 1);
        body = t.out;
        body.push_back((char)1);
    } else {
        int64_t groups = (n + 7) / 8;
        TC t;
        t.uvarint(((uint64_t)groups << 1) | 1);
        body = t.out;
        for (int64_t g = 0; g < groups; g++) {
            uint8_t b = 0;
            for (int j = 0; j < 8; j++) {
                int64_t i = g * 8 + j;
                if (i < n && valid[i]) b |= (uint8_t)(1u << j);
            }
            body.push_back((char)b);
        }
    }
    uint32_t len = (uint32_t)body.size();
    out.push_back((char)(len & 0xFF));
    out.push_back((char)((len >> 8) & 0xFF));
    out.push_back((char)((len >> 16) & 0xFF));
    out.push_back((char)((len >> 24) & 0xFF));
    out.append(body);
}

// PLAIN payload for rows [s, e): non-null values only, int8/16 widened to
// the INT32 physical type.
void encode_values(const PwCol &c, int64_t s, int64_t e, std::string &out) {
    for (int64_t i = s; i < e; i++) {
        if (c.valid && !c.valid[i]) continue;
        int32_t v32;
        switch (c.dtype) {
        case 1:
            v32 = (int32_t)((const int8_t *)c.data)[i];
            out.append((const char *)&v32, 4);
            break;
        case 2:
            v32 = (int32_t)((const int16_t *)c.data)[i];
            out.append((const char *)&v32, 4);
            break;
        case 3:
        case 5:
            out.append((const char *)c.data + i * 4, 4);
            break;
        case 4:
        case 6:
            out.append((const char *)c.data + i * 8, 8);
            break;
        }
    }
}

void page_header(int64_t n_vals, int32_t unc, int32_t comp, int encoding,
                 std::string &out) {
    TC t;
    int l0 = 0;
    t.i32(l0, 1, 0);     // type = DATA_PAGE
    t.i32(l0, 2, unc);   // uncompressed_page_size
    t.i32(l0, 3, comp);  // compressed_page_size
    t.field(l0, 5, 12);     // data_page_header: struct
    {
        int l1 = 0;
        t.i32(l1, 1, (int32_t)n_vals);
        t.i32(l1, 2, encoding);
        t.i32(l1, 3, ENC_RLE);    // definition_level_encoding
        t.i32(l1, 4, ENC_RLE);    // repetition_level_encoding
        t.stop();
    }
    t.stop();
    out.append(t.out);
}

}  // namespace

bool write_parquet(const std::vector<PwCol> &cols, int64_t n_rows,
                   const std::string &path, int64_t row_group_rows,
                   int64_t page_rows, int codec, std::string &err) {
    if (codec != CODEC_UNCOMPRESSED && codec != CODEC_ZSTD) {
        err = "write codec " + std::to_string(codec) +
              " not supported (v1: UNCOMPRESSED, ZSTD)";
        return false;
    }
    for (const auto &c : cols) {
        if (phys_of(c.dtype) < 0) {
            err = "unsupported dtype " + std::to_string(c.dtype) +
                  " for parquet write (v1 matrix: int8..int64/float/double/"
                  "decimal<=18/dictionary string)";
            return false;
        }
        if (c.dtype == 7 &&
            (!c.dict_data || !c.dict_offsets || c.dict_len <= 0)) {
            err = "string column '" + c.name + "' needs a dictionary";
            return false;
        }
    }
    if (row_group_rows <= 0) row_group_rows = 1 << 20;
    if (page_rows <= 0) page_rows = 1 << 16;
    FILE *f = fopen(path.c_str(), "wb");
    if (!f) {
        err = "cannot open " + path + " for write";
        return false;
    }
    std::string buf = "PAR1";

    struct CcInfo {
        int64_t data_page_offset;
        int64_t dict_page_offset;  // 0 = no dictionary page
        int64_t total_size;       // compressed (on-file) bytes
        int64_t total_unc_size;   // uncompressed payload + headers
        int64_t num_values;
    };
    struct RgInfo {
        std::vector<CcInfo> ccs;
        int64_t rows;
        int64_t bytes;
    };
    std::vector<RgInfo> rgs;

    for (int64_t rg0 = 0; rg0 < n_rows || (n_rows == 0 && rg0 == 0);
         rg0 += row_group_rows) {
        int64_t rg1 = rg0 + row_group_rows < n_rows ? rg0 + row_group_rows
                                                    : n_rows;
        RgInfo rg{};
        rg.rows = rg1 - rg0;
        for (const auto &c : cols) {
            CcInfo cc{};
            cc.num_values = rg1 - rg0;
            cc.data_page_offset = -1;  // set at the first data page
            int64_t unc_total = 0;
            const bool dict = c.dtype == 7;
            const int enc = dict ? ENC_RLE_DICTIONARY : ENC_PLAIN;
            // pass 1: build every page payload of this chunk (dictionary
            // page first), so ZSTD chunks can compress as ONE GPU batch
            // (k_zstd_compress — §8f.1; host libzstd is the fallback)
            std::vector<std::string> payloads;
            std::vector<int64_t> prows;  // -1 marks the dictionary page
            if (dict) {
                std::string dp;
                for (int32_t i = 0; i < c.dict_len; i++) {
                    uint32_t len =
                        (uint32_t)(c.dict_offsets[i + 1] - c.dict_offsets[i]);
                    dp.append((const char *)&len, 4);
                    dp.append(
                        (const char *)c.dict_data + c.dict_offsets[i], len);
                }
                payloads.push_back(std::move(dp));
                prows.push_back(-1);
            }
            for (int64_t p0 = rg0; p0 < rg1 || (rg1 == rg0 && p0 == rg0);
                 p0 += page_rows) {
                int64_t p1 = p0 + page_rows < rg1 ? p0 + page_rows : rg1;
                std::string payload;
                if (c.valid) encode_def_levels(c.valid + p0, p1 - p0, payload);
                if (dict) encode_dict_ids(c, p0, p1, payload);
                else encode_values(c, p0, p1, payload);
                payloads.push_back(std::move(payload));
                prows.push_back(p1 - p0);
                if (rg1 == rg0) break;  // single empty page for 0 rows
            }
            // pass 2: compress (ZSTD) — GPU batch, host fallback
            std::vector<std::vector<uint8_t>> comp;
            if (codec == CODEC_ZSTD) {
                if (!pw_gpu_zstd_enc_enabled() ||
                    !pw_gpu_zstd_compress(payloads, comp)) {
                    comp.resize(payloads.size());
                    for (size_t i = 0; i < payloads.size(); i++) {
                        if (!zstd_compress_buf(
                                (const uint8_t *)payloads[i].data(),
                                payloads[i].size(), comp[i], err)) {
                            fclose(f);
                            return false;
                        }
                    }
                }
            }
            // pass 3: assemble headers + page bytes
            for (size_t i = 0; i < payloads.size(); i++) {
                const std::string &payload = payloads[i];
                const bool is_dict = prows[i] < 0;
                int32_t on_file = codec == CODEC_ZSTD
                                      ? (int32_t)comp[i].size()
                                      : (int32_t)payload.size();
                if (is_dict)
                    cc.dict_page_offset = (int64_t)buf.size();
                else if (cc.data_page_offset < 0)
                    cc.data_page_offset = (int64_t)buf.size();
                size_t h0 = buf.size();
                if (is_dict)
                    dict_page_header(c.dict_len, (int32_t)payload.size(),
                                     on_file, buf);
                else
                    page_header(prows[i], (int32_t)payload.size(), on_file,
                                enc, buf);
                unc_total +=
                    (int64_t)(buf.size() - h0) + (int64_t)payload.size();
                if (codec == CODEC_ZSTD)
                    buf.append((const char *)comp[i].data(),
                               comp[i].size());
                else
                    buf.append(payload);
            }
            cc.total_size = (int64_t)buf.size() -
                            (dict ? cc.dict_page_offset
                                  : cc.data_page_offset);
            cc.total_unc_size = unc_total;
            rg.bytes += cc.total_size;
            rg.ccs.push_back(cc);
        }
        rgs.push_back(rg);
        if (n_rows == 0) break;
    }

    // footer: FileMetaData
    TC t;
    int l0 = 0;
    t.i32(l0, 1, 1);  // version
    t.list_begin(l0, 2, 12, cols.size() + 1);  // schema
    {
        int ls = 0;  // root group
        TC &g = t;
        g.i32(ls, 3, 0);  // repetition REQUIRED (ignored on root)
        g.str(ls, 4, "schema");
        g.i32(ls, 5, (int32_t)cols.size());  // num_children
        g.stop();
        for (const auto &c : cols) {
            int lf = 0;
            g.i32(lf, 1, phys_of(c.dtype));
            g.i32(lf, 3, c.valid ? 1 : 0);  // OPTIONAL : REQUIRED
            g.str(lf, 4, c.name);
            int ct = converted_of(c);
            if (ct >= 0) g.i32(lf, 6, ct);
            if (c.precision > 0) {
                g.i32(lf, 7, c.scale);
                g.i32(lf, 8, c.precision);
            }
            g.stop();
        }
    }
    t.i64(l0, 3, n_rows);
    t.list_begin(l0, 4, 12, rgs.size());
    for (const auto &rg : rgs) {
        int lr = 0;
        t.list_begin(lr, 1, 12, rg.ccs.size());
        for (size_t ci = 0; ci < rg.ccs.size(); ci++) {
            const CcInfo &cc = rg.ccs[ci];
            const PwCol &c = cols[ci];
            int lc = 0;
            t.i64(lc, 2, cc.dict_page_offset ? cc.dict_page_offset
                                              : cc.data_page_offset);
            t.field(lc, 3, 12);                 // meta_data: struct
            {
                int lm = 0;
                t.i32(lm, 1, phys_of(c.dtype));
                if (c.dtype == 7) {
                    t.list_begin(lm, 2, 5, 3);  // encodings
                    t.uvarint(TC::zz(ENC_PLAIN));
                    t.uvarint(TC::zz(ENC_RLE));
                    t.uvarint(TC::zz(ENC_RLE_DICTIONARY));
                } else {
                    t.list_begin(lm, 2, 5, 2);  // encodings
                    t.uvarint(TC::zz(ENC_PLAIN));
                    t.uvarint(TC::zz(ENC_RLE));
                }
                t.list_begin(lm, 3, 8, 1);  // path_in_schema
                t.uvarint(c.name.size());
                t.out.append(c.name);
                t.i32(lm, 4, codec);
                t.i64(lm, 5, cc.num_values);
                t.i64(lm, 6, cc.total_unc_size);  // total_uncompressed_size
                t.i64(lm, 7, cc.total_size);      // total_compressed_size
                t.i64(lm, 9, cc.data_page_offset);
                if (cc.dict_page_offset)
                    t.i64(lm, 11, cc.dict_page_offset);
                t.stop();
            }
            t.stop();
        }
        t.i64(lr, 2, rg.bytes);
        t.i64(lr, 3, rg.rows);
        t.stop();
    }
    t.str(l0, 6, "paimon-hip v1");
    t.stop();

    buf.append(t.out);
    uint32_t flen = (uint32_t)t.out.size();
    buf.append((const char *)&flen, 4);
    buf.append("PAR1");

    bool ok = fwrite(buf.data(), 1, buf.size(), f) == buf.size();
    fclose(f);
    if (!ok) err = "short write to " + path;
    return ok;
}

}  // namespace pmh
