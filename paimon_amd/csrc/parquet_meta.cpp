#include "parquet_meta.h"

#include <cstring>
#include <map>
#include <memory>
#include <stdexcept>

// Thrift compact protocol reader (the wire format of parquet footers and page
// headers; see parquet-format Thrift definitions consumed by the reference's
// vendored ParquetFileReader).
namespace pmh {
namespace {

enum TType {
    T_STOP = 0,
    T_BOOL_TRUE = 1,
    T_BOOL_FALSE = 2,
    T_BYTE = 3,
    T_I16 = 4,
    T_I32 = 5,
    T_I64 = 6,
    T_DOUBLE = 7,
    T_BINARY = 8,
    T_LIST = 9,
    T_SET = 10,
    T_MAP = 11,
    T_STRUCT = 12,
};

struct TValue;
using TStruct = std::map<int, TValue>;

struct TValue {
    int type = T_STOP;
    int64_t i = 0;
    double d = 0;
    std::string bin;
    std::vector<TValue> list;
    std::shared_ptr<TStruct> st;

    bool has(int fid) const { return st && st->count(fid); }
    const TValue &f(int fid) const {
        static TValue none;
        if (!st) return none;
        auto it = st->find(fid);
        return it == st->end() ? none : it->second;
    }
    int64_t fi(int fid, int64_t dflt = 0) const {
        return has(fid) ? f(fid).i : dflt;
    }
};

struct Reader {
    const uint8_t *p;
    const uint8_t *end;

    uint8_t u8() {
        if (p >= end) throw std::runtime_error("thrift: EOF");
        return *p++;
    }
    uint64_t uvarint() {
        uint64_t v = 0;
        int shift = 0;
        for (;;) {
            uint8_t b = u8();
            v |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) return v;
            shift += 7;
            if (shift > 63) throw std::runtime_error("thrift: varint overflow");
        }
    }
    int64_t zigzag() {
        uint64_t v = uvarint();
        return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
    }
    TValue value(int type) {
        TValue v;
        v.type = type;
        switch (type) {
        case T_BOOL_TRUE: v.i = 1; break;
        case T_BOOL_FALSE: v.i = 0; break;
        case T_BYTE: v.i = (int8_t)u8(); break;
        case T_I16:
        case T_I32:
        case T_I64: v.i = zigzag(); break;
        case T_DOUBLE: {
            uint64_t bits = 0;
            for (int k = 0; k < 8; k++) bits |= (uint64_t)u8() << (8 * k);
            memcpy(&v.d, &bits, 8);
            break;
        }
        case T_BINARY: {
            uint64_t n = uvarint();
            if (p + n > end) throw std::runtime_error("thrift: bad binary");
            v.bin.assign((const char *)p, n);
            p += n;
            break;
        }
        case T_LIST:
        case T_SET: {
            uint8_t hdr = u8();
            uint64_t size = hdr >> 4;
            int et = hdr & 0x0F;
            if (size == 15) size = uvarint();
            v.list.reserve(size);
            for (uint64_t k = 0; k < size; k++) {
                if (et == T_BOOL_TRUE || et == T_BOOL_FALSE) {
                    TValue e;
                    e.type = T_BOOL_TRUE;
                    e.i = u8() == 1;
                    v.list.push_back(e);
                } else {
                    v.list.push_back(value(et));
                }
            }
            break;
        }
        case T_STRUCT: v.st = std::make_shared<TStruct>(read_struct()); break;
        case T_MAP: {
            uint64_t size = uvarint();
            if (size > 0) {
                uint8_t kv = u8();
                int kt = kv >> 4, vt = kv & 0x0F;
                for (uint64_t k = 0; k < size; k++) {
                    value(kt);
                    value(vt);
                }
            }
            break;  // maps not needed; consumed and dropped
        }
        default: throw std::runtime_error("thrift: bad type");
        }
        return v;
    }
    TStruct read_struct() {
        TStruct s;
        int last_fid = 0;
        for (;;) {
            uint8_t b = u8();
            if (b == 0) return s;
            int delta = b >> 4;
            int type = b & 0x0F;
            int fid = delta ? last_fid + delta : (int)zigzag();
            last_fid = fid;
            s[fid] = value(type == T_BOOL_FALSE ? T_BOOL_FALSE : type);
        }
    }
};

}  // namespace

ParquetFileMeta parse_parquet_footer(const uint8_t *data, int64_t size) {
    ParquetFileMeta out;
    if (size < 12 || memcmp(data, "PAR1", 4) != 0 ||
        memcmp(data + size - 4, "PAR1", 4) != 0) {
        out.error = "not a parquet file";
        return out;
    }
    uint32_t meta_len;
    memcpy(&meta_len, data + size - 8, 4);
    if ((int64_t)meta_len + 12 > size) {
        out.error = "bad footer length";
        return out;
    }
    try {
        Reader r{data + size - 8 - meta_len, data + size - 8};
        TStruct fmd = r.read_struct();
        // FileMetaData: 2=schema, 3=num_rows, 4=row_groups
        out.num_rows = fmd.count(3) ? fmd[3].i : 0;
        const auto &schema = fmd[2].list;
        // flat schema: element 0 is root; leaves follow in order
        for (size_t i = 1; i < schema.size(); i++) {
            const auto &se = *schema[i].st;
            auto it = se.find(4);
            out.schema_names.push_back(it != se.end() ? it->second.bin : "");
            int rep = se.count(3) ? (int)se.at(3).i : 0;
            out.max_def_levels.push_back(rep == 1 ? 1 : 0);
            out.phys_types.push_back(se.count(1) ? (int)se.at(1).i : -1);
            if (se.count(5) && se.at(5).i > 0) {
                out.error = "nested parquet schemas unsupported (flat KeyValue rows only)";
                return out;
            }
        }
        for (const auto &rgv : fmd[4].list) {
            const TStruct &rg = *rgv.st;
            RowGroupMeta rgm;
            rgm.num_rows = rg.count(3) ? rg.at(3).i : 0;
            for (const auto &ccv : rg.at(1).list) {
                const TStruct &cc = *ccv.st;
                const TStruct &md = *cc.at(3).st;  // ColumnMetaData
                ColumnChunkMeta c;
                std::string path;
                for (const auto &part : md.at(3).list) {
                    if (!path.empty()) path += ".";
                    path += part.bin;
                }
                c.name = path;
                c.phys_type = (int)md.at(1).i;
                c.codec = (int)md.at(4).i;
                c.num_values = md.at(5).i;
                c.total_compressed_size = md.at(7).i;
                c.data_page_offset = md.at(9).i;
                c.dictionary_page_offset =
                    md.count(11) ? md.at(11).i : 0;
                for (const auto &e : md.at(2).list)
                    c.encodings.push_back((int)e.i);
                rgm.columns.push_back(std::move(c));
            }
            out.row_groups.push_back(std::move(rgm));
        }
    } catch (const std::exception &e) {
        out.error = std::string("footer parse: ") + e.what();
    }
    return out;
}

bool scan_chunk_pages(const uint8_t *data, int64_t size, ColumnChunkMeta &cc,
                      std::string &err) {
    int64_t start = cc.dictionary_page_offset ? cc.dictionary_page_offset
                                              : cc.data_page_offset;
    int64_t end = start + cc.total_compressed_size;
    if (start < 0 || end > size) {
        err = "column chunk out of bounds";
        return false;
    }
    int64_t p = start;
    int64_t row = 0;
    try {
        while (p < end) {
            Reader r{data + p, data + end};
            TStruct hdr = r.read_struct();
            int64_t body = r.p - data;  // absolute offset of payload
            PageMeta pg{};
            pg.page_type = (int)hdr.at(1).i;
            pg.header_off = p;
            pg.data_off = body;
            pg.uncompressed_size = (int32_t)hdr.at(2).i;
            pg.compressed_size = (int32_t)hdr.at(3).i;
            if (pg.page_type == 0) {  // DATA_PAGE v1
                const TStruct &d = *hdr.at(5).st;
                pg.num_values = (int32_t)d.at(1).i;
                pg.encoding = (int)d.at(2).i;
                pg.def_level_encoding = d.count(3) ? (int)d.at(3).i : ENC_RLE;
                pg.first_row = row;
                row += pg.num_values;
            } else if (pg.page_type == 2) {  // DICTIONARY_PAGE
                const TStruct &d = *hdr.at(7).st;
                pg.num_values = (int32_t)d.at(1).i;
                pg.encoding = (int)d.at(2).i;
                pg.first_row = -1;
            } else if (pg.page_type == 3) {  // DATA_PAGE_V2
                err = "parquet data page v2 not supported yet";
                return false;
            }
            cc.pages.push_back(pg);
            p = body + pg.compressed_size;
        }
    } catch (const std::exception &e) {
        err = std::string("page scan: ") + e.what();
        return false;
    }
    return true;
}

}  // namespace pmh
