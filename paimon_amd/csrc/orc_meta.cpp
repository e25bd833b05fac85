#include "orc_meta.h"

#include "codec.h"

#include <cstring>
#include <map>
#include <stdexcept>

namespace pmh {
namespace {

// minimal protobuf wire reader: {field -> values}; ints as uint64, bytes
// for length-delimited. Repeated packed uint32 handled by the caller.
struct PbValue {
    uint64_t i = 0;
    std::string bin;
    bool is_bin = false;
};

using PbMsg = std::map<int, std::vector<PbValue>>;

struct PbReader {
    const uint8_t *p;
    const uint8_t *end;

    uint64_t uvarint() {
        uint64_t v = 0;
        int shift = 0;
        for (;;) {
            if (p >= end) throw std::runtime_error("pb: EOF");
            uint8_t b = *p++;
            v |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) return v;
            shift += 7;
        }
    }
};

PbMsg pb_parse(const uint8_t *buf, int64_t len) {
    PbMsg out;
    PbReader r{buf, buf + len};
    while (r.p < r.end) {
        uint64_t hdr = r.uvarint();
        int fnum = (int)(hdr >> 3), wt = (int)(hdr & 7);
        PbValue v;
        if (wt == 0) {
            v.i = r.uvarint();
        } else if (wt == 2) {
            uint64_t n = r.uvarint();
            if (r.p + n > r.end) throw std::runtime_error("pb: bad len");
            v.bin.assign((const char *)r.p, n);
            v.is_bin = true;
            r.p += n;
        } else if (wt == 1) {
            if (r.p + 8 > r.end) throw std::runtime_error("pb: bad i64");
            memcpy(&v.i, r.p, 8);
            r.p += 8;
        } else if (wt == 5) {
            uint32_t x;
            if (r.p + 4 > r.end) throw std::runtime_error("pb: bad i32");
            memcpy(&x, r.p, 4);
            v.i = x;
            r.p += 4;
        } else {
            throw std::runtime_error("pb: bad wire type");
        }
        out[fnum].push_back(std::move(v));
    }
    return out;
}

uint64_t pb_int(const PbMsg &m, int f, uint64_t dflt = 0) {
    auto it = m.find(f);
    return it == m.end() || it->second.empty() ? dflt : it->second[0].i;
}

// repeated uint32: either individual varints or packed blobs
std::vector<uint64_t> pb_ints(const PbMsg &m, int f) {
    std::vector<uint64_t> out;
    auto it = m.find(f);
    if (it == m.end()) return out;
    for (const auto &v : it->second) {
        if (!v.is_bin) {
            out.push_back(v.i);
        } else {
            PbReader r{(const uint8_t *)v.bin.data(),
                       (const uint8_t *)v.bin.data() + v.bin.size()};
            while (r.p < r.end) out.push_back(r.uvarint());
        }
    }
    return out;
}

}  // namespace

bool is_orc_file(const uint8_t *data, int64_t size) {
    return size > 16 && memcmp(data, "ORC", 3) == 0;
}

const OrcStream *orc_find_stream(const OrcStripe &st, int column, int kind) {
    for (const auto &s : st.streams)
        if (s.column == column && s.kind == kind) return &s;
    return nullptr;
}

OrcFileMeta parse_orc_meta(const uint8_t *data, int64_t size) {
    OrcFileMeta out;
    try {
        if (!is_orc_file(data, size)) {
            out.error = "not an ORC file";
            return out;
        }
        int ps_len = data[size - 1];
        PbMsg ps = pb_parse(data + size - 1 - ps_len, ps_len);
        int64_t footer_len = (int64_t)pb_int(ps, 1);
        out.compression = (int)pb_int(ps, 2);
        out.compression_block_size = (int64_t)pb_int(ps, 3);
        // file footer and stripe footers share the stream compression
        // framing (ORC spec "Compression"); streams themselves decompress
        // at staging (plan.cpp)
        std::string cerr;
        std::vector<uint8_t> fdec;
        int64_t fstart = size - 1 - ps_len - footer_len;
        const uint8_t *fptr = data + fstart;
        int64_t flen = footer_len;
        if (out.compression != 0) {
            if (!orc_decompress(fptr, flen, out.compression,
                                out.compression_block_size, fdec, cerr)) {
                out.error = cerr;
                return out;
            }
            fptr = fdec.data();
            flen = (int64_t)fdec.size();
        }
        PbMsg footer = pb_parse(fptr, flen);
        out.num_rows = (int64_t)pb_int(footer, 6);
        // types (field 4, repeated Type)
        std::vector<PbMsg> types;
        auto tit = footer.find(4);
        if (tit != footer.end())
            for (const auto &t : tit->second)
                types.push_back(
                    pb_parse((const uint8_t *)t.bin.data(), t.bin.size()));
        if (types.empty() || pb_int(types[0], 1) != ORC_STRUCT) {
            out.error = "ORC schema root is not a flat struct";
            return out;
        }
        auto fit = types[0].find(3);
        if (fit != types[0].end())
            for (const auto &n : fit->second) out.column_names.push_back(n.bin);
        for (uint64_t sub : pb_ints(types[0], 2)) {
            if (sub >= types.size()) {
                out.error = "bad ORC subtype id";
                return out;
            }
            out.column_kinds.push_back((int)pb_int(types[sub], 1));
        }
        // stripes (field 3)
        auto sit = footer.find(3);
        if (sit != footer.end()) {
            for (const auto &sv : sit->second) {
                PbMsg sm = pb_parse((const uint8_t *)sv.bin.data(),
                                    sv.bin.size());
                OrcStripe st;
                st.offset = (int64_t)pb_int(sm, 1);
                st.index_length = (int64_t)pb_int(sm, 2);
                st.data_length = (int64_t)pb_int(sm, 3);
                st.footer_length = (int64_t)pb_int(sm, 4);
                st.num_rows = (int64_t)pb_int(sm, 5);
                int64_t sf_off = st.offset + st.index_length + st.data_length;
                std::vector<uint8_t> sdec;
                const uint8_t *sptr = data + sf_off;
                int64_t slen = st.footer_length;
                if (out.compression != 0) {
                    if (!orc_decompress(sptr, slen, out.compression,
                                        out.compression_block_size, sdec,
                                        cerr)) {
                        out.error = cerr;
                        return out;
                    }
                    sptr = sdec.data();
                    slen = (int64_t)sdec.size();
                }
                PbMsg spf = pb_parse(sptr, slen);
                int64_t pos = st.offset;
                auto stit = spf.find(1);
                if (stit != spf.end()) {
                    for (const auto &sb : stit->second) {
                        PbMsg strm = pb_parse(
                            (const uint8_t *)sb.bin.data(), sb.bin.size());
                        OrcStream s;
                        s.kind = (int)pb_int(strm, 1);
                        s.column = (int)pb_int(strm, 2);
                        s.length = (int64_t)pb_int(strm, 3);
                        s.offset = pos;
                        pos += s.length;
                        st.streams.push_back(s);
                    }
                }
                auto eit = spf.find(2);
                if (eit != spf.end())
                    for (const auto &eb : eit->second)
                        st.encodings.push_back((int)pb_int(
                            pb_parse((const uint8_t *)eb.bin.data(),
                                     eb.bin.size()),
                            1));
                out.stripes.push_back(std::move(st));
            }
        }
    } catch (const std::exception &e) {
        out.error = std::string("ORC parse: ") + e.what();
    }
    return out;
}

}  // namespace pmh
