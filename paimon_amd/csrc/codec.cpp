#include "codec.h"

#include <dlfcn.h>
#include <zlib.h>

#include <cstring>

namespace pmh {

namespace {
typedef size_t (*zstd_fn)(void *, size_t, const void *, size_t);
typedef unsigned (*zstd_iserror_fn)(size_t);

bool zstd_call(const uint8_t *src, size_t n, uint8_t *dst, size_t cap,
               size_t &got, std::string &err) {
    static zstd_fn fn = nullptr;
    static zstd_iserror_fn err_fn = nullptr;
    if (!fn) {
        void *h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
        if (!h) {
            err = "libzstd.so.1 not found for zstd-compressed input";
            return false;
        }
        fn = (zstd_fn)dlsym(h, "ZSTD_decompress");
        err_fn = (zstd_iserror_fn)dlsym(h, "ZSTD_isError");
        if (!fn || !err_fn) {
            err = "ZSTD_decompress symbol missing";
            return false;
        }
    }
    size_t r = fn(dst, cap, src, n);
    if (err_fn(r)) {
        err = "zstd decompress failed";
        return false;
    }
    got = r;
    return true;
}

bool inflate_raw(const uint8_t *src, size_t n, uint8_t *dst, size_t cap,
                 size_t &got, std::string &err) {
    z_stream zs;
    std::memset(&zs, 0, sizeof(zs));
    if (inflateInit2(&zs, -15) != Z_OK) {  // raw deflate (ORC ZLIB framing)
        err = "inflateInit2 failed";
        return false;
    }
    zs.next_in = const_cast<Bytef *>(src);
    zs.avail_in = (uInt)n;
    zs.next_out = dst;
    zs.avail_out = (uInt)cap;
    int rc = inflate(&zs, Z_FINISH);
    got = zs.total_out;
    inflateEnd(&zs);
    if (rc != Z_STREAM_END) {
        err = "raw deflate decode failed (rc " + std::to_string(rc) + ")";
        return false;
    }
    return true;
}
}  // namespace

bool zstd_compress_buf(const uint8_t *src, size_t n,
                       std::vector<uint8_t> &out, std::string &err) {
    typedef size_t (*compress_fn)(void *, size_t, const void *, size_t, int);
    typedef size_t (*bound_fn)(size_t);
    static compress_fn cfn = nullptr;
    static bound_fn bfn = nullptr;
    static zstd_iserror_fn efn = nullptr;
    if (!cfn) {
        void *h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
        if (!h) {
            err = "libzstd.so.1 not found for zstd-compressed output";
            return false;
        }
        cfn = (compress_fn)dlsym(h, "ZSTD_compress");
        bfn = (bound_fn)dlsym(h, "ZSTD_compressBound");
        efn = (zstd_iserror_fn)dlsym(h, "ZSTD_isError");
        if (!cfn || !bfn || !efn) {
            err = "ZSTD_compress symbols missing";
            return false;
        }
    }
    out.resize(bfn(n));
    size_t r = cfn(out.data(), out.size(), src, n, 3);  // default level
    if (efn(r)) {
        err = "zstd compress failed";
        return false;
    }
    out.resize(r);
    return true;
}

bool zstd_decompress_exact(const uint8_t *src, size_t n, uint8_t *dst,
                           size_t dst_n, std::string &err) {
    size_t got = 0;
    if (!zstd_call(src, n, dst, dst_n, got, err)) return false;
    if (got != dst_n) {
        err = "zstd decompressed size mismatch (" + std::to_string(got) +
              " != " + std::to_string(dst_n) + ")";
        return false;
    }
    return true;
}

// raw snappy block format (no framing): uvarint decompressed length, then
// literal/copy tags — implemented from scratch (no libsnappy in this image).
// Format per google/snappy format_description.txt.
bool snappy_decompress(const uint8_t *src, size_t n, uint8_t *dst,
                       size_t cap, size_t &got, std::string &err) {
    size_t p = 0;
    uint64_t ulen = 0;
    int shift = 0;
    for (;;) {
        if (p >= n) {
            err = "snappy: truncated length";
            return false;
        }
        uint8_t b = src[p++];
        ulen |= (uint64_t)(b & 0x7F) << shift;
        if (!(b & 0x80)) break;
        shift += 7;
    }
    if (ulen > cap) {
        err = "snappy: output exceeds capacity";
        return false;
    }
    size_t o = 0;
    while (p < n) {
        uint8_t tag = src[p++];
        size_t len;
        if ((tag & 3) == 0) {  // literal
            len = (tag >> 2) + 1;
            if (len > 60) {
                int extra = (int)len - 60;
                if (p + extra > n) {
                    err = "snappy: bad literal length";
                    return false;
                }
                len = 0;
                for (int i = 0; i < extra; i++)
                    len |= (size_t)src[p + i] << (8 * i);
                len += 1;
                p += extra;
            }
            if (p + len > n || o + len > ulen) {
                err = "snappy: literal overrun";
                return false;
            }
            std::memcpy(dst + o, src + p, len);
            p += len;
            o += len;
        } else {  // copy
            size_t off;
            if ((tag & 3) == 1) {
                if (p >= n) {
                    err = "snappy: bad copy1";
                    return false;
                }
                len = ((tag >> 2) & 7) + 4;
                off = ((size_t)(tag >> 5) << 8) | src[p++];
            } else if ((tag & 3) == 2) {
                if (p + 2 > n) {
                    err = "snappy: bad copy2";
                    return false;
                }
                len = (tag >> 2) + 1;
                off = (size_t)src[p] | ((size_t)src[p + 1] << 8);
                p += 2;
            } else {
                if (p + 4 > n) {
                    err = "snappy: bad copy4";
                    return false;
                }
                len = (tag >> 2) + 1;
                off = (size_t)src[p] | ((size_t)src[p + 1] << 8) |
                      ((size_t)src[p + 2] << 16) |
                      ((size_t)src[p + 3] << 24);
                p += 4;
            }
            if (off == 0 || off > o || o + len > ulen) {
                err = "snappy: bad copy offset/length";
                return false;
            }
            // copies may overlap (off < len): byte-by-byte semantics
            for (size_t i = 0; i < len; i++, o++) dst[o] = dst[o - off];
        }
    }
    if (o != ulen) {
        err = "snappy: decompressed size mismatch";
        return false;
    }
    got = o;
    return true;
}

bool gzip_decompress_exact(const uint8_t *src, size_t n, uint8_t *dst,
                           size_t dst_n, std::string &err) {
    z_stream zs;
    std::memset(&zs, 0, sizeof(zs));
    if (inflateInit2(&zs, 15 + 32) != Z_OK) {  // auto gzip/zlib wrapper
        err = "inflateInit2 failed";
        return false;
    }
    zs.next_in = const_cast<Bytef *>(src);
    zs.avail_in = (uInt)n;
    zs.next_out = dst;
    zs.avail_out = (uInt)dst_n;
    int rc = inflate(&zs, Z_FINISH);
    size_t got = zs.total_out;
    inflateEnd(&zs);
    if (rc != Z_STREAM_END || got != dst_n) {
        err = "gzip page decode failed (rc " + std::to_string(rc) + ", " +
              std::to_string(got) + " != " + std::to_string(dst_n) + ")";
        return false;
    }
    return true;
}

bool orc_decompress(const uint8_t *src, int64_t len, int kind,
                    int64_t block_size, std::vector<uint8_t> &out,
                    std::string &err) {
    if (kind != 1 && kind != 2 && kind != 5) {
        err = "ORC compression kind " + std::to_string(kind) +
              " not supported (v1: NONE, ZLIB, SNAPPY, ZSTD)";
        return false;
    }
    if (block_size <= 0) block_size = 256 * 1024;
    out.clear();
    int64_t p = 0;
    while (p < len) {
        if (p + 3 > len) {
            err = "truncated ORC compression chunk header";
            return false;
        }
        uint32_t h = (uint32_t)src[p] | ((uint32_t)src[p + 1] << 8) |
                     ((uint32_t)src[p + 2] << 16);
        bool original = h & 1;
        int64_t clen = h >> 1;
        p += 3;
        if (p + clen > len) {
            err = "ORC compression chunk overruns stream";
            return false;
        }
        if (original) {
            out.insert(out.end(), src + p, src + p + clen);
        } else {
            size_t old = out.size();
            out.resize(old + block_size);
            size_t got = 0;
            bool ok;
            if (kind == 1)
                ok = inflate_raw(src + p, clen, out.data() + old, block_size,
                                 got, err);
            else if (kind == 2)
                ok = snappy_decompress(src + p, clen, out.data() + old,
                                       block_size, got, err);
            else
                ok = zstd_call(src + p, clen, out.data() + old, block_size,
                               got, err);
            if (!ok) return false;
            out.resize(old + got);
        }
        p += clen;
    }
    return true;
}

}  // namespace pmh
