// Host-side decompression codecs for staging (the GPU consumes decoded or
// raw-encoded bytes; on-GPU decompression is roadmap §8f). zstd loads
// libzstd.so.1 via dlopen (exactly what the parquet zstd path always did);
// zlib links against the system libz.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace pmh {

// zstd-compress a buffer (parquet page write-back); level 3
bool zstd_compress_buf(const uint8_t *src, size_t n,
                       std::vector<uint8_t> &out, std::string &err);

// one-shot zstd frame with known decompressed size (parquet pages)
bool zstd_decompress_exact(const uint8_t *src, size_t n, uint8_t *dst,
                           size_t dst_n, std::string &err);

// one-shot gzip/zlib-wrapped member with known decompressed size
// (parquet GZIP pages)
bool gzip_decompress_exact(const uint8_t *src, size_t n, uint8_t *dst,
                           size_t dst_n, std::string &err);

// raw snappy block (parquet SNAPPY pages, ORC SNAPPY chunks); from-scratch
// decoder, no libsnappy dependency
bool snappy_decompress(const uint8_t *src, size_t n, uint8_t *dst,
                       size_t cap, size_t &got, std::string &err);

// ORC chunked stream framing: 3-byte LE header (len << 1 | isOriginal) per
// chunk, each decompressing to <= block_size bytes. kind: 1 = ZLIB (raw
// deflate), 5 = ZSTD (ORC proto CompressionKind).
bool orc_decompress(const uint8_t *src, int64_t len, int kind,
                    int64_t block_size, std::vector<uint8_t> &out,
                    std::string &err);

}  // namespace pmh
