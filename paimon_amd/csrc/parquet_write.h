// Host-side Parquet v1 PLAIN writer (compaction write-back). See
// parquet_write.cpp for the reference surfaces replaced.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace pmh {

struct PwCol {
    std::string name;
    int dtype;            // pmh_dtype
    const void *data;     // host pointer, output width (int8 = 1 B, ...)
    const uint8_t *valid; // byte validity or null (REQUIRED)
};

// codec: CODEC_UNCOMPRESSED or CODEC_ZSTD (parquet_meta.h values)
bool write_parquet(const std::vector<PwCol> &cols, int64_t n_rows,
                   const std::string &path, int64_t row_group_rows,
                   int64_t page_rows, int codec, std::string &err);

}  // namespace pmh
