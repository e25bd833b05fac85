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
    const void *data;     // host pointer, output width (int8 = 1 B, ...);
                          // dtype 7 (string): int32 dictionary ids
    const uint8_t *valid; // byte validity or null (REQUIRED)
    // dtype 7: the dictionary (entry i = dict_data[off[i], off[i+1]))
    const uint8_t *dict_data = nullptr;
    const int32_t *dict_offsets = nullptr;
    int32_t dict_len = 0;
    // DECIMAL(p,s) annotation on INT32/INT64 (0 = plain integer)
    int32_t precision = 0;
    int32_t scale = 0;
};

// codec: CODEC_UNCOMPRESSED or CODEC_ZSTD (parquet_meta.h values)
bool write_parquet(const std::vector<PwCol> &cols, int64_t n_rows,
                   const std::string &path, int64_t row_group_rows,
                   int64_t page_rows, int codec, std::string &err);

}  // namespace pmh

namespace pmh {
// GPU batch page compression (implemented in plan.cpp over
// k_zstd_compress); false = use the host codec.
bool pw_gpu_zstd_compress(const std::vector<std::string> &payloads,
                          std::vector<std::vector<uint8_t>> &outs);
bool pw_gpu_zstd_enc_enabled();  // PMH_GPU_ZSTD_ENC=1 opt-in
}  // namespace pmh
