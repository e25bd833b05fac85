// Host-side Parquet metadata reader: thrift-compact footer + page headers.
//
// Replaces, for this path, the reference's vendored ParquetFileReader
// (paimon-format/src/main/java/org/apache/parquet/hadoop/ParquetFileReader.java)
// and the page bookkeeping of VectorizedParquetRecordReader
// (paimon-format/.../parquet/reader/VectorizedParquetRecordReader.java:92-323).
// Footer / thrift parsing stays on the CPU in this build.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace pmh {

enum ParquetEncoding {
    ENC_PLAIN = 0,
    ENC_PLAIN_DICTIONARY = 2,
    ENC_RLE = 3,
    ENC_BIT_PACKED = 4,
    ENC_DELTA_BINARY_PACKED = 5,
    ENC_DELTA_LENGTH_BYTE_ARRAY = 6,
    ENC_DELTA_BYTE_ARRAY = 7,
    ENC_RLE_DICTIONARY = 8,
};

enum ParquetCodec {
    CODEC_UNCOMPRESSED = 0,
    CODEC_SNAPPY = 1,
    CODEC_GZIP = 2,
    CODEC_ZSTD = 6,
};

enum ParquetPhysType {
    PHYS_BOOLEAN = 0,
    PHYS_INT32 = 1,
    PHYS_INT64 = 2,
    PHYS_INT96 = 3,
    PHYS_FLOAT = 4,
    PHYS_DOUBLE = 5,
    PHYS_BYTE_ARRAY = 6,
    PHYS_FIXED_LEN_BYTE_ARRAY = 7,
};

struct PageMeta {
    int page_type;        // 0 data v1, 2 dictionary, 3 data v2
    int64_t header_off;   // absolute file offset of page header
    int64_t data_off;     // absolute offset of (compressed) payload
    int32_t compressed_size;
    int32_t uncompressed_size;
    int32_t num_values;
    int encoding;
    int def_level_encoding;
    int64_t first_row;    // first row index within the column chunk
};

struct ColumnChunkMeta {
    std::string name;
    int phys_type;
    int codec;
    int64_t num_values;
    int64_t data_page_offset;
    int64_t dictionary_page_offset;  // 0 if none
    int64_t total_compressed_size;
    std::vector<int> encodings;
    std::vector<PageMeta> pages;     // filled by scan_pages
};

struct RowGroupMeta {
    int64_t num_rows;
    std::vector<ColumnChunkMeta> columns;
};

struct ParquetFileMeta {
    int64_t num_rows;
    std::vector<std::string> schema_names;  // leaf columns, flat schema
    std::vector<int> max_def_levels;        // 1 for OPTIONAL, 0 for REQUIRED
    std::vector<int> phys_types;
    std::vector<RowGroupMeta> row_groups;
    std::string error;  // non-empty on failure

    bool ok() const { return error.empty(); }
};

// Parse footer from an in-memory file image.
ParquetFileMeta parse_parquet_footer(const uint8_t *data, int64_t size);

// Sequentially parse all page headers of one column chunk (sets first_row).
bool scan_chunk_pages(const uint8_t *data, int64_t size, ColumnChunkMeta &cc,
                      std::string &err);

}  // namespace pmh
