// Host-side ORC file-structure reader (protobuf wire format): postscript,
// footer, stripe footers, stream locations. Replaces, for this path, the
// stripe planning the reference delegates to orc-core 1.9.8
// (format/orc/OrcReaderFactory.java:108-171 + vendored
// org.apache.orc.impl.RecordReaderImpl overrides). Decode arithmetic
// (RLEv2 / byte-RLE / boolean-RLE) runs on the GPU (kernels.hip); this
// parser only locates streams. Uncompressed ORC only (v1).
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace pmh {

enum OrcTypeKind {
    ORC_BOOLEAN = 0,
    ORC_BYTE = 1,
    ORC_SHORT = 2,
    ORC_INT = 3,
    ORC_LONG = 4,
    ORC_FLOAT = 5,
    ORC_DOUBLE = 6,
    ORC_STRING = 7,
    ORC_STRUCT = 12,
};

enum OrcStreamKind {
    ORC_STREAM_PRESENT = 0,
    ORC_STREAM_DATA = 1,
    ORC_STREAM_LENGTH = 2,
    ORC_STREAM_DICTIONARY = 3,
};

struct OrcStream {
    int kind = 0;
    int column = 0;
    int64_t length = 0;
    int64_t offset = 0;  // absolute file offset
};

struct OrcStripe {
    int64_t offset = 0;
    int64_t index_length = 0;
    int64_t data_length = 0;
    int64_t footer_length = 0;
    int64_t num_rows = 0;
    std::vector<OrcStream> streams;
    std::vector<int> encodings;  // per column id
};

struct OrcFileMeta {
    int64_t num_rows = 0;
    int compression = 0;  // CompressionKind: 0 NONE, 1 ZLIB, 5 ZSTD (v1 set)
    int64_t compression_block_size = 0;
    std::vector<std::string> column_names;  // flat struct; col id = idx + 1
    std::vector<int> column_kinds;
    std::vector<OrcStripe> stripes;
    std::string error;
    bool ok() const { return error.empty(); }
};

OrcFileMeta parse_orc_meta(const uint8_t *data, int64_t size);

bool is_orc_file(const uint8_t *data, int64_t size);

const OrcStream *orc_find_stream(const OrcStripe &st, int column, int kind);

}  // namespace pmh
