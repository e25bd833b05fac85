"""Minimal Parquet footer/page-header parser (thrift compact protocol) —
TEST INFRASTRUCTURE for locating pages inside files written by pyarrow, so
the oracle decoders (and the GPU decode kernels) can be pinned page-by-page
against pyarrow's independent reads.

Restates the on-wire structures consumed by the reference's vendored
ParquetFileReader (paimon-format/src/main/java/org/apache/parquet/hadoop/
ParquetFileReader.java) and read by VectorizedParquetRecordReader
(paimon-format/.../parquet/reader/VectorizedParquetRecordReader.java:92-323).
"""

import struct
from dataclasses import dataclass, field
from typing import List

# thrift compact type codes
_T_BOOL_TRUE, _T_BOOL_FALSE, _T_BYTE, _T_I16, _T_I32, _T_I64 = 1, 2, 3, 4, 5, 6
_T_DOUBLE, _T_BINARY, _T_LIST, _T_SET, _T_MAP, _T_STRUCT = 7, 8, 9, 10, 11, 12

PLAIN, PLAIN_DICTIONARY, RLE, BIT_PACKED = 0, 2, 3, 4
DELTA_BINARY_PACKED, DELTA_LENGTH_BYTE_ARRAY, DELTA_BYTE_ARRAY = 5, 6, 7
RLE_DICTIONARY = 8

PAGE_DATA, PAGE_INDEX, PAGE_DICTIONARY, PAGE_DATA_V2 = 0, 1, 2, 3

PHYS_BOOLEAN, PHYS_INT32, PHYS_INT64, PHYS_INT96 = 0, 1, 2, 3
PHYS_FLOAT, PHYS_DOUBLE, PHYS_BYTE_ARRAY, PHYS_FIXED = 4, 5, 6, 7

CODEC_UNCOMPRESSED, CODEC_SNAPPY, CODEC_GZIP = 0, 1, 2
CODEC_LZO, CODEC_BROTLI, CODEC_LZ4, CODEC_ZSTD = 3, 4, 5, 6


def _uvarint(buf, p):
    v, shift = 0, 0
    while True:
        b = buf[p]
        p += 1
        v |= (b & 0x7F) << shift
        if not (b & 0x80):
            return v, p
        shift += 7


def _zigzag(v):
    return (v >> 1) ^ -(v & 1)


def _read_value(buf, p, t):
    if t == _T_BOOL_TRUE:
        return True, p
    if t == _T_BOOL_FALSE:
        return False, p
    if t == _T_BYTE:
        return struct.unpack_from("b", buf, p)[0], p + 1
    if t in (_T_I16, _T_I32, _T_I64):
        v, p = _uvarint(buf, p)
        return _zigzag(v), p
    if t == _T_DOUBLE:
        return struct.unpack_from("<d", buf, p)[0], p + 8
    if t == _T_BINARY:
        n, p = _uvarint(buf, p)
        return bytes(buf[p:p + n]), p + n
    if t in (_T_LIST, _T_SET):
        hdr = buf[p]
        p += 1
        size = hdr >> 4
        et = hdr & 0x0F
        if size == 15:
            size, p = _uvarint(buf, p)
        out = []
        for _ in range(size):
            if et == _T_BOOL_TRUE:  # list<bool> elements are full bytes
                out.append(buf[p] == 1)
                p += 1
            else:
                v, p = _read_value(buf, p, et)
                out.append(v)
        return out, p
    if t == _T_STRUCT:
        return _read_struct(buf, p)
    if t == _T_MAP:
        size, p = _uvarint(buf, p)
        if size == 0:
            return {}, p
        kt_vt = buf[p]
        p += 1
        kt, vt = kt_vt >> 4, kt_vt & 0x0F
        out = {}
        for _ in range(size):
            k, p = _read_value(buf, p, kt)
            v, p = _read_value(buf, p, vt)
            out[k] = v
        return out, p
    raise ValueError(f"unknown thrift compact type {t}")


def _read_struct(buf, p):
    fields = {}
    last_fid = 0
    while True:
        b = buf[p]
        p += 1
        if b == 0:
            return fields, p
        delta = b >> 4
        t = b & 0x0F
        if delta == 0:
            v, p2 = _uvarint(buf, p)
            fid = _zigzag(v)
            p = p2
        else:
            fid = last_fid + delta
        last_fid = fid
        v, p = _read_value(buf, p, t)
        fields[fid] = v


@dataclass
class ColumnChunkInfo:
    name: str
    phys_type: int
    codec: int
    num_values: int
    data_page_offset: int
    dictionary_page_offset: int
    total_compressed_size: int
    encodings: List[int]


@dataclass
class RowGroupInfo:
    num_rows: int
    columns: List[ColumnChunkInfo] = field(default_factory=list)


@dataclass
class PageInfo:
    page_type: int
    offset: int          # absolute file offset of the page header
    data_offset: int     # absolute offset of the (compressed) page payload
    compressed_size: int
    uncompressed_size: int
    num_values: int
    encoding: int
    def_level_encoding: int = RLE


@dataclass
class FileInfo:
    num_rows: int
    schema_names: List[str]
    max_def_levels: List[int]
    row_groups: List[RowGroupInfo] = field(default_factory=list)


def parse_footer(path) -> FileInfo:
    with open(path, "rb") as f:
        data = f.read()
    assert data[:4] == b"PAR1" and data[-4:] == b"PAR1", "not a parquet file"
    meta_len = struct.unpack_from("<I", data, len(data) - 8)[0]
    meta = memoryview(data)[len(data) - 8 - meta_len:len(data) - 8]
    fmd, _ = _read_struct(meta, 0)
    # FileMetaData: 2=schema list<SchemaElement>, 3=num_rows, 4=row_groups
    schema = fmd[2]
    names, max_defs, phys = [], [], []
    # flat schemas only (root + leaf children) — paimon KeyValue rows are flat
    for se in schema[1:]:
        names.append(se[4].decode())
        # repetition_type: 0=REQUIRED,1=OPTIONAL,2=REPEATED
        max_defs.append(1 if se.get(3, 0) == 1 else 0)
        phys.append(se.get(1, -1))
    fi = FileInfo(num_rows=fmd[3], schema_names=names, max_def_levels=max_defs)
    for rg in fmd[4]:
        rgi = RowGroupInfo(num_rows=rg[3])
        for i, cc in enumerate(rg[1]):
            md = cc[3]  # ColumnMetaData
            rgi.columns.append(ColumnChunkInfo(
                name=b".".join(md[3]).decode(),
                phys_type=md[1],
                codec=md[4],
                num_values=md[5],
                data_page_offset=md[9],
                dictionary_page_offset=md.get(11, 0),
                total_compressed_size=md[7],
                encodings=md[2],
            ))
        fi.row_groups.append(rgi)
    return fi


def scan_pages(path, chunk: ColumnChunkInfo) -> List[PageInfo]:
    """Sequentially parse page headers of one column chunk."""
    start = chunk.dictionary_page_offset or chunk.data_page_offset
    end = start + chunk.total_compressed_size
    with open(path, "rb") as f:
        f.seek(start)
        buf = memoryview(f.read(end - start))
    pages = []
    p = 0
    while p < len(buf):
        hdr, p2 = _read_struct(buf, p)
        ptype = hdr[1]
        unc, comp = hdr[2], hdr[3]
        if ptype == PAGE_DATA:
            dph = hdr[5]
            pages.append(PageInfo(ptype, start + p, start + p2, comp, unc,
                                  dph[1], dph[2], dph.get(3, RLE)))
        elif ptype == PAGE_DICTIONARY:
            dph = hdr[7]
            pages.append(PageInfo(ptype, start + p, start + p2, comp, unc,
                                  dph[1], dph[2]))
        elif ptype == PAGE_DATA_V2:
            dph = hdr[8]
            pages.append(PageInfo(ptype, start + p, start + p2, comp, unc,
                                  dph[1], dph[4]))
        p = p2 + comp
    return pages
