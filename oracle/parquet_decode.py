"""Oracle-side full column-chunk decode: footer/page walk (parquet_light) +
the C RLE/bit-packed restatement + numpy PLAIN/dictionary materialization.

Restates the per-column orchestration of VectorizedColumnReader.readBatch
(paimon-format/.../reader/VectorizedColumnReader.java:143-241):
  data page v1 = [def levels: 4-byte LE length + RLE stream]  (OPTIONAL cols;
                  VectorizedRleValuesReader.java:102-109)
                 [values: PLAIN | RLE_DICTIONARY (1-byte bit width prefix,
                  VectorizedRleValuesReader.java:110-114)]
  dictionary page = PLAIN values; ids gathered through ParquetDictionary
                  (paimon-format/.../reader/ParquetDictionary.java).
TEST INFRASTRUCTURE ONLY (oracle package rules apply).
"""

import struct

import numpy as np

from . import parquet_light as pl
from .oracle import rle_bp_decode

_PHYS_NP = {
    pl.PHYS_INT32: np.dtype("<i4"),
    pl.PHYS_INT64: np.dtype("<i8"),
    pl.PHYS_FLOAT: np.dtype("<f4"),
    pl.PHYS_DOUBLE: np.dtype("<f8"),
}


def _decompress(payload: bytes, codec: int, uncompressed_size: int) -> bytes:
    if codec == pl.CODEC_UNCOMPRESSED:
        return payload
    if codec == pl.CODEC_ZSTD:
        import ctypes
        z = ctypes.CDLL("libzstd.so.1")
        z.ZSTD_decompress.restype = ctypes.c_size_t
        out = ctypes.create_string_buffer(uncompressed_size)
        n = z.ZSTD_decompress(out, uncompressed_size, payload, len(payload))
        if n != uncompressed_size:
            raise ValueError("zstd decompress size mismatch")
        return out.raw
    raise NotImplementedError(f"codec {codec}")


def decode_chunk(path, chunk: "pl.ColumnChunkInfo", max_def_level: int):
    """Decode one column chunk to (values: np.ndarray, valid: bool array)."""
    pages = pl.scan_pages(path, chunk)
    dtype = _PHYS_NP[chunk.phys_type]
    with open(path, "rb") as f:
        raw = f.read()
    dictionary = None
    vals_out, valid_out = [], []
    for pg in pages:
        payload = _decompress(raw[pg.data_offset:pg.data_offset + pg.compressed_size],
                              chunk.codec, pg.uncompressed_size)
        if pg.page_type == pl.PAGE_DICTIONARY:
            dictionary = np.frombuffer(payload, dtype=dtype)
            continue
        if pg.page_type != pl.PAGE_DATA:
            raise NotImplementedError("data page v2")
        pos = 0
        n = pg.num_values
        if max_def_level > 0:
            dl_len = struct.unpack_from("<I", payload, 0)[0]
            pos = 4 + dl_len
            def_levels = rle_bp_decode(payload[4:pos], 1, n)
            valid = def_levels.astype(bool)
        else:
            valid = np.ones(n, dtype=bool)
        n_non_null = int(valid.sum())
        if pg.encoding == pl.PLAIN:
            dense = np.frombuffer(payload, dtype=dtype, count=n_non_null,
                                  offset=pos)
        elif pg.encoding in (pl.RLE_DICTIONARY, pl.PLAIN_DICTIONARY):
            bw = payload[pos]
            ids = rle_bp_decode(payload[pos + 1:], bw, n_non_null)
            dense = dictionary[ids]
        else:
            raise NotImplementedError(f"encoding {pg.encoding}")
        vals = np.zeros(n, dtype=dtype)
        vals[valid] = dense
        vals_out.append(vals)
        valid_out.append(valid)
    return np.concatenate(vals_out), np.concatenate(valid_out)


def read_file(path):
    """Decode all columns of a (flat-schema) parquet file via the oracle path.
    Returns dict name -> (values, valid)."""
    fi = pl.parse_footer(path)
    out = {}
    for ci, name in enumerate(fi.schema_names):
        cols, valids = [], []
        for rg in fi.row_groups:
            chunk = rg.columns[ci]
            v, m = decode_chunk(path, chunk, fi.max_def_levels[ci])
            cols.append(v)
            valids.append(m)
        out[name] = (np.concatenate(cols), np.concatenate(valids))
    return out
