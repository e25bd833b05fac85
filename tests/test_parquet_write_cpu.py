"""Native Parquet writer (parquet_write.cpp — the CompactRewriter write-back
half) pinned against TWO independent readers on CPU:
  1. pyarrow (an independent Parquet implementation);
  2. this repo's own reader stack — the C++ footer parser via
     pmh_debug_footer_json, and the oracle's pure-Python page decoder
     (oracle/parquet_decode.py, itself pinned to pyarrow) — which also proves
     the GPU read path can consume compaction output, since both share the
     same wire format.
"""

import os

import numpy as np
import pytest

import pyarrow.parquet as pq

from paimon_amd import debug_footer, write_parquet
from oracle.parquet_decode import read_file


def _rand_cols(rng, n):
    msk = rng.random(n) > 0.3
    return [
        ("k", np.sort(rng.choice(10 ** 12, n, replace=False)).astype(np.int64)),
        ("v32", rng.integers(-2 ** 31, 2 ** 31, n).astype(np.int32), msk),
        ("v8", rng.integers(-128, 128, n).astype(np.int8)),
        ("v16", rng.integers(-2 ** 15, 2 ** 15, n).astype(np.int16), msk),
        ("f32", rng.standard_normal(n).astype(np.float32)),
        ("f64", rng.standard_normal(n), msk),
    ]


def _check_pyarrow(path, cols, n):
    t = pq.read_table(path)
    assert t.num_rows == n
    for col in cols:
        name, vals = col[0], col[1]
        valid = col[2] if len(col) > 2 else None
        got_valid = t[name].is_valid().to_numpy(zero_copy_only=False)
        if valid is None:
            assert got_valid.all(), name
            got = t[name].to_numpy(zero_copy_only=False)
            assert got.dtype == vals.dtype, name
            assert (got == vals).all(), name
        else:
            assert (got_valid == valid).all(), name
            got = t[name].to_numpy(zero_copy_only=False)
            assert (got[valid] == vals[valid]).all(), name


class TestNativeParquetWriter:
    def test_pyarrow_roundtrip(self, tmp_path):
        rng = np.random.default_rng(7)
        n = 120_000
        cols = _rand_cols(rng, n)
        p = str(tmp_path / "t.parquet")
        write_parquet(p, cols, row_group_rows=50_000, page_rows=9_000)
        assert pq.ParquetFile(p).num_row_groups == 3
        _check_pyarrow(p, cols, n)

    def test_own_reader_roundtrip(self, tmp_path):
        rng = np.random.default_rng(8)
        n = 40_000
        cols = _rand_cols(rng, n)
        p = str(tmp_path / "t.parquet")
        write_parquet(p, cols, row_group_rows=16_000, page_rows=3_000)
        meta = debug_footer(p)  # the C++ thrift parser reads it
        assert meta["num_rows"] == n
        assert [c["name"] for c in meta["columns"]] == [c[0] for c in cols]
        decoded = read_file(p)  # the oracle's pure-python page decoder
        for col in cols:
            name, vals = col[0], col[1]
            valid = col[2] if len(col) > 2 else np.ones(n, bool)
            got, got_valid = decoded[name]
            assert (np.asarray(got_valid, bool) == valid).all(), name
            g = np.asarray(got)
            v = vals.astype(g.dtype)
            assert (g[valid] == v[valid]).all(), name

    def test_edge_cases(self, tmp_path):
        # zero rows, one row, all-null column, sub-page file
        p0 = str(tmp_path / "zero.parquet")
        write_parquet(p0, [("k", np.empty(0, np.int64))])
        assert pq.read_table(p0).num_rows == 0
        p1 = str(tmp_path / "one.parquet")
        write_parquet(p1, [("k", np.array([42], np.int64)),
                           ("v", np.array([7], np.int32),
                            np.array([True]))])
        t = pq.read_table(p1)
        assert t["k"].to_pylist() == [42] and t["v"].to_pylist() == [7]
        pn = str(tmp_path / "null.parquet")
        n = 5_000
        write_parquet(pn, [("k", np.arange(n, dtype=np.int64)),
                           ("v", np.zeros(n, np.int32),
                            np.zeros(n, bool))])
        t = pq.read_table(pn)
        assert t["v"].null_count == n

    def test_unsupported_dtype_rejected(self, tmp_path):
        with pytest.raises(ValueError, match="unsupported dtype"):
            write_parquet(str(tmp_path / "x.parquet"),
                          [("k", np.zeros(4, np.uint64))])

    def test_zstd_compressed_write(self, tmp_path):
        rng = np.random.default_rng(12)
        n = 80_000
        cols = _rand_cols(rng, n)
        p = str(tmp_path / "z.parquet")
        write_parquet(p, cols, row_group_rows=30_000, page_rows=8_000,
                      compression="zstd")
        md = pq.ParquetFile(p).metadata
        assert md.row_group(0).column(0).compression == "ZSTD"
        _check_pyarrow(p, cols, n)
        # the in-repo C++ footer parser reads it too
        meta = debug_footer(p)
        assert meta["num_rows"] == n

    def test_unknown_write_compression_rejected(self, tmp_path):
        with pytest.raises(RuntimeError, match="not supported"):
            write_parquet(str(tmp_path / "x.parquet"),
                          [("k", np.zeros(4, np.int64))], compression="lz77")


class TestDecimalStringWrite:
    """Native writer: decimal + dictionary-string columns, pinned against
    pyarrow (independent reader) — the C5 write-back matrix
    (ParquetSchemaConverter.java:153-171 decimal physical mapping;
    BYTE_ARRAY dictionary pages for strings)."""

    def _write(self, tmp_path, compression="NONE"):
        from paimon_amd.reader import write_parquet
        rng = np.random.default_rng(7)
        n = 5_000
        unscaled = rng.integers(-10**12, 10**12, n, dtype=np.int64)
        sdict = [f"val-{i:03d}" for i in range(117)]
        ids = rng.integers(0, len(sdict), n).astype(np.int32)
        msk = rng.random(n) > 0.2
        path = str(tmp_path / "ds.parquet")
        write_parquet(path, [
            ("k", np.arange(n, dtype=np.int64)),
            ("d", unscaled),
            ("d9", (unscaled % 10**7).astype(np.int32)),
            ("s", ids, msk),
        ], compression=compression,
            dicts={"s": sdict}, decimals={"d": (18, 2), "d9": (9, 3)})
        return path, unscaled, sdict, ids, msk

    def test_pyarrow_roundtrip(self, tmp_path):
        import decimal
        import pyarrow.parquet as pq
        path, unscaled, sdict, ids, msk = self._write(tmp_path)
        t = pq.read_table(path)
        assert str(t.schema.field("d").type) == "decimal128(18, 2)"
        assert str(t.schema.field("d9").type) == "decimal128(9, 3)"
        got_d = t.column("d").to_pylist()
        exp_d = [decimal.Decimal(int(u)).scaleb(-2) for u in unscaled]
        assert got_d == exp_d
        got_s = t.column("s").to_pylist()
        for i in range(len(ids)):
            if msk[i]:
                assert got_s[i] == sdict[ids[i]], i
            else:
                assert got_s[i] is None, i

    def test_pyarrow_roundtrip_zstd(self, tmp_path):
        import pyarrow.parquet as pq
        path, unscaled, sdict, ids, msk = self._write(tmp_path,
                                                      compression="zstd")
        t = pq.read_table(path)
        got = t.column("d9").to_pylist()
        import decimal
        exp = [decimal.Decimal(int(u) % 10**7).scaleb(-3) for u in unscaled]
        assert got == exp
        got_s = t.column("s").to_pylist()
        assert got_s[:10] == [sdict[ids[i]] if msk[i] else None
                              for i in range(10)]

    def test_footer_parses_with_annotations(self, tmp_path):
        from paimon_amd.reader import debug_footer
        path, *_ = self._write(tmp_path)
        meta = debug_footer(path)
        names = [c["name"] for c in meta["columns"]]
        assert names == ["k", "d", "d9", "s"]
        phys = [c["phys"] for c in meta["columns"]]
        assert phys == [2, 2, 1, 6]  # INT64, INT64, INT32, BYTE_ARRAY
        s_chunk = meta["row_groups"][0]["chunks"][3]
        assert s_chunk["dict_page_offset"] > 0
