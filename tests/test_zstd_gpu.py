"""GPU zstd page decoding (k_zstd_pages): device round trips vs libzstd
frames, and the staging integration (zstd parquet chunks decode on the GPU —
stats.gpu_zstd_pages counts them)."""
import os

import numpy as np
import pyarrow as pa
import pytest

from oracle import merge_dedup
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup, write_runs
from paimon_amd.reader import debug_zstd_gpu

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


class TestKernelRoundTrip:
    def _rt(self, raw, level=3):
        comp = pa.Codec("zstd", compression_level=level).compress(raw)
        got = debug_zstd_gpu(comp.to_pybytes(), len(raw))
        assert got == bytes(raw)

    def test_basic(self):
        self._rt(b"hello zstd on gfx950 " * 500)

    def test_structured(self):
        self._rt(np.arange(300_000, dtype=np.int32).tobytes())

    def test_incompressible(self):
        rng = np.random.default_rng(5)
        self._rt(rng.integers(0, 256, 250_000, dtype=np.uint8).tobytes())

    def test_multiblock(self):
        rng = np.random.default_rng(6)
        raw = np.cumsum(rng.integers(0, 9, 400_000, dtype=np.int64)).tobytes()
        self._rt(raw)  # ~3 MB, > 24 blocks

    def test_levels(self):
        raw = (b"abcdefgh" * 40_000)
        for lvl in (1, 3, 9, 19):
            self._rt(raw, level=lvl)


class TestStagingIntegration:
    def test_zstd_chunks_decode_on_gpu(self, tmp_path):
        runs = gen_runs_dedup(4, 80_000, n_value_cols=4, seed=421,
                              delete_frac=0.1)
        metas = write_runs(runs, str(tmp_path), compression="zstd")
        r, w = merge_dedup(runs)
        exp_keys = np.array([runs[a]["key"][b] for a, b in zip(r, w)],
                            np.int64)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_k", "type": "int64"}] +
                               [{"name": f"v_c{i}", "type": "int32"}
                                for i in range(3)]) as plan:
                got = []
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    got.append(b["_KEY_k"].copy())
                st = plan.stats()
        got = np.concatenate(got)
        assert (got == exp_keys).all()
        # the zstd pages must have decoded through k_zstd_pages, not libzstd
        assert st["gpu_zstd_pages"] > 0, st


class TestCompressKernel:
    """k_zstd_compress: frames compressed ON the GPU must decode with
    libzstd (pyarrow) bit-exactly, and the parquet write-back's zstd leg
    (pw_gpu_zstd_compress batches) must round trip through pyarrow."""

    def _rt(self, raw):
        from paimon_amd.reader import debug_zstd_enc_gpu
        comp = debug_zstd_enc_gpu(raw)
        back = pa.Codec("zstd").decompress(
            comp, decompressed_size=len(raw)).to_pybytes()
        assert back == bytes(raw)

    def test_text(self):
        self._rt(b"gpu compressed frame " * 20_000)

    def test_ascending_int64(self):
        self._rt(np.arange(400_000, dtype=np.int64).tobytes())

    def test_random(self):
        rng = np.random.default_rng(31)
        self._rt(rng.integers(0, 256, 300_000, dtype=np.uint8).tobytes())

    def test_write_back_pyarrow_reads(self, tmp_path):
        # write a zstd parquet file with pages compressed by the GPU batch
        # path (PMH_GPU_ZSTD_ENC=1 opt-in — host libzstd is the measured
        # write-side default) and read it back with PYARROW
        import pyarrow.parquet as pq
        from paimon_amd.reader import write_parquet
        os.environ["PMH_GPU_ZSTD_ENC"] = "1"
        n = 300_000
        k = np.arange(n, dtype=np.int64)
        v = (k * 31 + 7).astype(np.int32)
        path = str(tmp_path / "gpu_zstd.parquet")
        write_parquet(path, [("_KEY_k", k), ("v", v)], compression="zstd")
        t = pq.read_table(path)
        assert (t["_KEY_k"].to_numpy() == k).all()
        assert (t["v"].to_numpy() == v).all()


class TestDeviceResidentStaging:
    def test_mixed_nullable_and_plain_chunks(self, tmp_path):
        # nullable chunk (host-peeked def levels) and simple PLAIN chunks
        # (device-resident, D2D-staged) in the same zstd file
        import pyarrow.parquet as pq
        rng = np.random.default_rng(41)
        n = 120_000
        k = np.arange(n, dtype=np.int64) * 2
        v0 = rng.integers(0, 1 << 30, n).astype(np.int32)
        mask = rng.random(n) < 0.25
        tbl = pa.table({
            "_KEY_k": pa.array(k),
            "_SEQUENCE_NUMBER": pa.array(k.copy()),
            "_VALUE_KIND": pa.array(np.zeros(n, np.int8)),
            "v_c0": pa.array(v0, mask=mask),
            "v_c1": pa.array((k * 3).astype(np.int64)),
        })
        path = str(tmp_path / "mix.parquet")
        pq.write_table(tbl, path, compression="zstd",
                       use_dictionary=False, data_page_version="1.0",
                       store_schema=False, data_page_size=64 * 1024)
        metas = [{"path": path, "rowCount": n, "minKey": 0,
                  "maxKey": int(k[-1]), "level": 0}]
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_c0", "type": "int32"},
                                {"name": "v_c1", "type": "int64"}]) as plan:
                b = plan.read_next()
                st = plan.stats()
        assert st["gpu_zstd_pages"] > 0
        assert (b["_KEY_k"] == k).all()
        assert (b["v_c1"] == k * 3).all()
        assert (b["v_c0#valid"] == ~mask).all()
        assert (b["v_c0"][~mask] == v0[~mask]).all()
