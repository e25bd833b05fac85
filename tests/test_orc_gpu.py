"""GPU parity for the ORC read path (C3's format half): RLEv2 / byte-RLE /
PRESENT decode on device, feeding the same merge pipeline — against the CPU
oracle on ORC files written by pyarrow (the independent implementation)."""

import numpy as np
import pytest

from oracle import merge_dedup, partial_update_model
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup, gen_runs_partial_update, write_runs

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def _value_cols(n):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": "int32"} for i in range(n)])


def _read_all(plan):
    got = {}
    while True:
        b = plan.read_next()
        if b is None:
            break
        for kk, v in b.items():
            got.setdefault(kk, []).append(v.copy())
    return {kk: np.concatenate(v) for kk, v in got.items()}


class TestOrcDedup:
    def _run(self, tmp_path, runs, compression="NONE", **kw):
        metas = write_runs(runs, str(tmp_path), file_format="orc",
                           compression=compression)
        r, w = merge_dedup(runs, **kw)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(len(runs[0]["values"]) - 1),
                               **kw) as plan:
                got = _read_all(plan)
        names = (["_KEY_k", "_SEQUENCE_NUMBER", "_VALUE_KIND", "v_k"] +
                 [f"v_c{i}" for i in range(len(runs[0]["values"]) - 1)])
        srcs = (["key", "seq", "kind"] +
                [("values", c) for c in range(len(runs[0]["values"]))])
        for nm, src in zip(names, srcs):
            if isinstance(src, tuple):
                e = np.array([runs[a]["values"][src[1]][b]
                              for a, b in zip(r, w)])
            else:
                e = np.array([runs[a][src][b] for a, b in zip(r, w)])
            assert (got[nm] == e).all(), nm

    def test_orc_dedup_8_runs(self, tmp_path):
        runs = gen_runs_dedup(8, 40_000, n_value_cols=4, seed=81,
                              delete_frac=0.1)
        self._run(tmp_path, runs)

    def test_orc_dedup_monotonic_keys_delta(self, tmp_path):
        # dense monotonic keys exercise the DELTA sub-encoding
        rng = np.random.default_rng(82)
        runs = []
        seqs = rng.permutation(200_000).astype(np.int64)
        for r in range(4):
            keys = (np.arange(50_000, dtype=np.int64) * 2 + r)
            runs.append({"key": keys, "seq": seqs[r*50_000:(r+1)*50_000],
                         "kind": np.zeros(50_000, np.int8),
                         "values": [keys.copy(),
                                    rng.integers(-100, 100, 50_000).astype(np.int32)]})
        self._run(tmp_path, runs)

    def test_orc_dedup_outliers_patched(self, tmp_path):
        # small values + outliers exercise PATCHED_BASE in the values
        rng = np.random.default_rng(83)
        runs = []
        seqs = rng.permutation(80_000).astype(np.int64)
        for r in range(4):
            keys = np.sort(rng.choice(60_000, 20_000, replace=False)).astype(np.int64)
            v = rng.integers(0, 50, 20_000).astype(np.int32)
            v[rng.choice(20_000, 200, replace=False)] = rng.integers(
                2**28, 2**31 - 1, 200).astype(np.int32)
            runs.append({"key": keys, "seq": seqs[r*20_000:(r+1)*20_000],
                         "kind": rng.choice([0, 3], 20_000, p=[.9, .1]).astype(np.int8),
                         "values": [keys.copy(), v]})
        self._run(tmp_path, runs)

    def test_orc_zlib(self, tmp_path):
        # ORC's default codec: streams + footers arrive chunk-compressed
        # (host decompress at staging, ORC spec "Compression")
        runs = gen_runs_dedup(6, 30_000, n_value_cols=3, seed=88,
                              delete_frac=0.1)
        self._run(tmp_path, runs, compression="zlib")

    def test_orc_zstd(self, tmp_path):
        runs = gen_runs_dedup(5, 25_000, n_value_cols=2, seed=89,
                              delete_frac=0.15)
        self._run(tmp_path, runs, compression="zstd")

    def test_orc_snappy(self, tmp_path):
        runs = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=90,
                              delete_frac=0.1)
        self._run(tmp_path, runs, compression="snappy")

    def test_orc_keep_delete(self, tmp_path):
        runs = gen_runs_dedup(4, 15_000, n_value_cols=2, seed=84,
                              delete_frac=0.3)
        self._run(tmp_path, runs, drop_delete=False)


class TestOrcPartialUpdate:
    def test_orc_pu_with_nulls(self, tmp_path):
        # C3 proper: ORC + PartialUpdate + nulls (PRESENT streams on device)
        runs = gen_runs_partial_update(4, 25_000, n_value_cols=8, seed=85,
                                       update_frac=0.3, update_cols=3)
        metas = write_runs(runs, str(tmp_path), file_format="orc",
                           compression="zlib")
        exp = partial_update_model(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(8),
                               merge_engine="partial-update") as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(8)]
        for c, nm in enumerate(names):
            ev, em = exp["values"][c], exp["valid"][c]
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm


class TestOrcFloatDouble:
    def test_orc_float_double_columns(self, tmp_path):
        # FLOAT/DOUBLE DATA streams are raw IEEE754: staged as a copy, merged
        # bit-exactly
        rng = np.random.default_rng(86)
        runs = []
        seqs = rng.permutation(45_000).astype(np.int64)
        for i in range(3):
            keys = np.sort(rng.choice(70_000, 15_000,
                                      replace=False)).astype(np.int64)
            runs.append({
                "key": keys, "seq": seqs[i*15_000:(i+1)*15_000],
                "kind": rng.choice([0, 3], 15_000, p=[.9, .1]).astype(np.int8),
                "values": [keys.copy(),
                           rng.standard_normal(15_000).astype(np.float32),
                           rng.standard_normal(15_000)]})
        metas = write_runs(runs, str(tmp_path), file_format="orc",
                           compression="zlib")
        r, w = merge_dedup(runs, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_k", "type": "int64"},
                                {"name": "v_c0", "type": "float32"},
                                {"name": "v_c1", "type": "float64"}]) as plan:
                got = _read_all(plan)
        for nm, c, dt in (("v_c0", 1, np.float32), ("v_c1", 2, np.float64)):
            e = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)], dt)
            assert got[nm].dtype == dt
            assert (got[nm].view(np.uint8) == e.view(np.uint8)).all(), nm

    def test_orc_nullable_double_pu(self, tmp_path):
        rng = np.random.default_rng(87)
        runs = gen_runs_partial_update(3, 12_000, n_value_cols=2, seed=87,
                                       update_frac=0.5, update_cols=1)
        for r in runs:
            n = len(r["key"])
            r["values"][1] = rng.standard_normal(n)
            r["values"][2] = rng.standard_normal(n).astype(np.float32)
        metas = write_runs(runs, str(tmp_path), file_format="orc")
        exp = partial_update_model(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_k", "type": "int64"},
                                {"name": "v_c0", "type": "float64"},
                                {"name": "v_c1", "type": "float32"}],
                               merge_engine="partial-update") as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        for c, nm in ((1, "v_c0"), (2, "v_c1")):
            em = exp["valid"][c]
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == exp["values"][c][em]).all(), nm
