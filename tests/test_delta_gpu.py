"""GPU parity for Parquet DELTA_BINARY_PACKED columns
(VectorizedDeltaBinaryPackedReader.java; decoded by k_delta_sum/scan/emit):
pyarrow-written files (independent implementation), merged and compared
against the oracle over the original arrays."""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

from oracle import merge_dedup
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def _value_cols(n, t="int32"):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": t} for i in range(n)])


def _write_delta(runs, out_dir, delta_cols, page_kb=64):
    os.makedirs(out_dir, exist_ok=True)
    metas = []
    for i, r in enumerate(runs):
        arrays = [pa.array(r["key"]), pa.array(r["seq"]),
                  pa.array(r["kind"])]
        fields = [pa.field("_KEY_k", pa.int64(), nullable=False),
                  pa.field("_SEQUENCE_NUMBER", pa.int64(), nullable=False),
                  pa.field("_VALUE_KIND", pa.int8(), nullable=False)]
        names = ["v_k"] + [f"v_c{j}" for j in range(len(r["values"]) - 1)]
        for c, nm in enumerate(names):
            arrays.append(pa.array(r["values"][c]))
            fields.append(pa.field(nm, arrays[-1].type, nullable=False))
        tbl = pa.Table.from_arrays(arrays, schema=pa.schema(fields))
        path = os.path.join(out_dir, f"run-{i}.parquet")
        pq.write_table(tbl, path, compression=None, use_dictionary=False,
                       column_encoding={c: "DELTA_BINARY_PACKED"
                                        for c in delta_cols},
                       data_page_version="1.0", store_schema=False,
                       data_page_size=page_kb * 1024)
        metas.append({"path": path, "rowCount": len(r["key"]),
                      "minKey": int(r["key"][0]),
                      "maxKey": int(r["key"][-1]), "level": 0})
    return metas


def _expected(runs, names):
    r, w = merge_dedup(runs, drop_delete=True)
    exp = {"_KEY_k": np.array([runs[a]["key"][b] for a, b in zip(r, w)],
                              np.int64)}
    for c, nm in enumerate(names):
        exp[nm] = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)])
    return exp


def _run(tmp_path, runs, delta_cols, vtype="int32", page_kb=64):
    names = ["v_k"] + [f"v_c{i}" for i in range(len(runs[0]["values"]) - 1)]
    metas = _write_delta(runs, str(tmp_path), delta_cols, page_kb)
    exp = _expected(runs, names)
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                           _value_cols(len(names) - 1, vtype)) as plan:
            got = {}
            while True:
                b = plan.read_next()
                if b is None:
                    break
                for kk, v in b.items():
                    got.setdefault(kk, []).append(v.copy())
            got = {kk: np.concatenate(v) for kk, v in got.items()}
    assert (got["_KEY_k"] == exp["_KEY_k"]).all()
    for nm in names:
        assert (got[nm] == exp[nm]).all(), nm


class TestDeltaBinaryPacked:
    def test_delta_int32_values(self, tmp_path):
        runs = gen_runs_dedup(4, 60_000, n_value_cols=3, seed=501,
                              delete_frac=0.1)
        _run(tmp_path, runs, ["v_c0", "v_c1", "v_c2"])

    def test_delta_int64_and_key(self, tmp_path):
        # DELTA on the int64 key and pk-copy columns too (small deltas:
        # sorted keys compress hard); multiple pages per chunk
        runs = gen_runs_dedup(3, 120_000, n_value_cols=2, seed=502)
        _run(tmp_path, runs, ["_KEY_k", "v_k", "v_c0", "v_c1"], page_kb=16)

    def test_delta_wide_range(self, tmp_path):
        # extreme deltas: alternating min/max int64 forces wide miniblocks
        rng = np.random.default_rng(503)
        runs = gen_runs_dedup(2, 5_000, n_value_cols=1, seed=503)
        for r in runs:
            n = len(r["key"])
            r["values"][1] = (rng.integers(-2**62, 2**62, n)
                              .astype(np.int64))
        _run(tmp_path, runs, ["v_c0"], vtype="int64")

    def test_delta_nullable_single_run(self, tmp_path):
        # nullable DELTA columns are supported (dense decode + def scatter);
        # the richer multi-run cases live in TestNullableDelta below
        n = 1000
        tbl = pa.table({
            "_KEY_k": pa.array(np.arange(n, dtype=np.int64)),
            "_SEQUENCE_NUMBER": pa.array(np.arange(n, dtype=np.int64)),
            "_VALUE_KIND": pa.array(np.zeros(n, np.int8)),
            "v_c0": pa.array([None if i % 7 == 0 else i
                              for i in range(n)], pa.int32())})
        path = os.path.join(str(tmp_path), "nulls.parquet")
        pq.write_table(tbl, path, compression=None, use_dictionary=False,
                       column_encoding={"v_c0": "DELTA_BINARY_PACKED"},
                       data_page_version="1.0", store_schema=False)
        metas = [{"path": path, "rowCount": n, "minKey": 0,
                  "maxKey": n - 1, "level": 0}]
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_c0", "type": "int32"}]) as plan:
                b = plan.read_next()
                assert len(b["_KEY_k"]) == n
                valid = b["v_c0#valid"]
                exp_null = np.arange(n) % 7 == 0
                assert (valid == ~exp_null).all()
                assert (b["v_c0"][valid] ==
                        np.arange(n, dtype=np.int32)[valid]).all()


class TestNullableDelta:
    """Nullable DELTA_BINARY_PACKED columns: the stream encodes only
    non-null values; they decode to the dense buffer and k_level_scatter
    positions them by def levels (mirrors VectorizedDeltaBinaryPackedReader
    under VectorizedColumnReader's null handling)."""

    def _run(self, tmp_path, null_frac, rows=60_000, n_runs=3, seed=702,
             page_kb=48):
        rng = np.random.default_rng(seed)
        runs = gen_runs_dedup(n_runs, rows, n_value_cols=2, seed=seed,
                              delete_frac=0.1)
        masks = []
        for r in runs:
            m = rng.random(len(r["key"])) < null_frac
            masks.append(m)
        metas = []
        os.makedirs(str(tmp_path), exist_ok=True)
        for i, (r, m) in enumerate(zip(runs, masks)):
            vals = r["values"][1]  # int32 DELTA column, nullable
            arr = pa.array(np.where(m, 0, vals).astype(np.int32),
                           mask=m)
            arrays = [pa.array(r["key"]), pa.array(r["seq"]),
                      pa.array(r["kind"]),
                      pa.array(r["values"][0]), arr,
                      pa.array(r["values"][2])]
            fields = [pa.field("_KEY_k", pa.int64(), nullable=False),
                      pa.field("_SEQUENCE_NUMBER", pa.int64(),
                               nullable=False),
                      pa.field("_VALUE_KIND", pa.int8(), nullable=False),
                      pa.field("v_k", pa.int64(), nullable=False),
                      pa.field("v_c0", pa.int32(), nullable=True),
                      pa.field("v_c1", pa.int32(), nullable=False)]
            tbl = pa.Table.from_arrays(arrays, schema=pa.schema(fields))
            path = os.path.join(str(tmp_path), f"run-{i}.parquet")
            pq.write_table(tbl, path, compression=None,
                           use_dictionary=False,
                           column_encoding={"v_c0": "DELTA_BINARY_PACKED",
                                            "v_c1": "DELTA_BINARY_PACKED"},
                           data_page_version="1.0", store_schema=False,
                           data_page_size=page_kb * 1024)
            metas.append({"path": path, "rowCount": len(r["key"]),
                          "minKey": int(r["key"][0]),
                          "maxKey": int(r["key"][-1]), "level": 0})
        from oracle import merge_dedup
        rr, ww = merge_dedup(runs, drop_delete=True)
        exp_keys = np.array([runs[a]["key"][b] for a, b in zip(rr, ww)],
                            np.int64)
        exp_v = np.array([runs[a]["values"][1][b] for a, b in zip(rr, ww)],
                         np.int32)
        exp_null = np.array([masks[a][b] for a, b in zip(rr, ww)], bool)
        exp_v1 = np.array([runs[a]["values"][2][b] for a, b in zip(rr, ww)],
                          np.int32)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2)) as plan:
                got = {}
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    for k, v in b.items():
                        got.setdefault(k, []).append(v.copy())
                got = {k: np.concatenate(v) for k, v in got.items()}
        assert (got["_KEY_k"] == exp_keys).all()
        assert (got["v_c1"] == exp_v1).all()
        valid = got["v_c0#valid"]
        assert (valid == ~exp_null).all(), \
            np.flatnonzero(valid != ~exp_null)[:10]
        live = ~exp_null
        assert (got["v_c0"][live] == exp_v[live]).all()

    def test_sparse_nulls(self, tmp_path):
        self._run(tmp_path, 0.05, seed=702)

    def test_half_nulls(self, tmp_path):
        self._run(tmp_path, 0.5, seed=703)

    def test_dense_nulls(self, tmp_path):
        self._run(tmp_path, 0.95, seed=704, rows=30_000)

    def test_multi_page(self, tmp_path):
        self._run(tmp_path, 0.3, rows=200_000, page_kb=16, seed=705)


class TestDeltaByteArray:
    """DELTA_BYTE_ARRAY string pages (VectorizedDeltaByteArrayReader.java:
    prefix-shared byte arrays): values intern into the plan-level global
    dictionary at staging; the device representation is int32 global ids,
    like every other string path."""

    def _write(self, tmp_path, runs, strs, masks=None, page_kb=64):
        metas = []
        for i, r in enumerate(runs):
            arrays = {
                "_KEY_k": pa.array(r["key"]),
                "_SEQUENCE_NUMBER": pa.array(r["seq"]),
                "_VALUE_KIND": pa.array(r["kind"]),
                "v_s": pa.array(strs[i],
                                mask=masks[i] if masks else None),
                "v_c0": pa.array(r["values"][1]),
            }
            path = os.path.join(str(tmp_path), f"run-{i}.parquet")
            pq.write_table(pa.table(arrays), path, compression=None,
                           use_dictionary=False,
                           column_encoding={"v_s": "DELTA_BYTE_ARRAY"},
                           data_page_version="1.0", store_schema=False,
                           data_page_size=page_kb * 1024)
            metas.append({"path": path, "rowCount": len(r["key"]),
                          "minKey": int(r["key"][0]),
                          "maxKey": int(r["key"][-1]), "level": 0})
        return metas

    def _check(self, tmp_path, null_frac=0.0, n_runs=3, rows=30_000,
               seed=971, page_kb=64):
        rng = np.random.default_rng(seed)
        runs = gen_runs_dedup(n_runs, rows, n_value_cols=1, seed=seed,
                              delete_frac=0.1)
        strs, masks = [], []
        for r in runs:
            # shared prefixes make DELTA_BYTE_ARRAY effective
            strs.append([f"company/dept-{k % 37:02d}/user-{k % 1009:04d}"
                         for k in r["key"].tolist()])
            masks.append(rng.random(len(r["key"])) < null_frac
                         if null_frac else None)
        metas = self._write(tmp_path, runs, strs,
                            masks if null_frac else None, page_kb)
        rr, ww = merge_dedup(runs)
        exp_k = np.array([runs[a]["key"][b] for a, b in zip(rr, ww)],
                         np.int64)
        exp_s = [strs[a][b] for a, b in zip(rr, ww)]
        exp_null = (np.array([masks[a][b] for a, b in zip(rr, ww)], bool)
                    if null_frac else np.zeros(len(rr), bool))
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_s", "type": "string"},
                                {"name": "v_c0", "type": "int32"}]) as plan:
                got, dicts = {}, None
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    dicts = b["v_s#dict"]
                    for kk, v in b.items():
                        if kk.endswith("#dict"):
                            continue
                        got.setdefault(kk, []).append(v.copy())
                got = {kk: np.concatenate(v) for kk, v in got.items()}
        assert (got["_KEY_k"] == exp_k).all()
        live = ~exp_null
        if null_frac:
            assert (got["v_s#valid"] == live).all()
        dec = [dicts[i].decode() for i in got["v_s"][live]]
        assert dec == [x for x, lv in zip(exp_s, live) if lv]

    def test_basic(self, tmp_path):
        self._check(tmp_path)

    def test_with_nulls(self, tmp_path):
        self._check(tmp_path, null_frac=0.3, seed=972)

    def test_multi_page(self, tmp_path):
        self._check(tmp_path, rows=120_000, page_kb=16, seed=973)


class TestDeltaLengthByteArray:
    def test_basic(self, tmp_path):
        # DELTA_LENGTH_BYTE_ARRAY: lengths (DELTA) + raw bytes, no prefixes
        n = 40_000
        rng = np.random.default_rng(981)
        k = np.arange(n, dtype=np.int64)
        strs = [f"value-{int(x) % 500:03d}-{'y' * (int(x) % 5)}" for x in k]
        tbl = pa.table({"_KEY_k": pa.array(k),
                        "_SEQUENCE_NUMBER": pa.array(k),
                        "_VALUE_KIND": pa.array(np.zeros(n, np.int8)),
                        "v_s": pa.array(strs)})
        path = os.path.join(str(tmp_path), "dlba.parquet")
        pq.write_table(tbl, path, compression=None, use_dictionary=False,
                       column_encoding={"v_s": "DELTA_LENGTH_BYTE_ARRAY"},
                       data_page_version="1.0", store_schema=False,
                       data_page_size=32 * 1024)
        metas = [{"path": path, "rowCount": n, "minKey": 0,
                  "maxKey": n - 1, "level": 0}]
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_s", "type": "string"}]) as plan:
                b = plan.read_next()
                dec = [b["v_s#dict"][i].decode() for i in b["v_s"]]
        assert dec == strs
