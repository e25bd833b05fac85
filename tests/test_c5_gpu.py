"""GPU parity for the C5 compaction matrix (BASELINE configs[4]): mixed
types — decimal(18,2) riding INT64 (ParquetSchemaConverter.java:153-171) and
a dictionary-encoded string column (BYTE_ARRAY + dictionary pages) — through
read, merge and the compaction write-back round trip. Inputs are
pyarrow-written (independent implementation) unless the test pins the
native writer's output."""

import numpy as np
import pytest

from oracle import merge_dedup
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.compact import rewrite
from paimon_amd.datagen import C5_VALUE_COLS, gen_runs_c5, write_runs_c5

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]
VAL_NAMES = [c["name"] for c in C5_VALUE_COLS]


def _read_all(plan):
    got = {}
    while True:
        b = plan.read_next()
        if b is None:
            break
        for kk, v in b.items():
            got.setdefault(kk, []).append(v.copy())
    out = {}
    for kk, v in got.items():
        out[kk] = v[0] if kk.endswith("#dict") else np.concatenate(v)
    return out


def _strings(got, name):
    d = got[name + "#dict"]
    return d[got[name]]


def _expected(runs, drop_delete=True):
    r, w = merge_dedup(runs, drop_delete=drop_delete)
    exp = {
        "_KEY_k": np.array([runs[a]["key"][b] for a, b in zip(r, w)],
                           np.int64),
        "_SEQUENCE_NUMBER": np.array(
            [runs[a]["seq"][b] for a, b in zip(r, w)], np.int64),
        "_VALUE_KIND": np.array([runs[a]["kind"][b] for a, b in zip(r, w)],
                                np.int8),
    }
    for c, nm in enumerate(VAL_NAMES):
        exp[nm] = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)])
    sdict = np.array([s.encode() for s in runs[0]["str_dict"]], dtype=object)
    exp["v_str#strings"] = sdict[exp["v_str"]]
    return exp


def _check(got, exp):
    assert (got["_KEY_k"] == exp["_KEY_k"]).all()
    assert (got["_SEQUENCE_NUMBER"] == exp["_SEQUENCE_NUMBER"]).all()
    assert (got["_VALUE_KIND"] == exp["_VALUE_KIND"]).all()
    for nm in VAL_NAMES:
        if nm == "v_str":
            assert (_strings(got, "v_str") == exp["v_str#strings"]).all()
        else:
            assert (got[nm] == exp[nm]).all(), nm


class TestC5Read:
    def test_c5_dedup_parity(self, tmp_path):
        runs = gen_runs_c5(4, 20_000, seed=61, str_card=500)
        metas = write_runs_c5(runs, str(tmp_path))
        exp = _expected(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               C5_VALUE_COLS) as plan:
                got = _read_all(plan)
        _check(got, exp)

    def test_c5_zstd(self, tmp_path):
        runs = gen_runs_c5(3, 15_000, seed=62, str_card=200)
        metas = write_runs_c5(runs, str(tmp_path), compression="zstd")
        exp = _expected(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               C5_VALUE_COLS) as plan:
                got = _read_all(plan)
        _check(got, exp)

    def test_native_written_c5_readable(self, tmp_path):
        # the library's own writer output feeds the GPU reader (the rolling
        # compaction outputs are re-read exactly this way)
        runs = gen_runs_c5(3, 12_000, seed=63, str_card=300)
        metas = write_runs_c5(runs, str(tmp_path), writer="native")
        exp = _expected(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               C5_VALUE_COLS) as plan:
                got = _read_all(plan)
        _check(got, exp)

    def test_plain_string_rejected(self, tmp_path):
        # PLAIN byte-array pages fail loudly (dictionary-encoded only in v1)
        import os
        import pyarrow as pa
        import pyarrow.parquet as pq
        n = 1000
        tbl = pa.table({
            "_KEY_k": pa.array(np.arange(n, dtype=np.int64)),
            "_SEQUENCE_NUMBER": pa.array(np.arange(n, dtype=np.int64)),
            "_VALUE_KIND": pa.array(np.zeros(n, np.int8)),
            "v_str": pa.array([f"x{i}" for i in range(n)])})
        path = os.path.join(str(tmp_path), "plain.parquet")
        pq.write_table(tbl, path, compression=None, use_dictionary=False,
                       data_page_version="1.0", store_schema=False)
        metas = [{"path": path, "rowCount": n, "minKey": 0,
                  "maxKey": n - 1, "level": 0}]
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="dictionary-encoded"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              [{"name": "v_str", "type": "string"}])

    def test_phys_type_mismatch_rejected(self, tmp_path):
        # decimal(18,2) over a DOUBLE column fails loudly at staging
        import os
        import pyarrow as pa
        import pyarrow.parquet as pq
        n = 100
        tbl = pa.table({
            "_KEY_k": pa.array(np.arange(n, dtype=np.int64)),
            "_SEQUENCE_NUMBER": pa.array(np.arange(n, dtype=np.int64)),
            "_VALUE_KIND": pa.array(np.zeros(n, np.int8)),
            "v_dec": pa.array(np.zeros(n))})
        path = os.path.join(str(tmp_path), "bad.parquet")
        pq.write_table(tbl, path, compression=None, use_dictionary=False,
                       data_page_version="1.0", store_schema=False)
        metas = [{"path": path, "rowCount": n, "minKey": 0,
                  "maxKey": n - 1, "level": 0}]
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="physical type"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              [{"name": "v_dec", "type": "decimal(18,2)"}])


class TestC5Compact:
    def test_c5_compact_roundtrip(self, tmp_path):
        # the full C5 shape in miniature: 6 -> 1 mixed-type compaction,
        # outputs re-read by pyarrow (independent) and by the GPU reader
        import decimal
        import pyarrow.parquet as pq
        runs = gen_runs_c5(6, 10_000, seed=64, str_card=400)
        metas = write_runs_c5(runs, str(tmp_path / "in"))
        exp = _expected(runs, drop_delete=True)
        with Session(0) as s:
            res = rewrite(s, metas, KEY_COLS, C5_VALUE_COLS,
                          str(tmp_path / "out"), output_level=5,
                          drop_delete=True, target_file_rows=25_000)
            after = res["after"]
            assert sum(m["rowCount"] for m in after) == len(exp["_KEY_k"])
            # pyarrow re-read of the rolling outputs
            parts = [pq.read_table(m["path"]) for m in after]
            keys = np.concatenate(
                [p.column("_KEY_k").to_numpy() for p in parts])
            assert (keys == exp["_KEY_k"]).all()
            dec = sum((p.column("v_dec").to_pylist() for p in parts), [])
            exp_dec = [decimal.Decimal(int(u)).scaleb(-2)
                       for u in exp["v_dec"]]
            assert dec == exp_dec
            strs = np.array(
                sum((p.column("v_str").to_pylist() for p in parts), []),
                dtype=object)
            exp_strs = np.array([b.decode() for b in exp["v_str#strings"]],
                                dtype=object)
            assert (strs == exp_strs).all()
            # GPU re-read of the compacted outputs
            with MergeReadPlan(s, file_descs_from_metas(after), KEY_COLS,
                               C5_VALUE_COLS) as plan:
                got2 = _read_all(plan)
        _check(got2, exp)
