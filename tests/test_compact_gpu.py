"""Compaction rewrite round trip (CompactRewriter.rewrite semantics,
mergetree/compact/CompactRewriter.java:29-56): GPU merge + rolling Parquet
write-back; outputs re-readable by pyarrow AND by our own reader; per-file
DataFileMeta stats per KeyValueDataFileWriter (io/KeyValueDataFileWriter.java:
121-170)."""

import numpy as np
import pyarrow.parquet as pq
import pytest

from oracle import merge_dedup
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.compact import rewrite
from paimon_amd.datagen import gen_runs_dedup, write_runs

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def _value_cols(n):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": "int32"} for i in range(n)])


def test_compact_16_to_1(tmp_path):
    # C5 shape scaled down: 16 runs -> 1 level, rolling files
    runs = gen_runs_dedup(16, 8_000, n_value_cols=4, seed=91, delete_frac=0.1)
    metas = write_runs(runs, str(tmp_path / "in"), compression="NONE")
    with Session(0) as s:
        res = rewrite(s, metas, KEY_COLS, _value_cols(4),
                      str(tmp_path / "out"), output_level=5,
                      drop_delete=True, target_file_rows=40_000)
    after = res["after"]
    assert len(after) >= 2  # rolled
    # expected merge from the oracle
    r, w = merge_dedup(runs, drop_delete=True)
    exp_key = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
    exp_seq = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
    # stats + re-readability via pyarrow
    got_key, got_seq = [], []
    total = 0
    for m in after:
        t = pq.read_table(m["path"])
        k = np.asarray(t.column("_KEY_k"))
        sq = np.asarray(t.column("_SEQUENCE_NUMBER"))
        assert m["rowCount"] == len(k)
        assert m["minKey"] == k[0] and m["maxKey"] == k[-1]
        assert m["minSequenceNumber"] == sq.min()
        assert m["maxSequenceNumber"] == sq.max()
        assert m["level"] == 5
        assert m["deleteRowCount"] == 0  # drop_delete=True
        got_key.append(k)
        got_seq.append(sq)
        total += len(k)
    got_key = np.concatenate(got_key)
    got_seq = np.concatenate(got_seq)
    assert (got_key == exp_key).all()
    assert (got_seq == exp_seq).all()

    # the compacted level re-reads through our own reader (single run now)
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(after), KEY_COLS,
                           _value_cols(4)) as plan:
            got = {}
            while True:
                b = plan.read_next()
                if b is None:
                    break
                for kk, v in b.items():
                    got.setdefault(kk, []).append(v.copy())
            got = {kk: np.concatenate(v) for kk, v in got.items()}
    assert (got["_KEY_k"] == exp_key).all()


def test_compact_keep_deletes(tmp_path):
    # rewriting below the top level keeps retracts (dropDelete=false,
    # MergeTreeCompactManager.triggerCompaction semantics)
    runs = gen_runs_dedup(4, 5_000, n_value_cols=2, seed=92, delete_frac=0.3)
    metas = write_runs(runs, str(tmp_path / "in"), compression="NONE")
    with Session(0) as s:
        res = rewrite(s, metas, KEY_COLS, _value_cols(2),
                      str(tmp_path / "out"), output_level=2,
                      drop_delete=False, target_file_rows=1_000_000)
    r, w = merge_dedup(runs, drop_delete=False)
    exp_kind = np.array([runs[a]["kind"][b] for a, b in zip(r, w)], np.int8)
    t = pq.read_table(res["after"][0]["path"])
    kd = np.asarray(t.column("_VALUE_KIND"))
    assert (kd == exp_kind).all()
    n_del = int(np.count_nonzero((exp_kind != 0) & (exp_kind != 2)))
    assert res["after"][0]["deleteRowCount"] == n_del


def test_compact_partial_update_nullable(tmp_path):
    # PU compaction produces nullable output columns; the native writer must
    # carry validity through (def levels) and the result must re-read both
    # via pyarrow and via our own reader
    from oracle import partial_update_model
    from paimon_amd.datagen import gen_runs_partial_update
    runs = gen_runs_partial_update(4, 10_000, n_value_cols=4, seed=93,
                                   update_frac=0.4, update_cols=2)
    metas = write_runs(runs, str(tmp_path / "in"), compression="NONE")
    with Session(0) as s:
        res = rewrite(s, metas, KEY_COLS, _value_cols(4),
                      str(tmp_path / "out"), output_level=5,
                      drop_delete=True, merge_engine="partial-update",
                      target_file_rows=1_000_000)
    exp = partial_update_model(runs)
    t = pq.read_table(res["after"][0]["path"])
    assert (np.asarray(t.column("_KEY_k")) == exp["key"]).all()
    names = ["v_k"] + [f"v_c{i}" for i in range(4)]
    for c, nm in enumerate(names):
        col = t.column(nm)
        valid = col.is_valid().to_numpy(zero_copy_only=False)
        assert (valid == exp["valid"][c]).all(), nm
        vals = col.to_numpy(zero_copy_only=False)
        ev = exp["values"][c]
        assert (vals[valid] == ev[valid]).all(), nm
    # re-read the compacted file through the GPU reader
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(res["after"]), KEY_COLS,
                           _value_cols(4)) as plan:
            b = plan.read_next()
            assert (b["_KEY_k"] == exp["key"]).all()
            for c, nm in enumerate(names):
                gv = b.get(nm + "#valid")
                if gv is None:
                    gv = np.ones(len(b[nm]), bool)
                assert (gv.astype(bool) == exp["valid"][c]).all(), nm


def test_compact_zstd_write(tmp_path):
    # zstd write-back round-trips: pyarrow reads it AND the GPU reader
    # re-merges the compacted level
    runs = gen_runs_dedup(4, 10_000, n_value_cols=2, seed=94, delete_frac=0.1)
    metas = write_runs(runs, str(tmp_path / "in"), compression="NONE")
    with Session(0) as s:
        res = rewrite(s, metas, KEY_COLS, _value_cols(2),
                      str(tmp_path / "out"), output_level=5,
                      drop_delete=True, compression="zstd",
                      target_file_rows=1_000_000)
    r, w = merge_dedup(runs, drop_delete=True)
    exp_key = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
    m = res["after"][0]
    t = pq.read_table(m["path"])
    assert pq.ParquetFile(m["path"]).metadata.row_group(0).column(0)\
        .compression == "ZSTD"
    assert (np.asarray(t.column("_KEY_k")) == exp_key).all()
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(res["after"]), KEY_COLS,
                           _value_cols(2)) as plan:
            b = plan.read_next()
            assert (b["_KEY_k"] == exp_key).all()
