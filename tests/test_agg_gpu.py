"""GPU parity for the aggregation merge engine (AggregateMergeFunction,
per-field FieldAggregators) against the oracle model — which is itself pinned
to the reference's ported semantics in test_oracle_merge.TestAggregationModel."""

import numpy as np
import pytest

from oracle import aggregation_model
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_partial_update, write_runs

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]
AGGS = ["sum", "max", "min", "last_value", "first_value",
        "last_non_null_value", "first_non_null_value"]


def _value_cols(n, t="int32"):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": t} for i in range(n)])


def _read_all(plan):
    got = {}
    while True:
        b = plan.read_next()
        if b is None:
            break
        for kk, v in b.items():
            got.setdefault(kk, []).append(v.copy())
    return {kk: np.concatenate(v) for kk, v in got.items()}


def _check(got, exp, names):
    assert (got["_KEY_k"] == exp["key"]).all()
    assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
    assert (got["_VALUE_KIND"] == exp["kind"]).all()
    for c, nm in enumerate(names):
        ev, evalid = exp["values"][c], exp["valid"][c]
        gvalid = got.get(nm + "#valid")
        if gvalid is None:
            gvalid = np.ones(len(got[nm]), dtype=bool)
        assert (gvalid == evalid).all(), nm
        if ev.dtype.kind == "f":
            # sums run in the same member order on both sides: bit-exact
            assert (got[nm][evalid].view(np.uint8).reshape(evalid.sum(), -1)
                    == ev[evalid].view(np.uint8).reshape(evalid.sum(), -1)
                    ).all(), nm
        else:
            assert (got[nm][evalid] == ev[evalid]).all(), nm


class TestAggregationEngine:
    def _run(self, tmp_path, runs, aggs_by_col, file_format="parquet",
             col_type="int32", **write_kwargs):
        write_kwargs.setdefault("compression", "NONE")
        metas = write_runs(runs, str(tmp_path), file_format=file_format,
                           **write_kwargs)
        n_vals = len(runs[0]["values"])
        names = ["v_k"] + [f"v_c{i}" for i in range(n_vals - 1)]
        exp = aggregation_model(
            runs, ["last_non_null_value"] +
            [aggs_by_col.get(nm, "last_non_null_value") for nm in names[1:]])
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(n_vals - 1, col_type),
                               merge_engine="aggregation",
                               aggregations=aggs_by_col) as plan:
                got = _read_all(plan)
        _check(got, exp, names)

    def test_all_aggregators_int32(self, tmp_path):
        runs = gen_runs_partial_update(4, 30_000, n_value_cols=7, seed=91,
                                       update_frac=0.4, update_cols=4)
        for r in runs:  # bound so int32 sums cannot overflow
            for c in range(1, 8):
                r["values"][c] = (r["values"][c] % 10_000).astype(np.int32)
        aggs = {f"v_c{i}": a for i, a in enumerate(AGGS)}
        self._run(tmp_path, runs, aggs)

    def test_default_is_last_non_null(self, tmp_path):
        # no aggregations map at all -> every column last_non_null_value
        runs = gen_runs_partial_update(3, 20_000, n_value_cols=4, seed=92)
        self._run(tmp_path, runs, {})

    def test_orc_aggregation(self, tmp_path):
        runs = gen_runs_partial_update(4, 25_000, n_value_cols=4, seed=93,
                                       update_frac=0.3, update_cols=2)
        for r in runs:
            for c in range(1, 5):
                r["values"][c] = (r["values"][c] % 10_000).astype(np.int32)
        aggs = {"v_c0": "sum", "v_c1": "max", "v_c2": "min",
                "v_c3": "first_non_null_value"}
        self._run(tmp_path, runs, aggs, file_format="orc")

    def test_float_sum_max_min(self, tmp_path):
        rng = np.random.default_rng(94)
        runs = gen_runs_partial_update(4, 20_000, n_value_cols=3, seed=95,
                                       update_frac=0.5, update_cols=2)
        for r in runs:
            n = len(r["key"])
            for c in (1, 2, 3):
                r["values"][c] = rng.standard_normal(n).astype(np.float64)
        aggs = {"v_c0": "sum", "v_c1": "max", "v_c2": "min"}
        self._run(tmp_path, runs, aggs, col_type="float64")

    def test_float32_sum(self, tmp_path):
        rng = np.random.default_rng(96)
        runs = gen_runs_partial_update(3, 15_000, n_value_cols=2, seed=97,
                                       update_frac=0.5, update_cols=1)
        for r in runs:
            n = len(r["key"])
            for c in (1, 2):
                r["values"][c] = rng.standard_normal(n).astype(np.float32)
        self._run(tmp_path, runs, {"v_c0": "sum", "v_c1": "max"},
                  col_type="float32")

    def test_agg_zstd_multi_rowgroup(self, tmp_path):
        # chunked pages + zstd staging + >1 row group under the member lists
        runs = gen_runs_partial_update(3, 60_000, n_value_cols=5, seed=101,
                                       update_frac=0.4, update_cols=3)
        for r in runs:
            for c in range(1, 6):
                r["values"][c] = (r["values"][c] % 10_000).astype(np.int32)
        aggs = {"v_c0": "sum", "v_c1": "max", "v_c2": "min",
                "v_c3": "last_non_null_value", "v_c4": "first_value"}
        self._run(tmp_path, runs, aggs, compression="zstd",
                  row_group_rows=16_384, data_page_rows=4_096)

    def test_singleton_retract_bypass(self, tmp_path):
        # singleton groups bypass the merge function (ReducerMergeFunction
        # Wrapper.java:53-73): lone retracts are legal — dropped under
        # drop-delete, served as-is (own RowKind) under keep-delete
        for drop_delete in (True, False):
            rng = np.random.default_rng(94)
            runs = []
            n = 6_000
            for i in range(2):
                key = np.arange(n, dtype=np.int64) * 2 + i  # disjoint
                kind = np.where(rng.random(n) < 0.25, 3, 0).astype(np.int8)
                vals = rng.integers(-1000, 1000, n).astype(np.int32)
                msk = rng.random(n) > 0.3
                runs.append({"key": key,
                             "seq": np.arange(i * n, (i + 1) * n, dtype=np.int64),
                             "kind": kind, "values": [key.copy(), vals],
                             "valid": [np.ones(n, bool), msk]})
            metas = write_runs(runs, str(tmp_path / str(drop_delete)),
                               compression="NONE")
            exp = aggregation_model(runs, ["last_non_null_value", "sum"],
                                    drop_delete=drop_delete)
            with Session(0) as s:
                with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                                   _value_cols(1),
                                   merge_engine="aggregation",
                                   aggregations={"v_c0": "sum"},
                                   drop_delete=drop_delete) as plan:
                    got = _read_all(plan)
            _check(got, exp, ["v_k", "v_c0"])

    def test_retract_rejected(self, tmp_path):
        runs = gen_runs_partial_update(2, 5_000, n_value_cols=2, seed=98)
        # turn some rows of run 1 into DELETEs
        runs[1]["kind"][::7] = 3
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2), merge_engine="aggregation",
                               aggregations={"v_c0": "sum"}) as plan:
                with pytest.raises(RuntimeError, match="aggregation"):
                    _read_all(plan)

    def test_unknown_aggregator_rejected(self, tmp_path):
        runs = gen_runs_partial_update(1, 100, n_value_cols=2, seed=99)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="not on the GPU path"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(2), merge_engine="aggregation",
                              aggregations={"v_c0": "hll_sketch"})
            with pytest.raises(RuntimeError, match="not a value column"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(2), merge_engine="aggregation",
                              aggregations={"nope": "sum"})


class TestAggRemoveRecordOnDelete:
    def test_agg_rrod(self, tmp_path):
        from oracle import aggregation_rrod_model
        rng = np.random.default_rng(102)
        runs = gen_runs_partial_update(4, 20_000, n_value_cols=4, seed=102,
                                       update_frac=0.4, update_cols=2)
        for r in runs:
            n = len(r["key"])
            r["kind"] = np.where(rng.random(n) < 0.2, 3, 0).astype(np.int8)
            for c in range(1, 5):
                r["values"][c] = (r["values"][c] % 10_000).astype(np.int32)
        aggs_map = {"v_c0": "sum", "v_c1": "max", "v_c2": "min",
                    "v_c3": "last_non_null_value"}
        aggs_list = ["last_non_null_value", "sum", "max", "min",
                     "last_non_null_value"]
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        for dd in (True, False):
            exp = aggregation_rrod_model(runs, aggs_list, drop_delete=dd)
            with Session(0) as s:
                with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                                   _value_cols(4),
                                   merge_engine="aggregation",
                                   aggregations=aggs_map, drop_delete=dd,
                                   remove_record_on_delete=True) as plan:
                    got = _read_all(plan)
            assert (got["_KEY_k"] == exp["key"]).all(), dd
            assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all(), dd
            assert (got["_VALUE_KIND"] == exp["kind"]).all(), dd
            names = ["v_k"] + [f"v_c{i}" for i in range(4)]
            for c, nm in enumerate(names):
                ev, em = exp["values"][c], exp["valid"][c]
                gm = got.get(nm + "#valid")
                if gm is None:
                    gm = np.ones(len(got[nm]), dtype=bool)
                assert (gm == em).all(), (dd, nm)
                assert (got[nm][em] == ev[em]).all(), (dd, nm)

    def test_first_agg_rejected_with_rrod(self, tmp_path):
        runs = gen_runs_partial_update(1, 200, n_value_cols=2, seed=103)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="remove-record-on-delete"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(2), merge_engine="aggregation",
                              aggregations={"v_c0": "first_value"},
                              remove_record_on_delete=True)


class TestAggregationRetracts:
    """Retract records through retract-capable aggregators
    (AggregateMergeFunction.add :80-101; FieldSumAgg.retract;
    FieldPrimaryKeyAgg; FieldIgnoreRetractAgg) vs the oracle port of
    getExpectedForAggSum's non-RROD branch (MergeFunctionTestUtils.java:
    99-110)."""

    def _gen(self, n_runs, rows, seed, n_value_cols=4, nulls=True):
        rng = np.random.default_rng(seed)
        total = n_runs * rows
        seqs = rng.permutation(total).astype(np.int64)
        runs = []
        for r in range(n_runs):
            space = max(int(rows * n_runs * 0.6), rows)
            key = np.sort(rng.choice(space, rows,
                                     replace=False)).astype(np.int64)
            kind = rng.choice([0, 0, 0, 2, 1, 3], rows).astype(np.int8)
            vals = [key.copy()]
            msks = [np.ones(rows, bool)]
            for _ in range(n_value_cols):
                vals.append(rng.integers(0, 1000, rows).astype(np.int32))
                msks.append(rng.random(rows) > 0.25 if nulls
                            else np.ones(rows, bool))
            runs.append({"key": key, "seq": seqs[r * rows:(r + 1) * rows],
                         "kind": kind, "values": vals, "valid": msks})
        return runs

    def test_sum_retracts(self, tmp_path):
        from oracle import aggregation_retract_model
        runs = self._gen(5, 8_000, seed=401)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        aggs = ["primary_key", "sum", "sum", "sum", "sum"]
        exp = aggregation_retract_model(runs, aggs, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(4), merge_engine="aggregation",
                               aggregations={"v_k": "primary_key",
                                             "v_c0": "sum", "v_c1": "sum",
                                             "v_c2": "sum", "v_c3": "sum"},
                               ) as plan:
                got = _read_all(plan)
        _check(got, exp, ["v_k", "v_c0", "v_c1", "v_c2", "v_c3"])

    def test_ignore_retract_wrappers(self, tmp_path):
        from oracle import aggregation_retract_model
        runs = self._gen(4, 6_000, seed=402)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        aggs = ["primary_key", "sum", "last_non_null_value", "max",
                "first_value"]
        ign = {2, 3, 4}  # value-list indices of wrapped columns
        exp = aggregation_retract_model(runs, aggs, ignore_retract=ign,
                                        drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(4), merge_engine="aggregation",
                               aggregations={"v_k": "primary_key",
                                             "v_c0": "sum",
                                             "v_c1": "last_non_null_value",
                                             "v_c2": "max",
                                             "v_c3": "first_value"},
                               ignore_retract=["v_c1", "v_c2", "v_c3"],
                               ) as plan:
                got = _read_all(plan)
        _check(got, exp, ["v_k", "v_c0", "v_c1", "v_c2", "v_c3"])

    def test_retract_without_capability_still_rejected(self, tmp_path):
        # default last_non_null (no wrapper) + retract data -> loud error
        runs = self._gen(2, 3_000, seed=403)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(4), merge_engine="aggregation",
                               aggregations={"v_c0": "sum"}) as plan:
                with pytest.raises(RuntimeError, match="aggregation"):
                    _read_all(plan)
