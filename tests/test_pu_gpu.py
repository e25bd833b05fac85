"""GPU parity for the PartialUpdate engine and for null-carrying columns
(def-level decode + dense scatter on device), against the numpy oracle model
(which is itself pinned to pypaimon golden fixtures in test_oracle_merge)."""

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


class TestPartialUpdate:
    def _run(self, tmp_path, runs):
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        exp = partial_update_model(runs)
        n_vals = len(runs[0]["values"])
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(n_vals - 1),
                               merge_engine="partial-update") as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
        assert (got["_VALUE_KIND"] == exp["kind"]).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(n_vals - 1)]
        for c, nm in enumerate(names):
            ev, evalid = exp["values"][c], exp["valid"][c]
            gvalid = got.get(nm + "#valid")
            if gvalid is None:
                gvalid = np.ones(len(got[nm]), dtype=bool)
            assert (gvalid == evalid).all(), nm
            assert (got[nm][evalid] == ev[evalid]).all(), nm

    def test_pu_c3_shape(self, tmp_path):
        runs = gen_runs_partial_update(4, 30_000, n_value_cols=12, seed=71,
                                       update_frac=0.3, update_cols=4)
        self._run(tmp_path, runs)

    def test_pu_heavy_updates(self, tmp_path):
        runs = gen_runs_partial_update(8, 10_000, n_value_cols=6, seed=72,
                                       update_frac=0.9, update_cols=2)
        self._run(tmp_path, runs)

    def test_pu_no_nulls(self, tmp_path):
        runs = gen_runs_partial_update(3, 15_000, n_value_cols=4, seed=73,
                                       update_frac=0.0)
        self._run(tmp_path, runs)

    def test_singleton_retract_bypass(self, tmp_path):
        # ReducerMergeFunctionWrapper.java:53-73: singleton groups bypass the
        # merge function entirely, so a lone retract is legal in PU mode —
        # dropped under drop-delete, served with its own RowKind otherwise
        for drop_delete in (True, False):
            rng = np.random.default_rng(93)
            runs = []
            n = 8_000
            for i in range(3):
                # keys disjoint across runs: every group is a singleton
                key = (np.arange(n, dtype=np.int64) * 3 + i)
                kind = np.where(rng.random(n) < 0.2, 3, 0).astype(np.int8)
                kind[rng.random(n) < 0.05] = 1  # lone UPDATE_BEFOREs too
                vals = rng.integers(-1000, 1000, n).astype(np.int32)
                msk = rng.random(n) > 0.3
                runs.append({"key": key,
                             "seq": np.arange(i * n, (i + 1) * n, dtype=np.int64),
                             "kind": kind, "values": [key.copy(), vals],
                             "valid": [np.ones(n, bool), msk]})
            metas = write_runs(runs, str(tmp_path / str(drop_delete)),
                               compression="NONE")
            exp = partial_update_model(runs, drop_delete=drop_delete)
            with Session(0) as s:
                with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                                   _value_cols(1),
                                   merge_engine="partial-update",
                                   drop_delete=drop_delete) as plan:
                    got = _read_all(plan)
            assert (got["_KEY_k"] == exp["key"]).all()
            assert (got["_VALUE_KIND"] == exp["kind"]).all()
            em = exp["valid"][1]
            gm = got["v_c0#valid"]
            assert (gm == em).all()
            assert (got["v_c0"][em] == exp["values"][1][em]).all()

    def test_pu_rejects_retracts(self, tmp_path):
        runs = gen_runs_dedup(2, 5_000, n_value_cols=2, seed=74,
                              delete_frac=0.2)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2),
                               merge_engine="partial-update") as plan:
                with pytest.raises(RuntimeError, match="retract"):
                    plan.read_next()


class TestDedupWithNulls:
    def test_dedup_nullable_values(self, tmp_path):
        # dedup over runs whose value columns carry nulls: output validity
        # must equal the winning record's validity
        runs = gen_runs_partial_update(4, 20_000, n_value_cols=5, seed=75,
                                       update_frac=0.5, update_cols=2)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        r, w = merge_dedup(runs, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(5)) as plan:
                got = _read_all(plan)
        exp_key = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
        assert (got["_KEY_k"] == exp_key).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(5)]
        for c, nm in enumerate(names):
            ev = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)])
            em = np.array([runs[a]["valid"][c][b] for a, b in zip(r, w)])
            gvalid = got.get(nm + "#valid")
            if gvalid is None:
                gvalid = np.ones(len(got[nm]), dtype=bool)
            assert (gvalid == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm


class TestDictionaryWithNulls:
    def test_dict_encoded_nullable_columns(self, tmp_path):
        # dictionary-encoded value columns that also carry nulls: ids decode
        # to dense positions, dict-gather fills the dense buffer, and the
        # level scatter positions rows (previously an unsupported-path error)
        import os
        import pyarrow.parquet as pq
        from paimon_amd.datagen import run_to_arrow
        runs = gen_runs_partial_update(3, 20_000, n_value_cols=3, seed=77,
                                       update_frac=0.5, update_cols=2)
        for r in runs:  # low cardinality so pyarrow keeps dictionary pages
            for c in range(1, 4):
                r["values"][c] = (r["values"][c] % 50).astype(np.int32)
        metas = []
        for i, r in enumerate(runs):
            tbl = run_to_arrow(r)
            path = os.path.join(str(tmp_path), f"run-{i}.parquet")
            pq.write_table(tbl, path, compression=None,
                           use_dictionary=[f"v_c{c}" for c in range(3)],
                           data_page_version="1.0", store_schema=False,
                           data_page_size=16_384)
            metas.append({"path": path, "rowCount": len(r["key"]),
                          "minKey": int(r["key"][0]),
                          "maxKey": int(r["key"][-1]), "level": 0})
        exp = partial_update_model(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(3),
                               merge_engine="partial-update") as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(3)]
        for c, nm in enumerate(names):
            ev, em = exp["values"][c], exp["valid"][c]
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm


class TestRemoveRecordOnDelete:
    def _run(self, tmp_path, runs, drop_delete=True):
        from oracle import partial_update_rrod_model
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        exp = partial_update_rrod_model(runs, drop_delete=drop_delete)
        n_vals = len(runs[0]["values"])
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(n_vals - 1),
                               merge_engine="partial-update",
                               drop_delete=drop_delete,
                               remove_record_on_delete=True) as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
        assert (got["_VALUE_KIND"] == exp["kind"]).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(n_vals - 1)]
        for c, nm in enumerate(names):
            ev, em = exp["values"][c], exp["valid"][c]
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm

    def test_rrod_basic(self, tmp_path):
        rng = np.random.default_rng(88)
        runs = gen_runs_partial_update(4, 25_000, n_value_cols=5, seed=88,
                                       update_frac=0.4, update_cols=2)
        for r in runs:  # ~15% deletes, fields per the existing masks
            r["kind"] = np.where(rng.random(len(r["key"])) < 0.15, 3,
                                 0).astype(np.int8)
        self._run(tmp_path, runs, drop_delete=True)

    def test_rrod_keep_delete(self, tmp_path):
        rng = np.random.default_rng(89)
        runs = gen_runs_partial_update(3, 15_000, n_value_cols=3, seed=89,
                                       update_frac=0.5, update_cols=2)
        for r in runs:
            r["kind"] = np.where(rng.random(len(r["key"])) < 0.25, 3,
                                 0).astype(np.int8)
        self._run(tmp_path, runs, drop_delete=False)

    def test_rrod_rejects_update_before(self, tmp_path):
        runs = gen_runs_partial_update(2, 4_000, n_value_cols=2, seed=90)
        runs[0]["kind"][::9] = 1  # UPDATE_BEFORE
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2), merge_engine="partial-update",
                               remove_record_on_delete=True) as plan:
                with pytest.raises(RuntimeError, match="UPDATE_BEFORE"):
                    _read_all(plan)

    def test_rrod_conflicts_with_ignore_delete(self, tmp_path):
        runs = gen_runs_partial_update(1, 100, n_value_cols=1, seed=91)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="ignore-delete"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1), merge_engine="partial-update",
                              remove_record_on_delete=True,
                              ignore_delete=True)


class TestCompositeKeyPartialUpdate:
    def test_two_key_cols_with_pu(self, tmp_path):
        # composite keys x member-list engines: the packed comparand drives
        # grouping while the emit returns both key columns
        import os
        import pyarrow as pa
        import pyarrow.parquet as pq
        rng = np.random.default_rng(105)
        runs = []
        metas = []
        seqs = rng.permutation(30_000).astype(np.int64)
        for i in range(3):
            key = np.sort(rng.choice(40_000, 10_000,
                                     replace=False)).astype(np.int64)
            k1 = (key >> 6).astype(np.int32)
            k2 = (key & 63).astype(np.int32)
            n = len(key)
            vals = rng.integers(0, 1000, n).astype(np.int32)
            msk = rng.random(n) > 0.4
            runs.append({"key": key, "seq": seqs[i*n:(i+1)*n],
                         "kind": np.zeros(n, np.int8),
                         "values": [vals], "valid": [msk],
                         "k1": k1, "k2": k2})
            fields = [pa.field("_KEY_k1", pa.int32(), nullable=False),
                      pa.field("_KEY_k2", pa.int32(), nullable=False),
                      pa.field("_SEQUENCE_NUMBER", pa.int64(),
                               nullable=False),
                      pa.field("_VALUE_KIND", pa.int8(), nullable=False),
                      pa.field("v_c0", pa.int32())]
            tbl = pa.Table.from_arrays(
                [pa.array(k1), pa.array(k2), pa.array(runs[i]["seq"]),
                 pa.array(runs[i]["kind"]), pa.array(vals, mask=~msk)],
                schema=pa.schema(fields))
            path = os.path.join(str(tmp_path), f"run-{i}.parquet")
            pq.write_table(tbl, path, compression=None, use_dictionary=False,
                           data_page_version="1.0", store_schema=False)

            def comp(a, b):
                u = ((int(a) + 2**31) << 32) | (int(b) + 2**31)
                u ^= 2**63
                return u - 2**64 if u >= 2**63 else u
            metas.append({"path": path, "rowCount": n,
                          "minKey": comp(k1[0], k2[0]),
                          "maxKey": comp(k1[-1], k2[-1]), "level": 0})
        exp = partial_update_model(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas),
                               [{"name": "_KEY_k1", "type": "int32"},
                                {"name": "_KEY_k2", "type": "int32"}],
                               [{"name": "v_c0", "type": "int32"}],
                               merge_engine="partial-update") as plan:
                got = _read_all(plan)
        # expected key split from the logical int64 keys
        assert (got["_KEY_k1"].astype(np.int64) == (exp["key"] >> 6)).all()
        assert (got["_KEY_k2"].astype(np.int64) == (exp["key"] & 63)).all()
        assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
        gm = got.get("v_c0#valid")
        if gm is None:
            gm = np.ones(len(got["v_c0"]), dtype=bool)
        assert (gm == exp["valid"][0]).all()
        assert (got["v_c0"][gm] == exp["values"][0][gm]).all()


class TestSequenceGroups:
    """PartialUpdate with sequence groups (PartialUpdateMergeFunction.java:
    219-377) against the oracle sequential port — which is itself pinned to
    the reference's PartialUpdateMergeFunctionTest vectors in
    tests/test_seqgroup_cpu.py. Streams carry INSERT/UPDATE_AFTER/
    UPDATE_BEFORE/DELETE records."""

    def _gen(self, n_runs, rows, seed, n_value_cols=6, retract_frac=0.25):
        rng = np.random.default_rng(seed)
        total = n_runs * rows
        seqs = rng.permutation(total).astype(np.int64)
        runs = []
        for r in range(n_runs):
            space = max(int(rows * n_runs * 0.6), rows)
            key = np.sort(rng.choice(space, rows,
                                     replace=False)).astype(np.int64)
            kind = rng.choice([0, 0, 0, 2, 1, 3], rows,
                              p=[.45, .1, .1, .1, .1, .15]).astype(np.int8)
            if retract_frac == 0:
                kind[:] = 0
            vals = [key.copy()]
            msks = [np.ones(rows, bool)]
            for _ in range(n_value_cols):
                vals.append(rng.integers(0, 50, rows).astype(np.int32))
                msks.append(rng.random(rows) > 0.35)
            runs.append({"key": key, "seq": seqs[r * rows:(r + 1) * rows],
                         "kind": kind, "values": vals, "valid": msks})
        return runs

    def _check(self, tmp_path, runs, sgs_idx, sgs_named, drop_delete):
        from oracle import partial_update_seqgroup_model
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        exp = partial_update_seqgroup_model(runs, sgs_idx,
                                            drop_delete=drop_delete)
        n_vals = len(runs[0]["values"])
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(n_vals - 1),
                               merge_engine="partial-update",
                               drop_delete=drop_delete,
                               sequence_groups=sgs_named) as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
        assert (got["_VALUE_KIND"] == exp["kind"]).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(n_vals - 1)]
        for c, nm in enumerate(names):
            ev, em = exp["values"][c], exp["valid"][c]
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm

    # indices into the runs' values list (values[0] = v_k);
    # names as the plan sees them
    SG_IDX = [{"sequence_fields": [1], "group_fields": [2, 3]},
              {"sequence_fields": [4, 5], "group_fields": [6]}]
    SG_NAMED = [{"sequence_fields": ["v_c0"],
                 "group_fields": ["v_c1", "v_c2"]},
                {"sequence_fields": ["v_c3", "v_c4"],
                 "group_fields": ["v_c5"]}]

    def test_seqgroup_with_retracts(self, tmp_path):
        runs = self._gen(5, 8_000, seed=201)
        self._check(tmp_path, runs, self.SG_IDX, self.SG_NAMED, True)

    def test_seqgroup_keep_delete(self, tmp_path):
        runs = self._gen(4, 6_000, seed=202)
        self._check(tmp_path / "kd", runs, self.SG_IDX, self.SG_NAMED,
                    False)

    def test_seqgroup_insert_only(self, tmp_path):
        runs = self._gen(4, 6_000, seed=203, retract_frac=0.0)
        self._check(tmp_path, runs, self.SG_IDX, self.SG_NAMED, True)

    def test_seqgroup_validation(self, tmp_path):
        runs = self._gen(1, 100, seed=204, retract_frac=0.0)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="two sequence groups"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(6),
                              merge_engine="partial-update",
                              sequence_groups=[
                                  {"sequence_fields": ["v_c0"],
                                   "group_fields": ["v_c1"]},
                                  {"sequence_fields": ["v_c1"],
                                   "group_fields": ["v_c2"]}])
