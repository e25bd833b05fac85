"""Full-compaction changelog producer on the GPU
(FullChangelogMergeFunctionWrapper.java:74-130 through k_merge_tiles'
changelog entries + k_cl_finalize + k_cl_emit) vs the oracle model that is
itself pinned to the reference's test vectors (tests/test_changelog_cpu.py)."""
import numpy as np
import pytest

from oracle import full_changelog_model
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup, write_runs

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]
MAX_LEVEL = 5


def _value_cols(n):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": "int32"} for i in range(n)])


def _read_all(plan):
    main, cl = [], []
    while True:
        b = plan.read_next()
        if b is None:
            break
        main.append({k: v.copy() for k, v in b.items()})
        cl.append({k: v.copy() for k, v in plan.read_changelog().items()})
    cat = lambda parts: {k: np.concatenate([p[k] for p in parts])
                         for k in parts[0]} if parts else {}
    return cat(main), cat(cl)


def _expected(runs, levels, row_dedup=False):
    (cr, cw, ck), (rr, rw) = full_changelog_model(
        runs, levels, MAX_LEVEL, row_dedup=row_dedup)
    exp_cl = {
        "_KEY_k": np.array([runs[a]["key"][b] for a, b in zip(cr, cw)],
                           np.int64),
        "_SEQUENCE_NUMBER": np.array(
            [runs[a]["seq"][b] for a, b in zip(cr, cw)], np.int64),
        "_VALUE_KIND": ck.astype(np.int8),
        "v_k": np.array([runs[a]["values"][0][b] for a, b in zip(cr, cw)]),
    }
    exp_res_keys = np.array([runs[a]["key"][b] for a, b in zip(rr, rw)],
                            np.int64)
    return exp_cl, exp_res_keys


def _run_case(tmp_path, runs, levels, row_dedup=False):
    metas = write_runs(runs, str(tmp_path), compression="NONE")
    for m, lvl in zip(metas, levels):
        m["level"] = lvl
    exp_cl, exp_res_keys = _expected(runs, levels, row_dedup)
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                           _value_cols(len(runs[0]["values"]) - 1),
                           changelog_producer="full-compaction",
                           changelog_row_dedup=row_dedup,
                           max_level=MAX_LEVEL) as plan:
            main, cl = _read_all(plan)
    assert len(main["_KEY_k"]) == len(exp_res_keys)
    assert (main["_KEY_k"] == exp_res_keys).all()
    assert len(cl["_KEY_k"]) == len(exp_cl["_KEY_k"]), \
        (len(cl["_KEY_k"]), len(exp_cl["_KEY_k"]))
    for name, e in exp_cl.items():
        g = cl[name]
        assert (g == e).all(), \
            f"changelog col {name}: {np.flatnonzero(g != e)[:10]}"


def _with_top_run(n_runs, rows, seed, delete_frac=0.2, top_frac=0.6,
                  n_value_cols=3, value_card=None):
    """n_runs level-0 runs + one top-level run holding earlier compaction
    results (INSERT only, lowest sequence numbers)."""
    rng = np.random.default_rng(seed)
    runs = gen_runs_dedup(n_runs, rows, n_value_cols=n_value_cols, seed=seed,
                          delete_frac=delete_frac)
    keyspace = int(max(r["key"].max() for r in runs)) + 1
    n_top = max(1, int(rows * top_frac))
    keys = np.sort(rng.choice(keyspace, n_top, replace=False))
    top = {
        "key": keys.astype(np.int64),
        "seq": np.arange(n_top, dtype=np.int64) - n_top,  # oldest
        "kind": np.zeros(n_top, dtype=np.int8),
        "values": [rng.integers(0, value_card or 1 << 30, n_top
                                ).astype(np.int64)] +
                  [rng.integers(0, value_card or 1 << 30, n_top
                                ).astype(np.int32)
                   for _ in range(n_value_cols)],
    }
    if value_card:
        for r in runs:
            for c in range(len(r["values"])):
                r["values"][c] = (r["values"][c] % value_card).astype(
                    r["values"][c].dtype)
    runs.append(top)
    return runs, [0] * n_runs + [MAX_LEVEL]


class TestFullCompactionChangelog:
    def test_basic(self, tmp_path):
        runs, levels = _with_top_run(4, 30_000, seed=501)
        _run_case(tmp_path, runs, levels)

    def test_no_top_level(self, tmp_path):
        runs = gen_runs_dedup(4, 25_000, n_value_cols=3, seed=502,
                              delete_frac=0.15)
        _run_case(tmp_path, runs, [0, 0, 1, 1])

    def test_row_dedup_suppression(self, tmp_path):
        # tiny value cardinality: many top/merged pairs compare equal
        runs, levels = _with_top_run(3, 20_000, seed=503, delete_frac=0.1,
                                     value_card=2)
        _run_case(tmp_path, runs, levels, row_dedup=True)
        _run_case(tmp_path, runs, levels, row_dedup=False)

    def test_delete_heavy(self, tmp_path):
        runs, levels = _with_top_run(4, 20_000, seed=504, delete_frac=0.6)
        _run_case(tmp_path, runs, levels)

    def test_sixteen_runs(self, tmp_path):
        runs, levels = _with_top_run(15, 8_000, seed=505)
        _run_case(tmp_path, runs, levels)

    def test_duplicate_top_rejected(self, tmp_path):
        runs, levels = _with_top_run(2, 5_000, seed=506)
        # second copy of the top run at max level -> same key in two
        # max-level runs -> checkState error
        runs.append({k: (v.copy() if not isinstance(v, list) else
                         [x.copy() for x in v])
                     for k, v in runs[-1].items()})
        runs[-1]["seq"] = runs[-1]["seq"] - 100_000
        levels = levels + [MAX_LEVEL]
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        for m, lvl in zip(metas, levels):
            m["level"] = lvl
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(len(runs[0]["values"]) - 1),
                               changelog_producer="full-compaction",
                               max_level=MAX_LEVEL) as plan:
                with pytest.raises(RuntimeError, match="Top level"):
                    while plan.read_next() is not None:
                        pass

    def test_missing_max_level_rejected(self, tmp_path):
        runs, levels = _with_top_run(2, 2_000, seed=507)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="max_level"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(len(runs[0]["values"]) - 1),
                              changelog_producer="full-compaction")


class TestChangelogWithDeletionVectors:
    """Cross-feature: DV tombstones remove members BEFORE the wrapper sees
    them (ApplyDeletionVectorReader feeds the merge), so a deleted top-level
    record means "no top" for the changelog decision."""

    def test_dv_filtered_top(self, tmp_path):
        from scripts.gen_dv_golden import serialize_roaring32, wrap_dv
        rng = np.random.default_rng(911)
        runs, levels = _with_top_run(3, 15_000, seed=911, delete_frac=0.15)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        for m, lvl in zip(metas, levels):
            m["level"] = lvl
        # delete ~20% of the TOP run's rows and ~10% of run 0's
        dels = {
            len(runs) - 1: sorted(rng.choice(
                len(runs[-1]["key"]),
                len(runs[-1]["key"]) // 5, replace=False).tolist()),
            0: sorted(rng.choice(
                len(runs[0]["key"]),
                len(runs[0]["key"]) // 10, replace=False).tolist()),
        }
        blob = b""
        for fi, pos in dels.items():
            ser = wrap_dv(serialize_roaring32(pos))
            metas[fi]["deletionVector"] = {
                "file": str(tmp_path / "index.dv"),
                "offset": len(blob), "length": len(ser)}
            blob += ser
        (tmp_path / "index.dv").write_bytes(blob)
        # expected: run the oracle over the FILTERED runs
        fruns = []
        for i, r in enumerate(runs):
            keep = np.ones(len(r["key"]), dtype=bool)
            if i in dels:
                keep[np.array(dels[i], dtype=np.int64)] = False
            fruns.append({"key": r["key"][keep], "seq": r["seq"][keep],
                          "kind": r["kind"][keep],
                          "values": [v[keep] for v in r["values"]]})
        exp_cl, exp_res_keys = _expected(fruns, levels)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(len(runs[0]["values"]) - 1),
                               changelog_producer="full-compaction",
                               max_level=MAX_LEVEL) as plan:
                main, cl = _read_all(plan)
        assert (main["_KEY_k"] == exp_res_keys).all()
        assert len(cl["_KEY_k"]) == len(exp_cl["_KEY_k"])
        for name, e in exp_cl.items():
            assert (cl[name] == e).all(), name


class TestRowDedupWithNulls:
    """changelog-producer.row-deduplicate over NULLABLE value columns: the
    equaliser treats null==null as equal and null!=value as different
    (k_cl_finalize compares validity bytes before values)."""

    def test_nullable_equality(self, tmp_path):
        import pyarrow as pa
        import pyarrow.parquet as pq
        rng = np.random.default_rng(921)
        runs, levels = _with_top_run(2, 12_000, seed=921, delete_frac=0.05,
                                     value_card=3)
        masks = []
        for r in runs:
            m = [np.zeros(len(r["key"]), bool)]  # v_k non-null
            for _ in range(len(r["values"]) - 1):
                m.append(rng.random(len(r["key"])) < 0.4)
            masks.append(m)
        metas = []
        for i, r in enumerate(runs):
            cols = {"_KEY_k": pa.array(r["key"]),
                    "_SEQUENCE_NUMBER": pa.array(r["seq"]),
                    "_VALUE_KIND": pa.array(r["kind"]),
                    "v_k": pa.array(r["values"][0])}
            for c in range(1, len(r["values"])):
                cols[f"v_c{c-1}"] = pa.array(r["values"][c],
                                             mask=masks[i][c])
            tbl = pa.table(cols)
            path = str(tmp_path / f"run-{i}.parquet")
            pq.write_table(tbl, path, compression=None,
                           use_dictionary=False, data_page_version="1.0",
                           store_schema=False)
            metas.append({"path": path, "rowCount": len(r["key"]),
                          "minKey": int(r["key"][0]),
                          "maxKey": int(r["key"][-1]),
                          "level": levels[i]})
        (cr, cw, ck), (rr2, rw2) = full_changelog_model(
            runs, levels, MAX_LEVEL, row_dedup=True, masks=masks)
        exp_keys = np.array([runs[a]["key"][b] for a, b in zip(cr, cw)],
                            np.int64)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(len(runs[0]["values"]) - 1),
                               changelog_producer="full-compaction",
                               changelog_row_dedup=True,
                               max_level=MAX_LEVEL) as plan:
                main, cl = _read_all(plan)
        assert len(cl["_KEY_k"]) == len(exp_keys), \
            (len(cl["_KEY_k"]), len(exp_keys))
        assert (cl["_KEY_k"] == exp_keys).all()
        assert (cl["_VALUE_KIND"] == ck).all()
