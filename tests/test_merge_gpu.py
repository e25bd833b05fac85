"""GPU parity tests: the full product path (pmh_plan_create + pmh_read_next
over real Parquet files) against the CPU oracle, on a real MI355X.

These tests call through the C-ABI only; /root/reference is NOT read (the
oracle golden fixtures are committed)."""

import os

import numpy as np
import pytest

from oracle import merge_dedup
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup, write_runs

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def _value_cols(n):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": "int32"} for i in range(n)])


def _expected_dedup(runs, drop_delete=True, ignore_delete=False):
    r, w = merge_dedup(runs, ignore_delete=ignore_delete,
                       drop_delete=drop_delete)
    exp = {
        "_KEY_k": np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64),
        "_SEQUENCE_NUMBER": np.array([runs[a]["seq"][b] for a, b in zip(r, w)],
                                     np.int64),
        "_VALUE_KIND": np.array([runs[a]["kind"][b] for a, b in zip(r, w)],
                                np.int8),
    }
    n_vals = len(runs[0]["values"])
    names = ["v_k"] + [f"v_c{i}" for i in range(n_vals - 1)]
    for c, nm in enumerate(names):
        exp[nm] = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)])
    return exp


def _run_and_compare(tmp_path, runs, compression="NONE", drop_delete=True,
                     ignore_delete=False):
    metas = write_runs(runs, str(tmp_path), compression=compression)
    exp = _expected_dedup(runs, drop_delete, ignore_delete)
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                           _value_cols(len(runs[0]["values"]) - 1),
                           drop_delete=drop_delete,
                           ignore_delete=ignore_delete) as plan:
            got = {}
            while True:
                b = plan.read_next()
                if b is None:
                    break
                for k, v in b.items():
                    got.setdefault(k, []).append(v.copy())
            got = {k: np.concatenate(v) for k, v in got.items()}
    assert len(got["_KEY_k"]) == len(exp["_KEY_k"]), \
        (len(got["_KEY_k"]), len(exp["_KEY_k"]))
    for name, e in exp.items():
        g = got[name]
        assert (g == e).all(), f"column {name} mismatch: " \
            f"{np.flatnonzero(g != e)[:10]}"


class TestDedupParity:
    def test_small_2x100k(self, tmp_path):
        # C1 plumbing parity shape: 2 runs x 100k rows, int64 PK + 4 int32
        runs = gen_runs_dedup(2, 100_000, n_value_cols=4, seed=42)
        _run_and_compare(tmp_path, runs)

    def test_8_runs_collisions(self, tmp_path):
        runs = gen_runs_dedup(8, 50_000, n_value_cols=8, seed=43,
                              delete_frac=0.1)
        _run_and_compare(tmp_path, runs)

    def test_keep_delete(self, tmp_path):
        runs = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=44,
                              delete_frac=0.3)
        _run_and_compare(tmp_path, runs, drop_delete=False)

    def test_ignore_delete(self, tmp_path):
        runs = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=45,
                              delete_frac=0.3)
        _run_and_compare(tmp_path, runs, ignore_delete=True)

    def test_single_run(self, tmp_path):
        runs = gen_runs_dedup(1, 30_000, n_value_cols=2, seed=46)
        _run_and_compare(tmp_path, runs)

    def test_sixteen_runs(self, tmp_path):
        runs = gen_runs_dedup(16, 8_000, n_value_cols=2, seed=47)
        _run_and_compare(tmp_path, runs)

    def test_twentyfour_runs(self, tmp_path):
        # beyond the reference's default spill threshold shapes: the widened
        # run:5|row:27 winner packing takes sections up to 32 runs
        runs = gen_runs_dedup(24, 5_000, n_value_cols=2, seed=147,
                              delete_frac=0.1)
        _run_and_compare(tmp_path, runs)

    def test_thirtytwo_runs(self, tmp_path):
        runs = gen_runs_dedup(32, 3_000, n_value_cols=2, seed=148)
        _run_and_compare(tmp_path, runs)

    def test_heavy_collision_tiny_keyspace(self, tmp_path):
        # many groups span tile boundaries relative to key space
        rng = np.random.default_rng(48)
        runs = []
        total = 8 * 5000
        seqs = rng.permutation(total).astype(np.int64)
        for r in range(8):
            keys = np.sort(rng.choice(6000, size=5000, replace=False)).astype(np.int64)
            runs.append({
                "key": keys,
                "seq": seqs[r * 5000:(r + 1) * 5000],
                "kind": rng.choice([0, 3], 5000, p=[.85, .15]).astype(np.int8),
                "values": [keys.copy(),
                           rng.integers(-2**31, 2**31, 5000).astype(np.int32)],
            })
        _run_and_compare(tmp_path, runs)

    def test_zstd_compressed(self, tmp_path):
        runs = gen_runs_dedup(4, 30_000, n_value_cols=4, seed=49)
        _run_and_compare(tmp_path, runs, compression="zstd")

    def test_non_overlapping_sections(self, tmp_path):
        # two disjoint key ranges -> two sections -> two batches
        rng = np.random.default_rng(50)
        runs = []
        seqs = rng.permutation(40_000).astype(np.int64)
        for r in range(2):
            keys = np.sort(rng.choice(30_000, 10_000, replace=False)).astype(np.int64)
            base = 0 if r == 0 else 1_000_000  # disjoint ranges
            keys = keys + base
            runs.append({"key": keys, "seq": seqs[r*10_000:(r+1)*10_000],
                         "kind": np.zeros(10_000, np.int8),
                         "values": [keys.copy()]})
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        exp = _expected_dedup(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_k", "type": "int64"}]) as plan:
                batches = []
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    batches.append({k: v.copy() for k, v in b.items()})
        assert len(batches) == 2  # one batch per section
        got_keys = np.concatenate([b["_KEY_k"] for b in batches])
        assert (got_keys == exp["_KEY_k"]).all()

    def test_dictionary_encoded_values(self, tmp_path):
        # force dictionary encoding on value columns
        import pyarrow.parquet as pq
        from paimon_amd.datagen import run_to_arrow
        rng = np.random.default_rng(51)
        runs = []
        seqs = rng.permutation(60_000).astype(np.int64)
        import os
        metas = []
        for r in range(3):
            keys = np.sort(rng.choice(60_000, 20_000, replace=False)).astype(np.int64)
            run = {"key": keys, "seq": seqs[r*20_000:(r+1)*20_000],
                   "kind": np.zeros(20_000, np.int8),
                   "values": [keys.copy(),
                              rng.integers(0, 100, 20_000).astype(np.int32)]}
            runs.append(run)
            tbl = run_to_arrow(run)
            path = os.path.join(str(tmp_path), f"run-{r}.parquet")
            pq.write_table(tbl, path, compression=None,
                           use_dictionary=["v_c0"], data_page_version="1.0",
                           store_schema=False)
            metas.append({"path": path, "rowCount": 20_000,
                          "minKey": int(keys[0]), "maxKey": int(keys[-1]),
                          "level": 0})
        exp = _expected_dedup(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(1)) as plan:
                got = plan.read_next()
                assert (got["_KEY_k"] == exp["_KEY_k"]).all()
                assert (got["v_c0"] == exp["v_c0"]).all()

    def test_stats_populated(self, tmp_path):
        runs = gen_runs_dedup(2, 10_000, n_value_cols=1, seed=52)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(1)) as plan:
                plan.read_next()
                st = plan.stats()
                assert st["rows_in"] == 20_000
                assert st["rows_out"] > 0
                assert st["total_device_ms"] > 0


class TestMultiFileRuns:
    def test_files_chain_into_runs(self, tmp_path):
        # disjoint-range files chain into one SortedRun (IntervalPartition
        # min-heap packing); an overlapping file forms its own run. The
        # staged run is the concatenation of its files (SortedRun.fromSorted).
        import pyarrow.parquet as pq
        from paimon_amd.datagen import run_to_arrow
        rng = np.random.default_rng(60)
        seqs = rng.permutation(10_000).astype(np.int64)

        def mk(keys, seq):
            keys = np.sort(np.asarray(keys, np.int64))
            return {"key": keys, "seq": seq[:len(keys)],
                    "kind": np.zeros(len(keys), np.int8),
                    "values": [keys.copy()]}

        fa = mk(rng.choice(1000, 800, replace=False), seqs[:800])
        fb = mk(rng.choice(1000, 800, replace=False) + 2000, seqs[800:1600])
        fc = mk(rng.choice(2500, 1500, replace=False) + 400, seqs[1600:3100])
        metas = []
        for name, f in (("a", fa), ("b", fb), ("c", fc)):
            path = str(tmp_path / f"{name}.parquet")
            pq.write_table(run_to_arrow(f), path, compression=None,
                           use_dictionary=False, data_page_version="1.0",
                           store_schema=False)
            metas.append({"path": path, "rowCount": len(f["key"]),
                          "minKey": int(f["key"][0]),
                          "maxKey": int(f["key"][-1]), "level": 0})
        # oracle: run1 = concat(A, B) (chained), run2 = C
        runs = [{k: np.concatenate([fa[k], fb[k]]) for k in ("key", "seq", "kind")},
                {k: fc[k] for k in ("key", "seq", "kind")}]
        runs[0]["values"] = [np.concatenate([fa["values"][0], fb["values"][0]])]
        runs[1]["values"] = [fc["values"][0]]
        r, w = merge_dedup(runs, drop_delete=True)
        exp_key = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
        exp_seq = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_k", "type": "int64"}]) as plan:
                got = {}
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    for kk, v in b.items():
                        got.setdefault(kk, []).append(v.copy())
                got = {kk: np.concatenate(v) for kk, v in got.items()}
        assert (got["_KEY_k"] == exp_key).all()
        assert (got["_SEQUENCE_NUMBER"] == exp_seq).all()


class TestScaleAndTypes:
    def test_size_representative_8x1M(self, tmp_path):
        # 1/10th of the C2 shape, bit-exact against the oracle
        runs = gen_runs_dedup(8, 1_000_000, n_value_cols=4, seed=1042,
                              delete_frac=0.05)
        _run_and_compare(tmp_path, runs)

    def test_float_double_columns(self, tmp_path):
        import pyarrow as pa
        import pyarrow.parquet as pq
        rng = np.random.default_rng(53)
        n = 30_000
        runs = []
        metas = []
        seqs = rng.permutation(3 * n).astype(np.int64)
        for r in range(3):
            keys = np.sort(rng.choice(3 * n, n, replace=False)).astype(np.int64)
            f32 = rng.standard_normal(n).astype(np.float32)
            f64 = rng.standard_normal(n)
            runs.append({"key": keys, "seq": seqs[r * n:(r + 1) * n],
                         "kind": np.zeros(n, np.int8),
                         "values": [keys.copy(), f32, f64]})
            fields = [pa.field("_KEY_k", pa.int64(), nullable=False),
                      pa.field("_SEQUENCE_NUMBER", pa.int64(), nullable=False),
                      pa.field("_VALUE_KIND", pa.int8(), nullable=False),
                      pa.field("v_k", pa.int64()),
                      pa.field("f", pa.float32()),
                      pa.field("d", pa.float64())]
            tbl = pa.Table.from_arrays(
                [pa.array(keys), pa.array(runs[r]["seq"]),
                 pa.array(runs[r]["kind"]), pa.array(keys), pa.array(f32),
                 pa.array(f64)], schema=pa.schema(fields))
            path = str(tmp_path / f"run-{r}.parquet")
            pq.write_table(tbl, path, compression=None, use_dictionary=False,
                           data_page_version="1.0", store_schema=False)
            metas.append({"path": path, "rowCount": n,
                          "minKey": int(keys[0]), "maxKey": int(keys[-1]),
                          "level": 0})
        r_, w_ = merge_dedup(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_k", "type": "int64"},
                                {"name": "f", "type": "float32"},
                                {"name": "d", "type": "float64"}]) as plan:
                got = plan.read_next()
        # floats pass through the gather bit-exactly
        ef = np.array([runs[a]["values"][1][b] for a, b in zip(r_, w_)])
        ed = np.array([runs[a]["values"][2][b] for a, b in zip(r_, w_)])
        assert got["f"].dtype == np.float32
        assert (got["f"].view(np.int32) == ef.view(np.int32)).all()
        assert (got["d"].view(np.int64) == ed.view(np.int64)).all()

    def test_multi_row_group_files(self, tmp_path):
        runs = gen_runs_dedup(4, 50_000, n_value_cols=3, seed=54)
        metas = write_runs(runs, str(tmp_path), compression="NONE",
                           row_group_rows=8_192)
        exp = _expected_dedup(runs)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(3)) as plan:
                got = _read_all_batches(plan)
        assert (got["_KEY_k"] == exp["_KEY_k"]).all()
        assert (got["v_c0"] == exp["v_c0"]).all()


def _read_all_batches(plan):
    got = {}
    while True:
        b = plan.read_next()
        if b is None:
            break
        for k, v in b.items():
            got.setdefault(k, []).append(v.copy())
    return {k: np.concatenate(v) for k, v in got.items()}


class TestConcurrentPlans:
    def test_two_plans_interleaved(self, tmp_path):
        # one plan per bucket, interleaved batches (the C-ABI threading
        # contract: one plan = one HIP stream; concurrent plans allowed)
        runs_a = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=61)
        runs_b = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=62)
        metas_a = write_runs(runs_a, str(tmp_path / "a"), compression="NONE")
        metas_b = write_runs(runs_b, str(tmp_path / "b"), compression="NONE")
        ra, wa = merge_dedup(runs_a)
        rb, wb = merge_dedup(runs_b)
        with Session(0) as s:
            pa_ = MergeReadPlan(s, file_descs_from_metas(metas_a), KEY_COLS,
                                _value_cols(2))
            pb_ = MergeReadPlan(s, file_descs_from_metas(metas_b), KEY_COLS,
                                _value_cols(2))
            ga = pa_.read_next()
            gb = pb_.read_next()
            ea = np.array([runs_a[x]["key"][y] for x, y in zip(ra, wa)])
            eb = np.array([runs_b[x]["key"][y] for x, y in zip(rb, wb)])
            assert (ga["_KEY_k"] == ea).all()
            assert (gb["_KEY_k"] == eb).all()
            pa_.close()
            pb_.close()


class TestFirstRowEngine:
    def _run(self, tmp_path, runs, file_format="parquet", **kw):
        from oracle import merge_first_row_model
        metas = write_runs(runs, str(tmp_path), compression="NONE",
                           file_format=file_format)
        r, w = merge_first_row_model(runs, **kw)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(len(runs[0]["values"]) - 1),
                               merge_engine="first-row", **kw) as plan:
                got = _read_all_batches(plan)
        ek = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
        es = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
        assert (got["_KEY_k"] == ek).all()
        assert (got["_SEQUENCE_NUMBER"] == es).all()

    def test_first_row_insert_only(self, tmp_path):
        runs = gen_runs_dedup(6, 25_000, n_value_cols=2, seed=63,
                              delete_frac=0.0)
        self._run(tmp_path, runs)

    def test_first_row_ignore_delete(self, tmp_path):
        runs = gen_runs_dedup(5, 15_000, n_value_cols=2, seed=64,
                              delete_frac=0.25)
        self._run(tmp_path, runs, ignore_delete=True)

    def test_first_row_orc(self, tmp_path):
        runs = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=66,
                              delete_frac=0.0)
        self._run(tmp_path, runs, file_format="orc")

    def test_first_row_rejects_retracts(self, tmp_path):
        runs = gen_runs_dedup(3, 8_000, n_value_cols=1, seed=65,
                              delete_frac=0.3)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(1),
                               merge_engine="first-row") as plan:
                with pytest.raises(RuntimeError, match="first-row"):
                    plan.read_next()


class TestNarrowIntColumns:
    def test_int8_int16_columns(self, tmp_path):
        # TINYINT/SMALLINT travel as INT32 physical and truncate on emit
        runs = gen_runs_dedup(4, 20_000, n_value_cols=2, seed=67,
                              delete_frac=0.1)
        for r in runs:
            r["values"][1] = (r["values"][1] % 127).astype(np.int8)
            r["values"][2] = (r["values"][2] % 32_000).astype(np.int16)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        r, w = merge_dedup(runs, drop_delete=True)
        vcols = [{"name": "v_k", "type": "int64"},
                 {"name": "v_c0", "type": "int8"},
                 {"name": "v_c1", "type": "int16"}]
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               vcols) as plan:
                got = _read_all_batches(plan)
        for nm, c in (("v_c0", 1), ("v_c1", 2)):
            e = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)])
            assert got[nm].dtype == e.dtype, nm
            assert (got[nm] == e).all(), nm

    def test_gzip_compressed(self, tmp_path):
        runs = gen_runs_dedup(4, 25_000, n_value_cols=3, seed=54)
        _run_and_compare(tmp_path, runs, compression="gzip")

    def test_snappy_compressed(self, tmp_path):
        runs = gen_runs_dedup(4, 25_000, n_value_cols=3, seed=55)
        _run_and_compare(tmp_path, runs, compression="snappy")


class TestErrorPaths:
    # everything outside the matrix fails loudly at plan create with a
    # descriptive error so a Java-side provider can fall
    # back to the stock reader per split
    def test_too_many_runs_unsupported_engine(self, tmp_path):
        # >32 runs merge hierarchically for deduplicate/first-row
        # (TestHierarchicalSections); engines whose fold is NOT associative
        # still reject (the reference spills to disk, MergeSorter.java)
        runs = gen_runs_dedup(33, 500, n_value_cols=1, seed=70,
                              delete_frac=0.0)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="32"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1), merge_engine="partial-update")

    def test_run_row_overflow_rejected(self, tmp_path):
        # a run is the concatenation of its files: two non-overlapping files
        # whose TOTAL claims >= 2^27 rows must be rejected before the packed
        # (run | row) winner format could overflow (the oversized files are
        # never opened — the guard fires on the declared row counts; only
        # the tiny bridge run, which stages first, exists on disk)
        from paimon_amd.reader import write_parquet
        keys = np.arange(5, 26, dtype=np.int64)
        write_parquet(str(tmp_path / "c.parquet"), [
            ("_KEY_k", keys), ("_SEQUENCE_NUMBER", keys),
            ("_VALUE_KIND", np.zeros(len(keys), np.int8)),
            ("v_k", keys), ("v_c0", np.zeros(len(keys), np.int32))])
        metas = [{"path": str(tmp_path / "a.parquet"), "rowCount": 70_000_000,
                  "minKey": 0, "maxKey": 10, "level": 0},
                 {"path": str(tmp_path / "b.parquet"), "rowCount": 70_000_000,
                  "minKey": 20, "maxKey": 30, "level": 0},
                 # bridge file: overlaps both, forcing one section where a+b
                 # concatenate into a single run
                 {"path": str(tmp_path / "c.parquet"), "rowCount": 21,
                  "minKey": 5, "maxKey": 25, "level": 0}]
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="per-run limit"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1))

    def test_missing_column(self, tmp_path):
        runs = gen_runs_dedup(2, 500, n_value_cols=1, seed=71)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="not found"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              [{"name": "nope", "type": "int32"}])

    def test_unsupported_parquet_codec(self, tmp_path):
        runs = gen_runs_dedup(2, 500, n_value_cols=1, seed=72)
        metas = write_runs(runs, str(tmp_path), compression="lz4")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="codec"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1))

    def test_unsupported_orc_codec(self, tmp_path):
        runs = gen_runs_dedup(2, 500, n_value_cols=1, seed=73)
        metas = write_runs(runs, str(tmp_path), compression="lz4",
                           file_format="orc")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="compression kind"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1))

    def test_unknown_merge_engine(self, tmp_path):
        runs = gen_runs_dedup(1, 100, n_value_cols=1, seed=74)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="merge engine"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1), merge_engine="lookup")

    def test_int32_key_column(self, tmp_path):
        # integer key widths: TINYINT..INT keys stage as INT32; the
        # partition/merge key loads are width-aware
        runs = gen_runs_dedup(4, 30_000, n_value_cols=2, seed=75,
                              delete_frac=0.1)
        for r in runs:
            r["key"] = r["key"].astype(np.int32)
            r["values"][0] = r["values"][0].astype(np.int32)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        oruns = [{**r, "key": r["key"].astype(np.int64)} for r in runs]
        r, w = merge_dedup(oruns, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas),
                               [{"name": "_KEY_k", "type": "int32"}],
                               [{"name": "v_k", "type": "int32"},
                                {"name": "v_c0", "type": "int32"},
                                {"name": "v_c1", "type": "int32"}]) as plan:
                got = _read_all_batches(plan)
        ek = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int32)
        es = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
        assert got["_KEY_k"].dtype == np.int32
        assert (got["_KEY_k"] == ek).all()
        assert (got["_SEQUENCE_NUMBER"] == es).all()

    def test_int16_key_column_orc(self, tmp_path):
        rng = np.random.default_rng(76)
        runs = []
        seqs = rng.permutation(40_000).astype(np.int64)
        for i in range(4):
            keys = np.sort(rng.choice(20_000, 10_000,
                                      replace=False)).astype(np.int16)
            runs.append({
                "key": keys, "seq": seqs[i * 10_000:(i + 1) * 10_000],
                "kind": np.zeros(10_000, np.int8),
                "values": [keys.astype(np.int32)]})
        metas = write_runs(runs, str(tmp_path), compression="NONE",
                           file_format="orc")
        oruns = [{**r, "key": r["key"].astype(np.int64)} for r in runs]
        r, w = merge_dedup(oruns, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas),
                               [{"name": "_KEY_k", "type": "int16"}],
                               [{"name": "v_k", "type": "int32"}]) as plan:
                got = _read_all_batches(plan)
        ek = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int16)
        assert got["_KEY_k"].dtype == np.int16
        assert (got["_KEY_k"] == ek).all()


class TestCompositeKeys:
    # multi-column integer PKs: the partition/merge comparand is an
    # order-preserving pack of the biased sub-keys (<= 64 bits total);
    # minKey/maxKey in the plan carry the same encoding
    @staticmethod
    def _comp(k1, k2):
        u = ((int(k1) + 2**31) << 32) | (int(k2) + 2**31)
        u ^= 2**63
        return u - 2**64 if u >= 2**63 else u

    def test_two_int32_key_columns(self, tmp_path):
        import os
        import pyarrow as pa
        import pyarrow.parquet as pq
        rng = np.random.default_rng(78)
        runs = []
        metas = []
        seqs = rng.permutation(60_000).astype(np.int64)
        for i in range(3):
            key = np.sort(rng.choice(90_000, 20_000,
                                     replace=False)).astype(np.int64)
            k1 = (key >> 8).astype(np.int32) - 1000   # negatives included
            k2 = (key & 0xFF).astype(np.int32)
            vals = rng.integers(-2**31, 2**31, 20_000).astype(np.int32)
            runs.append({"key": key, "seq": seqs[i*20_000:(i+1)*20_000],
                         "kind": np.zeros(20_000, np.int8),
                         "values": [vals], "k1": k1, "k2": k2})
            fields = [pa.field("_KEY_k1", pa.int32(), nullable=False),
                      pa.field("_KEY_k2", pa.int32(), nullable=False),
                      pa.field("_SEQUENCE_NUMBER", pa.int64(),
                               nullable=False),
                      pa.field("_VALUE_KIND", pa.int8(), nullable=False),
                      pa.field("v_c0", pa.int32())]
            tbl = pa.Table.from_arrays(
                [pa.array(k1), pa.array(k2), pa.array(runs[i]["seq"]),
                 pa.array(runs[i]["kind"]), pa.array(vals)],
                schema=pa.schema(fields))
            path = os.path.join(str(tmp_path), f"run-{i}.parquet")
            pq.write_table(tbl, path, compression=None, use_dictionary=False,
                           data_page_version="1.0", store_schema=False)
            metas.append({"path": path, "rowCount": 20_000,
                          "minKey": self._comp(k1[0], k2[0]),
                          "maxKey": self._comp(k1[-1], k2[-1]), "level": 0})
        r, w = merge_dedup(runs, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas),
                               [{"name": "_KEY_k1", "type": "int32"},
                                {"name": "_KEY_k2", "type": "int32"}],
                               [{"name": "v_c0", "type": "int32"}]) as plan:
                got = _read_all_batches(plan)
        e1 = np.array([runs[a]["k1"][b] for a, b in zip(r, w)], np.int32)
        e2 = np.array([runs[a]["k2"][b] for a, b in zip(r, w)], np.int32)
        es = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
        ev = np.array([runs[a]["values"][0][b] for a, b in zip(r, w)],
                      np.int32)
        assert (got["_KEY_k1"] == e1).all()
        assert (got["_KEY_k2"] == e2).all()
        assert (got["_SEQUENCE_NUMBER"] == es).all()
        assert (got["v_c0"] == ev).all()

    def test_key_bits_overflow_rejected(self, tmp_path):
        runs = gen_runs_dedup(1, 100, n_value_cols=1, seed=79)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="64"):
                MergeReadPlan(s, file_descs_from_metas(metas),
                              [{"name": "_KEY_k", "type": "int64"},
                               {"name": "x", "type": "int32"}],
                              _value_cols(1))


class TestProjections:
    """Column projection (MergeFileSplitRead.java:485-540 pushdown read
    type): the plan's value_cols ARE the projection — the reader stages and
    emits only the requested columns, in the requested order."""

    def test_value_projection_subset(self, tmp_path):
        runs = gen_runs_dedup(4, 20_000, n_value_cols=6, seed=171,
                              delete_frac=0.1)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        exp = _expected_dedup(runs)
        # read only two of the seven value columns, reordered
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               [{"name": "v_c3", "type": "int32"},
                                {"name": "v_c0", "type": "int32"}]) as plan:
                got = {}
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    for kk, v in b.items():
                        got.setdefault(kk, []).append(v.copy())
                got = {kk: np.concatenate(v) for kk, v in got.items()}
        assert (got["_KEY_k"] == exp["_KEY_k"]).all()
        assert (got["v_c3"] == exp["v_c3"]).all()
        assert (got["v_c0"] == exp["v_c0"]).all()
        assert "v_c1" not in got and "v_k" not in got
        assert list(got)[:2] != []  # column order follows the read type


class TestFusedPathAB:
    """The fused single-pass merge path (k_merge_emit [+ k_emit_dense]) is
    non-default for plain dedup/first-row since the same-box A/B found the
    classic chain faster — it remains the sequence-fields
    path and must stay parity-green. PMH_FUSED=1 selects it explicitly."""

    def _with_env(self, tmp_path, env, seed):
        runs = gen_runs_dedup(6, 60_000, n_value_cols=4, seed=seed,
                              delete_frac=0.15)
        saved = {k: os.environ.get(k) for k in env}
        os.environ.update(env)
        try:
            _run_and_compare(tmp_path, runs)
        finally:
            for k, v in saved.items():
                if v is None:
                    os.environ.pop(k, None)
                else:
                    os.environ[k] = v

    def test_fused_split_pair(self, tmp_path):
        self._with_env(tmp_path, {"PMH_FUSED": "1"}, seed=311)

    def test_fused_inkernel_emit(self, tmp_path):
        self._with_env(tmp_path, {"PMH_FUSED": "1", "PMH_FSPLIT": "0"},
                       seed=312)

    def test_default_is_legacy_chain(self, tmp_path):
        # path_mode 0 = chain, 1 = fused in-kernel, 2 = fused split pair
        runs = gen_runs_dedup(3, 30_000, n_value_cols=2, seed=313)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        assert "PMH_FUSED" not in os.environ
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2)) as plan:
                while plan.read_next() is not None:
                    pass
                assert plan.stats().get("path_mode", 0) == 0


class TestSortEngineOption:
    """sort-engine = min-heap (SortMergeReader.java:41-57): both host
    algorithms have identical merge semantics; the GPU merge-path replaces
    them — the option is accepted and results are identical."""

    def test_min_heap_identical(self, tmp_path):
        runs = gen_runs_dedup(4, 40_000, n_value_cols=3, seed=601,
                              delete_frac=0.2)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        outs = []
        for se in ("loser-tree", "min-heap"):
            with Session(0) as s:
                with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                                   _value_cols(3), sort_engine=se) as plan:
                    ks = []
                    while True:
                        b = plan.read_next()
                        if b is None:
                            break
                        ks.append(np.concatenate(
                            [b["_KEY_k"].copy(), b["_SEQUENCE_NUMBER"].copy()]))
                    outs.append(np.concatenate(ks))
        assert (outs[0] == outs[1]).all()

    def test_unknown_engine_rejected(self, tmp_path):
        runs = gen_runs_dedup(1, 1_000, n_value_cols=1, seed=602)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="sort-engine"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(1), sort_engine="quick-sort")


class TestHierarchicalSections:
    """> PMH_MAX_RUNS (32) overlapping runs: batches of runs merge into
    VIRTUAL runs first (winner-of-winners — the deduplicate/first-row fold
    is associative; deletes survive the batch pass and drop only at the
    final chain), then the normal chain merges the virtual runs. The
    reference spills to disk here (MergeSorter.spillMergeSort)."""

    def test_40_runs_dedup(self, tmp_path):
        runs = gen_runs_dedup(40, 4_000, n_value_cols=3, seed=901,
                              delete_frac=0.25)
        _run_and_compare(tmp_path, runs)

    def test_64_runs_dedup(self, tmp_path):
        runs = gen_runs_dedup(64, 1_500, n_value_cols=2, seed=902,
                              delete_frac=0.3)
        _run_and_compare(tmp_path, runs)

    def test_33_runs_keep_delete(self, tmp_path):
        runs = gen_runs_dedup(33, 2_000, n_value_cols=2, seed=903,
                              delete_frac=0.4)
        _run_and_compare(tmp_path, runs, drop_delete=False)

    def test_first_row_40_runs(self, tmp_path):
        from oracle import merge_first_row_model
        runs = gen_runs_dedup(40, 2_000, n_value_cols=2, seed=904,
                              delete_frac=0.0)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        r, w = merge_first_row_model(runs)
        exp = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
        exp_s = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2),
                               merge_engine="first-row") as plan:
                got_k, got_s = [], []
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    got_k.append(b["_KEY_k"].copy())
                    got_s.append(b["_SEQUENCE_NUMBER"].copy())
        assert (np.concatenate(got_k) == exp).all()
        assert (np.concatenate(got_s) == exp_s).all()

    def test_hier_with_nullable_columns(self, tmp_path):
        # nulls flow through the virtual runs' validity bytes
        import pyarrow as pa
        import pyarrow.parquet as pq
        rng = np.random.default_rng(905)
        runs = gen_runs_dedup(36, 3_000, n_value_cols=2, seed=905,
                              delete_frac=0.1)
        masks = [rng.random(len(r["key"])) < 0.3 for r in runs]
        metas = []
        for i, (r, m) in enumerate(zip(runs, masks)):
            tbl = pa.table({
                "_KEY_k": pa.array(r["key"]),
                "_SEQUENCE_NUMBER": pa.array(r["seq"]),
                "_VALUE_KIND": pa.array(r["kind"]),
                "v_k": pa.array(r["values"][0]),
                "v_c0": pa.array(r["values"][1], mask=m),
                "v_c1": pa.array(r["values"][2]),
            })
            path = str(tmp_path / f"run-{i}.parquet")
            pq.write_table(tbl, path, compression=None,
                           use_dictionary=False, data_page_version="1.0",
                           store_schema=False)
            metas.append({"path": path, "rowCount": len(r["key"]),
                          "minKey": int(r["key"][0]),
                          "maxKey": int(r["key"][-1]), "level": 0})
        rr, ww = merge_dedup(runs)
        exp_k = np.array([runs[a]["key"][b] for a, b in zip(rr, ww)],
                         np.int64)
        exp_v = np.array([runs[a]["values"][1][b] for a, b in zip(rr, ww)])
        exp_null = np.array([masks[a][b] for a, b in zip(rr, ww)], bool)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2)) as plan:
                got = {}
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    for kk, v in b.items():
                        got.setdefault(kk, []).append(v.copy())
                got = {kk: np.concatenate(v) for kk, v in got.items()}
        assert (got["_KEY_k"] == exp_k).all()
        assert (got["v_c0#valid"] == ~exp_null).all()
        live = ~exp_null
        assert (got["v_c0"][live] == exp_v[live]).all()


class TestHierarchicalCrosses:
    """Hierarchical sections x other features: deletion vectors consume in
    the batch pass (rtombs sub-array), zstd inputs decode through
    k_zstd_pages before the batch merges."""

    def test_hier_with_deletion_vectors(self, tmp_path):
        from scripts.gen_dv_golden import serialize_roaring32, wrap_dv
        rng = np.random.default_rng(991)
        runs = gen_runs_dedup(36, 3_000, n_value_cols=2, seed=991,
                              delete_frac=0.2)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        dels = {}
        for fi in (0, 7, 35):
            n = len(runs[fi]["key"])
            dels[fi] = sorted(rng.choice(n, n // 6, replace=False).tolist())
        blob = b""
        for fi, pos in dels.items():
            ser = wrap_dv(serialize_roaring32(pos))
            metas[fi]["deletionVector"] = {
                "file": str(tmp_path / "idx.dv"),
                "offset": len(blob), "length": len(ser)}
            blob += ser
        (tmp_path / "idx.dv").write_bytes(blob)
        fruns = []
        for i, r in enumerate(runs):
            keep = np.ones(len(r["key"]), dtype=bool)
            if i in dels:
                keep[np.array(dels[i], dtype=np.int64)] = False
            fruns.append({"key": r["key"][keep], "seq": r["seq"][keep],
                          "kind": r["kind"][keep],
                          "values": [v[keep] for v in r["values"]]})
        rr, ww = merge_dedup(fruns)
        exp = np.array([fruns[a]["key"][b] for a, b in zip(rr, ww)],
                       np.int64)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(2)) as plan:
                got = []
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    got.append(b["_KEY_k"].copy())
        assert (np.concatenate(got) == exp).all()

    def test_hier_with_zstd_inputs(self, tmp_path):
        runs = gen_runs_dedup(34, 4_000, n_value_cols=2, seed=992,
                              delete_frac=0.15)
        _run_and_compare(tmp_path, runs, compression="zstd")
