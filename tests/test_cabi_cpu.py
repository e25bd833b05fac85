"""CPU-side tests of libpaimon_hip.so: the library loads, exports every
symbol include/paimon_hip.h declares, the native thrift footer parser
matches pyarrow metadata, and the IntervalPartition restatement matches the
reference semantics (IntervalPartition.java:67-125). No GPU calls."""

import ctypes
import os
import re
import subprocess

import numpy as np
import pyarrow.parquet as pq
import pytest

from paimon_amd import LIB_PATH, debug_footer, interval_partition
from paimon_amd.datagen import gen_runs_dedup, write_runs

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module", autouse=True)
def built():
    if not os.path.exists(LIB_PATH):
        subprocess.run(["make", "-C", os.path.join(REPO, "paimon_amd", "csrc")],
                       check=True, capture_output=True)


def test_all_header_symbols_exported():
    hdr = open(os.path.join(REPO, "include", "paimon_hip.h")).read()
    declared = re.findall(r"\b(pmh_[a-z_0-9]+)\s*\(", hdr)
    declared = sorted(set(declared))
    assert declared, "no symbols found in header"
    lib = ctypes.CDLL(LIB_PATH)
    for sym in declared:
        assert hasattr(lib, sym), f"symbol {sym} missing from libpaimon_hip.so"


def test_footer_parser_vs_pyarrow(tmp_path):
    runs = gen_runs_dedup(2, 4000, n_value_cols=3, seed=21)
    metas = write_runs(runs, str(tmp_path), compression="NONE")
    for m in metas:
        got = debug_footer(m["path"])
        md = pq.ParquetFile(m["path"]).metadata
        assert got["num_rows"] == md.num_rows
        assert [c["name"] for c in got["columns"]] == \
            [md.schema.column(i).name for i in range(md.num_columns)]
        assert len(got["row_groups"]) == md.num_row_groups
        for g, rg in enumerate(got["row_groups"]):
            assert rg["num_rows"] == md.row_group(g).num_rows
            for c, ch in enumerate(rg["chunks"]):
                ref = md.row_group(g).column(c)
                assert ch["num_values"] == ref.num_values
                assert ch["data_page_offset"] == ref.data_page_offset


def test_footer_parser_zstd(tmp_path):
    runs = gen_runs_dedup(1, 1000, n_value_cols=1, seed=22)
    metas = write_runs(runs, str(tmp_path), compression="zstd")
    got = debug_footer(metas[0]["path"])
    md = pq.ParquetFile(metas[0]["path"]).metadata
    assert got["num_rows"] == md.num_rows
    assert got["row_groups"][0]["chunks"][0]["codec"] == 6  # ZSTD


class TestIntervalPartition:
    # semantics from IntervalPartition.java:67-125 and IntervalPartitionTest
    def test_non_overlapping_one_run_each_section(self):
        # disjoint files -> each its own section (single run)
        sec, run, ns = interval_partition([0, 10, 20], [5, 15, 25])
        assert ns == 3
        assert sec == [0, 1, 2]
        assert run == [0, 0, 0]

    def test_overlapping_single_section(self):
        sec, run, ns = interval_partition([0, 3, 4], [5, 8, 10])
        assert ns == 1
        assert sec == [0, 0, 0]
        # 3 mutually overlapping intervals cannot share runs pairwise:
        # [0,5] vs [3,8] overlap, [3,8] vs [4,10] overlap; but [0,5] and
        # [4,10] overlap too -> 3 runs? greedy: f2(min 3) <= max 5 -> new run;
        # f3(min 4) <= max 5 and <= 8 -> new run => 3 runs
        assert sorted(run) == [0, 1, 2]

    def test_chain_packs_into_runs(self):
        # [0,5],[6,10] chain into one run; [3,8] overlaps both -> own run
        sec, run, ns = interval_partition([0, 6, 3], [5, 10, 8])
        assert ns == 1
        assert run[0] == run[1]  # chained
        assert run[2] != run[0]

    def test_touching_bounds_overlap(self):
        # minKey == bound is NOT greater -> same section (compare > 0 rule)
        sec, run, ns = interval_partition([0, 5], [5, 9])
        assert ns == 1
        # equal boundary keys overlap -> two runs
        assert run[0] != run[1]

    def test_matches_oracle_model(self):
        # randomized cross-check against a python restatement
        rng = np.random.default_rng(17)
        for _ in range(50):
            n = int(rng.integers(1, 20))
            mins = rng.integers(0, 1000, n)
            lens = rng.integers(0, 100, n)
            maxs = mins + lens
            sec, run, ns = interval_partition(mins.tolist(), maxs.tolist())
            # model: sort by (min,max); section break when min > bound
            order = np.lexsort((maxs, mins))
            bound = None
            exp_sec = {}
            cur = -1
            import heapq
            heap = []  # (last_max, run_id) per run of current section
            next_run = 0
            for i in order:
                if bound is not None and mins[i] > bound:
                    cur += 1
                    heap = []
                    next_run = 0
                    bound = None
                if cur < 0:
                    cur = 0
                if not heap:
                    heapq.heappush(heap, (maxs[i], next_run))
                    exp_sec[i] = (cur, next_run)
                    next_run += 1
                else:
                    last_max, rid = heapq.heappop(heap)
                    if mins[i] > last_max:
                        heapq.heappush(heap, (maxs[i], rid))
                        exp_sec[i] = (cur, rid)
                    else:
                        heapq.heappush(heap, (last_max, rid))
                        heapq.heappush(heap, (maxs[i], next_run))
                        exp_sec[i] = (cur, next_run)
                        next_run += 1
                bound = maxs[i] if bound is None or maxs[i] > bound else bound
            # compare section ids and run-set sizes (run numbering may differ)
            for i in range(n):
                assert sec[i] == exp_sec[i][0], (mins.tolist(), maxs.tolist())
            # same number of runs per section
            for s in set(sec):
                got_runs = len({run[i] for i in range(n) if sec[i] == s})
                exp_runs = len({exp_sec[i][1] for i in range(n)
                                if exp_sec[i][0] == s})
                assert got_runs == exp_runs
