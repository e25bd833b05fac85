"""GPU parity for deletion vectors: rows listed in a file's
DV never reach the merge (ApplyDeletionVectorReader,
io/KeyValueFileReaderFactory.java:139-143). Expected outputs come from the
oracle over numpy-PREFILTERED runs — end-to-end row filtering is verified
independently of the bitmap format (which test_dv_cpu pins)."""

import json
import os

import numpy as np
import pytest

from oracle import merge_dedup, partial_update_model
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import (gen_runs_dedup, gen_runs_partial_update,
                                write_runs)
from scripts.gen_dv_golden import serialize_roaring32, wrap_dv

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


def _attach_dvs(metas, dels, tmp_path):
    """dels: {file_index: sorted positions}. Writes one DV index file."""
    blob = b""
    for fi, pos in dels.items():
        ser = wrap_dv(serialize_roaring32(pos))
        metas[fi]["deletionVector"] = {
            "file": str(tmp_path / "index.dv"),
            "offset": len(blob), "length": len(ser)}
        blob += ser
    (tmp_path / "index.dv").write_bytes(blob)
    return metas


def _filter_runs(runs, dels):
    out = []
    for i, r in enumerate(runs):
        keep = np.ones(len(r["key"]), dtype=bool)
        if i in dels:
            keep[np.array(dels[i], dtype=np.int64)] = False
        fr = {"key": r["key"][keep], "seq": r["seq"][keep],
              "kind": r["kind"][keep],
              "values": [v[keep] for v in r["values"]]}
        if "valid" in r:
            fr["valid"] = [m[keep] for m in r["valid"]]
        out.append(fr)
    return out


class TestDeletionVectors:
    def test_dedup_with_dvs(self, tmp_path):
        rng = np.random.default_rng(301)
        runs = gen_runs_dedup(5, 20_000, n_value_cols=3, seed=301,
                              delete_frac=0.1)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        dels = {0: sorted(rng.choice(20_000, 3_000, replace=False).tolist()),
                2: sorted(rng.choice(20_000, 500, replace=False).tolist()),
                4: list(range(0, 20_000, 7))}
        metas = _attach_dvs(metas, dels, tmp_path)
        fruns = _filter_runs(runs, dels)
        r, w = merge_dedup(fruns, drop_delete=True)
        exp_key = np.array([fruns[a]["key"][b] for a, b in zip(r, w)],
                           np.int64)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(3)) as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp_key).all()
        for c, nm in enumerate(["v_k", "v_c0", "v_c1", "v_c2"]):
            ev = np.array([fruns[a]["values"][c][b] for a, b in zip(r, w)])
            assert (got[nm] == ev).all(), nm

    def test_partial_update_with_dvs(self, tmp_path):
        rng = np.random.default_rng(302)
        runs = gen_runs_partial_update(4, 15_000, n_value_cols=5, seed=302,
                                       update_frac=0.4, update_cols=2)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        dels = {1: sorted(rng.choice(15_000, 2_000, replace=False).tolist()),
                3: [0, 1, 2, 14_999]}
        metas = _attach_dvs(metas, dels, tmp_path)
        exp = partial_update_model(_filter_runs(runs, dels))
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(5),
                               merge_engine="partial-update") as plan:
                got = _read_all(plan)
        assert (got["_KEY_k"] == exp["key"]).all()
        assert (got["_SEQUENCE_NUMBER"] == exp["seq"]).all()
        names = ["v_k"] + [f"v_c{i}" for i in range(5)]
        for c, nm in enumerate(names):
            ev, em = exp["values"][c], exp["valid"][c]
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm

    def test_fully_deleted_file(self, tmp_path):
        runs = gen_runs_dedup(3, 5_000, n_value_cols=1, seed=303,
                              delete_frac=0.0)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        dels = {1: list(range(5_000))}
        metas = _attach_dvs(metas, dels, tmp_path)
        fruns = _filter_runs(runs, dels)
        r, w = merge_dedup(fruns, drop_delete=True)
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                               _value_cols(1)) as plan:
                got = _read_all(plan)
        assert len(got["_KEY_k"]) == len(r)
