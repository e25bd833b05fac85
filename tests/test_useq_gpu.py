"""GPU parity for user-defined sequence fields (sequence.field;
utils/UserDefinedSeqComparator.java:38-80, wired at
MergeFileSplitRead.java:543-545): the merge order becomes ascending
(key, sequence fields..., sequenceNumber, isAdd), nulls first."""

import numpy as np
import pytest

from oracle import merge_dedup_useq_model
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup, write_runs

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def _value_cols(n):
    return ([{"name": "v_k", "type": "int64"}] +
            [{"name": f"v_c{i}", "type": "int32"} for i in range(n)])


def _run(tmp_path, runs, useq_idx, useq_names, engine="deduplicate",
         drop_delete=True, ignore_delete=False):
    metas = write_runs(runs, str(tmp_path), compression="NONE")
    r, w = merge_dedup_useq_model(runs, useq_idx, drop_delete=drop_delete,
                                  ignore_delete=ignore_delete,
                                  first_row=engine == "first-row")
    names = ["v_k"] + [f"v_c{i}" for i in range(len(runs[0]["values"]) - 1)]
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                           _value_cols(len(names) - 1),
                           merge_engine=engine, drop_delete=drop_delete,
                           ignore_delete=ignore_delete,
                           sequence_fields=useq_names) as plan:
            got = {}
            while True:
                b = plan.read_next()
                if b is None:
                    break
                for kk, v in b.items():
                    got.setdefault(kk, []).append(v.copy())
            got = {kk: np.concatenate(v) for kk, v in got.items()}
    exp_key = np.array([runs[a]["key"][b] for a, b in zip(r, w)], np.int64)
    assert (got["_KEY_k"] == exp_key).all()
    exp_seq = np.array([runs[a]["seq"][b] for a, b in zip(r, w)], np.int64)
    assert (got["_SEQUENCE_NUMBER"] == exp_seq).all()
    for c, nm in enumerate(names):
        ev = np.array([runs[a]["values"][c][b] for a, b in zip(r, w)])
        if "valid" in runs[0]:
            em = np.array([runs[a]["valid"][c][b] for a, b in zip(r, w)])
            gm = got.get(nm + "#valid")
            if gm is None:
                gm = np.ones(len(got[nm]), dtype=bool)
            assert (gm == em).all(), nm
            assert (got[nm][em] == ev[em]).all(), nm
        else:
            assert (got[nm] == ev).all(), nm


class TestSequenceFields:
    def _gen(self, n_runs, rows, seed, low_card=True, nulls=False):
        rng = np.random.default_rng(seed)
        runs = gen_runs_dedup(n_runs, rows, n_value_cols=3, seed=seed,
                              delete_frac=0.1)
        for r in runs:
            n = len(r["key"])
            # low-cardinality sequence fields force real tie-breaks down
            # to (seq, isAdd)
            r["values"][1] = rng.integers(0, 4 if low_card else 10**6,
                                          n).astype(np.int32)
            r["values"][2] = rng.integers(0, 3, n).astype(np.int32)
            if nulls:
                r["valid"] = [np.ones(n, bool),
                              rng.random(n) > 0.3,
                              rng.random(n) > 0.3,
                              np.ones(n, bool)]
        return runs

    def test_single_sequence_field(self, tmp_path):
        runs = self._gen(5, 15_000, seed=601)
        _run(tmp_path, runs, [1], ["v_c0"])

    def test_two_sequence_fields_with_nulls(self, tmp_path):
        runs = self._gen(4, 10_000, seed=602, nulls=True)
        _run(tmp_path, runs, [1, 2], ["v_c0", "v_c1"])

    def test_first_row_with_sequence_field(self, tmp_path):
        runs = self._gen(3, 8_000, seed=603)
        for r in runs:
            r["kind"][:] = 0  # first-row rejects retracts
        _run(tmp_path, runs, [1], ["v_c0"], engine="first-row")

    def test_keep_delete(self, tmp_path):
        runs = self._gen(4, 8_000, seed=604)
        _run(tmp_path, runs, [1], ["v_c0"], drop_delete=False)

    def test_pu_with_sequence_field_rejected(self, tmp_path):
        runs = self._gen(1, 500, seed=605)
        metas = write_runs(runs, str(tmp_path), compression="NONE")
        with Session(0) as s:
            with pytest.raises(RuntimeError, match="later round"):
                MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                              _value_cols(3),
                              merge_engine="partial-update",
                              sequence_fields=["v_c0"])
