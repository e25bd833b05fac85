"""Full-scale value verification of the headline config: the GPU product
path's 8x10M (C2) merged output is checksummed row-by-row (order-sensitive)
against a golden computed in-container by the CPU oracle
(scripts/gen_fullscale_golden.py, committed under tests/golden/) — the
headline shape is value-verified on hardware, not just count-verified."""

import json
import os

import numpy as np
import pytest

from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup, write_runs

pytestmark = pytest.mark.gpu

C1 = np.uint64(0x9E3779B97F4A7C15)
C2 = np.uint64(0xC2B2AE3D27D4EB4F)


def checksum(cols):
    n = len(cols[0])
    idx = np.arange(1, n + 1, dtype=np.uint64)
    acc = np.zeros(n, dtype=np.uint64)
    with np.errstate(over="ignore"):
        for c in cols:
            v = c.astype(np.int64).view(np.uint64)
            acc ^= (v * C1) ^ ((acc >> np.uint64(7)) + C2)
            acc = acc * C2 + np.uint64(1)
        return int((acc * idx).sum(dtype=np.uint64))


def test_c2_fullscale_checksum(tmp_path):
    golden = json.load(open(os.path.join(
        os.path.dirname(__file__), "golden", "fullscale_checksums.json")))
    g = golden["c2"]
    runs = gen_runs_dedup(8, 10_000_000, n_value_cols=8, seed=42)
    metas = write_runs(runs, str(tmp_path), compression="NONE")
    key_cols = [{"name": "_KEY_k", "type": "int64"}]
    value_cols = ([{"name": "v_k", "type": "int64"}] +
                  [{"name": f"v_c{i}", "type": "int32"} for i in range(8)])
    got = {}
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(metas), key_cols,
                           value_cols, output="host") as plan:
            while True:
                b = plan.read_next()
                if b is None:
                    break
                for kk, v in b.items():
                    got.setdefault(kk, []).append(v.copy())
    got = {kk: np.concatenate(v) for kk, v in got.items()}
    assert len(got["_KEY_k"]) == g["rows"]
    cols = [got["_KEY_k"], got["_SEQUENCE_NUMBER"], got["_VALUE_KIND"],
            got["v_k"]] + [got[f"v_c{i}"] for i in range(8)]
    assert checksum(cols) == g["checksum"]
