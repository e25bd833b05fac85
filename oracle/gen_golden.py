"""Generate committed golden vectors under tests/golden/ by running pypaimon
(the reference's own Python implementation) on seeded synthetic runs.

Run in the build container only (needs /root/reference):
    python3 -m oracle.gen_golden
The fixtures travel with the repo; GPU-box tests read ONLY the fixtures.
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oracle.pypaimon_ref import merge_with_pypaimon  # noqa: E402
from paimon_amd.datagen import gen_runs_dedup, gen_runs_partial_update  # noqa: E402

GOLDEN_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests", "golden")


def _save_runs(d, prefix, runs):
    d[f"{prefix}_n_runs"] = np.array(len(runs))
    for i, r in enumerate(runs):
        d[f"{prefix}_key_{i}"] = r["key"]
        d[f"{prefix}_seq_{i}"] = r["seq"]
        d[f"{prefix}_kind_{i}"] = r["kind"]
        if "values" in r:
            d[f"{prefix}_nvals_{i}"] = np.array(len(r["values"]))
            for c, v in enumerate(r["values"]):
                d[f"{prefix}_val_{i}_{c}"] = v
                if "valid" in r:
                    d[f"{prefix}_msk_{i}_{c}"] = r["valid"][c]
    return d


def main():
    os.makedirs(GOLDEN_DIR, exist_ok=True)

    # Case A: dedup with deletes, 5 runs x 2000 rows, drop-delete both ways
    runs = gen_runs_dedup(5, 2000, n_value_cols=2, seed=7, delete_frac=0.15)
    d = {}
    _save_runs(d, "in", runs)
    for dd in (True, False):
        out = merge_with_pypaimon(runs, "deduplicate", drop_delete=dd)
        d[f"dedup_dd{int(dd)}_key"] = np.array([o[0] for o in out], np.int64)
        d[f"dedup_dd{int(dd)}_seq"] = np.array([o[1] for o in out], np.int64)
        d[f"dedup_dd{int(dd)}_kind"] = np.array([o[2] for o in out], np.int8)
        vals = np.array([[v if v is not None else -2**31 for v in o[3]]
                         for o in out], np.int64)
        d[f"dedup_dd{int(dd)}_vals"] = vals
    np.savez_compressed(os.path.join(GOLDEN_DIR, "dedup_5x2000.npz"), **d)
    print("dedup golden:", len(d["dedup_dd1_key"]), "and",
          len(d["dedup_dd0_key"]), "records")

    # Case B: partial-update, 4 runs x 1500 rows, 8 cols, INSERT-only
    runs = gen_runs_partial_update(4, 1500, n_value_cols=8, seed=11,
                                   update_frac=0.5, update_cols=3)
    d = {}
    _save_runs(d, "in", runs)
    out = merge_with_pypaimon(runs, "partial-update", drop_delete=True)
    d["pu_key"] = np.array([o[0] for o in out], np.int64)
    d["pu_seq"] = np.array([o[1] for o in out], np.int64)
    d["pu_kind"] = np.array([o[2] for o in out], np.int8)
    nv = len(out[0][3])
    vals = np.full((len(out), nv), -2**31, np.int64)
    msk = np.zeros((len(out), nv), bool)
    for i, o in enumerate(out):
        for c, v in enumerate(o[3]):
            if v is not None:
                vals[i, c] = v
                msk[i, c] = True
    d["pu_vals"] = vals
    d["pu_valid"] = msk
    np.savez_compressed(os.path.join(GOLDEN_DIR, "partial_update_4x1500.npz"), **d)
    print("partial-update golden:", len(out), "records")

    # Case C: tiny edge cases incl. empty runs and heavy collisions
    rng = np.random.default_rng(3)
    runs = []
    seqs = rng.permutation(600).astype(np.int64)
    off = 0
    for n in (0, 1, 200, 37, 0, 150):
        keys = np.sort(rng.choice(120, size=min(n, 120), replace=False)).astype(np.int64)
        n2 = len(keys)
        runs.append({"key": keys, "seq": seqs[off:off + n2],
                     "kind": rng.choice([0, 3], n2, p=[.7, .3]).astype(np.int8),
                     "values": [keys.copy()]})
        off += n2
    d = {}
    _save_runs(d, "in", runs)
    for dd in (True, False):
        out = merge_with_pypaimon(runs, "deduplicate", drop_delete=dd)
        d[f"dedup_dd{int(dd)}_key"] = np.array([o[0] for o in out], np.int64)
        d[f"dedup_dd{int(dd)}_seq"] = np.array([o[1] for o in out], np.int64)
        d[f"dedup_dd{int(dd)}_kind"] = np.array([o[2] for o in out], np.int8)
    np.savez_compressed(os.path.join(GOLDEN_DIR, "dedup_edge.npz"), **d)
    print("edge golden:", len(d["dedup_dd1_key"]), "records")

    # Case D: first-row, 5 runs x 1200 rows, INSERT-only (the pypaimon
    # FirstRowMergeFunction raises on retracts, like the Java reference)
    runs = gen_runs_dedup(5, 1200, n_value_cols=2, seed=23, delete_frac=0.0)
    d = {}
    _save_runs(d, "in", runs)
    out = merge_with_pypaimon(runs, "first-row", drop_delete=True)
    d["fr_key"] = np.array([o[0] for o in out], np.int64)
    d["fr_seq"] = np.array([o[1] for o in out], np.int64)
    d["fr_kind"] = np.array([o[2] for o in out], np.int8)
    d["fr_vals"] = np.array([o[3] for o in out], np.int64)
    np.savez_compressed(os.path.join(GOLDEN_DIR, "first_row_5x1200.npz"), **d)
    print("first-row golden:", len(out), "records")


if __name__ == "__main__":
    main()
