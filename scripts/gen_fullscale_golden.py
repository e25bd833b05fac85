#!/usr/bin/env python3
"""Compute the ORDER-SENSITIVE checksum of the C2 (8x10M) merged output with
the CPU oracle and commit it as a golden fixture. Runs IN-CONTAINER (no GPU,
no /root/reference); tests/test_fullscale_gpu.py re-generates the identical
seeded inputs on the GPU box and compares the product path's checksum —
value-verifying the headline config, not just count-verifying it
(VERDICT r01 "what's weak" item: GPU parity tests topped out at 8x50k)."""
import json
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from oracle import merge_dedup  # noqa: E402
from paimon_amd.datagen import gen_runs_dedup  # noqa: E402

C1 = np.uint64(0x9E3779B97F4A7C15)
C2 = np.uint64(0xC2B2AE3D27D4EB4F)


def checksum(cols):
    """Order-sensitive 64-bit fold: rank-weighted mix of every column."""
    n = len(cols[0])
    idx = np.arange(1, n + 1, dtype=np.uint64)
    acc = np.zeros(n, dtype=np.uint64)
    for c in cols:
        v = c.astype(np.int64).view(np.uint64) if c.dtype != np.uint64 else c
        acc ^= (v * C1) ^ ((acc >> np.uint64(7)) + C2)
        acc = acc * C2 + np.uint64(1)
    with np.errstate(over="ignore"):
        return int((acc * idx).sum(dtype=np.uint64))


def main():
    runs_cfg = [(8, 10_000_000, 42, "c2")]
    out = {}
    for n_runs, rows, seed, tag in runs_cfg:
        runs = gen_runs_dedup(n_runs, rows, n_value_cols=8, seed=seed)
        r, w = merge_dedup(runs, drop_delete=True)
        # index-based gather, vectorized per run
        r = np.asarray(r)
        w = np.asarray(w)

        def gather(getter, dtype):
            out_arr = np.empty(len(r), dtype=dtype)
            for a in range(n_runs):
                m = r == a
                out_arr[m] = getter(runs[a])[w[m]]
            return out_arr
        cols = [gather(lambda rr: rr["key"], np.int64),
                gather(lambda rr: rr["seq"], np.int64),
                gather(lambda rr: rr["kind"], np.int8)]
        for c in range(len(runs[0]["values"])):
            cols.append(gather(lambda rr, c=c: rr["values"][c],
                               runs[0]["values"][c].dtype))
        out[tag] = {"rows": int(len(r)), "checksum": checksum(cols),
                    "config": f"{n_runs}x{rows} seed {seed} dedup "
                              "drop-delete, int64 pk + 8 int32 + v_k"}
    path = os.path.join(REPO, "tests", "golden", "fullscale_checksums.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
