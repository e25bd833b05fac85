#!/usr/bin/env python3
"""C5 bench — BASELINE.json configs[4]: full compaction rewrite on 1 MI355X,
16 -> 1 sorted runs x 20M rows, mixed types incl. decimal(18,2) and a
dictionary-encoded string (cardinality 1k), Parquet read + merge + write
round trip through the CompactRewriter surface (paimon_amd.compact.rewrite).

Inputs are written with the library's native writer (fast path; the on-disk
form is pinned against pyarrow at test scale in tests/test_c5_gpu.py /
test_parquet_write_cpu.py). Runs are generated one at a time to bound host
memory; per-run sequence ranges are disjoint (bucket-unique sequence
numbers, as MergeTreeWriter guarantees).
"""
import argparse
import json
import os
import shutil
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from paimon_amd import Session  # noqa: E402
from paimon_amd.compact import rewrite  # noqa: E402
from paimon_amd.datagen import C5_VALUE_COLS, gen_runs_c5  # noqa: E402
from paimon_amd.reader import write_parquet  # noqa: E402

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--runs", type=int, default=16)
    ap.add_argument("--rows", type=int, default=20_000_000)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--compression", default="NONE")
    ap.add_argument("--data-dir", default=os.path.join(REPO, "data", "c5"))
    args = ap.parse_args()

    t0 = time.perf_counter()
    os.makedirs(args.data_dir, exist_ok=True)
    metas = []
    names = [c["name"] for c in C5_VALUE_COLS]
    for r in range(args.runs):
        path = os.path.join(args.data_dir, f"run-{r}.parquet")
        run = gen_runs_c5(1, args.rows, seed=args.seed + r,
                          str_card=1000)[0]
        run["seq"] = run["seq"] + np.int64(r * args.rows)
        cols = [("_KEY_k", run["key"]), ("_SEQUENCE_NUMBER", run["seq"]),
                ("_VALUE_KIND", run["kind"])]
        for c, nm in enumerate(names):
            cols.append((nm, run["values"][c]))
        write_parquet(path, cols, compression=args.compression,
                      dicts={"v_str": run["str_dict"]},
                      decimals={"v_dec": (18, 2)})
        metas.append({"path": path, "rowCount": len(run["key"]),
                      "minKey": int(run["key"][0]),
                      "maxKey": int(run["key"][-1]), "level": 0})
        del run
    t_gen = time.perf_counter() - t0

    out_dir = os.path.join(args.data_dir, "out")
    rows_in = args.runs * args.rows
    reps = []
    with Session(0) as s:
        for rep in range(args.reps):
            shutil.rmtree(out_dir, ignore_errors=True)
            t1 = time.perf_counter()
            res = rewrite(s, metas, KEY_COLS, C5_VALUE_COLS, out_dir,
                          output_level=5, drop_delete=True,
                          compression=args.compression,
                          target_file_rows=args.rows)
            t2 = time.perf_counter()
            reps.append(t2 - t1)
    after = res["after"]
    rows_out = sum(m["rowCount"] for m in after)
    best = min(reps)
    print(json.dumps({
        "bench": "C5 compaction round trip (BASELINE configs[4])",
        "config": {"runs": args.runs, "rows_per_run": args.rows,
                   "schema": "int64 pk + 4 int32 + decimal(18,2) + "
                             "dictionary string (card 1k)",
                   "compression": args.compression, "seed": args.seed},
        "rows_in": rows_in,
        "rows_out": rows_out,
        "files_out": len(after),
        "gen_seconds": round(t_gen, 2),
        "round_trip_seconds": [round(x, 3) for x in reps],
        "best_rows_per_s": round(rows_in / best, 0),
        "note": "read+merge on GPU through the C-ABI plan; encode on host "
                "(GPU encode is roadmap SURVEY 8f.1); times include file "
                "write-back",
    }))


if __name__ == "__main__":
    main()
