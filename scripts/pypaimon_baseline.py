#!/usr/bin/env python3
"""Time pypaimon — the reference's own Python implementation — on a bounded
sample of the C2 workload. CONTAINER-ONLY (/root/reference required; it does
not travel to the GPU box, so this leg is measured here and committed to
profiles/pypaimon_baseline.json; the bench's live cpu_baseline leg is the C
restatement). BASELINE.md CPU-baseline leg 1."""
import json
import os
import platform
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from oracle import pypaimon_ref  # noqa: E402
from paimon_amd.datagen import gen_runs_dedup  # noqa: E402


def main():
    if not pypaimon_ref.available():
        print("pypaimon reference not available", file=sys.stderr)
        return 1
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 50_000
    runs = gen_runs_dedup(8, rows, n_value_cols=1, seed=42)
    n = sum(len(r["key"]) for r in runs)
    t0 = time.perf_counter()
    out = pypaimon_ref.merge_with_pypaimon(runs, "deduplicate")
    dt = time.perf_counter() - t0
    res = {
        "impl": "pypaimon SortMergeReaderWithMinHeap + "
                "DeduplicateMergeFunction "
                "(paimon-python/pypaimon/read/reader/sort_merge_reader.py)",
        "rows_in": n,
        "rows_out": len(out[0]) if isinstance(out, tuple) else len(out),
        "seconds": round(dt, 3),
        "rows_per_s": round(n / dt, 0),
        "cores": 1,
        "host": f"in-container {platform.processor() or 'x86_64'} "
                f"({os.cpu_count()} cores) — NOT the GPU box; "
                "/root/reference does not travel (SURVEY.md §8c)",
        "workload": f"8 runs x {rows} rows prefix of the C2 shape, seed 42",
    }
    path = os.path.join(REPO, "profiles", "pypaimon_baseline.json")
    with open(path, "w") as f:
        json.dump(res, f, indent=1)
    print(json.dumps(res))
    return 0


if __name__ == "__main__":
    sys.exit(main())
