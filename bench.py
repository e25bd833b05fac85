#!/usr/bin/env python3
"""Benchmark: Paimon merge-on-read hot path on MI355X (BASELINE.json metric:
merged rows/sec at 8-run x 10M-row merge-on-read, Parquet, Deduplicate).

A "step" is one full merge-on-read pass (partition + tiled k-way merge +
dedup + emit) over one bucket's sorted runs, with the encoded column chunks
already resident in HBM (staged once at plan creation, untimed; the
PCIe-inclusive staging rate is reported separately and noted in DESIGN.md).

Workload (default = BASELINE.json configs[1], the single-GPU metric config):
8 sorted runs x 10M rows, int64 PK + 8 int32 value columns, Parquet v1
pages, PLAIN (dictionary off), uncompressed, Deduplicate merge with
drop-delete — synthetic seeded data with ~50% cross-run key collisions.

Multi-GPU: buckets shard one-per-GPU (weak scaling, no collective —
SURVEY.md §8e); each rank merges its own bucket; value aggregates all ranks.

cpu_baseline: the CPU oracle (faithful single-thread C restatement of
SortMergeReaderWithLoserTree + Deduplicate, oracle/merge_oracle.c) timed on
this box's host cores over a bounded prefix sample of the same workload —
a reported baseline, not the optimization target.
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

HBM_PEAK_GBS = 8000.0  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def pmc_traffic(dom, args):
    """Measured HBM bytes/launch for the dominant kernel, from the committed
    PMC calibration (profiles/pmc_traffic.json — rocprofv3 --pmc FETCH_SIZE /
    WRITE_SIZE passes over this same command, corrections per
    MI355X_MICROARCH.md §HBM; scripts/pmc_traffic.sh regenerates it). Only
    valid at the calibrated workload; None otherwise."""
    try:
        with open(os.path.join(REPO, "profiles", "pmc_traffic.json")) as f:
            cal = json.load(f)
    except (OSError, ValueError):
        return None
    w = cal.get("workload", {})
    if (w.get("runs") == args.runs and w.get("rows") == args.rows
            and w.get("vals") == args.vals
            and w.get("engine") == args.engine
            and w.get("format") == args.file_format
            and w.get("compression") == args.compression):
        ks = cal.get("kernels", {})
        if dom == "merge_emit+emit_dense":
            a = ks.get("merge_emit")
            b = ks.get("emit_dense")
            if a and b:
                return a["bytes_per_step"] + b["bytes_per_step"]
            return None
        e = ks.get(dom)
        if e:
            return e.get("bytes_per_step", e.get("bytes_per_launch"))
    return None


def get_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--runs", type=int, default=8)
    ap.add_argument("--rows", type=int, default=10_000_000)
    ap.add_argument("--vals", type=int, default=8)
    ap.add_argument("--compression", default="NONE")
    ap.add_argument("--format", dest="file_format", default="parquet",
                    choices=["parquet", "orc"])
    ap.add_argument("--engine", default="deduplicate",
                    choices=["deduplicate", "partial-update", "aggregation",
                             "first-row"],
                    help="supplementary merge-engine variants; the default "
                    "metric line is deduplicate per BASELINE.json configs[1]")
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--data-dir", default=os.path.join(REPO, "data"))
    ap.add_argument("--cpu-baseline-rows", type=int, default=2_000_000,
                    help="prefix rows per run for the CPU-oracle baseline; "
                    "0 disables")
    return ap.parse_args()


def ensure_data(args, rank):
    from paimon_amd.datagen import (gen_runs_dedup, gen_runs_partial_update,
                                    write_runs)
    pu = args.engine in ("partial-update", "aggregation")
    tagname = {"partial-update": "c3pu", "aggregation": "c3pu",
               "first-row": "c2fr"}.get(args.engine, "c2")
    tag = (f"{tagname}_{args.runs}x{args.rows}v{args.vals}_"
           f"{args.file_format}_{args.compression}_seed{args.seed}_rank{rank}")
    out_dir = os.path.join(args.data_dir, tag)
    manifest = os.path.join(out_dir, "files.json")
    if os.path.exists(manifest):
        with open(manifest) as f:
            return json.load(f), out_dir
    if pu:
        runs = gen_runs_partial_update(args.runs, args.rows,
                                       n_value_cols=args.vals,
                                       seed=args.seed + rank,
                                       update_frac=0.3, update_cols=6)
    elif args.engine == "first-row":
        # first-row rejects retracts (FirstRowMergeFunction.java:49-59)
        runs = gen_runs_dedup(args.runs, args.rows, n_value_cols=args.vals,
                              seed=args.seed + rank, delete_frac=0.0)
    else:
        runs = gen_runs_dedup(args.runs, args.rows, n_value_cols=args.vals,
                              seed=args.seed + rank)
    metas = write_runs(runs, out_dir, compression=args.compression,
                       file_format=args.file_format)
    return metas, out_dir


def cpu_baseline(args, metas_dir):
    """Time the C oracle loser-tree dedup on a bounded prefix sample:
    N-thread (key-space-sliced loser trees, BASELINE.md leg 2) headline,
    single-thread rate in the sample note. The pypaimon leg (BASELINE.md
    leg 1) cannot run on the GPU box (/root/reference does not travel);
    its in-container measurement is committed in
    profiles/pypaimon_baseline.json."""
    from oracle import merge_dedup, merge_dedup_count_mt
    from paimon_amd.datagen import gen_runs_dedup
    rows = min(args.cpu_baseline_rows, args.rows)
    runs = gen_runs_dedup(args.runs, args.rows, n_value_cols=0,
                          seed=args.seed)
    sample = [{"key": r["key"][:rows], "seq": r["seq"][:rows],
               "kind": r["kind"][:rows]} for r in runs]
    n = sum(len(r["key"]) for r in sample)
    t0 = time.perf_counter()
    merge_dedup(sample, drop_delete=True)
    dt1 = time.perf_counter() - t0
    cores = min(os.cpu_count() or 1, 64)  # the C slicer caps at 64 threads
    t0 = time.perf_counter()
    merge_dedup_count_mt(sample, cores)
    dtn = time.perf_counter() - t0
    return {
        "value": n / dtn,
        "unit": "rows/s",
        "cores": cores,
        "kind": "port",
        "sample": (f"{args.runs} runs x {rows}-row prefixes of the same "
                   f"workload ({n} rows; {cores}-thread key-space-sliced C "
                   f"loser-tree restatement {dtn:.2f}s; single-thread "
                   f"{n / dt1 / 1e6:.1f} M rows/s in {dt1:.2f}s)"),
    }


def main():
    args = get_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = world if world > 1 else args.gpus
    if world == 1 and args.gpus > 1:
        # first-class launcher: re-exec under torch.distributed.run so a
        # direct `bench.py --gpus N` measures N ranks (the driver's own
        # torchrun invocation takes the world>1 path and never gets here)
        import subprocess
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr=127.0.0.1",
               "--master-port=29517", os.path.abspath(__file__)] +               sys.argv[1:]
        print(f"[bench] relaunching {args.gpus} ranks via "
              "torch.distributed.run", file=sys.stderr)
        raise SystemExit(subprocess.call(cmd))

    import torch
    dist = None
    if world > 1:
        import torch.distributed as tdist
        tdist.init_process_group("nccl")
        dist = tdist
        torch.cuda.set_device(local_rank)

    from paimon_amd import Session, MergeReadPlan, file_descs_from_metas

    t_gen0 = time.perf_counter()
    metas, out_dir = ensure_data(args, rank)
    t_gen = time.perf_counter() - t_gen0

    sess = Session(local_rank)
    key_cols = [{"name": "_KEY_k", "type": "int64"}]
    value_cols = ([{"name": "v_k", "type": "int64"}] +
                  [{"name": f"v_c{i}", "type": "int32"}
                   for i in range(args.vals)])
    aggs = None
    if args.engine == "aggregation":
        cycle = ["sum", "max", "min", "last_non_null_value"]
        aggs = {f"v_c{i}": cycle[i % 4] for i in range(args.vals)}
    t_stage0 = time.perf_counter()
    plan = MergeReadPlan(sess, file_descs_from_metas(metas), key_cols,
                         value_cols, merge_engine=args.engine,
                         drop_delete=True, output="device",
                         aggregations=aggs)
    t_stage = time.perf_counter() - t_stage0

    def one_step():
        plan.reset()
        rows = 0
        while True:
            b = plan.read_next()
            if b is None:
                break
            rows += b.n_rows
        return rows

    # warmup
    rows_out = 0
    for _ in range(args.warmup):
        rows_out = one_step()
    stats0 = plan.stats()

    if dist:
        dist.barrier()
    torch.cuda.synchronize(local_rank)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        rows_out = one_step()
    torch.cuda.synchronize(local_rank)
    if dist:
        dist.barrier()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=f"cuda:{local_rank}")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    stats1 = plan.stats()
    path_mode = int(stats1.pop("path_mode", 0))
    stats0.pop("path_mode", None)
    d = {k: stats1[k] - stats0[k] for k in stats1}
    K = args.steps
    rows_in = args.runs * args.rows  # per step per rank, by construction
    U = rows_out

    # per-step kernel times (HIP events accumulated in libpaimon_hip)
    kms = {k: d[k] / K for k in
           ("decode_ms", "partition_ms", "merge_ms", "scan_ms", "emit_ms",
            "total_device_ms")}

    # whole-pipeline algorithmic denominator: encoded input +
    # merged output
    in_bytes = rows_in * (8 + 8 + 4 + 8 + args.vals * 4)
    out_bytes = U * (8 + 8 + 1 + 8 + args.vals * 4)

    # algorithmic bytes per kernel launch (DESIGN.md "Measurement").
    # Fused path (emit_ms == 0: k_merge_emit does merge + emission in one
    # launch): reads N*(key 8 + seq 8 + kind-as-stored 4) for the merge
    # staging + N*(8 + vals*4) value-column staging, writes the merged
    # output — i.e. exactly the whole-pipeline algorithmic bytes.
    # Legacy 3-kernel path: per-kernel models as in round 1.
    n_tiles = (rows_in + 3583) // 3584
    if path_mode == 1:
        # fused in-kernel emission: one kernel moves the whole pipeline
        kernels = {"merge_emit": (kms["merge_ms"], in_bytes + out_bytes)}
    elif path_mode == 2:
        # split fused pair: A (merge staging + key/seq/kind emit + dense
        # winners) and B (value gather) are the two halves of one fused
        # emission pipeline and overlap on two streams — the wall between
        # the partition and batch-done events is merge_ms + scan_ms +
        # emit_ms, and the PAIR is the dominant unit the roofline tracks
        a_bytes = rows_in * 20 + U * (17 + 4)
        b_bytes = U * (4 + 2 * (8 + args.vals * 4))
        pair_ms = kms["merge_ms"] + kms["scan_ms"] + kms["emit_ms"]
        kernels = {"merge_emit+emit_dense": (pair_ms, a_bytes + b_bytes)}
    else:
        merge_bytes = rows_in * 20 + U * 4 + n_tiles * 4
        emit_bytes = U * (4 + (28 + args.vals * 4) + (25 + args.vals * 4))
        kernels = {
            "merge": (kms["merge_ms"], merge_bytes),
            "emit": (kms["emit_ms"], emit_bytes),
        }
    dom = max(kernels, key=lambda k: kernels[k][0])
    dom_ms, dom_bytes = kernels[dom]
    achieved = dom_bytes / (dom_ms / 1e3) / 1e9 if dom_ms > 0 else 0.0
    pipe_gbs = ((in_bytes + out_bytes) / (kms["total_device_ms"] / 1e3) / 1e9
                if kms["total_device_ms"] > 0 else 0.0)

    value_per_rank = rows_in * K / elapsed
    value = value_per_rank * n_gpus if world > 1 else value_per_rank

    result = {
        "metric": "merged rows/sec",
        "value": value,
        "unit": "rows/s",
        "n_gpus": n_gpus,
        "steps": K,
        "warmup": args.warmup,
        "ms_per_step": elapsed / K * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "int64",
        "data": "synthetic",
        "config": {
            "workload": (f"merge-on-read: {args.runs} sorted runs x "
                         f"{args.rows} rows, int64 PK + {args.vals} int32, "
                         f"{args.file_format} {args.compression}, "
                         + {"partial-update":
                            "PartialUpdate (30% updates, 6-col subsets; "
                            "variant of configs[2])",
                            "aggregation":
                            "Aggregation (sum/max/min/last_non_null cycle, "
                            "30% update rows)",
                            "first-row": "FirstRow (insert-only)",
                            }.get(args.engine,
                                  "Deduplicate, drop-delete "
                                  "(BASELINE.json configs[1])")),
            "n_runs": args.runs,
            "rows_per_run": args.rows,
            "value_cols": args.vals,
            "compression": args.compression,
            "seed": args.seed,
            "rows_out_per_step": int(U),
            "parallelism": f"bucket-parallel dp{n_gpus}, no collectives",
        },
        "roofline": {
            "bound": "hbm",
            "kernel": dom,
            "achieved": round(achieved, 1),
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved / HBM_PEAK_GBS, 4),
            "traffic": pmc_traffic(dom, args),
        },
        "kernels_ms_per_step": {k: round(v, 3) for k, v in kms.items()},
        "pipeline_algorithmic_GBs": round(pipe_gbs, 1),
        "stage_seconds": round(t_stage, 2),
        "gen_seconds": round(t_gen, 2),
    }

    if rank == 0 and n_gpus == 1 and args.cpu_baseline_rows > 0:
        result["cpu_baseline"] = cpu_baseline(args, out_dir)

    plan.close()
    sess.close()
    if dist:
        dist.destroy_process_group()
    if rank == 0:
        print(json.dumps(result))


if __name__ == "__main__":
    main()
