#!/bin/bash
# Round-2 consolidation on the GPU box: full suite, smoke, all bench lines,
# kernel trace + PMC refresh (counters never combined with trace domains).
set -x
cd /root/repo
mkdir -p gpurun_out/prof gpurun_out/tprof
timeout 700 python3 -m pytest tests -m gpu -q 2>&1 | tail -2
python3 -c "import __graft_entry__ as g; g.smoke()" && echo SMOKE_OK
# default metric line (BASELINE.json configs[1]) with CPU baselines
timeout 300 python3 bench.py --steps 10 --warmup 3 \
    > gpurun_out/bench_r02_final.json 2> gpurun_out/bench_r02_final.err
echo BENCH_RC=$?
# compressed-table variant (reference default file.compression = zstd;
# pages decompress on host at staging — stage_seconds carries that cost)
timeout 300 python3 bench.py --steps 6 --warmup 2 --compression zstd \
    --cpu-baseline-rows 0 > gpurun_out/bench_r02_zstd.json 2>&1
# supplementary engine lines
timeout 300 python3 bench.py --steps 6 --warmup 2 --engine partial-update \
    --vals 12 --cpu-baseline-rows 0 > gpurun_out/bench_r02_pu.json 2>&1
timeout 300 python3 bench.py --steps 6 --warmup 2 --engine partial-update \
    --vals 20 --format orc --cpu-baseline-rows 0 \
    > gpurun_out/bench_r02_c3orc.json 2>&1
timeout 300 python3 bench.py --steps 6 --warmup 2 --engine aggregation \
    --cpu-baseline-rows 0 > gpurun_out/bench_r02_agg.json 2>&1
timeout 300 python3 bench.py --steps 6 --warmup 2 --engine first-row \
    --cpu-baseline-rows 0 > gpurun_out/bench_r02_fr.json 2>&1
# capacity: 16 runs x 20M rows
timeout 400 python3 bench.py --steps 3 --warmup 1 --runs 16 \
    --rows 20000000 --cpu-baseline-rows 0 \
    > gpurun_out/bench_r02_16x20M.json 2>&1
# C5 as specced (mixed types, read+merge+write round trip)
timeout 900 python3 scripts/bench_c5.py --reps 3 \
    > gpurun_out/bench_r02_c5.json 2>&1
echo C5_RC=$?
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof \
    -o r02 --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/prof_r02.log 2>&1
echo TRACE_RC=$?
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof \
    -o r02_c3 --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 \
    --engine partial-update --vals 20 --format orc --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/prof_r02_c3.log 2>&1
timeout 420 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAVE_CYCLES SQ_INSTS_VALU \
    -d /root/repo/gpurun_out/prof -o r02_sq --output-format csv -- \
    python3 /root/repo/bench.py --steps 2 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/pmc_r02_sq.log 2>&1
echo PMC_RC=$?
timeout 420 rocprofv3 --pmc FETCH_SIZE \
    -d /root/repo/gpurun_out/tprof -o pmc_fetch --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/tprof/fetch.log 2>&1
timeout 420 rocprofv3 --pmc WRITE_SIZE \
    -d /root/repo/gpurun_out/tprof -o pmc_write --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/tprof/write.log 2>&1
python3 /root/repo/scripts/pmc_traffic.py /root/repo/gpurun_out/tprof \
    8 10000000 8 4 > /root/repo/gpurun_out/pmc_traffic_r02.json
echo TRAFFIC_RC=$?
# reduce the SQ pass before discarding the raw CSVs
python3 - << 'PYEOF'
import csv, glob, json
tot, disp = {}, {}
for f in glob.glob("/root/repo/gpurun_out/prof/*r02_sq*counter*.csv"):
    for row in csv.DictReader(open(f)):
        kn = row["Kernel_Name"].split("(")[0]
        key = (kn, row["Counter_Name"])
        tot[key] = tot.get(key, 0.0) + float(row["Counter_Value"])
        disp.setdefault(key, set()).add(
            (row.get("Dispatch_Id"), row.get("Correlation_Id")))
out = {}
for (kn, cn), v in tot.items():
    out.setdefault(kn, {})[cn] = v / len(disp[(kn, cn)])
json.dump(out, open("/root/repo/gpurun_out/pmc_r02_sq.json", "w"), indent=1)
for kn, d in out.items():
    wc = d.get("SQ_WAVE_CYCLES", 0) or 1
    print(kn[:60], "WAIT", round(d.get("SQ_WAIT_ANY", 0)/wc, 3),
          "LDSconf", round(d.get("SQ_LDS_BANK_CONFLICT", 0) /
                           max(d.get("SQ_LDS_IDX_ACTIVE", 1), 1), 3))
PYEOF
rm -f /root/repo/gpurun_out/tprof/*counter*.csv \
      /root/repo/gpurun_out/prof/*counter_collection*.csv \
      /root/repo/gpurun_out/prof/*kernel_trace*.csv
for f in /root/repo/gpurun_out/bench_r02_*.json; do
  echo "== $f"; tail -c 700 "$f"; echo
done
head -c 1200 /root/repo/gpurun_out/pmc_traffic_r02.json
