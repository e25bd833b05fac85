#!/bin/bash
# Round-end consolidation on the GPU box: full suite, smoke, all bench lines,
# kernel trace + PMC refresh (counters never combined with trace domains).
set -x
cd /root/repo
mkdir -p gpurun_out/prof
python3 -m pytest tests -m gpu -q 2>&1 | tail -2
python3 -c "import __graft_entry__ as g; g.smoke()"
# default metric line (BASELINE.json configs[1]) with CPU baseline
python3 bench.py --steps 10 --warmup 3 > gpurun_out/bench_r01_final.json \
    2> gpurun_out/bench_r01_final.err
echo BENCH_RC=$?
# supplementary engine lines
python3 bench.py --steps 6 --warmup 2 --engine partial-update --vals 12 \
    --cpu-baseline-rows 0 > gpurun_out/bench_r01_pu.json \
    2> gpurun_out/bench_r01_pu.err
echo BENCH_PU_RC=$?
python3 bench.py --steps 6 --warmup 2 --engine partial-update --vals 20 \
    --format orc --cpu-baseline-rows 0 > gpurun_out/bench_r01_c3orc.json \
    2> gpurun_out/bench_r01_c3orc.err
echo BENCH_C3_RC=$?
python3 bench.py --steps 6 --warmup 2 --engine aggregation \
    --cpu-baseline-rows 0 > gpurun_out/bench_r01_agg.json \
    2> gpurun_out/bench_r01_agg.err
echo BENCH_AGG_RC=$?
python3 bench.py --steps 6 --warmup 2 --engine first-row \
    --cpu-baseline-rows 0 > gpurun_out/bench_r01_fr.json \
    2> gpurun_out/bench_r01_fr.err
echo BENCH_FR_RC=$?
# capacity: 16 runs x 20M rows (320M input rows, near the C5 shape)
python3 bench.py --steps 3 --warmup 1 --runs 16 --rows 20000000 \
    --cpu-baseline-rows 0 > gpurun_out/bench_r01_c5_16x20M.json \
    2> gpurun_out/bench_r01_c5.err
echo BENCH_C5_RC=$?
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof \
    -o final --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/prof_final.log 2>&1
echo TRACE_RC=$?
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof \
    -o final_c3 --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 \
    --engine partial-update --vals 20 --format orc --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/prof_final_c3.log 2>&1
echo TRACE_C3_RC=$?
timeout 420 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAVE_CYCLES SQ_INSTS_VALU \
    -d /root/repo/gpurun_out/prof -o final_sq --output-format csv -- \
    python3 /root/repo/bench.py --steps 2 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/pmc_final_sq.log 2>&1
echo PMC_RC=$?
# refresh the committed traffic calibration (FETCH/WRITE in separate passes)
mkdir -p /root/repo/gpurun_out/tprof
timeout 420 rocprofv3 --pmc FETCH_SIZE \
    -d /root/repo/gpurun_out/tprof -o pmc_fetch --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/tprof/fetch.log 2>&1
timeout 420 rocprofv3 --pmc WRITE_SIZE \
    -d /root/repo/gpurun_out/tprof -o pmc_write --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/tprof/write.log 2>&1
python3 /root/repo/scripts/pmc_traffic.py /root/repo/gpurun_out/tprof \
    8 10000000 8 > /root/repo/gpurun_out/pmc_traffic.json
echo TRAFFIC_RC=$?
rm -f /root/repo/gpurun_out/tprof/*counter*.csv
# raw per-dispatch counter CSVs are large; keep stats summaries only
rm -f /root/repo/gpurun_out/prof/*counter_collection*.csv \
      /root/repo/gpurun_out/prof/*kernel_trace*.csv
ls /root/repo/gpurun_out/prof/ | head
tail -c 600 /root/repo/gpurun_out/bench_r01_final.json
