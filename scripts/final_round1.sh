#!/bin/bash
# Round-end consolidation on the GPU box: full suite, default bench (with
# CPU baseline), PU supplementary bench, kernel trace + PMC refresh.
set -x
cd /root/repo
mkdir -p gpurun_out/prof
python3 -m pytest tests -m gpu -q 2>&1 | tail -2
python3 -c "import __graft_entry__ as g; g.smoke()"
# default metric line (BASELINE.json configs[1])
python3 bench.py --steps 10 --warmup 3 > gpurun_out/bench_r01_final.json \
    2> gpurun_out/bench_r01_final.err
echo BENCH_RC=$?
# supplementary: PartialUpdate on the same shape (parquet variant of C3's
# merge-function half)
python3 bench.py --steps 6 --warmup 2 --engine partial-update --vals 12 \
    --cpu-baseline-rows 0 > gpurun_out/bench_r01_pu.json \
    2> gpurun_out/bench_r01_pu.err
echo BENCH_PU_RC=$?
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof \
    -o final --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/prof_final.log 2>&1
echo TRACE_RC=$?
timeout 420 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAVE_CYCLES SQ_INSTS_VALU \
    -d /root/repo/gpurun_out/prof -o final_sq --output-format csv -- \
    python3 /root/repo/bench.py --steps 2 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/pmc_final_sq.log 2>&1
echo PMC_RC=$?
ls /root/repo/gpurun_out/prof/ | head
tail -c 600 /root/repo/gpurun_out/bench_r01_final.json
