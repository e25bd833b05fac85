#!/bin/bash
# PMC profiling of the C2 bench kernels on the GPU box (run via gpurun).
# Counters collected in their own passes (never combined with trace domains
# — see the gpurun/rocprofv3 combination rule).
set -x
cd /root/repo
mkdir -p gpurun_out/prof
# one un-profiled run generates+caches the data and gives a clean timing
python3 bench.py --steps 5 --warmup 2 --cpu-baseline-rows 0 \
    > gpurun_out/bench_latest.json 2> gpurun_out/bench_latest.err
echo BENCH_RC=$?
export TMPDIR=/tmp
cd /tmp
# pass 1: LDS + wait breakdown (SQ block)
timeout 360 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAVE_CYCLES SQ_INSTS_VALU \
    -d /root/repo/gpurun_out/prof -o pmc_sq --output-format csv -- \
    python3 /root/repo/bench.py --steps 2 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/pmc_sq.log 2>&1
echo PMC1_RC=$?
# pass 2: HBM fetch (TCC block; FETCH_SIZE costs 3 of 4 slots)
timeout 360 rocprofv3 --pmc FETCH_SIZE \
    -d /root/repo/gpurun_out/prof -o pmc_fetch --output-format csv -- \
    python3 /root/repo/bench.py --steps 2 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/pmc_fetch.log 2>&1
echo PMC2_RC=$?
# pass 3: HBM writes
timeout 360 rocprofv3 --pmc WRITE_SIZE \
    -d /root/repo/gpurun_out/prof -o pmc_write --output-format csv -- \
    python3 /root/repo/bench.py --steps 2 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/pmc_write.log 2>&1
echo PMC3_RC=$?
ls -la /root/repo/gpurun_out/prof/
