#!/bin/bash
# PMC traffic calibration for the default bench workload (run via gpurun).
# Counters collected in their own passes, never combined with trace domains.
set -x
cd /root/repo
mkdir -p gpurun_out/tprof
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --pmc FETCH_SIZE \
    -d /root/repo/gpurun_out/tprof -o pmc_fetch --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/tprof/fetch.log 2>&1
echo FETCH_RC=$?
timeout 420 rocprofv3 --pmc WRITE_SIZE \
    -d /root/repo/gpurun_out/tprof -o pmc_write --output-format csv -- \
    python3 /root/repo/bench.py --steps 3 --warmup 1 --cpu-baseline-rows 0 \
    > /root/repo/gpurun_out/tprof/write.log 2>&1
echo WRITE_RC=$?
python3 /root/repo/scripts/pmc_traffic.py /root/repo/gpurun_out/tprof \
    8 10000000 8 4 > /root/repo/gpurun_out/pmc_traffic.json
echo PARSE_RC=$?
head -c 2000 /root/repo/gpurun_out/pmc_traffic.json
# keep only the reduced json + logs; the raw counter CSVs can exceed the
# gpurun_out merge budget
rm -f /root/repo/gpurun_out/tprof/*counter*.csv
