#!/bin/bash
# emit-kernel A/B sweep on one box: by-column slices + block-count knobs.
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
B="--steps 6 --warmup 2 --cpu-baseline-rows 0"
run() {
  name=$1; shift
  env "$@" timeout 200 python bench.py $B > gpurun_out/es_$name.json 2> gpurun_out/es_$name.err
  python -c "
import json
try:
    d=json.load(open('gpurun_out/es_$name.json'))
    k=d['kernels_ms_per_step']
    print('$name', round(d['ms_per_step'],3),'ms  emit',k['emit_ms'],' merge',k['merge_ms'],' rows_out',d['config']['rows_out_per_step'])
except Exception as e: print('$name ERR',e)
"
}
run base PMH_X=0
run bycol PMH_EMIT_BYCOL=1
run bycol512 PMH_EMIT_BYCOL=1 PMH_EMIT_BLOCKS=512
run bycol2048 PMH_EMIT_BYCOL=1 PMH_EMIT_BLOCKS=2048
run blk4096 PMH_EMIT_BLOCKS=4096
run blk8192 PMH_EMIT_BLOCKS=8192
# parity guard: by-column emit against the oracle path in the GPU suite
PMH_EMIT_BYCOL=1 timeout 300 python -m pytest tests/test_merge_gpu.py -x -q 2>&1 | tail -1
