#!/bin/bash
# Same-box A/B: new default (legacy chain + improved partition) vs round-1
# tree (_r01) and vs the fused split path (PMH_FUSED=1), C2 and 16x20M.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/ab_gputests.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/ab_gputests.log
B="--steps 8 --warmup 3 --cpu-baseline-rows 0"
B16="--steps 4 --warmup 2 --runs 16 --rows 20000000 --cpu-baseline-rows 0"
timeout 300 python bench.py $B               > gpurun_out/ab_new_c2.json 2> gpurun_out/ab_new_c2.err
timeout 300 python bench.py $B16             > gpurun_out/ab_new_16.json 2> gpurun_out/ab_new_16.err
PMH_FUSED=1 timeout 300 python bench.py $B   > gpurun_out/ab_fused_c2.json 2> gpurun_out/ab_fused_c2.err
(cd _r01 && timeout 300 python bench.py $B   > ../gpurun_out/ab_r01_c2.json 2> ../gpurun_out/ab_r01_c2.err)
(cd _r01 && timeout 300 python bench.py $B16 > ../gpurun_out/ab_r01_16.json 2> ../gpurun_out/ab_r01_16.err)
for f in gpurun_out/ab_*.json; do echo "== $f"; python -c "
import json,sys
try:
    d=json.load(open('$f'))
    print(round(d['ms_per_step'],3),'ms ', d.get('kernels_ms_per_step'), ' rows_out', d['config']['rows_out_per_step'], ' roofline_frac', d['roofline']['frac'])
except Exception as e: print('ERR',e)
"; done
tail -3 gpurun_out/ab_gputests.log
