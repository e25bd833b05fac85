#!/bin/bash
# Fused-kernel (k_merge_emit) profiling round: SQ counters + HBM traffic +
# phase ablations. Runs ON THE GPU BOX via gpurun. Counters collected in
# dedicated --pmc passes, never combined with trace domains.
set -x
cd /root/repo
mkdir -p gpurun_out/fprof
export TMPDIR=/tmp
cd /tmp

B="python3 /root/repo/bench.py --warmup 2 --cpu-baseline-rows 0"

# phase ablations (HIP-event timing only, no profiler)
for abl in 0 1 2 3; do
  PMH_FABL=$abl timeout 240 $B --steps 8 \
      > /root/repo/gpurun_out/fprof/fabl$abl.json 2>&1
done

timeout 420 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAVE_CYCLES SQ_INSTS_VALU \
    --output-format csv -d /root/repo/gpurun_out/fprof -o sq -- \
    timeout 200 $B --steps 3 > /root/repo/gpurun_out/fprof/sq.log 2>&1
echo SQ_RC=$?
timeout 420 rocprofv3 --pmc FETCH_SIZE \
    --output-format csv -d /root/repo/gpurun_out/fprof -o fetch -- \
    timeout 200 $B --steps 3 > /root/repo/gpurun_out/fprof/fetch.log 2>&1
echo FETCH_RC=$?
timeout 420 rocprofv3 --pmc WRITE_SIZE \
    --output-format csv -d /root/repo/gpurun_out/fprof -o write -- \
    timeout 200 $B --steps 3 > /root/repo/gpurun_out/fprof/write.log 2>&1
echo WRITE_RC=$?

python3 - << 'EOF'
import csv, glob, json
def per_dispatch(pattern, counters):
    tot, disp = {}, {}
    for f in glob.glob(pattern):
        for row in csv.DictReader(open(f)):
            cn = row.get("Counter_Name")
            if cn not in counters:
                continue
            kn = row["Kernel_Name"].split("(")[0]
            key = (kn, cn)
            tot[key] = tot.get(key, 0.0) + float(row["Counter_Value"])
            disp.setdefault(key, set()).add(
                (row.get("Dispatch_Id"), row.get("Correlation_Id")))
    return {k: tot[k] / len(disp[k]) for k in tot}

out = {}
sq = per_dispatch("/root/repo/gpurun_out/fprof/*sq*counter*.csv",
                  {"SQ_LDS_BANK_CONFLICT", "SQ_LDS_IDX_ACTIVE", "SQ_WAIT_ANY",
                   "SQ_WAIT_INST_ANY", "SQ_WAVE_CYCLES", "SQ_INSTS_VALU"})
tr = per_dispatch("/root/repo/gpurun_out/fprof/*fetch*counter*.csv",
                  {"FETCH_SIZE"})
tr.update(per_dispatch("/root/repo/gpurun_out/fprof/*write*counter*.csv",
                       {"WRITE_SIZE"}))
for (kn, cn), v in list(sq.items()) + list(tr.items()):
    out.setdefault(kn, {})[cn] = v
json.dump(out, open("/root/repo/gpurun_out/fprof/reduced.json", "w"),
          indent=1)
for kn, d in out.items():
    if "merge_emit" in kn or "partition" in kn:
        wc = d.get("SQ_WAVE_CYCLES", 0) or 1
        print(kn[:60], {k: round(v/wc, 3) if k.startswith("SQ_WAIT") else v
                        for k, v in d.items()})
EOF
# keep only the reduced json; raw counter CSVs can exceed the merge budget
rm -f /root/repo/gpurun_out/fprof/*counter*.csv
