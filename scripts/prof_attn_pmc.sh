#!/bin/bash
# PMC profile of the standalone attention bench (run on GPU box).
set -u
export TMPDIR=/tmp
cd /root/repo
hipcc --offload-arch=gfx950 -O3 -I photon_amd/ops/hip scripts/attn_bench.hip -o /tmp/ab 2>/dev/null
cd /tmp
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT \
  --output-format csv -d /root/repo/gpurun_out/pmc_attn -o pmc -- /tmp/ab > /tmp/pmc.log 2>&1 || { echo FAILED; tail -5 /tmp/pmc.log; exit 1; }
f=$(find /root/repo/gpurun_out/pmc_attn -name "*counter*" | head -1)
python3 - "$f" << 'PY'
import csv, sys
from collections import defaultdict
rows = list(csv.DictReader(open(sys.argv[1])))
agg = defaultdict(lambda: defaultdict(float))
for r in rows:
    name = r.get("Kernel_Name") or r.get("Kernel Name") or ""
    short = name.split("(")[0].split("<")[0].split("::")[-1]
    agg[short][r["Counter_Name"]] += float(r["Counter_Value"])
for k, c in agg.items():
    wc = c.get("SQ_WAVE_CYCLES", 0)
    if wc < 1e6: continue
    print(k)
    for n, v in sorted(c.items()):
        qc = v
        print(f"  {n:28s} {qc:,.0f}  ({qc/wc*100:5.1f}% of wave cycles)" if n != "SQ_WAVE_CYCLES" else f"  {n:28s} {qc:,.0f}")
PY
