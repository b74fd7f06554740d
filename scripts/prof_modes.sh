#!/bin/bash
# Profile bench.py kernel stats for PHOTON_LT_MODE variants (run on GPU box).
set -u
export TMPDIR=/tmp
cd /tmp
for MODE in "$@"; do
  PHOTON_LT_MODE=$MODE rocprofv3 --kernel-trace --stats --output-format csv \
    -d /root/repo/gpurun_out/prof_$MODE -o st -- \
    /bin/bash -c "cd /root/repo && timeout 240 python bench.py --gpus 1 --steps 4 --warmup 2 > /dev/null 2>&1" \
    > /tmp/prof_$MODE.log 2>&1 || { echo "prof $MODE failed"; tail -3 /tmp/prof_$MODE.log; }
done
cd /root/repo
for MODE in "$@"; do
  echo "=== mode=$MODE top kernels ==="
  f=$(find gpurun_out/prof_$MODE -name "*kernel_stats*" 2>/dev/null | head -1)
  [ -n "$f" ] && python3 scripts/topk.py "$f" 14
done
