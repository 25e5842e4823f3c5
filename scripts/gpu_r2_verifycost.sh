#!/bin/bash
# Verification cost accounting on one box: off vs chunked vs gpu-digest.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
for v in off chunked off chunked; do
  n=$(ls gpurun_out/vc_${v}_*.json 2>/dev/null | wc -l)
  timeout 420 python bench.py --steps 4 --warmup 1 --verify $v \
      > gpurun_out/vc_${v}_$n.json 2>/dev/null
  echo "$v#$n: $(tail -1 gpurun_out/vc_${v}_$n.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(d["value"], d["ms_per_step"])')"
done
echo DONE
