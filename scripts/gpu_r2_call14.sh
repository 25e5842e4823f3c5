#!/bin/bash
# Round-2 GPU call 14: pinned-slab geometry + worker sweep (one box,
# flagship), interleaved with the baseline to beat drift.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
run() {
  local name=$1; shift
  timeout 420 python bench.py --steps 4 --warmup 1 "$@" \
      > gpurun_out/sw_$name.json 2>/dev/null
  echo "$name: $(tail -1 gpurun_out/sw_$name.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(d["value"], d["ms_per_step"])')"
}
run base1
run s64n4 --slab-mib 64 --n-slabs 4
run base2
run s64n8 --slab-mib 64 --n-slabs 8
run s16n8 --slab-mib 16 --n-slabs 8
run base3
run w12 --workers 12
run s64n4w12 --slab-mib 64 --n-slabs 4 --workers 12
echo DONE
