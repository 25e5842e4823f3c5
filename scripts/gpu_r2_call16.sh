#!/bin/bash
# gguf single-file stream tuning: seg pool is max(workers,4) threads.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
run() {
  local name=$1; shift
  timeout 420 env "$@" > gpurun_out/gg_$name.json 2>/dev/null
  echo "$name: $(tail -1 gpurun_out/gg_$name.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(d["value"], d["ms_per_step"])')"
}
run base python bench.py --model gguf-8b --steps 4 --warmup 1
run w12 python bench.py --model gguf-8b --steps 4 --warmup 1 --workers 12
run w16s16 DEMODEL_MAX_SEGMENTS=16 python bench.py --model gguf-8b --steps 4 --warmup 1 --workers 16
run base2 python bench.py --model gguf-8b --steps 4 --warmup 1
run s12w12 DEMODEL_MAX_SEGMENTS=12 python bench.py --model gguf-8b --steps 4 --warmup 1 --workers 12
echo DONE
