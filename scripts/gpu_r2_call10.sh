#!/bin/bash
# Round-2 GPU call 10: segment-count sweep for single-file pulls
# (gguf blobs cap at MAX_SEGMENTS=8 streams today) + flagship worker
# sweep.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

run() {  # name, env..., args...
  local name=$1; shift
  timeout 420 env "$@" > "gpurun_out/seg_$name.json" \
      2> "gpurun_out/seg_$name.log"
  tail -1 "gpurun_out/seg_$name.json"
}

run gguf_base python bench.py --model gguf-8b --steps 3 --warmup 1
run gguf_s16 DEMODEL_MAX_SEGMENTS=16 DEMODEL_SEGMENT_MIN_MB=256 \
    python bench.py --model gguf-8b --steps 3 --warmup 1
run gguf_s24w16 DEMODEL_MAX_SEGMENTS=24 DEMODEL_SEGMENT_MIN_MB=128 \
    python bench.py --model gguf-8b --steps 3 --warmup 1 --workers 16
run flag_base python bench.py --steps 3 --warmup 1
run flag_s16w16 DEMODEL_MAX_SEGMENTS=16 \
    python bench.py --steps 3 --warmup 1 --workers 16
run gguf70_s16 DEMODEL_MAX_SEGMENTS=16 \
    python bench.py --model gguf-70b --virtual --steps 2 --warmup 1

echo DONE
