#!/bin/bash
# rocprof CSV evidence for gguf-70b (dequant at scale) + dataset (zstd).
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 900 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof17 -o gguf70b -- python bench.py --model gguf-70b \
    --virtual --steps 1 --warmup 1 > gpurun_out/prof17_g.json \
    2> gpurun_out/prof17_g.log
tail -1 gpurun_out/prof17_g.json
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof17 -o dataset -- python bench.py --model dataset \
    --steps 2 --warmup 1 > gpurun_out/prof17_d.json \
    2> gpurun_out/prof17_d.log
tail -1 gpurun_out/prof17_d.json
ls gpurun_out/prof17/
echo DONE
