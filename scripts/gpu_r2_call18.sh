#!/bin/bash
# Re-measure gguf-70b WITH dequant (the real BASELINE config 4) +
# profile proof that the dequant kernels run.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 900 python bench.py --model gguf-70b --virtual --steps 2 --warmup 1 \
    > gpurun_out/gguf70b_fixed.json 2> gpurun_out/gguf70b_fixed.log
tail -1 gpurun_out/gguf70b_fixed.json
grep -E "dequant overlap|warmup" gpurun_out/gguf70b_fixed.log | tail -4
timeout 900 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof18 -o gguf70b -- python bench.py --model gguf-70b \
    --virtual --steps 1 --warmup 1 > gpurun_out/prof18_g.json \
    2> gpurun_out/prof18_g.log
tail -1 gpurun_out/prof18_g.json
head -6 gpurun_out/prof18/gguf70b_kernel_stats.csv
# also re-check gguf-8b unchanged
timeout 420 python bench.py --model gguf-8b --steps 3 --warmup 1 \
    > gpurun_out/gguf8b_recheck.json 2> gpurun_out/gguf8b_recheck.log
tail -1 gpurun_out/gguf8b_recheck.json
echo DONE
