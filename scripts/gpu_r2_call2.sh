#!/bin/bash
# Round-2 GPU call 2: head-fix validation, scatter timing, threaded
# relay, zstd x2 kernel A/B, 70B GGUF virtual.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 1200 python -m pytest tests -m gpu -q -x \
    > gpurun_out/pytest_gpu2.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu2.log
tail -3 gpurun_out/pytest_gpu2.log

# zstd decode probe (x2 kernel engages at >=768 streams)
timeout 420 python scripts/zstd_probe.py \
    > gpurun_out/zstd_probe_x2.log 2>&1
grep zstd_decode gpurun_out/zstd_probe_x2.log

# flagship dp with scatter timing in stderr log
timeout 420 python bench.py --steps 3 --warmup 1 \
    > gpurun_out/bench_dp2.json 2> gpurun_out/bench_dp2.log
tail -1 gpurun_out/bench_dp2.json
grep scatter gpurun_out/bench_dp2.log | tail -4

# threaded-relay proxy miss path
timeout 600 python bench.py --steps 2 --warmup 1 --via proxy-miss \
    > gpurun_out/bench_proxymiss2.json 2> gpurun_out/bench_proxymiss2.log
tail -1 gpurun_out/bench_proxymiss2.json

# 70B GGUF at nameplate scale (41 GB q4_K -> 141 GB bf16), virtual
timeout 900 python bench.py --model gguf-70b --virtual --steps 2 \
    --warmup 1 > gpurun_out/bench_gguf70b.json \
    2> gpurun_out/bench_gguf70b.log
tail -1 gpurun_out/bench_gguf70b.json
grep -E "dequant|warmup" gpurun_out/bench_gguf70b.log | tail -5

echo DONE
