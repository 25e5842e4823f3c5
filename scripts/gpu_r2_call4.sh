#!/bin/bash
# Round-2 GPU call 4: zstd 8K-window A/B, rocprof CSV evidence refresh,
# full gpu pytest.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# 1. full GPU suite
timeout 1500 python -m pytest tests -m gpu -q -x \
    > gpurun_out/pytest_gpu4.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu4.log
tail -3 gpurun_out/pytest_gpu4.log

# 2. zstd w8 A/B (vs x1 16K baseline measured in call 3)
for m in x1 x1w8; do
  DEMODEL_ZSTD_MODE=$m timeout 300 python -c "
import sys; sys.path.insert(0, 'scripts')
from gpu_probe import zstd_bench
print('MODE=$m')
for p in ('words', 'text', 'random'):
    zstd_bench(payload=p)
" >> gpurun_out/zstd_w8.log 2>&1
done
grep -E "MODE|zstd_decode" gpurun_out/zstd_w8.log

# also the end-to-end dataset bench under w8
DEMODEL_ZSTD_MODE=x1w8 timeout 420 python bench.py --model dataset \
    --steps 3 --warmup 1 > gpurun_out/bench_ds_w8.json \
    2> gpurun_out/bench_ds_w8.log
tail -1 gpurun_out/bench_ds_w8.json
timeout 420 python bench.py --model dataset --steps 3 --warmup 1 \
    > gpurun_out/bench_ds_x1.json 2> gpurun_out/bench_ds_x1.log
tail -1 gpurun_out/bench_ds_x1.json

# 3. rocprof CSV kernel stats for the flagship + gguf (evidence)
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof4 -o dp -- python bench.py --steps 2 --warmup 1 \
    > gpurun_out/prof4_dp.json 2> gpurun_out/prof4_dp.log
tail -1 gpurun_out/prof4_dp.json
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof4 -o gguf -- python bench.py --model gguf-8b \
    --steps 2 --warmup 1 > gpurun_out/prof4_gguf.json \
    2> gpurun_out/prof4_gguf.log
tail -1 gpurun_out/prof4_gguf.json
ls gpurun_out/prof4/

echo DONE
