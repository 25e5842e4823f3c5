#!/bin/bash
# Round-2 GPU call 3: zstd mode A/B (x1/x1nf/x2/x2nf), scatter fix
# validation, rocprof evidence refresh.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# 1. zstd kernel variant correctness (new tests) + full zstd suite
timeout 600 python -m pytest tests/test_gpu_zstd.py tests/test_datasets.py \
    -q -m gpu > gpurun_out/pytest_zstd.log 2>&1
echo "zstd tests rc=$?" | tee -a gpurun_out/pytest_zstd.log
tail -2 gpurun_out/pytest_zstd.log

# 2. zstd mode A/B
for m in x1 x1nf x2 x2nf; do
  DEMODEL_ZSTD_MODE=$m timeout 300 python -c "
import sys; sys.path.insert(0, 'scripts')
from gpu_probe import zstd_bench
print('MODE=$m')
for p in ('words', 'text', 'random'):
    zstd_bench(payload=p)
" >> gpurun_out/zstd_ab.log 2>&1
done
grep -E "MODE|zstd_decode" gpurun_out/zstd_ab.log

# 3. flagship dp with the scatter fix
timeout 420 python bench.py --steps 3 --warmup 1 \
    > gpurun_out/bench_dp3.json 2> gpurun_out/bench_dp3.log
tail -1 gpurun_out/bench_dp3.json
grep scatter gpurun_out/bench_dp3.log | tail -4

# 4. rocprof kernel stats for the flagship (evidence refresh)
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_dp \
    -o dp -- python bench.py --steps 2 --warmup 1 \
    > gpurun_out/bench_dp_prof.json 2> gpurun_out/bench_dp_prof.log
tail -1 gpurun_out/bench_dp_prof.json
find gpurun_out/prof_dp -name "*stats*" | head -5

echo DONE
