#!/bin/bash
# Round-2 GPU call 1: full gpu pytest + bench sanity on the new paths.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# 1. full GPU test suite (includes new: rccl world-1, torchrun smokes,
#    segmented-resume head, GPU prefetch)
timeout 1200 python -m pytest tests -m gpu -q -x \
    > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu.log
tail -5 gpurun_out/pytest_gpu.log

# 2. flagship dp bench (now with scatter in the timed step)
timeout 420 python bench.py --steps 3 --warmup 1 \
    > gpurun_out/bench_dp.json 2> gpurun_out/bench_dp.log
tail -1 gpurun_out/bench_dp.json

# 3. upstream-exact verified first pull (digest chain, hasher thread)
timeout 420 python bench.py --steps 2 --warmup 1 --verify digest \
    > gpurun_out/bench_digest.json 2> gpurun_out/bench_digest.log
tail -1 gpurun_out/bench_digest.json

# 4. proxy data plane: HIT path and relay path
timeout 600 python bench.py --steps 2 --warmup 1 --via proxy \
    > gpurun_out/bench_proxy.json 2> gpurun_out/bench_proxy.log
tail -1 gpurun_out/bench_proxy.json
timeout 600 python bench.py --steps 2 --warmup 1 --via proxy-miss \
    > gpurun_out/bench_proxymiss.json 2> gpurun_out/bench_proxymiss.log
tail -1 gpurun_out/bench_proxymiss.json

echo DONE
