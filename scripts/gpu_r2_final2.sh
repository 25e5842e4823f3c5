#!/bin/bash
# Final validation after the gguf routing fix: full suite + smoke +
# driver-shaped bench runs.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 1500 python -m pytest tests -m gpu -q \
    > gpurun_out/pytest_final2.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_final2.log
tail -3 gpurun_out/pytest_final2.log
timeout 600 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/smoke_final2.log 2>&1
echo "smoke rc=$?"
# driver-shaped flagship (long run, recycling stress)
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 2 \
    > gpurun_out/final2_flagship.json 2> gpurun_out/final2_flagship.log
tail -1 gpurun_out/final2_flagship.json
timeout 420 python bench.py --model gguf-8b --steps 4 --warmup 1 \
    > gpurun_out/final2_gguf8b.json 2> gpurun_out/final2_gguf8b.log
tail -1 gpurun_out/final2_gguf8b.json
timeout 600 python bench.py --model gguf-70b --virtual --steps 2 --warmup 1 \
    > gpurun_out/final2_gguf70b.json 2> gpurun_out/final2_gguf70b.log
tail -1 gpurun_out/final2_gguf70b.json
echo DONE
