#!/bin/bash
# Round-2 GPU call 6: consistent-box measurement sweep for the record
# (all bench modes back-to-back on ONE box) + full suite.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 1500 python -m pytest tests -m gpu -q \
    > gpurun_out/pytest_gpu6.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu6.log
tail -3 gpurun_out/pytest_gpu6.log

for args in "" "--via proxy" "--via proxy-miss" "--via peer" \
            "--verify digest"; do
  name=$(echo "bench6${args}" | tr ' -' '__')
  timeout 600 python bench.py --steps 3 --warmup 1 $args \
      > "gpurun_out/${name}.json" 2> "gpurun_out/${name}.log"
  tail -1 "gpurun_out/${name}.json"
done

timeout 600 python bench.py --model gguf-70b --virtual --steps 2 \
    --warmup 1 > gpurun_out/bench6_gguf70b.json \
    2> gpurun_out/bench6_gguf70b.log
tail -1 gpurun_out/bench6_gguf70b.json
timeout 420 python bench.py --model gguf-8b --steps 3 --warmup 1 \
    > gpurun_out/bench6_gguf8b.json 2> gpurun_out/bench6_gguf8b.log
tail -1 gpurun_out/bench6_gguf8b.json

echo DONE
