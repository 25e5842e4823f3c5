#!/bin/bash
# Round-2 final validation: full GPU suite + smoke + every bench config
# back-to-back on one box.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 1500 python -m pytest tests -m gpu -q \
    > gpurun_out/pytest_final.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_final.log
tail -3 gpurun_out/pytest_final.log

timeout 600 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/smoke_final.log 2>&1
echo "smoke rc=$?"
tail -2 gpurun_out/smoke_final.log

bench() {
  local name=$1; shift
  timeout 600 python bench.py "$@" > gpurun_out/final_$name.json \
      2> gpurun_out/final_$name.log
  echo "$name: $(tail -1 gpurun_out/final_$name.json)"
}
bench flagship --steps 5 --warmup 1
bench gguf8b --model gguf-8b --steps 4 --warmup 1
bench gguf70b --model gguf-70b --virtual --steps 2 --warmup 1
bench st70b --model llama3-70b --virtual --steps 2 --warmup 1
bench dataset --model dataset --steps 3 --warmup 1
bench parquet --model parquet --steps 3 --warmup 1
bench pq_lz4 --model parquet --parquet-codec lz4 --steps 3 --warmup 1
bench viaproxy --steps 3 --warmup 1 --via proxy
bench viamiss --steps 3 --warmup 1 --via proxy-miss
bench viapeer --steps 3 --warmup 1 --via peer
bench digest --steps 2 --warmup 1 --verify digest
echo DONE
