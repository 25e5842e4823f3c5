#!/bin/bash
# Round-2 GPU call 13: same-box A/B — incremental vs tail verification,
# interleaved runs to beat box drift.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
for rep in 1 2; do
  for mode in 1 0; do
    DEMODEL_INC_VERIFY=$mode timeout 420 python bench.py --steps 4 \
        --warmup 1 > gpurun_out/ab_flag_inc${mode}_r${rep}.json \
        2> /dev/null
    echo "flag inc=$mode rep=$rep: $(tail -1 gpurun_out/ab_flag_inc${mode}_r${rep}.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(d["value"], d["ms_per_step"])')"
    DEMODEL_INC_VERIFY=$mode timeout 420 python bench.py --model gguf-8b \
        --steps 4 --warmup 1 > gpurun_out/ab_gguf_inc${mode}_r${rep}.json \
        2> /dev/null
    echo "gguf inc=$mode rep=$rep: $(tail -1 gpurun_out/ab_gguf_inc${mode}_r${rep}.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(d["value"], d["ms_per_step"])')"
  done
done
echo DONE
