#!/bin/bash
# Sustained-stability soak: 30-step flagship with buffer recycling,
# tracking HBM usage for drift/leaks.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
(while true; do rocm-smi --showmeminfo vram --csv 2>/dev/null | tail -1; sleep 5; done) > gpurun_out/soak_vram.log 2>&1 &
MONPID=$!
timeout 900 python bench.py --steps 30 --warmup 2 \
    > gpurun_out/soak_flagship.json 2> gpurun_out/soak_flagship.log
rc=$?
kill $MONPID 2>/dev/null
echo "bench rc=$rc"
tail -1 gpurun_out/soak_flagship.json
# per-step times from the log
grep -c "landed model" gpurun_out/soak_flagship.log
# vram trace: first / last samples
head -3 gpurun_out/soak_vram.log; tail -3 gpurun_out/soak_vram.log
echo DONE
