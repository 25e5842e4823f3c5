#!/bin/bash
# Round-2 GPU call 11: DEFLATE literal-batching validation + A/B, new
# segment defaults sanity.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# correctness: inflate + parquet gzip matrix + zstd (deflate shares
# nothing but be safe)
timeout 900 python -m pytest tests/test_gpu_inflate.py \
    tests/test_parquet.py tests/test_gpu_snappy.py -q -m gpu \
    > gpurun_out/pytest_inflate.log 2>&1
echo "rc=$?" | tee -a gpurun_out/pytest_inflate.log
tail -3 gpurun_out/pytest_inflate.log

# throughput: the standard probe + a literal-heavy payload (word salad
# gzip: mostly literal symbols -> the batching's target)
timeout 420 python - > gpurun_out/inflate_ab.log 2>&1 <<'PYEOF'
import sys, os, json, zlib, ctypes, time
sys.path.insert(0, "scripts")
from gpu_probe import inflate_bench, bench
inflate_bench()

import numpy as np
from demodel_amd.engine.formats.compress import inflate_gpu
from demodel_amd.gpu import hip

h = hip()
s = h.Stream(0)
rng = np.random.default_rng(5)
words = [f"w{i:04d}" for i in range(20000)]
idx = rng.integers(0, len(words), size=(4 << 20) // 6)
data = " ".join(words[i] for i in idx).encode()[:4 << 20]
comp = zlib.compressobj(6, zlib.DEFLATED, -15)
blob = comp.compress(data) + comp.flush()
src = h.DeviceBuffer(len(blob))
carr = (ctypes.c_char * len(blob)).from_buffer_copy(blob)
h.h2d_async(src.ptr, ctypes.addressof(carr), len(blob), s.handle)
s.sync()
n = 256
dsts = [h.DeviceBuffer(len(data)) for _ in range(n)]
streams = [(src.ptr, len(blob), d.ptr, len(data)) for d in dsts]
t = bench(lambda: inflate_gpu(streams), iters=2, warmup=1)
print(json.dumps({"op": "inflate_words", "streams": n,
                  "GBps_out": round(len(data) * n / t / 1e9, 2),
                  "ratio": round(len(data) / len(blob), 2)}))
PYEOF
grep -E "inflate" gpurun_out/inflate_ab.log

# new segment defaults: gguf-8b + flagship quick
timeout 420 python bench.py --model gguf-8b --steps 3 --warmup 1 \
    > gpurun_out/bench_gguf_segdef.json 2> gpurun_out/bench_gguf_segdef.log
tail -1 gpurun_out/bench_gguf_segdef.json
timeout 420 python bench.py --steps 3 --warmup 1 \
    > gpurun_out/bench_dp_segdef.json 2> gpurun_out/bench_dp_segdef.log
tail -1 gpurun_out/bench_dp_segdef.json

echo DONE
