#!/bin/bash
# Round-2 GPU call 8: PMC evidence for the new kernels + digest-shards
# scaling datapoint.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# PMC run (no trace domains alongside --pmc, pool rule): lz4 + q5_K
# dequant probes under counters
cat > /tmp/pmc_probe.py <<'PYEOF'
import sys
sys.path.insert(0, "/root/repo")
import ctypes
import numpy as np
import pyarrow as pa
from demodel_amd.gpu import hip
from demodel_amd.engine.formats.compress import lz4_gpu
from demodel_amd.engine.formats import gguf

h = hip()
s = h.Stream(0)
rng = np.random.default_rng(5)
words = [f"w{i:04d}" for i in range(20000)]
idx = rng.integers(0, len(words), size=(1 << 20) // 6)
base = " ".join(words[i] for i in idx).encode()[:1 << 20]
comp = bytes(pa.Codec("lz4_raw").compress(base))
src = h.DeviceBuffer(len(comp))
carr = ctypes.create_string_buffer(comp, len(comp))
h.h2d_async(src.ptr, ctypes.addressof(carr), len(comp), s.handle)
s.sync()
n = 1024
dst = h.DeviceBuffer(n * len(base))
streams = [(src.ptr, len(comp), dst.ptr + i * len(base), len(base))
           for i in range(n)]
assert all(r.ok for r in lz4_gpu(streams))

# q5_K dequant under counters
bb = gguf.GGML_TYPES[13][2]
n_sup = (512 << 20) // bb
raw = rng.integers(0, 256, size=n_sup * bb, dtype=np.uint8).tobytes()
qsrc = h.DeviceBuffer(len(raw))
qarr = (ctypes.c_char * len(raw)).from_buffer_copy(raw)
h.h2d_async(qsrc.ptr, ctypes.addressof(qarr), len(raw), s.handle)
s.sync()
qdst = h.DeviceBuffer(n_sup * 256 * 2)
h.gguf_dequant(13, qsrc.ptr, qdst.ptr, n_sup, s.handle)
s.sync()
print("pmc probe done")
PYEOF
timeout 600 rocprofv3 --pmc SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT \
    SQ_WAVES -d gpurun_out/pmc8 --output-format csv -o newk \
    -- python /tmp/pmc_probe.py > gpurun_out/pmc8.log 2>&1
echo "pmc rc=$?"
find gpurun_out/pmc8 -name "*.csv" | head

# digest-verify scaling with shard count (more files = more parallel
# sha256 chains)
for sh in 4 8 16; do
  timeout 600 python bench.py --steps 2 --warmup 1 --verify digest \
      --shards $sh > gpurun_out/bench_digest_s$sh.json \
      2> gpurun_out/bench_digest_s$sh.log
  tail -1 gpurun_out/bench_digest_s$sh.json
done

echo DONE
