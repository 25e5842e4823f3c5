#!/bin/bash
# Round-2 GPU call 5: new dequant kernels (q4_1/q5_0/q5_1/q2_K/q3_K/q5_K)
# correctness vs CPU references + throughput probe.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_pipeline.py \
    -q -m gpu > gpurun_out/pytest_dequant.log 2>&1
echo "rc=$?" | tee -a gpurun_out/pytest_dequant.log
tail -3 gpurun_out/pytest_dequant.log

timeout 300 python - > gpurun_out/dequant_probe.log 2>&1 <<'EOF'
import ctypes, json, time
import numpy as np
from demodel_amd.engine.formats import gguf
from demodel_amd.gpu import hip

h = hip()
s = h.Stream(0)
rng = np.random.default_rng(3)
for qtype in (2, 3, 6, 7, 8, 10, 11, 12, 13, 14):
    name, be, bb = gguf.GGML_TYPES[qtype]
    n_sup = (1 << 30) // bb          # ~1 GiB of quant input
    raw = rng.integers(0, 256, size=n_sup * bb, dtype=np.uint8).tobytes()
    src = h.DeviceBuffer(len(raw))
    carr = (ctypes.c_char * len(raw)).from_buffer_copy(raw)
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(raw), s.handle)
    s.sync()
    dst = h.DeviceBuffer(n_sup * be * 2)
    h.gguf_dequant(qtype, src.ptr, dst.ptr, n_sup, s.handle)  # warm
    s.sync()
    t0 = time.perf_counter()
    for _ in range(3):
        h.gguf_dequant(qtype, src.ptr, dst.ptr, n_sup, s.handle)
    s.sync()
    dt = (time.perf_counter() - t0) / 3
    print(json.dumps({
        "op": f"dequant_{name}", "in_GBps": round(len(raw) / dt / 1e9, 1),
        "out_GBps": round(n_sup * be * 2 / dt / 1e9, 1),
        "ms": round(dt * 1e3, 2)}), flush=True)
    del src, dst, carr
EOF
cat gpurun_out/dequant_probe.log

echo DONE
