#!/bin/bash
# Round-2 GPU call 7: LZ4 kernel validation + parquet-lz4 bench.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/test_gpu_lz4.py tests/test_parquet.py \
    tests/test_datasets.py -q -m gpu > gpurun_out/pytest_lz4.log 2>&1
echo "rc=$?" | tee -a gpurun_out/pytest_lz4.log
tail -3 gpurun_out/pytest_lz4.log

timeout 420 python bench.py --model parquet --parquet-codec lz4 \
    --steps 3 --warmup 1 > gpurun_out/bench_pq_lz4.json \
    2> gpurun_out/bench_pq_lz4.log
tail -1 gpurun_out/bench_pq_lz4.json

# lz4 decode throughput probe (same harness as snappy)
timeout 300 python - > gpurun_out/lz4_probe.log 2>&1 <<'PYEOF'
import ctypes, json, time
import numpy as np
import pyarrow as pa
from demodel_amd.gpu import hip
from demodel_amd.engine.formats.compress import lz4_gpu

h = hip()
s = h.Stream(0)
rng = np.random.default_rng(5)
for payload in ("words", "text", "random"):
    if payload == "words":
        words = [f"w{i:04d}" for i in range(20000)]
        idx = rng.integers(0, len(words), size=(1 << 20) // 6)
        base = " ".join(words[i] for i in idx).encode()[:1 << 20]
    elif payload == "text":
        base = (b"some plainly compressible text payload flows here "
                * 200)[:64 << 10] * 16
    else:
        base = rng.integers(0, 256, size=1 << 20,
                            dtype=np.uint8).tobytes()
    comp = bytes(pa.Codec("lz4_raw").compress(base))
    src = h.DeviceBuffer(len(comp))
    carr = ctypes.create_string_buffer(comp, len(comp))
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(comp), s.handle)
    s.sync()
    n_streams = 2048
    dst = h.DeviceBuffer(n_streams * len(base))
    streams = [(src.ptr, len(comp), dst.ptr + i * len(base), len(base))
               for i in range(n_streams)]
    lz4_gpu(streams)  # warm
    t0 = time.perf_counter()
    results = lz4_gpu(streams)
    dt = time.perf_counter() - t0
    assert all(r.ok for r in results)
    print(json.dumps({"op": "lz4_decode", "payload": payload,
                      "streams": n_streams, "mib_each": 1,
                      "s": round(dt, 3),
                      "GBps_out": round(n_streams * len(base) / dt / 1e9, 2),
                      "ratio": round(len(base) / len(comp), 2)}),
          flush=True)
    del src, dst
PYEOF
cat gpurun_out/lz4_probe.log

echo DONE
