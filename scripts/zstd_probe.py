"""Per-payload zstd decode probe (words/text/random) — quick A/B.

Run from anywhere: python scripts/zstd_probe.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from gpu_probe import zstd_bench  # noqa: E402



def snappy_bench(n_streams=2048, payload="words"):
    import ctypes
    import json
    import time

    import numpy as np
    import pyarrow as pa

    import demodel_amd.gpu as g
    from demodel_amd.engine.formats.compress import snappy_gpu

    h = g.hip()
    rng = np.random.default_rng(5)
    if payload == "words":
        words = [f"w{i:04d}" for i in range(20000)]
        idx = rng.integers(0, len(words), size=(1 << 20) // 6)
        base = " ".join(words[i] for i in idx).encode()[:1 << 20]
    elif payload == "text":
        base = (b"some plainly compressible text payload flows here " * 200
                )[:64 << 10] * 16
    else:
        base = rng.integers(0, 256, size=1 << 20,
                            dtype=np.uint8).tobytes()
    comp = bytes(pa.Codec("snappy").compress(base))
    src = h.DeviceBuffer(len(comp))
    s = h.Stream(0)
    carr = ctypes.create_string_buffer(comp, len(comp))
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(comp), s.handle)
    s.sync()
    dst = h.DeviceBuffer(n_streams * len(base))
    streams = [(src.ptr, len(comp), dst.ptr + i * len(base), len(base))
               for i in range(n_streams)]
    snappy_gpu(streams)  # warm
    t0 = time.perf_counter()
    results = snappy_gpu(streams)
    dt = time.perf_counter() - t0
    assert all(r.ok for r in results)
    out = n_streams * len(base)
    print(json.dumps({"op": "snappy_decode", "payload": payload,
                      "streams": n_streams, "mib_each": 1,
                      "s": round(dt, 3),
                      "GBps_out": round(out / dt / 1e9, 2),
                      "MBps_per_wave": round(len(base) / dt / 1e6, 2),
                      "ratio": round(len(base) / len(comp), 2)}),
          flush=True)


if __name__ == "__main__":
    for payload in ("words", "text", "random"):
        zstd_bench(payload=payload)
    for payload in ("words", "text", "random"):
        snappy_bench(payload=payload)
