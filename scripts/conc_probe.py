"""Concurrent-zstd-launch probe: is N small launches on N streams as
fast as one big launch?  Separates three suspects: (a) kernel-side
concurrency across streams / HW queues, (b) hipMalloc/hipFree syncs in
the per-job path, (c) the streaming orchestration itself.

Run on a GPU box:  python scripts/conc_probe.py
"""

import ctypes
import struct
import sys
import time

sys.path.insert(0, ".")

import demodel_amd.gpu as g  # noqa: E402  (torch-first import)

h = g.hip()

N_FRAMES = 4096
FRAME_RAW = 1 << 20          # 1 MiB decompressed per frame
WS = 144 << 10
DESC_WORDS = 8


def make_frames():
    """One compressed word-salad frame, replicated."""
    import numpy as np
    import pyarrow as pa

    rng = np.random.default_rng(7)
    words = [f"w{i:04d}" for i in range(30000)]
    idx = rng.integers(0, len(words), size=FRAME_RAW // 6)
    raw = " ".join(words[i] for i in idx).encode()[:FRAME_RAW]
    comp = bytes(pa.Codec("zstd", compression_level=1).compress(raw))
    return comp, len(raw)


def main():
    comp, rawlen = make_frames()
    src = h.DeviceBuffer(len(comp))
    s0 = h.Stream(0)
    carr = ctypes.create_string_buffer(comp, len(comp))
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(comp), s0.handle)
    s0.sync()

    out = h.DeviceBuffer(N_FRAMES * rawlen)
    ws = h.DeviceBuffer(N_FRAMES * WS)

    def pack_desc(lo, n):
        desc = bytearray(n * DESC_WORDS * 8)
        for i in range(n):
            j = lo + i
            struct.pack_into("<8Q", desc, i * DESC_WORDS * 8,
                             src.ptr, len(comp), out.ptr + j * rawlen,
                             rawlen, 0, 0, 0, ws.ptr + j * WS)
        return desc

    def run_one_big():
        desc = pack_desc(0, N_FRAMES)
        dbuf = h.DeviceBuffer(len(desc))
        ca = (ctypes.c_char * len(desc)).from_buffer(desc)
        t0 = time.perf_counter()
        h.h2d_async(dbuf.ptr, ctypes.addressof(ca), len(desc), s0.handle)
        h.zstd_frames(dbuf.ptr, N_FRAMES, s0.handle, window=16 << 10)
        s0.sync()
        return time.perf_counter() - t0

    def run_split(n_jobs, prealloc, window=16 << 10):
        per = N_FRAMES // n_jobs
        streams = [h.Stream(0) for _ in range(n_jobs)]
        descs = [pack_desc(k * per, per) for k in range(n_jobs)]
        if prealloc:
            dbufs = [h.DeviceBuffer(len(d)) for d in descs]
        t0 = time.perf_counter()
        if not prealloc:
            dbufs = [h.DeviceBuffer(len(d)) for d in descs]
        cas = [(ctypes.c_char * len(d)).from_buffer(d) for d in descs]
        for k in range(n_jobs):
            h.h2d_async(dbufs[k].ptr, ctypes.addressof(cas[k]),
                        len(descs[k]), streams[k].handle)
            h.zstd_frames(dbufs[k].ptr, per, streams[k].handle,
                          window=window)
        for s in streams:
            s.sync()
        dt = time.perf_counter() - t0
        if not prealloc:
            t1 = time.perf_counter()
            del dbufs          # hipFree cost
            free_s = time.perf_counter() - t1
        else:
            free_s = 0.0
        return dt, free_s

    total_gb = N_FRAMES * rawlen / 1e9
    for name, fn in [
        ("one-big-4096", lambda: (run_one_big(), 0.0)),
        ("16x256-prealloc", lambda: run_split(16, True)),
        ("16x256-alloc-in-loop", lambda: run_split(16, False)),
        ("16x256-prealloc-64K", lambda: run_split(16, True, 64 << 10)),
        ("4x1024-prealloc", lambda: run_split(4, True)),
    ]:
        fn()  # warm
        dt, free_s = fn()
        print({"case": name, "s": round(dt, 3),
               "GBps_out": round(total_gb / dt, 2),
               "free_s": round(free_s, 3)}, flush=True)


if __name__ == "__main__":
    main()
