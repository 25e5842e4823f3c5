#!/usr/bin/env python3
"""Microbenchmarks of the landing-pipeline primitives on one MI355X:
H2D pinned copy, sha256_batch throughput (by chunk size), sha256 chain
rate, scatter bandwidth, dequant bandwidth.  Prints JSON lines."""

import ctypes
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

from demodel_amd.gpu import hip  # noqa: E402


def bench(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    return (time.perf_counter() - t0) / iters


def main():
    h = hip()
    h.set_device(0)
    s = h.Stream(0)
    N = 1 << 30  # 1 GiB

    pool = h.PinnedPool(N, 1)
    buf = h.DeviceBuffer(N)

    def h2d():
        h.h2d_async(buf.ptr, pool.slab_ptr(0), N, s.handle)
        s.sync()

    t = bench(h2d)
    print(json.dumps({"op": "h2d_pinned", "gib": 1, "s": round(t, 4),
                      "GBps": round(N / t / 1e9, 2)}), flush=True)

    for chunk in (16 << 10, 64 << 10, 256 << 10, 1 << 20):
        n_chunks = N // chunk
        dig = h.DeviceBuffer(n_chunks * 32)

        def hash_():
            h.sha256_batch(buf.ptr, N, chunk, dig.ptr, n_chunks, s.handle)
            s.sync()

        t = bench(hash_, iters=3)
        print(json.dumps({"op": "sha256_batch", "chunk_kib": chunk >> 10,
                          "s": round(t, 4),
                          "GBps": round(N / t / 1e9, 2)}), flush=True)

    # chain rate over 64 MiB
    M = 64 << 20
    state = h.DeviceBuffer(32)
    h.sha256_chain_init(state.ptr, s.handle)

    def chain():
        h.sha256_chain_update(state.ptr, buf.ptr, M // 64, s.handle)
        s.sync()

    t = bench(chain, iters=2, warmup=1)
    print(json.dumps({"op": "sha256_chain", "mib": 64, "s": round(t, 3),
                      "MBps": round(M / t / 1e6, 1)}), flush=True)

    # scatter: 1 GiB in 4 MiB ranges to a second buffer
    dst = h.DeviceBuffer(N)
    n_desc = N // (4 << 20)
    desc = bytearray()
    import struct

    for i in range(n_desc):
        desc += struct.pack("<4Q", i * (4 << 20), dst.ptr + i * (4 << 20),
                            4 << 20, 0)
    dbuf = h.DeviceBuffer(len(desc))
    src_c = (ctypes.c_char * len(desc)).from_buffer(desc)
    h.h2d_async(dbuf.ptr, ctypes.addressof(src_c), len(desc), s.handle)
    s.sync()

    def scat():
        h.scatter_ranges(buf.ptr, dbuf.ptr, n_desc, s.handle)
        s.sync()

    t = bench(scat)
    print(json.dumps({"op": "scatter_4mib_ranges", "s": round(t, 4),
                      "GBps_moved": round(2 * N / t / 1e9, 2)}), flush=True)

    # q4_K dequant bandwidth: 1 GiB of superblocks
    n_sb = N // 144
    out = h.DeviceBuffer(n_sb * 256 * 2)

    def dq():
        h.gguf_dequant(12, buf.ptr, out.ptr, n_sb, s.handle)
        s.sync()

    t = bench(dq)
    print(json.dumps({"op": "dequant_q4K", "in_gib": 1, "s": round(t, 4),
                      "GBps_in": round(N / t / 1e9, 2),
                      "GBps_out": round(n_sb * 512 / t / 1e9, 2)}),
          flush=True)


def inflate_bench():
    import zlib

    from demodel_amd.engine.formats.compress import inflate_gpu
    from demodel_amd.gpu import hip

    h = hip()
    s = h.Stream(0)
    # 256 streams x 4 MiB of mixed text/random payload
    base = (b"some plainly compressible text payload " * 1000
            + os.urandom(1 << 20))
    data = (base * ((4 << 20) // len(base) + 1))[:(4 << 20)]
    comp = zlib.compressobj(6, zlib.DEFLATED, -15)
    blob = comp.compress(data) + comp.flush()
    import ctypes

    n_streams = 256
    src = h.DeviceBuffer(len(blob))
    carr = (ctypes.c_char * len(blob)).from_buffer_copy(blob)
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(blob), s.handle)
    s.sync()
    dsts = [h.DeviceBuffer(len(data)) for _ in range(n_streams)]
    streams = [(src.ptr, len(blob), d.ptr, len(data)) for d in dsts]

    t = bench(lambda: inflate_gpu(streams), iters=2, warmup=1)
    out_bytes = len(data) * n_streams
    print(json.dumps({"op": "inflate_deflate", "streams": n_streams,
                      "mib_out_each": len(data) >> 20, "s": round(t, 3),
                      "GBps_out": round(out_bytes / t / 1e9, 2),
                      "ratio": round(len(data) / len(blob), 2)}),
          flush=True)


def zstd_bench(n_frames=2048, frame_mb=2, payload="text"):
    import ctypes

    import pyarrow as pa

    from demodel_amd.engine.formats.compress import zstd_gpu
    from demodel_amd.gpu import hip

    h = hip()
    s = h.Stream(0)
    n = frame_mb << 20
    if payload == "text":
        base = (b"some plainly compressible text payload flows here " * 200
                + os.urandom(1 << 14))
    elif payload == "words":
        import numpy as np

        rng = np.random.default_rng(5)
        words = [f"w{i:04d}" for i in range(20000)]
        idx = rng.integers(0, len(words), size=(2 << 20) // 6)
        base = " ".join(words[i] for i in idx).encode()
    else:
        base = os.urandom(1 << 20)
    data = (base * (n // len(base) + 1))[:n]
    frame = bytes(pa.Codec("zstd", compression_level=1).compress(data))
    src = h.DeviceBuffer(len(frame))
    carr = (ctypes.c_char * len(frame)).from_buffer_copy(frame)
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(frame), s.handle)
    s.sync()
    ring = h.DeviceBuffer(n_frames * n)
    frames = [(src.ptr, len(frame), ring.ptr + i * n, n)
              for i in range(n_frames)]

    t = bench(lambda: zstd_gpu(frames), iters=2, warmup=1)
    print(json.dumps({"op": "zstd_decode", "payload": payload,
                      "frames": n_frames, "mib_each": frame_mb,
                      "s": round(t, 3),
                      "GBps_out": round(n_frames * n / t / 1e9, 2),
                      "MBps_per_wave": round(n / t / 1e6, 2),
                      "ratio": round(n / len(frame), 2)}), flush=True)


if __name__ == "__main__":
    main()
    inflate_bench()
    zstd_bench()
    zstd_bench(n_frames=256)
    zstd_bench(payload="random")
