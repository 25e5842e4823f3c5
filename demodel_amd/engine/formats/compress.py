"""GPU decompression front-end: gzip containers + raw DEFLATE streams.

Drives csrc/inflate.hip (wave-per-stream DEFLATE) over batches of
compressed regions already resident in HBM: cached response bodies in
their original Content-Encoding (reference CONTRIBUTING.md:116), gzip
members of dataset streams, parquet page payloads.
"""

from __future__ import annotations

import ctypes
import struct
from dataclasses import dataclass

DESC_WORDS = 8  # u64s per InflateDesc

ERR = {0: "ok", -1: "magic", -2: "format", -3: "overflow",
       -4: "underrun", -5: "dictionary-unsupported"}


def gzip_deflate_offset(head: bytes) -> int:
    """Offset of the raw DEFLATE stream inside a gzip member (RFC 1952)."""
    if len(head) < 10 or head[:2] != b"\x1f\x8b":
        raise ValueError("not gzip (bad magic)")
    if head[2] != 8:
        raise ValueError(f"unsupported gzip method {head[2]}")
    flg = head[3]
    off = 10
    if flg & 0x04:  # FEXTRA
        xlen = struct.unpack_from("<H", head, off)[0]
        off += 2 + xlen
    if flg & 0x08:  # FNAME
        off = head.index(b"\0", off) + 1
    if flg & 0x10:  # FCOMMENT
        off = head.index(b"\0", off) + 1
    if flg & 0x02:  # FHCRC
        off += 2
    return off


@dataclass
class InflateResult:
    written: int
    status: int
    consumed: int

    @property
    def ok(self) -> bool:
        return self.status == 0

    @property
    def error(self) -> str | None:
        return None if self.ok else ERR.get(self.status, str(self.status))


def inflate_gpu(streams: list[tuple[int, int, int, int]],
                stream_handle=None) -> list[InflateResult]:
    """Inflate raw DEFLATE streams on the GPU.

    streams: (src_ptr, src_len, dst_ptr, dst_cap) device addresses.
    Returns per-stream InflateResult; raises on any failed stream only if
    the caller doesn't inspect (callers should check .ok).
    """
    from ...gpu import hip

    h = hip()
    n = len(streams)
    if n == 0:
        return []
    own = stream_handle is None
    s = h.Stream(0) if own else None
    handle = s.handle if own else stream_handle
    desc = bytearray(n * DESC_WORDS * 8)
    for i, (src, slen, dst, cap) in enumerate(streams):
        struct.pack_into("<8Q", desc, i * DESC_WORDS * 8,
                         src, slen, dst, cap, 0, 0, 0, 0)
    dbuf = h.DeviceBuffer(len(desc))
    carr = (ctypes.c_char * len(desc)).from_buffer(desc)
    h.h2d_async(dbuf.ptr, ctypes.addressof(carr), len(desc), handle)
    h.inflate_streams(dbuf.ptr, n, handle)
    h.d2h_async(ctypes.addressof(carr), dbuf.ptr, len(desc), handle)
    if own:
        s.sync()
    else:
        h.device_sync()
    out = []
    for i in range(n):
        vals = struct.unpack_from("<8Q", desc, i * DESC_WORDS * 8)
        written, status_u, consumed = vals[4], vals[5], vals[6]
        status = status_u - (1 << 64) if status_u >= (1 << 63) else status_u
        out.append(InflateResult(written=written, status=int(status),
                                 consumed=consumed))
    return out


ZSTD_WS_BYTES = 144 << 10


def _parse_results(desc: bytearray, n: int) -> list[InflateResult]:
    out = []
    for i in range(n):
        vals = struct.unpack_from("<8Q", desc, i * DESC_WORDS * 8)
        written, status_u, consumed = vals[4], vals[5], vals[6]
        status = status_u - (1 << 64) if status_u >= (1 << 63) else status_u
        out.append(InflateResult(written=written, status=int(status),
                                 consumed=consumed))
    return out


class ZstdJob:
    """An in-flight GPU decompression batch on its own HIP stream: the
    launch returns immediately so several batches (e.g. dataset shards
    landing at different times) decode CONCURRENTLY, which keeps wave
    occupancy high even when each batch alone has fewer frames than the
    chip has wave slots.  pre_launch(stream_handle), when given, queues
    extra async work (device copies) on the same stream before the
    kernels.  snappy_frames run through the snappy kernel on the same
    stream (parquet's default page codec); their results surface as
    .snappy_results after wait()."""

    def __init__(self, frames: list[tuple[int, int, int, int]],
                 pre_launch=None, window: int = 0,
                 snappy_frames: list[tuple[int, int, int, int]] | None
                 = None,
                 deflate_frames: list[tuple[int, int, int, int]] | None
                 = None,
                 lz4_frames: list[tuple[int, int, int, int]] | None
                 = None):
        from ...gpu import hip

        h = hip()
        n = self._n = len(frames)
        sn = self._sn = len(snappy_frames or ())
        dn = self._dn = len(deflate_frames or ())
        ln = self._ln = len(lz4_frames or ())
        self._s = h.Stream(0)
        if pre_launch is not None:
            pre_launch(self._s.handle)
        if dn:
            nd = dn * DESC_WORDS * 8
            self._dpin = h.PinnedPool(nd, 1)
            ddesc = self._dpin.slab_view(0)
            for i, (src, slen, dst, cap) in enumerate(deflate_frames):
                struct.pack_into("<8Q", ddesc, i * DESC_WORDS * 8,
                                 src, slen, dst, cap, 0, 0, 0, 0)
            self._ddbuf = h.DeviceBuffer(nd)
            addr = self._dpin.slab_ptr(0)
            h.h2d_async(self._ddbuf.ptr, addr, nd, self._s.handle)
            h.inflate_streams(self._ddbuf.ptr, dn, self._s.handle)
            h.d2h_async(addr, self._ddbuf.ptr, nd, self._s.handle)
        if ln:
            nd = ln * DESC_WORDS * 8
            self._lpin = h.PinnedPool(nd, 1)
            ldesc = self._lpin.slab_view(0)
            for i, (src, slen, dst, cap) in enumerate(lz4_frames):
                struct.pack_into("<8Q", ldesc, i * DESC_WORDS * 8,
                                 src, slen, dst, cap, 0, 0, 0, 0)
            self._ldbuf = h.DeviceBuffer(nd)
            addr = self._lpin.slab_ptr(0)
            h.h2d_async(self._ldbuf.ptr, addr, nd, self._s.handle)
            h.lz4_streams(self._ldbuf.ptr, ln, self._s.handle)
            h.d2h_async(addr, self._ldbuf.ptr, nd, self._s.handle)
        if sn:
            nd = sn * DESC_WORDS * 8
            self._spin = h.PinnedPool(nd, 1)
            sdesc = self._spin.slab_view(0)
            for i, (src, slen, dst, cap) in enumerate(snappy_frames):
                struct.pack_into("<8Q", sdesc, i * DESC_WORDS * 8,
                                 src, slen, dst, cap, 0, 0, 0, 0)
            self._sdbuf = h.DeviceBuffer(nd)
            addr = self._spin.slab_ptr(0)
            h.h2d_async(self._sdbuf.ptr, addr, nd, self._s.handle)
            h.snappy_streams(self._sdbuf.ptr, sn, self._s.handle)
            h.d2h_async(addr, self._sdbuf.ptr, nd, self._s.handle)
        if n:
            self._ws = h.DeviceBuffer(n * ZSTD_WS_BYTES)
            nd = n * DESC_WORDS * 8
            # PINNED staging for the descriptor block: hipMemcpyAsync
            # to/from pageable memory BLOCKS the host thread until every
            # prior op on the stream (the kernel!) completes, which would
            # make this "async" launch synchronous
            self._pin = h.PinnedPool(nd, 1)
            desc = self._pin.slab_view(0)
            for i, (src, slen, dst, cap) in enumerate(frames):
                struct.pack_into("<8Q", desc, i * DESC_WORDS * 8,
                                 src, slen, dst, cap, 0, 0, 0,
                                 self._ws.ptr + i * ZSTD_WS_BYTES)
            self._dbuf = h.DeviceBuffer(nd)
            addr = self._pin.slab_ptr(0)
            h.h2d_async(self._dbuf.ptr, addr, nd, self._s.handle)
            h.zstd_frames(self._dbuf.ptr, n, self._s.handle,
                          window=window)
            h.d2h_async(addr, self._dbuf.ptr, nd, self._s.handle)
        self._ev = h.Event()
        self._ev.record(self._s.handle)

    def done(self) -> bool:
        return self._ev.query()

    def wait(self) -> list[InflateResult]:
        if not hasattr(self, "_results"):
            self._s.sync()
            self._results = (_parse_results(
                bytearray(self._pin.slab_view(0)), self._n)
                if self._n else [])
            self.snappy_results = (_parse_results(
                bytearray(self._spin.slab_view(0)), self._sn)
                if self._sn else [])
            self.deflate_results = (_parse_results(
                bytearray(self._dpin.slab_view(0)), self._dn)
                if self._dn else [])
            self.lz4_results = (_parse_results(
                bytearray(self._lpin.slab_view(0)), self._ln)
                if self._ln else [])
        return self._results

    def view(self, lo: int, n: int, slo: int = 0, sn: int = 0,
             dlo: int = 0, dn: int = 0, llo: int = 0,
             ln: int = 0) -> "ZstdJobView":
        return ZstdJobView(self, lo, n, slo, sn, dlo, dn, llo, ln)


class ZstdJobView:
    """A frame-range of a (possibly shared) ZstdJob — several shards
    coalesced into ONE launch each hold a view of it.  wait() returns
    this view's zstd results followed by its snappy results."""

    def __init__(self, job: ZstdJob, lo: int, n: int, slo: int = 0,
                 sn: int = 0, dlo: int = 0, dn: int = 0,
                 llo: int = 0, ln: int = 0):
        self._job, self._lo, self._vn = job, lo, n
        self._slo, self._svn = slo, sn
        self._dlo, self._dvn = dlo, dn
        self._llo, self._lvn = llo, ln

    def done(self) -> bool:
        return self._job.done()

    def wait(self) -> list[InflateResult]:
        res = self._job.wait()[self._lo:self._lo + self._vn]
        if self._svn:
            res = res + self._job.snappy_results[
                self._slo:self._slo + self._svn]
        if self._dvn:
            res = res + self._job.deflate_results[
                self._dlo:self._dlo + self._dvn]
        if self._lvn:
            res = res + self._job.lz4_results[
                self._llo:self._llo + self._lvn]
        return res


def zstd_gpu(frames: list[tuple[int, int, int, int]],
             stream_handle=None, window: int = 0) -> list[InflateResult]:
    """Decompress zstd frames on the GPU (csrc/zstd_kernel.hip).

    frames: (src_ptr, src_len, dst_ptr, dst_cap) device addresses.
    A 144 KiB workspace per frame is allocated here.
    """
    from ...gpu import hip

    n = len(frames)
    if n == 0:
        return []
    if stream_handle is None:
        return ZstdJob(frames, window=window).wait()
    h = hip()
    ws = h.DeviceBuffer(n * ZSTD_WS_BYTES)
    desc = bytearray(n * DESC_WORDS * 8)
    for i, (src, slen, dst, cap) in enumerate(frames):
        struct.pack_into("<8Q", desc, i * DESC_WORDS * 8,
                         src, slen, dst, cap, 0, 0, 0,
                         ws.ptr + i * ZSTD_WS_BYTES)
    dbuf = h.DeviceBuffer(len(desc))
    carr = (ctypes.c_char * len(desc)).from_buffer(desc)
    h.h2d_async(dbuf.ptr, ctypes.addressof(carr), len(desc), stream_handle)
    h.zstd_frames(dbuf.ptr, n, stream_handle, window=window)
    h.d2h_async(ctypes.addressof(carr), dbuf.ptr, len(desc), stream_handle)
    h.device_sync()
    return _parse_results(desc, n)


def gunzip_blob_gpu(blob, out_size: int | None = None):
    """Decompress a gzip blob landed in HBM -> new DeviceBuffer.

    Uses the gzip ISIZE trailer for sizing (mod 2^32; capacity doubles on
    overflow).  Single-member fast path; multi-member walks `consumed`.
    """
    from ...gpu import hip

    h = hip()
    off = gzip_deflate_offset(blob.head)
    if out_size is None:
        tail = bytearray(8)
        addr = ctypes.addressof((ctypes.c_char * 8).from_buffer(tail))
        s = h.Stream(0)
        h.d2h_async(addr, blob.buffer.ptr + blob.nbytes - 8, 8, s.handle)
        s.sync()
        out_size = struct.unpack("<I", bytes(tail[4:8]))[0]
        if out_size == 0:
            out_size = 1
    cap = out_size
    for _ in range(3):
        dst = h.DeviceBuffer(cap)
        res = inflate_gpu([(blob.buffer.ptr + off,
                            blob.nbytes - off - 8, dst.ptr, cap)])[0]
        if res.ok:
            return dst, res
        if res.status == -3:  # overflow: ISIZE wrapped (>4 GiB payloads)
            cap *= 4
            continue
        raise IOError(f"GPU inflate failed: {res.error}")
    raise IOError("GPU inflate: capacity growth exhausted")


def lz4_gpu(streams: list[tuple[int, int, int, int]]
            ) -> list[InflateResult]:
    """Decompress raw LZ4 blocks on the GPU (csrc/lz4.hip) — parquet's
    LZ4/LZ4_RAW page codecs."""
    job = ZstdJob([], lz4_frames=streams)
    job.wait()
    return job.lz4_results


def snappy_gpu(streams: list[tuple[int, int, int, int]]
               ) -> list[InflateResult]:
    """Decompress raw snappy streams on the GPU (csrc/snappy.hip).

    streams: (src_ptr, src_len, dst_ptr, dst_cap) device addresses."""
    if not streams:
        return []
    job = ZstdJob([], snappy_frames=streams)
    job.wait()
    return job.snappy_results
