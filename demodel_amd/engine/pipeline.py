"""The HBM landing pipeline (SURVEY.md §2.3 K6 + K1).

Replaces the reference proxy's hot byte loop (goproxy io.Copy,
SURVEY.md §3.2) with the MI355X path:

    socket/file ──drain (no GIL)──▶ pinned ring slab ──hipMemcpyAsync──▶
      HBM blob buffer ──sha256_batch (GPU, chunk-parallel)──▶ verified

* The ring has N slabs; slab reuse waits on that slab's H2D event, so
  network receive of chunk k+N overlaps the copy of chunk k.
* Verification is chunk-parallel on the GPU (64 KiB sub-chunks by
  default) — digests land in the cache sidecar and are compared on
  replay.  An exact whole-blob sha256 (upstream etag/digest identity) is
  available as `host_chain=True` (hashlib pipelined on the fill thread) or
  `gpu_chain=True` (on-device sequential chain kernel, honest-but-slow;
  SURVEY.md §7 hard part (b)).
* On CPU-only machines HostLander lands into page-cache memory with the
  same interface, so the whole engine is testable without a GPU.
"""

from __future__ import annotations

import hashlib
import struct
import time
from dataclasses import dataclass, field

from ..gpu import have_gpu, hip

VERIFY_CHUNK = 64 << 10


@dataclass
class LandedBlob:
    nbytes: int
    device: str
    buffer: object                    # _hip.DeviceBuffer | bytearray
    digest_blob: bytes = b""          # n_chunks x 32B raw sha256 digests
    verify_chunk: int = VERIFY_CHUNK
    sha256: str | None = None         # exact whole-blob digest if computed
    head: bytes = b""                 # first bytes (for header parsing)
    timings: dict = field(default_factory=dict)
    shared: bool = False              # registry-owned: recycle must skip

    @property
    def chunk_digests(self) -> list[str]:
        return [self.digest_blob[i:i + 32].hex()
                for i in range(0, len(self.digest_blob), 32)]

    def torch_u8(self):
        """The landed bytes as a zero-copy 1-D torch.uint8 tensor."""
        import torch

        if self.device == "cpu":
            return torch.frombuffer(self.buffer, dtype=torch.uint8)
        return torch.from_dlpack(self.buffer.to_dlpack())


class LandingError(IOError):
    """A fill source died mid-landing; `landed` bytes (from the segment's
    start) are safely in HBM — a resume can continue from there with a
    Range request."""

    def __init__(self, landed: int, cause: BaseException | None = None):
        self.landed = landed
        self.cause = cause
        super().__init__(f"landing interrupted after {landed} bytes"
                         + (f": {cause!r}" if cause else ""))


class DigestMismatch(IOError):
    def __init__(self, chunk_index: int, verify_chunk: int):
        self.chunk_index = chunk_index
        super().__init__(
            f"chunk digest mismatch at chunk {chunk_index} "
            f"(byte offset {chunk_index * verify_chunk})")


def check_digests(got: bytes, expected: bytes, verify_chunk: int) -> None:
    if got == expected:
        return
    if len(got) != len(expected):
        raise IOError(
            f"digest count mismatch: {len(got) // 32} vs "
            f"{len(expected) // 32} chunks")
    for i in range(0, len(got), 32):
        if got[i:i + 32] != expected[i:i + 32]:
            raise DigestMismatch(i // 32, verify_chunk)


class BufferPool:
    """Exact-size HBM buffer recycling.

    Freeing a large DeviceBuffer and re-allocating it is NOT cheap when
    the device is near capacity: the driver defers reclaim of the freed
    pages, and the next big hipMalloc pays it (measured: 16x 8.8 GB
    alloc 0.07 s cold, 3.96 s RE-alloc after a free at ~2x141 GB of
    traffic on a 288 GB device).  Steady-state repeated pulls of the
    same model (benchmarks, refresh loops) hand buffers back here and
    take them again for free.  Caller contract: recycle only buffers
    with no outstanding views (DLPack tensors)."""

    def __init__(self):
        import threading

        self._lock = threading.Lock()
        self._free: dict[int, list] = {}

    def take(self, nbytes: int):
        with self._lock:
            lst = self._free.get(nbytes)
            if lst:
                return lst.pop()
        return None

    def put(self, buf, nbytes: int) -> None:
        with self._lock:
            self._free.setdefault(nbytes, []).append(buf)

    def clear(self) -> None:
        with self._lock:
            self._free.clear()


class Lander:
    """GPU landing pipeline; one instance per device + stream pair."""

    def __init__(self, device_index: int = 0, slab_bytes: int = 32 << 20,
                 n_slabs: int = 4, verify_chunk: int = VERIFY_CHUNK,
                 head_bytes: int = 8 << 20, buffer_pool=None):
        self._h = hip()
        self._h.set_device(device_index)
        self.device_index = device_index
        self.slab_bytes = slab_bytes
        self.verify_chunk = verify_chunk
        self.head_bytes = head_bytes
        self.pool = self._h.PinnedPool(slab_bytes, n_slabs)
        self.buffer_pool = buffer_pool
        self.copy_stream = self._h.Stream(0)
        self.verify_stream = self._h.Stream(0)
        self._slab_events = [self._h.Event() for _ in range(n_slabs)]
        self._slab_busy = [False] * n_slabs

    def land(self, fill, nbytes: int, verify: bool = True,
             host_chain: bool = False, gpu_chain: bool = False,
             keep_head: bool = True,
             expected_digests: bytes | None = None,
             verify_chunk: int | None = None) -> LandedBlob:
        """Land exactly `nbytes` from `fill` into a fresh HBM buffer.

        fill(view: memoryview) -> int: write up to len(view) bytes into the
        pinned slab view, return bytes written (0 = EOF/underrun -> error).
        expected_digests: raw 32B-per-chunk sha256s to verify against
        (raises DigestMismatch).
        """
        h = self._h
        t0 = time.perf_counter()
        buf = self.alloc(nbytes)
        gpu_state = None
        if gpu_chain:
            gpu_state = h.DeviceBuffer(32)
            h.sha256_chain_init(gpu_state.ptr, self.verify_stream.handle)
        chain = hashlib.sha256() if host_chain else None
        head, fill_s = self.land_into(
            buf, 0, fill, nbytes, file_size=nbytes, chain=chain,
            gpu_state=gpu_state, keep_head=keep_head)
        vc = verify_chunk or self.verify_chunk
        blob = LandedBlob(nbytes=nbytes,
                          device=f"cuda:{self.device_index}", buffer=buf,
                          verify_chunk=vc, head=bytes(head))
        t_fill_done = time.perf_counter()

        if (verify or expected_digests is not None) and nbytes > 0:
            blob.digest_blob = self._gpu_chunk_digests(buf, nbytes, vc)
            if expected_digests is not None:
                check_digests(blob.digest_blob, expected_digests, vc)
        else:
            self.copy_stream.sync()

        if chain is not None:
            blob.sha256 = chain.hexdigest()
        elif gpu_state is not None:
            blob.sha256 = self._finalize_gpu_chain(gpu_state, buf, nbytes)

        t1 = time.perf_counter()
        blob.timings = {
            "total_s": t1 - t0,
            "fill_s": fill_s,
            "land_s": t_fill_done - t0,
            "verify_s": t1 - t_fill_done,
            "gbps": nbytes / max(t1 - t0, 1e-9) / 1e9,
        }
        return blob

    def land_into(self, buf, base_off: int, fill, nbytes: int,
                  file_size: int | None = None, chain=None, gpu_state=None,
                  keep_head: bool = False) -> tuple[bytearray, float]:
        """Land `nbytes` from `fill` into buf at byte offset `base_off`
        through this lander's pinned ring.  Segment primitive: several
        landers (one per worker thread) can fill disjoint ranges of one
        DeviceBuffer concurrently — range-parallel pulls of a single blob.

        Returns (head_bytes, fill_seconds)."""
        h = self._h
        head = bytearray()
        n_slabs = self.pool.n_slabs
        file_size = file_size if file_size is not None else base_off + nbytes
        off = 0
        i = 0
        fill_s = 0.0
        # Exact-digest mode: the sequential host SHA-256 chain runs on a
        # DEDICATED thread, gated per slab, so network receive of slab
        # k+1 overlaps hashing of slab k (hashlib releases the GIL on
        # >2 KiB updates).  Round 1 hashed inline on the fill thread,
        # serializing recv with the chain.
        hasher = hq = hash_done = None
        if chain is not None:
            import queue as _q
            import threading as _t

            hq = _q.Queue()
            hash_done = [_t.Event() for _ in range(n_slabs)]
            for e in hash_done:
                e.set()

            def _hash_loop():
                while True:
                    item = hq.get()
                    if item is None:
                        return
                    s_i, v = item
                    chain.update(v)
                    hash_done[s_i].set()

            hasher = _t.Thread(target=_hash_loop, daemon=True,
                               name="sha256-chain")
            hasher.start()
        try:
            while off < nbytes:
                slab = i % n_slabs
                if self._slab_busy[slab]:
                    self._slab_events[slab].sync()
                if hash_done is not None:
                    hash_done[slab].wait()
                want = min(self.slab_bytes, nbytes - off)
                view = self.pool.slab_view(slab)[:want]
                tf = time.perf_counter()
                got = 0
                while got < want:
                    try:
                        n = fill(view[got:])
                    except Exception as e:
                        raise LandingError(off, e) from e
                    if n <= 0:
                        raise LandingError(off)
                    got += n
                fill_s += time.perf_counter() - tf
                if keep_head and len(head) < self.head_bytes:
                    take = min(want, self.head_bytes - len(head))
                    head += bytes(view[:take])
                if chain is not None:
                    hash_done[slab].clear()
                    hq.put((slab, view))
                h.h2d_async(buf.ptr + base_off + off,
                            self.pool.slab_ptr(slab),
                            want, self.copy_stream.handle)
                self._slab_events[slab].record(self.copy_stream.handle)
                self._slab_busy[slab] = True
                if gpu_state is not None:
                    # chain over the landed region (whole 64B blocks only;
                    # the ragged tail is folded in at finalize)
                    self._slab_events[slab].wait(self.verify_stream.handle)
                    abs_off = base_off + off
                    nblk = (want if abs_off + want < file_size
                            else file_size - (file_size % 64) - abs_off
                            ) // 64
                    h.sha256_chain_update(gpu_state.ptr, buf.ptr + abs_off,
                                          max(nblk, 0),
                                          self.verify_stream.handle)
                off += want
                i += 1
        finally:
            if hasher is not None:
                # drain: completed slabs hash before we return/raise, so
                # the chain state always equals the landed byte count
                hq.put(None)
                hasher.join()
        return head, fill_s

    def sync(self) -> None:
        self.copy_stream.sync()
        self.verify_stream.sync()

    def read_head(self, buf, nbytes: int) -> bytes:
        """First min(nbytes, head_bytes) bytes of a landed buffer, read
        back D2H — the authoritative head after Range-resumed landings
        (a resumed stream starts past byte 0, so the head captured
        in-flight is incomplete)."""
        import ctypes

        n = min(nbytes, self.head_bytes)
        if n <= 0:
            return b""
        out = bytearray(n)
        addr = ctypes.addressof((ctypes.c_char * n).from_buffer(out))
        self.copy_stream.sync()
        self._h.d2h_async(addr, buf.ptr, n, self.copy_stream.handle)
        self.copy_stream.sync()
        return bytes(out)

    def alloc(self, nbytes: int):
        if self.buffer_pool is not None:
            buf = self.buffer_pool.take(max(nbytes, 1))
            if buf is not None:
                return buf
        return self._h.DeviceBuffer(max(nbytes, 1))

    def finish_verify(self, buf, nbytes: int,
                      verify_chunk: int | None = None) -> bytes:
        return self._gpu_chunk_digests(buf, nbytes, verify_chunk)

    def collect_digests(self, dig_dev, n_chunks: int, events) -> bytes:
        """Wait the given hash events on this verify stream, read the
        digest array back, and return canonical sha256 bytes."""
        import ctypes

        import numpy as np

        for e in events:
            e.wait(self.verify_stream.handle)
        host = bytearray(n_chunks * 32)
        addr = ctypes.addressof(
            (ctypes.c_char * len(host)).from_buffer(host))
        self._h.d2h_async(addr, dig_dev.ptr, n_chunks * 32,
                          self.verify_stream.handle)
        self.verify_stream.sync()
        return np.frombuffer(bytes(host), dtype="<u4").byteswap().tobytes()

    def hash_range_into(self, buf, lo: int, hi: int, vc: int, dig_dev,
                        file_size: int):
        """Launch chunk digests for the chunk-ALIGNED part of [lo, hi)
        into dig_dev (32 B per chunk, indexed by absolute chunk id), on
        this lander's verify stream, ordered after its copies.

        Incremental verification: a segment hashes WHILE other segments
        still download, so the end-of-blob verify tail shrinks to the
        few boundary chunks + one D2H (round-1 hashed the whole buffer
        after the last byte landed).  Returns (event, lo_chunk,
        hi_chunk) or None when no full chunk lies inside."""
        h = self._h
        lo_c = (lo + vc - 1) // vc
        hi_c = (file_size + vc - 1) // vc if hi >= file_size else hi // vc
        if hi_c <= lo_c:
            return None
        done = h.Event()
        done.record(self.copy_stream.handle)
        done.wait(self.verify_stream.handle)
        span = min(file_size, hi_c * vc) - lo_c * vc
        h.sha256_batch(buf.ptr + lo_c * vc, span, vc,
                       dig_dev.ptr + lo_c * 32, hi_c - lo_c,
                       self.verify_stream.handle)
        ev = h.Event()
        ev.record(self.verify_stream.handle)
        return ev, lo_c, hi_c

    def _gpu_chunk_digests(self, buf, nbytes: int,
                           verify_chunk: int | None = None) -> bytes:
        h = self._h
        vc = verify_chunk or self.verify_chunk
        n_chunks = (nbytes + vc - 1) // vc
        dig_dev = h.DeviceBuffer(n_chunks * 32)
        # hash must see completed copies
        done = h.Event()
        done.record(self.copy_stream.handle)
        done.wait(self.verify_stream.handle)
        h.sha256_batch(buf.ptr, nbytes, vc, dig_dev.ptr,
                       n_chunks, self.verify_stream.handle)
        host = bytearray(n_chunks * 32)
        import ctypes

        addr = ctypes.addressof(
            (ctypes.c_char * len(host)).from_buffer(host))
        h.d2h_async(addr, dig_dev.ptr, n_chunks * 32,
                    self.verify_stream.handle)
        self.verify_stream.sync()
        # kernel writes big-endian words as native u32 -> swap to the
        # canonical sha256 byte order
        import numpy as np

        return np.frombuffer(bytes(host), dtype="<u4").byteswap().tobytes()

    def _finalize_gpu_chain(self, state_buf, buf, nbytes: int) -> str:
        """Fold the ragged tail + padding into the device chain state."""
        h = self._h
        import ctypes

        raw = bytearray(32)
        addr = ctypes.addressof((ctypes.c_char * 32).from_buffer(raw))
        self.verify_stream.sync()
        h.d2h_async(addr, state_buf.ptr, 32, self.verify_stream.handle)
        tail_off = nbytes - (nbytes % 64)
        tail = bytearray(nbytes % 64)
        if tail:
            taddr = ctypes.addressof(
                (ctypes.c_char * len(tail)).from_buffer(tail))
            h.d2h_async(taddr, buf.ptr + tail_off, len(tail),
                        self.verify_stream.handle)
        self.verify_stream.sync()
        state = list(struct.unpack("<8I", raw))
        return _host_sha256_finish(state, nbytes, bytes(tail))


# --- pure-python SHA-256 tail finisher (for the GPU chain) ---------------

_K = [
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2,
]


def _compress(state: list[int], block: bytes) -> None:
    w = list(struct.unpack(">16I", block))
    for t in range(16, 64):
        w15, w2 = w[t - 15], w[t - 2]
        s0 = (_rotr(w15, 7) ^ _rotr(w15, 18) ^ (w15 >> 3))
        s1 = (_rotr(w2, 17) ^ _rotr(w2, 19) ^ (w2 >> 10))
        w.append((w[t - 16] + s0 + w[t - 7] + s1) & 0xffffffff)
    a, b, c, d, e, f, g, hh = state
    for t in range(64):
        S1 = _rotr(e, 6) ^ _rotr(e, 11) ^ _rotr(e, 25)
        ch = (e & f) ^ (~e & g) & 0xffffffff
        t1 = (hh + S1 + ch + _K[t] + w[t]) & 0xffffffff
        S0 = _rotr(a, 2) ^ _rotr(a, 13) ^ _rotr(a, 22)
        maj = (a & b) ^ (a & c) ^ (b & c)
        t2 = (S0 + maj) & 0xffffffff
        hh, g, f, e = g, f, e, (d + t1) & 0xffffffff
        d, c, b, a = c, b, a, (t1 + t2) & 0xffffffff
    for i, v in enumerate((a, b, c, d, e, f, g, hh)):
        state[i] = (state[i] + v) & 0xffffffff


def _rotr(x: int, n: int) -> int:
    return ((x >> n) | (x << (32 - n))) & 0xffffffff


def _host_sha256_finish(state: list[int], total_len: int,
                        tail: bytes) -> str:
    pad = bytearray(tail)
    pad.append(0x80)
    while (len(pad) % 64) != 56:
        pad.append(0)
    pad += struct.pack(">Q", total_len * 8)
    for i in range(0, len(pad), 64):
        _compress(state, bytes(pad[i:i + 64]))
    return "".join(f"{w:08x}" for w in state)


# --- CPU fallback landing target (tests, no-GPU plumbing config) ---------

class HostLander:
    def __init__(self, slab_bytes: int = 32 << 20,
                 verify_chunk: int = VERIFY_CHUNK, head_bytes: int = 8 << 20):
        self.slab_bytes = slab_bytes
        self.verify_chunk = verify_chunk
        self.head_bytes = head_bytes

    def alloc(self, nbytes: int):
        return bytearray(nbytes)

    def land_into(self, buf, base_off: int, fill, nbytes: int,
                  file_size: int | None = None, chain=None, gpu_state=None,
                  keep_head: bool = False) -> tuple[bytearray, float]:
        mv = memoryview(buf)
        off = 0
        t0 = time.perf_counter()
        while off < nbytes:
            want = min(self.slab_bytes, nbytes - off)
            got = 0
            while got < want:
                lo = base_off + off + got
                try:
                    n = fill(mv[lo:base_off + off + want])
                except Exception as e:
                    raise LandingError(off, e) from e
                if n <= 0:
                    raise LandingError(off)
                got += n
            if chain is not None:
                chain.update(mv[base_off + off:base_off + off + want])
            off += want
        head = bytearray(
            mv[base_off:base_off + min(nbytes, self.head_bytes)]) \
            if keep_head else bytearray()
        return head, time.perf_counter() - t0

    def finish_verify(self, buf, nbytes: int,
                      verify_chunk: int | None = None) -> bytes:
        vc = verify_chunk or self.verify_chunk
        mv = memoryview(buf)
        return b"".join(
            hashlib.sha256(mv[o:o + vc]).digest()
            for o in range(0, nbytes, vc))

    def read_head(self, buf, nbytes: int) -> bytes:
        return bytes(memoryview(buf)[:min(nbytes, self.head_bytes)])

    def sync(self) -> None:
        pass

    def land(self, fill, nbytes: int, verify: bool = True,
             host_chain: bool = False, gpu_chain: bool = False,
             keep_head: bool = True,
             expected_digests: bytes | None = None,
             verify_chunk: int | None = None) -> LandedBlob:
        t0 = time.perf_counter()
        buf = bytearray(nbytes)
        mv = memoryview(buf)
        chain = hashlib.sha256() if (host_chain or gpu_chain) else None
        off = 0
        while off < nbytes:
            want = min(self.slab_bytes, nbytes - off)
            got = 0
            while got < want:
                n = fill(mv[off + got: off + want])
                if n <= 0:
                    raise IOError(
                        f"blob underrun at {off + got}/{nbytes} bytes")
                got += n
            if chain is not None:
                chain.update(mv[off:off + want])
            off += want
        vc = verify_chunk or self.verify_chunk
        blob = LandedBlob(nbytes=nbytes, device="cpu", buffer=buf,
                          verify_chunk=vc,
                          head=bytes(buf[:min(nbytes, self.head_bytes)]))
        t_land = time.perf_counter()
        if verify or expected_digests is not None:
            blob.digest_blob = b"".join(
                hashlib.sha256(mv[o:o + vc]).digest()
                for o in range(0, nbytes, vc))
            if expected_digests is not None:
                check_digests(blob.digest_blob, expected_digests, vc)
        if chain is not None:
            blob.sha256 = chain.hexdigest()
        t1 = time.perf_counter()
        blob.timings = {
            "total_s": t1 - t0, "fill_s": t_land - t0,
            "land_s": t_land - t0, "verify_s": t1 - t_land,
            "gbps": nbytes / max(t1 - t0, 1e-9) / 1e9,
        }
        return blob


def make_lander(device_index: int = 0, gpu: bool | None = None, **kw):
    """gpu=True forces HBM landing (raises without a device), gpu=False
    forces host RAM, None auto-selects."""
    if gpu is None:
        gpu = have_gpu()
    elif gpu and not have_gpu():
        raise RuntimeError("GPU landing requested but no AMD GPU present")
    if gpu:
        return Lander(device_index=device_index, **kw)
    return HostLander(**{k: v for k, v in kw.items()
                         if k in ("slab_bytes", "verify_chunk",
                                  "head_bytes")})  # n_slabs: GPU-only
