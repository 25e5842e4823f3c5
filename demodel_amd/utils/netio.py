"""Async-friendly high-throughput socket IO helpers.

sendfile_threaded: a blocking os.sendfile loop on a worker thread, used
for blob bodies by both the proxy's cache-hit serving and the test/bench
origin.  asyncio's loop.sendfile serializes all streams through the event
loop thread and tops out ~3x lower at 4+ parallel blob streams
(scripts/net_probe.py); a thread per in-flight body keeps every byte in
kernel space and scales with cores.
"""

from __future__ import annotations

import asyncio
import concurrent.futures as cf
import os
import select as sel

_POOL: cf.ThreadPoolExecutor | None = None


def _pool() -> cf.ThreadPoolExecutor:
    global _POOL
    if _POOL is None:
        # sized for the worst co-residency: origin sendfile bodies +
        # proxy relay pumps + cache-writer threads can all be in flight
        # at once in one process (the bench runs all three); threads
        # here spend their lives in syscalls, so over-provisioning is
        # cheap and starvation is not
        _POOL = cf.ThreadPoolExecutor(max_workers=96,
                                      thread_name_prefix="netio")
    return _POOL


_MEMFDS: dict[int, int] = {}


def _pattern_memfd(pattern: bytes) -> int | None:
    """A memfd holding `pattern` (created once per pattern object):
    sendfile from it serves tiled virtual blobs with ZERO user-space
    copies — a plain sock.send loop pays a user->kernel copy per byte
    and capped the 141 GB bench ~27 GB/s vs sendfile's 39."""
    key = id(pattern)
    fd = _MEMFDS.get(key)
    if fd is not None:
        return fd
    try:
        fd = os.memfd_create("demodel-virtual-pattern")
        off = 0
        while off < len(pattern):
            off += os.pwrite(fd, pattern[off:off + (8 << 20)], off)
    except (AttributeError, OSError):
        return None
    _MEMFDS[key] = fd
    return fd


async def send_pattern_threaded(writer: asyncio.StreamWriter,
                                pattern: bytes, start: int,
                                length: int) -> None:
    """Send `length` bytes of an infinitely-tiled `pattern` beginning at
    absolute offset `start`, on a worker thread (virtual blobs:
    benchmarks larger than the disk).  Uses sendfile from a pattern
    memfd (page-cache -> socket, no user-space copy); falls back to a
    sock.send loop."""
    sock = writer.transport.get_extra_info("socket")
    tls = writer.transport.get_extra_info("sslcontext")
    if sock is None or tls is not None:
        raise NotImplementedError
    await writer.drain()
    writer.transport.pause_reading()
    mv = memoryview(pattern)
    plen = len(pattern)
    memfd = _pattern_memfd(pattern)
    out_fd = sock.fileno()

    def run_sendfile():
        sent_total = 0
        while sent_total < length:
            phase = (start + sent_total) % plen
            want = min(plen - phase, length - sent_total)
            try:
                n = os.sendfile(out_fd, memfd, phase, want)
            except BlockingIOError:
                sel.select([], [out_fd], [], 10)
                continue
            if n == 0:
                raise ConnectionResetError("peer went away")
            sent_total += n

    def run_send():
        sent_total = 0
        while sent_total < length:
            phase = (start + sent_total) % plen
            piece = mv[phase:min(plen, phase + (length - sent_total))]
            off = 0
            while off < len(piece):
                try:
                    n = sock.send(piece[off:])
                except BlockingIOError:
                    sel.select([], [sock.fileno()], [], 10)
                    continue
                if n == 0:
                    raise ConnectionResetError("peer went away")
                off += n
            sent_total += len(piece)

    loop = asyncio.get_running_loop()
    try:
        await loop.run_in_executor(
            _pool(), run_sendfile if memfd is not None else run_send)
    finally:
        try:
            writer.transport.resume_reading()
        except Exception:
            pass


def _sendall_nb(fd: int, view) -> None:
    """sendall on a non-blocking socket fd (select-backed)."""
    off = 0
    while off < len(view):
        try:
            n = os.write(fd, view[off:])
        except BlockingIOError:
            sel.select([], [fd], [], 10)
            continue
        if n == 0:
            raise ConnectionResetError("peer went away")
        off += n


async def relay_body_threaded(up_r: asyncio.StreamReader,
                              up_w: asyncio.StreamWriter,
                              writer: asyncio.StreamWriter,
                              length: int, tee=None) -> None:
    """Relay a length-delimited body upstream->client on a worker
    thread — the proxy MISS hot loop (reference: goproxy's io.Copy,
    start.go:201-204).  The asyncio relay pays event-loop wakeups and
    userspace copies per 256 KiB chunk and tops out ~3 GB/s aggregate;
    this path moves the socket->socket pump off-loop (os.splice
    kernel-side when there is no cache tee, else an 8 MiB recv/send
    loop) and scales with cores like the sendfile HIT path.

    Raises NotImplementedError (before moving any bytes) when a
    precondition fails — TLS on either side, no raw socket — so the
    caller falls back to the asyncio loop.
    tee(bytes) is called on the worker thread for every chunk, in
    order (cache fill).
    """
    up_sock = up_w.transport.get_extra_info("socket")
    cl_sock = writer.transport.get_extra_info("socket")
    if (up_sock is None or cl_sock is None
            or up_w.transport.get_extra_info("sslcontext") is not None
            or writer.transport.get_extra_info("sslcontext") is not None
            # CPython StreamReader internal; absent -> asyncio fallback
            or not isinstance(getattr(up_r, "_buffer", None), bytearray)):
        raise NotImplementedError
    await writer.drain()
    up_w.transport.pause_reading()
    up_fd = up_sock.fileno()
    cl_fd = cl_sock.fileno()
    # bytes asyncio already consumed past the response head
    buf = up_r._buffer  # CPython StreamReader internal (3.10)
    take = min(len(buf), length)
    pre = bytes(buf[:take])
    del buf[:take]
    can_splice = hasattr(os, "splice") and tee is None

    def run():
        if pre:
            if tee:
                tee(pre)
            _sendall_nb(cl_fd, memoryview(pre))
        left = length - len(pre)
        if left <= 0:
            return
        if can_splice:
            rfd, wfd = os.pipe()
            # a splice bigger than the pipe would BLOCK with the pipe
            # full (we drain only after it returns): grow the pipe and
            # never ask for more than its capacity
            pipe_sz = 1 << 16
            try:
                import fcntl

                pipe_sz = fcntl.fcntl(wfd, 1031, 1 << 20)  # F_SETPIPE_SZ
            except OSError:
                pass
            try:
                while left > 0:
                    try:
                        n = os.splice(up_fd, wfd, min(left, pipe_sz))
                    except BlockingIOError:
                        sel.select([up_fd], [], [], 10)
                        continue
                    if n == 0:
                        raise ConnectionResetError("upstream EOF")
                    m = 0
                    while m < n:
                        try:
                            m += os.splice(rfd, cl_fd, n - m)
                        except BlockingIOError:
                            sel.select([], [cl_fd], [], 10)
                    left -= n
            finally:
                os.close(rfd)
                os.close(wfd)
            return
        big = bytearray(8 << 20)
        mv = memoryview(big)
        while left > 0:
            want = min(left, len(big))
            try:
                n = os.readv(up_fd, [mv[:want]])
            except BlockingIOError:
                sel.select([up_fd], [], [], 10)
                continue
            if n == 0:
                raise ConnectionResetError("upstream EOF")
            if tee:
                tee(bytes(mv[:n]))
            _sendall_nb(cl_fd, mv[:n])
            left -= n

    loop = asyncio.get_running_loop()
    try:
        await loop.run_in_executor(_pool(), run)
    finally:
        try:
            up_w.transport.resume_reading()
        except Exception:
            pass


async def sendfile_threaded(writer: asyncio.StreamWriter, f,
                            start: int, length: int) -> None:
    """Send [start, start+length) of file f on writer's socket.

    Raises NotImplementedError when the transport has no plain socket
    (TLS) — caller falls back to chunked writes.
    """
    sock = writer.transport.get_extra_info("socket")
    tls = writer.transport.get_extra_info("sslcontext")
    if sock is None or tls is not None:
        raise NotImplementedError
    await writer.drain()
    writer.transport.pause_reading()
    fd = sock.fileno()
    infd = f.fileno()

    def run():
        off = start
        remaining = length
        while remaining > 0:
            try:
                sent = os.sendfile(fd, infd, off, remaining)
            except BlockingIOError:
                sel.select([], [fd], [], 10)
                continue
            if sent == 0:
                raise ConnectionResetError("peer went away")
            off += sent
            remaining -= sent

    loop = asyncio.get_running_loop()
    try:
        await loop.run_in_executor(_pool(), run)
    finally:
        try:
            writer.transport.resume_reading()
        except Exception:
            pass
