#!/usr/bin/env python3
"""Loopback throughput probes: where is the pull pipeline's network wall?

1. raw ceiling: blocking-thread server doing os.sendfile of a page-cached
   file vs client recv_into(MSG_WAITALL), K parallel connections.
2. FakeOrigin (asyncio loop.sendfile) GET path, K parallel engine fetches.

Prints JSON lines; no GPU needed.
"""

import json
import os
import socket
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests"))

FILE_MB = int(os.environ.get("NET_PROBE_MB", "2048"))


def make_file(tmp):
    path = os.path.join(tmp, "blob.bin")
    blk = os.urandom(16 << 20)
    with open(path, "wb") as f:
        for _ in range(FILE_MB // 16):
            f.write(blk)
    # warm page cache
    with open(path, "rb") as f:
        while f.read(64 << 20):
            pass
    return path


def raw_server(path, port_box):
    srv = socket.socket()
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("127.0.0.1", 0))
    srv.listen(64)
    port_box.append(srv.getsockname()[1])

    def handle(conn):
        with open(path, "rb") as f:
            size = os.fstat(f.fileno()).st_size
            off = 0
            while off < size:
                sent = os.sendfile(conn.fileno(), f.fileno(), off,
                                   size - off)
                if sent == 0:
                    break
                off += sent
        conn.close()

    def loop():
        while True:
            try:
                conn, _ = srv.accept()
            except OSError:
                return
            threading.Thread(target=handle, args=(conn,),
                             daemon=True).start()

    threading.Thread(target=loop, daemon=True).start()
    return srv


def raw_client(port, nbytes, results, i):
    s = socket.create_connection(("127.0.0.1", port))
    s.setsockopt(socket.SOL_SOCKET, socket.SO_RCVBUF, 8 << 20)
    buf = bytearray(64 << 20)
    mv = memoryview(buf)
    got = 0
    t0 = time.perf_counter()
    while got < nbytes:
        n = s.recv_into(mv, len(mv), socket.MSG_WAITALL)
        if n == 0:
            break
        got += n
    results[i] = (got, time.perf_counter() - t0)
    s.close()


def probe_raw(path):
    size = os.path.getsize(path)
    for k in (1, 2, 4, 8):
        port_box = []
        srv = raw_server(path, port_box)
        port = port_box[0]
        results = [None] * k
        ts = [threading.Thread(target=raw_client,
                               args=(port, size, results, i))
              for i in range(k)]
        t0 = time.perf_counter()
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        el = time.perf_counter() - t0
        total = sum(r[0] for r in results)
        print(json.dumps({"op": "raw_sendfile_loopback", "streams": k,
                          "GBps": round(total / el / 1e9, 2)}), flush=True)
        srv.close()


def probe_origin(path):
    from demodel_amd.engine import fetch
    from demodel_amd.testing.origin import FakeOrigin
    from helpers import LoopThread

    lt = LoopThread()
    origin = FakeOrigin(os.path.dirname(path))
    origin.add_hf_repo("p/p", {"blob.bin": path})
    port = lt.call(origin.start())
    size = os.path.getsize(path)

    def client(i, results):
        src = fetch.http_get(
            f"http://127.0.0.1:{port}/p/p/resolve/main/blob.bin")
        buf = bytearray(64 << 20)
        mv = memoryview(buf)
        got = 0
        t0 = time.perf_counter()
        while got < size:
            n = src.fill(mv)
            if n == 0:
                break
            got += n
        results[i] = (got, time.perf_counter() - t0)
        src.close()

    for k in (1, 2, 4, 8):
        results = [None] * k
        ts = [threading.Thread(target=client, args=(i, results))
              for i in range(k)]
        t0 = time.perf_counter()
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        el = time.perf_counter() - t0
        total = sum(r[0] for r in results)
        print(json.dumps({"op": "fakeorigin_asyncio", "streams": k,
                          "GBps": round(total / el / 1e9, 2)}), flush=True)
    lt.call(origin.close())
    lt.stop()


if __name__ == "__main__":
    tmp = tempfile.mkdtemp(prefix="netprobe-",
                           dir=os.environ.get("TMPDIR", "/tmp"))
    path = make_file(tmp)
    probe_raw(path)
    probe_origin(path)
    os.unlink(path)
