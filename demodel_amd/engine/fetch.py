"""Synchronous HTTP blob fetcher for the pull engine.

One GET per blob, body drained straight into the caller's buffers
(pinned ring slabs on GPU machines) with MSG_WAITALL — one syscall per
slab on plain TCP.  TLS uses the same interface via recv_into on the
wrapped socket.  Redirects are followed here (the HF hub -> CDN hop).

This replaces the client->goproxy->origin relay of the reference with an
engine-native client path; the proxy (proxy/server.py) remains the
compatibility surface for external clients.
"""

from __future__ import annotations

import gzip
import json
import socket
import ssl
import zlib
from dataclasses import dataclass, field
from urllib.parse import urlsplit

MAX_REDIRECTS = 5
HEAD_LIMIT = 256 * 1024


class FetchError(Exception):
    pass


@dataclass
class Response:
    url: str
    status: int
    reason: str
    headers: list[tuple[str, str]] = field(default_factory=list)

    def get(self, name: str, default=None):
        ln = name.lower()
        for k, v in self.headers:
            if k.lower() == ln:
                return v
        return default


class BlobSource:
    """Streaming body of one HTTP response.

    fill(view) -> int: writes into `view`, returns bytes written
    (0 on clean EOF).  `length` is the content length (-1 if unknown).
    """

    def __init__(self, sock, resp: Response, leftover: bytes,
                 is_tls: bool):
        self.sock = sock
        self.resp = resp
        self.status = resp.status
        self.headers = resp.headers
        self._left = memoryview(bytearray(leftover)) if leftover else None
        self._tls = is_tls
        te = (resp.get("transfer-encoding") or "").lower()
        self._chunked = "chunked" in te
        cl = resp.get("content-length")
        self.length = int(cl) if (cl is not None and not self._chunked) \
            else -1
        self._remaining = self.length
        self._chunk_rem = 0
        self._eof = False

    # ---- raw reads ----------------------------------------------------

    def _raw_into(self, view: memoryview) -> int:
        if self._left is not None and len(self._left):
            n = min(len(view), len(self._left))
            view[:n] = self._left[:n]
            self._left = self._left[n:] if n < len(self._left) else None
            return n
        if self._tls:
            return self.sock.recv_into(view)
        return self.sock.recv_into(view, len(view))

    def _raw_exact(self, n: int) -> bytes:
        out = bytearray(n)
        mv = memoryview(out)
        got = 0
        while got < n:
            r = self._raw_into(mv[got:])
            if r <= 0:
                raise FetchError("connection closed mid-body")
            got += r
        return bytes(out)

    def _raw_line(self) -> bytes:
        # only used for chunked framing (small reads)
        line = bytearray()
        one = bytearray(1)
        while True:
            r = self._raw_into(memoryview(one))
            if r <= 0:
                raise FetchError("connection closed in chunk header")
            line += one
            if line.endswith(b"\r\n"):
                return bytes(line[:-2])

    # ---- public fill --------------------------------------------------

    def fill(self, view: memoryview) -> int:
        if self._eof:
            return 0
        if self._chunked:
            return self._fill_chunked(view)
        if self.length >= 0:
            if self._remaining <= 0:
                self._eof = True
                return 0
            want = min(len(view), self._remaining)
            if not self._tls and (self._left is None or not len(self._left)):
                got = self.sock.recv_into(view[:want], want,
                                          socket.MSG_WAITALL)
            else:
                got = self._raw_into(view[:want])
            if got <= 0:
                raise FetchError("connection closed mid-body")
            self._remaining -= got
            if self._remaining == 0:
                self._eof = True
            return got
        # EOF-delimited
        got = self._raw_into(view)
        if got == 0:
            self._eof = True
        return got

    def _fill_chunked(self, view: memoryview) -> int:
        while self._chunk_rem == 0:
            line = self._raw_line()
            size = int(line.split(b";")[0] or b"0", 16)
            if size == 0:
                while True:  # trailers
                    t = self._raw_line()
                    if not t:
                        break
                self._eof = True
                return 0
            self._chunk_rem = size
        want = min(len(view), self._chunk_rem)
        got = self._raw_into(view[:want])
        if got <= 0:
            raise FetchError("connection closed mid-chunk")
        self._chunk_rem -= got
        if self._chunk_rem == 0:
            tail = self._raw_exact(2)
            if tail != b"\r\n":
                raise FetchError("bad chunk terminator")
        return got

    def read_all(self, limit: int = 1 << 30) -> bytes:
        out = bytearray()
        buf = bytearray(1 << 18)
        mv = memoryview(buf)
        while True:
            n = self.fill(mv)
            if n == 0:
                break
            out += mv[:n]
            if len(out) > limit:
                raise FetchError("body exceeds limit")
        body = bytes(out)
        enc = (self.resp.get("content-encoding") or "").lower()
        if enc == "gzip":
            body = gzip.decompress(body)
        elif enc == "deflate":
            body = zlib.decompress(body)
        return body

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass


def _connect(host: str, port: int, tls: bool, cafile, insecure,
             timeout: float):
    sock = socket.create_connection((host, port), timeout=timeout)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_RCVBUF, 8 << 20)
    sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
    if not tls:
        return sock, False
    ctx = ssl.create_default_context()
    if cafile:
        ctx.load_verify_locations(cafile=cafile)
    if insecure:
        ctx.check_hostname = False
        ctx.verify_mode = ssl.CERT_NONE
    return ctx.wrap_socket(sock, server_hostname=host), True


def _read_head(sock) -> tuple[Response, bytes]:
    buf = bytearray()
    while b"\r\n\r\n" not in buf:
        if len(buf) > HEAD_LIMIT:
            raise FetchError("response head too large")
        data = sock.recv(65536)
        if not data:
            raise FetchError("connection closed before response head")
        buf += data
    head, _, leftover = bytes(buf).partition(b"\r\n\r\n")
    lines = head.decode("latin-1").split("\r\n")
    parts = lines[0].split(" ", 2)
    resp = Response(url="", status=int(parts[1]),
                    reason=parts[2] if len(parts) > 2 else "")
    for line in lines[1:]:
        name, _, value = line.partition(":")
        resp.headers.append((name.strip(), value.strip()))
    return resp, leftover


def http_get(url: str, headers: dict | None = None, cafile=None,
             insecure=False, method: str = "GET", timeout: float = 60.0,
             follow_redirects: bool = True) -> BlobSource:
    seen = 0
    while True:
        u = urlsplit(url)
        tls = u.scheme == "https"
        port = u.port or (443 if tls else 80)
        path = u.path or "/"
        if u.query:
            path += "?" + u.query
        sock, is_tls = _connect(u.hostname, port, tls, cafile, insecure,
                                timeout)
        hdrs = {"Host": u.hostname if port in (80, 443)
                else f"{u.hostname}:{port}",
                "Accept-Encoding": "identity",
                "Connection": "close",
                "User-Agent": "demodel-amd/0.1"}
        if headers:
            hdrs.update(headers)
        req = f"{method} {path} HTTP/1.1\r\n" + "".join(
            f"{k}: {v}\r\n" for k, v in hdrs.items()) + "\r\n"
        sock.sendall(req.encode("latin-1"))
        resp, leftover = _read_head(sock)
        resp.url = url
        if (follow_redirects and resp.status in (301, 302, 303, 307, 308)
                and seen < MAX_REDIRECTS):
            loc = resp.get("location")
            if loc:
                seen += 1
                sock.close()
                if loc.startswith("http://") or loc.startswith("https://"):
                    url = loc
                elif loc.startswith("/"):
                    url = f"{u.scheme}://{u.netloc}{loc}"
                else:
                    base = u.path.rsplit("/", 1)[0]
                    url = f"{u.scheme}://{u.netloc}{base}/{loc}"
                # hub -> CDN hop: never forward credentials to a
                # DIFFERENT host (presigned CDN URLs reject them)
                if headers and urlsplit(url).netloc != u.netloc:
                    headers = {k: v for k, v in headers.items()
                               if k.lower() not in ("authorization",
                                                    "cookie")}
                continue
        src = BlobSource(sock, resp, leftover, is_tls)
        src.resp.url = url
        return src


def get_json(url: str, **kw):
    src = http_get(url, **kw)
    try:
        if src.status != 200:
            raise FetchError(f"GET {url} -> {src.status} {src.resp.reason}")
        return json.loads(src.read_all(limit=256 << 20))
    finally:
        src.close()
