"""Minimal HTTP/1.1 wire handling for the proxy and origin servers.

Replaces what goproxy + net/http did for the reference (the whole request/
response parse-and-stream machinery under cmd/demodel/start.go).  Only what
the proxy needs: request/response heads, content-length / chunked / EOF
bodies, streaming without buffering whole blobs.
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass, field

MAX_HEAD = 256 * 1024
CHUNK = 1 << 18  # 256 KiB read granularity for streamed bodies


class ProtocolError(Exception):
    pass


@dataclass
class RequestHead:
    method: str
    target: str          # as received: origin-form, absolute-form, or host:port
    version: str
    headers: list[tuple[str, str]] = field(default_factory=list)

    def get(self, name: str, default: str | None = None) -> str | None:
        ln = name.lower()
        for k, v in self.headers:
            if k.lower() == ln:
                return v
        return default

    def replace(self, name: str, value: str) -> None:
        ln = name.lower()
        self.headers = [(k, v) for k, v in self.headers if k.lower() != ln]
        self.headers.append((name, value))

    def remove(self, name: str) -> None:
        ln = name.lower()
        self.headers = [(k, v) for k, v in self.headers if k.lower() != ln]


@dataclass
class ResponseHead:
    version: str
    status: int
    reason: str
    headers: list[tuple[str, str]] = field(default_factory=list)

    get = RequestHead.get
    replace = RequestHead.replace
    remove = RequestHead.remove


async def _read_head_lines(reader: asyncio.StreamReader) -> list[str]:
    raw = await reader.readuntil(b"\r\n\r\n")
    if len(raw) > MAX_HEAD:
        raise ProtocolError("header block too large")
    text = raw.decode("latin-1")
    return text.split("\r\n")[:-2]  # drop the two trailing empties


def _parse_headers(lines: list[str]) -> list[tuple[str, str]]:
    headers: list[tuple[str, str]] = []
    for line in lines:
        if not line:
            continue
        if line[0] in " \t" and headers:  # obs-fold continuation
            k, v = headers[-1]
            headers[-1] = (k, v + " " + line.strip())
            continue
        name, _, value = line.partition(":")
        headers.append((name.strip(), value.strip()))
    return headers


async def read_request_head(reader: asyncio.StreamReader) -> RequestHead | None:
    try:
        lines = await _read_head_lines(reader)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None  # clean EOF between requests
    except asyncio.LimitOverrunError as e:
        raise ProtocolError("request head too large") from e
    parts = lines[0].split(" ", 2)
    if len(parts) != 3:
        raise ProtocolError(f"bad request line: {lines[0]!r}")
    return RequestHead(parts[0].upper(), parts[1], parts[2],
                       _parse_headers(lines[1:]))


async def read_response_head(reader: asyncio.StreamReader) -> ResponseHead:
    lines = await _read_head_lines(reader)
    parts = lines[0].split(" ", 2)
    if len(parts) < 2:
        raise ProtocolError(f"bad status line: {lines[0]!r}")
    reason = parts[2] if len(parts) == 3 else ""
    return ResponseHead(parts[0], int(parts[1]), reason,
                        _parse_headers(lines[1:]))


def serialize_request(head: RequestHead) -> bytes:
    out = [f"{head.method} {head.target} {head.version}\r\n"]
    out += [f"{k}: {v}\r\n" for k, v in head.headers]
    out.append("\r\n")
    return "".join(out).encode("latin-1")


def serialize_response(head: ResponseHead) -> bytes:
    out = [f"{head.version} {head.status} {head.reason}\r\n"]
    out += [f"{k}: {v}\r\n" for k, v in head.headers]
    out.append("\r\n")
    return "".join(out).encode("latin-1")


def body_mode(headers_obj, method: str | None = None,
              status: int | None = None) -> tuple[str, int]:
    """Return (mode, length): mode in {none, length, chunked, eof}."""
    if method is not None:  # request side
        te = headers_obj.get("transfer-encoding")
        if te and "chunked" in te.lower():
            return "chunked", -1
        cl = headers_obj.get("content-length")
        if cl is not None:
            return ("length", int(cl)) if int(cl) > 0 else ("none", 0)
        return "none", 0
    # response side
    assert status is not None
    if status < 200 or status in (204, 304):
        return "none", 0
    te = headers_obj.get("transfer-encoding")
    if te and "chunked" in te.lower():
        return "chunked", -1
    cl = headers_obj.get("content-length")
    if cl is not None:
        return ("length", int(cl)) if int(cl) > 0 else ("none", 0)
    return "eof", -1


async def iter_body(reader: asyncio.StreamReader, mode: str, length: int):
    """Yield raw body byte chunks (chunked framing removed)."""
    if mode == "none":
        return
    if mode == "length":
        remaining = length
        while remaining > 0:
            data = await reader.read(min(CHUNK, remaining))
            if not data:
                raise ProtocolError("body truncated")
            remaining -= len(data)
            yield data
        return
    if mode == "eof":
        while True:
            data = await reader.read(CHUNK)
            if not data:
                return
            yield data
        return
    if mode == "chunked":
        while True:
            size_line = await reader.readline()
            if not size_line:
                raise ProtocolError("chunked body truncated")
            s = size_line.split(b";")[0].strip()
            if not s:
                raise ProtocolError(f"bad chunk size line {size_line!r}")
            try:
                size = int(s, 16)
            except ValueError as e:
                raise ProtocolError(
                    f"bad chunk size line {size_line!r}") from e
            if size == 0:
                # trailers until blank line
                while True:
                    line = await reader.readline()
                    if line in (b"\r\n", b"\n", b""):
                        return
            remaining = size
            while remaining > 0:
                data = await reader.read(min(CHUNK, remaining))
                if not data:
                    raise ProtocolError("chunked body truncated")
                remaining -= len(data)
                yield data
            # the CRLF that terminates this chunk's data
            crlf = await reader.readexactly(2)
            if crlf != b"\r\n":
                raise ProtocolError("bad chunk terminator")
        return
    raise ProtocolError(f"unknown body mode {mode}")
