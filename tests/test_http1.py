"""HTTP/1.1 wire-layer unit tests (proxy/http1.py): chunked framing,
length/eof bodies, header parsing edge cases."""

import asyncio

import pytest

from demodel_amd.proxy import http1


def _drain(mode, length, payload: bytes) -> bytes:
    async def run():
        reader = asyncio.StreamReader()
        reader.feed_data(payload)
        reader.feed_eof()
        out = b""
        async for chunk in http1.iter_body(reader, mode, length):
            out += chunk
        return out

    return asyncio.run(run())


def test_chunked_multi_chunk_with_trailers():
    body = (b"5\r\nhello\r\n"
            b"6\r\n world\r\n"
            b"0\r\nX-Trailer: 1\r\n\r\n")
    assert _drain("chunked", -1, body) == b"hello world"


def test_chunked_with_extensions():
    body = b"4;ext=1\r\nabcd\r\n0\r\n\r\n"
    assert _drain("chunked", -1, body) == b"abcd"


def test_chunked_data_containing_crlf():
    data = b"ab\r\ncd"
    body = b"%x\r\n%s\r\n0\r\n\r\n" % (len(data), data)
    assert _drain("chunked", -1, body) == data


def test_chunked_truncated_raises():
    with pytest.raises(http1.ProtocolError):
        _drain("chunked", -1, b"10\r\nonly-a-few")


def test_chunked_bad_terminator_raises():
    with pytest.raises(http1.ProtocolError):
        _drain("chunked", -1, b"3\r\nabcXX\r\n0\r\n\r\n")


def test_length_body_exact_and_truncated():
    assert _drain("length", 4, b"wxyz-extra") == b"wxyz"
    with pytest.raises(http1.ProtocolError):
        _drain("length", 10, b"short")


def test_eof_body():
    assert _drain("eof", -1, b"everything until close") == \
        b"everything until close"


def test_request_head_parse_and_obs_fold():
    async def run():
        reader = asyncio.StreamReader()
        reader.feed_data(b"GET /x HTTP/1.1\r\n"
                         b"Host: h\r\n"
                         b"X-Long: part1\r\n"
                         b"  part2\r\n"
                         b"\r\n")
        return await http1.read_request_head(reader)

    head = asyncio.run(run())
    assert head.method == "GET" and head.target == "/x"
    assert head.get("host") == "h"
    assert head.get("x-long") == "part1 part2"


def test_body_mode_rules():
    r = http1.RequestHead("POST", "/", "HTTP/1.1",
                          [("Content-Length", "5")])
    assert http1.body_mode(r, method="POST") == ("length", 5)
    resp = http1.ResponseHead("HTTP/1.1", 204, "No Content", [])
    assert http1.body_mode(resp, status=204) == ("none", 0)
    resp2 = http1.ResponseHead("HTTP/1.1", 200, "OK",
                               [("Transfer-Encoding", "chunked")])
    assert http1.body_mode(resp2, status=200) == ("chunked", -1)
    resp3 = http1.ResponseHead("HTTP/1.1", 200, "OK", [])
    assert http1.body_mode(resp3, status=200) == ("eof", -1)
