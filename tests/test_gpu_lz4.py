"""GPU LZ4 raw-block decoder vs pyarrow's lz4_raw (reference codec)."""

import ctypes
import os

import pytest

pytestmark = pytest.mark.gpu

pa = pytest.importorskip("pyarrow")


@pytest.fixture(scope="module")
def hipmod():
    from demodel_amd.gpu import have_gpu, hip

    assert have_gpu()
    return hip()


def _upload(h, data: bytes, stream):
    buf = h.DeviceBuffer(max(len(data), 1))
    src = (ctypes.c_char * len(data)).from_buffer_copy(data)
    h.h2d_async(buf.ptr, ctypes.addressof(src), len(data), stream.handle)
    stream.sync()
    return buf


def _download(h, buf, n, stream) -> bytes:
    out = bytearray(n)
    addr = ctypes.addressof((ctypes.c_char * n).from_buffer(out))
    h.d2h_async(addr, buf.ptr, n, stream.handle)
    stream.sync()
    return bytes(out)


def _payloads():
    text = (b"lz4 raw blocks carry literal runs and 16-bit offset "
            b"matches; this sentence repeats to create them. " * 400)
    return {
        "text": text,
        "random": os.urandom(80_000),
        "runs": b"\x55" * 40_000,            # overlapping matches (dist 1)
        "mixed": text + os.urandom(20_000) + b"Q" * 9_000 + text[:4_000],
        "tiny": b"hi",
        "big": (text + os.urandom(700)) * 30,
    }


@pytest.mark.parametrize("name", list(_payloads()))
def test_lz4_roundtrip(hipmod, name):
    from demodel_amd.engine.formats.compress import lz4_gpu

    h = hipmod
    s = h.Stream(0)
    data = _payloads()[name]
    codec = pa.Codec("lz4_raw")
    block = bytes(codec.compress(data))
    src = _upload(h, block, s)
    dst = h.DeviceBuffer(max(len(data), 1))
    res = lz4_gpu([(src.ptr, len(block), dst.ptr, max(len(data), 1))])[0]
    assert res.ok, (name, res.error, res.written)
    assert res.written == len(data), name
    got = _download(h, dst, len(data), s)
    assert got == data, name


def test_lz4_many_streams(hipmod):
    from demodel_amd.engine.formats.compress import lz4_gpu

    h = hipmod
    s = h.Stream(0)
    codec = pa.Codec("lz4_raw")
    payloads = list(_payloads().values()) * 12
    streams, dsts = [], []
    for data in payloads:
        b = bytes(codec.compress(data))
        src = _upload(h, b, s)
        dst = h.DeviceBuffer(max(len(data), 1))
        dsts.append((src, dst, data))
        streams.append((src.ptr, len(b), dst.ptr, max(len(data), 1)))
    results = lz4_gpu(streams)
    for i, (res, (_, dst, data)) in enumerate(zip(results, dsts)):
        assert res.ok, (i, res.error)
        assert res.written == len(data), i
        assert _download(h, dst, len(data), s) == data, i


def test_lz4_handcrafted_and_errors(hipmod):
    """Hand-built sequences: extended literal/match lengths, overlap
    copies, last-sequence-literals-only; plus rejection of bad offsets
    and output overflow."""
    from demodel_amd.engine.formats.compress import lz4_gpu

    h = hipmod
    s = h.Stream(0)

    def run(block, cap):
        src = _upload(h, block, s)
        dst = h.DeviceBuffer(max(cap, 1))
        res = lz4_gpu([(src.ptr, len(block), dst.ptr, cap)])[0]
        out = _download(h, dst, res.written, s) if res.ok else b""
        return res, out

    # literals(4)="abcd", match dist=4 len=8 (overlap tiling), then
    # final literals "XY"
    blk = bytes([0x44, *b"abcd", 0x04, 0x00]) + bytes([0x20, *b"XY"])
    res, out = run(blk, 64)
    assert res.ok and out == b"abcd" + b"abcd" * 2 + b"XY", out

    # extended literal length: 15 + 255 + 3 = 273 literals, no match
    lits = bytes(range(256)) + b"Z" * 17
    blk = bytes([0xF0, 255, 3]) + lits
    res, out = run(blk, 400)
    assert res.ok and out == lits

    # extended match length: 4 literals + match len 4+15+255+6
    blk = bytes([0x4F, *b"wxyz", 0x04, 0x00, 255, 6]) + bytes([0x10, b"!"[0]])
    res, out = run(blk, 600)
    want = b"wxyz" + (b"wxyz" * ((4 + 15 + 255 + 6) // 4 + 1)
                      )[:4 + 15 + 255 + 6] + b"!"
    assert res.ok and out == want, (len(out), len(want))

    # bad offset (dist > produced)
    blk = bytes([0x14, *b"a", 0x09, 0x00])
    res, _ = run(blk, 64)
    assert not res.ok

    # zero offset
    blk = bytes([0x14, *b"a", 0x00, 0x00])
    res, _ = run(blk, 64)
    assert not res.ok

    # overflow: output larger than cap
    data = b"spam" * 100
    blk = bytes(pa.Codec("lz4_raw").compress(data))
    res, _ = run(blk, 10)
    assert not res.ok
