"""GPU snappy decompressor vs pyarrow (CPU reference) — parquet's
default page codec (csrc/snappy.hip)."""

import ctypes
import os

import pytest

pa = pytest.importorskip("pyarrow")

pytestmark = pytest.mark.gpu


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
    rng = os.urandom
    text = (b"the quick brown fox jumps over the lazy dog " * 500)
    return {
        "tiny": b"x",
        "short-text": b"hello hello hello hello world",
        "overlap-run": b"ab" * 50_000,           # dist-2 overlapping copies
        "text": text,
        "random-64k": rng(64 << 10),             # long literals (len>60 tags)
        "mixed-1m": (text + rng(200_000) + b"Z" * 100_000
                     + bytes(range(256)) * 512)[:1 << 20],
        "far-matches": (rng(40 << 10) * 8),      # offsets up to 40 KiB
    }


@pytest.mark.parametrize("name", sorted(_payloads()))
def test_snappy_kernel_matches_pyarrow(hipmod, name):
    from demodel_amd.engine.formats.compress import snappy_gpu

    h = hipmod
    s = h.Stream(0)
    data = _payloads()[name]
    codec = pa.Codec("snappy")
    comp = bytes(codec.compress(data))
    src = _upload(h, comp, s)
    dst = h.DeviceBuffer(max(len(data), 1))
    res = snappy_gpu([(src.ptr, len(comp), dst.ptr, len(data))])[0]
    assert res.ok, (name, res.status)
    assert res.written == len(data)
    assert _download(h, dst, len(data), s) == data


def test_snappy_kernel_batch(hipmod):
    """Many streams in one launch, mixed payloads, exact round-trip."""
    from demodel_amd.engine.formats.compress import snappy_gpu

    h = hipmod
    s = h.Stream(0)
    codec = pa.Codec("snappy")
    payloads = list(_payloads().values()) * 40
    comps = [bytes(codec.compress(d)) for d in payloads]
    total_c = sum(len(c) for c in comps)
    total_u = sum(len(d) for d in payloads)
    src = h.DeviceBuffer(total_c)
    dst = h.DeviceBuffer(total_u)
    streams = []
    co = uo = 0
    blob = b"".join(comps)
    carr = (ctypes.c_char * len(blob)).from_buffer_copy(blob)
    h.h2d_async(src.ptr, ctypes.addressof(carr), len(blob), s.handle)
    s.sync()
    for c, d in zip(comps, payloads):
        streams.append((src.ptr + co, len(c), dst.ptr + uo, len(d)))
        co += len(c)
        uo += len(d)
    results = snappy_gpu(streams)
    assert all(r.ok for r in results)
    got = _download(h, dst, total_u, s)
    assert got == b"".join(payloads)


def test_snappy_corrupt_fails_loudly(hipmod):
    from demodel_amd.engine.formats.compress import snappy_gpu

    h = hipmod
    s = h.Stream(0)
    codec = pa.Codec("snappy")
    comp = bytearray(codec.compress(b"some reasonable input " * 100))
    comp = comp[: len(comp) // 2]      # truncate mid-stream
    src = _upload(h, bytes(comp), s)
    dst = h.DeviceBuffer(4096)
    res = snappy_gpu([(src.ptr, len(comp), dst.ptr, 4096)])[0]
    assert not res.ok


def _varint(n: int) -> bytes:
    out = b""
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def _lit(data: bytes) -> bytes:
    n = len(data) - 1
    if n < 60:
        return bytes([n << 2]) + data
    nb = (n.bit_length() + 7) // 8            # 1..4 extended length bytes
    return bytes([(59 + nb) << 2]) + n.to_bytes(nb, "little") + data


def test_snappy_handcrafted_copy_tags(hipmod):
    """Streams pyarrow's encoder never emits: copy2/copy4 with large
    offsets, max-length copies, back-to-back overlapping runs."""
    from demodel_amd.engine.formats.compress import snappy_gpu

    h = hipmod
    s = h.Stream(0)
    base = bytes(range(256)) * 300           # 76800 bytes of literals
    # copy2: 64 bytes from distance 60000; copy4: 33 from distance
    # 76000 (offsets are distances BACK from the write position)
    stream = (_lit(base)
              + bytes([(63 << 2) | 2, 60000 & 0xFF, 60000 >> 8])
              + bytes([(32 << 2) | 3]) + (76000).to_bytes(4, "little")
              + bytes([(7 << 2) | 1 | (0 << 5), 1]))   # copy1 d=1 l=11
    expect = bytearray(base)
    expect += expect[len(expect) - 60000:len(expect) - 60000 + 64]
    expect += expect[len(expect) - 76000:len(expect) - 76000 + 33]
    expect += bytes([expect[-1]]) * 11       # overlapping d=1 run
    payload = _varint(len(expect)) + stream
    src = _upload(h, payload, s)
    dst = h.DeviceBuffer(len(expect))
    res = snappy_gpu([(src.ptr, len(payload), dst.ptr, len(expect))])[0]
    assert res.ok, res.status
    assert res.written == len(expect)
    assert _download(h, dst, len(expect), s) == bytes(expect)


def test_snappy_rejects_bad_offset(hipmod):
    from demodel_amd.engine.formats.compress import snappy_gpu

    h = hipmod
    s = h.Stream(0)
    # copy1 with offset 5 after only 4 bytes of output
    payload = _varint(15) + _lit(b"abcd") + bytes([(7 << 2) | 1, 5])
    src = _upload(h, payload, s)
    dst = h.DeviceBuffer(64)
    res = snappy_gpu([(src.ptr, len(payload), dst.ptr, 64)])[0]
    assert not res.ok
