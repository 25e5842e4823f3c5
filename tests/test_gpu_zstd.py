"""GPU zstd decoder vs pyarrow's zstd (CPU reference encoder/decoder)."""

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
    text = (b"zstd frames carry huffman literals and fse sequences; "
            b"this sentence repeats to create matches. " * 300)
    return {
        "text": text,
        "random": os.urandom(100_000),      # raw/uncompressible blocks
        "runs": b"\x42" * 50_000,           # RLE paths
        "mixed": text + os.urandom(30_000) + b"Z" * 10_000 + text[:5000],
        "tiny": b"hello",
        "empty": b"",
        "big": (text + os.urandom(1000)) * 40,  # multiple 128K blocks
    }


@pytest.mark.parametrize("level", [1, 3, 9, 19])
@pytest.mark.parametrize("name", list(_payloads()))
def test_zstd_roundtrip(hipmod, level, name):
    from demodel_amd.engine.formats.compress import zstd_gpu

    h = hipmod
    s = h.Stream(0)
    data = _payloads()[name]
    codec = pa.Codec("zstd", compression_level=level)
    frame = bytes(codec.compress(data))
    src = _upload(h, frame, s)
    dst = h.DeviceBuffer(max(len(data), 1))
    res = zstd_gpu([(src.ptr, len(frame), dst.ptr, max(len(data), 1))])[0]
    assert res.ok, (name, level, res.error, res.written)
    assert res.written == len(data), (name, level)
    got = _download(h, dst, len(data), s)
    if got != data:
        # locate first mismatch for debuggability
        for i, (a, b) in enumerate(zip(got, data)):
            if a != b:
                raise AssertionError(
                    f"{name} l{level}: first mismatch at {i}: "
                    f"{got[max(0,i-5):i+5]!r} vs {data[max(0,i-5):i+5]!r}")
        raise AssertionError("length mismatch")


def test_zstd_many_frames(hipmod):
    from demodel_amd.engine.formats.compress import zstd_gpu

    h = hipmod
    s = h.Stream(0)
    codec = pa.Codec("zstd", compression_level=3)
    payloads = list(_payloads().values()) * 10
    frames, dsts = [], []
    for data in payloads:
        f = bytes(codec.compress(data))
        src = _upload(h, f, s)
        dst = h.DeviceBuffer(max(len(data), 1))
        dsts.append((src, dst, data))
        frames.append((src.ptr, len(f), dst.ptr, max(len(data), 1)))
    results = zstd_gpu(frames)
    for i, (res, (_, dst, data)) in enumerate(zip(results, dsts)):
        assert res.ok, (i, res.error)
        assert res.written == len(data), i
        assert _download(h, dst, len(data), s) == data, i


def test_zstd_rejects_garbage(hipmod):
    from demodel_amd.engine.formats.compress import zstd_gpu

    h = hipmod
    s = h.Stream(0)
    junk = os.urandom(300)
    src = _upload(h, junk, s)
    dst = h.DeviceBuffer(1024)
    res = zstd_gpu([(src.ptr, len(junk), dst.ptr, 1024)])[0]
    assert not res.ok


@pytest.mark.parametrize("mode", ["x2", "x1nf", "x2nf"])
def test_zstd_kernel_variants(hipmod, mode, monkeypatch):
    """The paired-interleave (x2) and fence-free (nf) kernel variants
    (DEMODEL_ZSTD_MODE) must be byte-exact vs pyarrow — they share the
    decode machinery but different scheduling/synchronization."""
    from demodel_amd.engine.formats.compress import zstd_gpu

    monkeypatch.setenv("DEMODEL_ZSTD_MODE", mode)
    h = hipmod
    s = h.Stream(0)
    codec = pa.Codec("zstd", compression_level=3)
    payloads = list(_payloads().values()) * 3  # odd count: unpaired tail
    frames, dsts = [], []
    for data in payloads:
        f = bytes(codec.compress(data))
        src = _upload(h, f, s)
        dst = h.DeviceBuffer(max(len(data), 1))
        dsts.append((src, dst, data))
        frames.append((src.ptr, len(f), dst.ptr, max(len(data), 1)))
    results = zstd_gpu(frames, window=16 * 1024)
    for i, (res, (_, dst, data)) in enumerate(zip(results, dsts)):
        assert res.ok, (mode, i, res.error)
        assert res.written == len(data), (mode, i)
        assert _download(h, dst, len(data), s) == data, (mode, i)
