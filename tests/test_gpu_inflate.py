"""GPU DEFLATE inflate vs zlib (CPU reference), per SURVEY.md §4."""

import ctypes
import gzip
import os
import zlib

import pytest

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


def _mixed_payload(n):
    """Compressible-but-not-trivial payload: text + random + runs."""
    parts = []
    rnd = os.urandom(n // 3)
    parts.append((b"the quick brown fox jumps over the lazy dog. " * 200))
    parts.append(rnd)
    parts.append(b"A" * (n // 3))
    parts.append(bytes(range(256)) * 100)
    data = b"".join(parts)
    return data[:n] if len(data) >= n else data + os.urandom(n - len(data))


@pytest.mark.parametrize("level", [0, 1, 6, 9])
@pytest.mark.parametrize("nbytes", [0, 1, 1000, 300_000])
def test_inflate_raw_deflate(hipmod, level, nbytes):
    from demodel_amd.engine.formats.compress import inflate_gpu

    h = hipmod
    s = h.Stream(0)
    data = _mixed_payload(nbytes) if nbytes else b""
    comp = zlib.compressobj(level, zlib.DEFLATED, -15)
    blob = comp.compress(data) + comp.flush()
    src = _upload(h, blob, s)
    dst = h.DeviceBuffer(max(nbytes, 1))
    res = inflate_gpu([(src.ptr, len(blob), dst.ptr, max(nbytes, 1))])[0]
    assert res.ok, res.error
    assert res.written == len(data)
    assert _download(h, dst, len(data), s) == data


def test_inflate_many_streams_parallel(hipmod):
    from demodel_amd.engine.formats.compress import inflate_gpu

    h = hipmod
    s = h.Stream(0)
    rng_sizes = [17, 1000, 65536, 250_000, 4096, 0]
    datas, streams, dsts, blobs = [], [], [], []
    for i, n in enumerate(rng_sizes * 20):  # 120 streams
        data = _mixed_payload(n) if n else b""
        comp = zlib.compressobj(i % 9 + 1, zlib.DEFLATED, -15)
        blob = comp.compress(data) + comp.flush()
        src = _upload(h, blob, s)
        dst = h.DeviceBuffer(max(n, 1))
        datas.append(data)
        blobs.append(src)
        dsts.append(dst)
        streams.append((src.ptr, len(blob), dst.ptr, max(n, 1)))
    results = inflate_gpu(streams)
    for i, (res, data) in enumerate(zip(results, datas)):
        assert res.ok, (i, res.error)
        assert res.written == len(data), i
        assert _download(h, dsts[i], len(data), s) == data, i


def test_gunzip_blob_gpu(hipmod):
    from demodel_amd.engine.formats.compress import gunzip_blob_gpu
    from demodel_amd.engine.pipeline import Lander

    data = _mixed_payload(500_000)
    gz = gzip.compress(data, compresslevel=6)
    pos = [0]

    def fill(view):
        n = min(len(view), len(gz) - pos[0])
        view[:n] = gz[pos[0]:pos[0] + n]
        pos[0] += n
        return n

    lander = Lander(slab_bytes=1 << 20, n_slabs=2)
    blob = lander.land(fill, len(gz))
    dst, res = gunzip_blob_gpu(blob)
    assert res.ok and res.written == len(data)
    h = hipmod
    s = h.Stream(0)
    assert _download(h, dst, len(data), s) == data


def test_inflate_error_on_garbage(hipmod):
    from demodel_amd.engine.formats.compress import inflate_gpu

    h = hipmod
    s = h.Stream(0)
    junk = b"\x07" + os.urandom(500)  # btype=3 -> format error
    src = _upload(h, junk, s)
    dst = h.DeviceBuffer(1024)
    res = inflate_gpu([(src.ptr, len(junk), dst.ptr, 1024)])[0]
    assert not res.ok
