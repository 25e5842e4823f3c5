"""GPU kernel numerics tests (MI355X): every CDNA4 kernel vs a plain CPU
reference (hashlib / numpy), per SURVEY.md §4's test plan."""

import ctypes
import hashlib
import os
import struct

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hipmod():
    from demodel_amd.gpu import have_gpu, hip

    assert have_gpu(), "gpu marker requires a GPU"
    return hip()


def _upload(h, data: bytes, stream):
    buf = h.DeviceBuffer(len(data))
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


@pytest.mark.parametrize("nbytes,chunk", [
    (64 << 10, 64 << 10),       # single exact chunk
    (1 << 20, 64 << 10),        # many exact chunks
    (1000003, 64 << 10),        # ragged tail chunk
    (777, 64 << 10),            # single short chunk
    (3 << 20, 256 << 10),
])
def test_sha256_batch_matches_hashlib(hipmod, nbytes, chunk):
    h = hipmod
    s = h.Stream(0)
    data = os.urandom(nbytes)
    buf = _upload(h, data, s)
    n_chunks = (nbytes + chunk - 1) // chunk
    dig = h.DeviceBuffer(n_chunks * 32)
    h.sha256_batch(buf.ptr, nbytes, chunk, dig.ptr, n_chunks, s.handle)
    raw = _download(h, dig, n_chunks * 32, s)
    for c in range(n_chunks):
        # kernel stores digest words as native u32; canonical sha256 bytes
        # are those words big-endian
        words = struct.unpack_from("<8I", raw, c * 32)
        got = "".join(f"{w:08x}" for w in words)
        want = hashlib.sha256(data[c * chunk:(c + 1) * chunk]).hexdigest()
        assert got == want, f"chunk {c}"


def test_sha256_chain_matches_hashlib(hipmod):
    h = hipmod
    s = h.Stream(0)
    data = os.urandom(64 * 1000)  # whole blocks only for the kernel
    buf = _upload(h, data, s)
    state = h.DeviceBuffer(32)
    h.sha256_chain_init(state.ptr, s.handle)
    # two updates to exercise state carry
    h.sha256_chain_update(state.ptr, buf.ptr, 400, s.handle)
    h.sha256_chain_update(state.ptr, buf.ptr + 400 * 64, 600, s.handle)
    raw = _download(h, state, 32, s)
    st = list(struct.unpack("<8I", raw))
    from demodel_amd.engine.pipeline import _host_sha256_finish

    got = _host_sha256_finish(st, len(data), b"")
    assert got == hashlib.sha256(data).hexdigest()


def test_scatter_ranges(hipmod):
    h = hipmod
    s = h.Stream(0)
    src_data = os.urandom(1 << 20)
    src = _upload(h, src_data, s)
    # three destination buffers with assorted offsets/lengths (all inside
    # the 1 MiB source)
    specs = [(0, 1000), (1000, 65536), (66536 + 3, 900000)]
    dsts = [h.DeviceBuffer(ln + 64) for _, ln in specs]
    desc = b""
    for (off, ln), d in zip(specs, dsts):
        desc += struct.pack("<4Q", off, d.ptr + 16, ln, 0)  # unaligned dst
    dbuf = _upload(h, desc, s)
    h.scatter_ranges(src.ptr, dbuf.ptr, len(specs), s.handle)
    for (off, ln), d in zip(specs, dsts):
        got = _download(h, d, ln + 16, s)[16:16 + ln]
        assert got == src_data[off:off + ln]


def test_cast_f32_to_bf16(hipmod):
    import torch

    h = hipmod
    s = h.Stream(0)
    x = np.random.randn(100003).astype(np.float32)
    src = _upload(h, x.tobytes(), s)
    dst = h.DeviceBuffer(x.size * 2)
    h.cast_f32_to_bf16(src.ptr, dst.ptr, x.size, s.handle)
    raw = _download(h, dst, x.size * 2, s)
    got = torch.frombuffer(bytearray(raw), dtype=torch.bfloat16)
    want = torch.from_numpy(x).to(torch.bfloat16)
    assert torch.equal(got, want)


@pytest.mark.parametrize("qtype,n_sup", [(2, 1000), (3, 1000), (6, 1000), (7, 1000),
                                         (8, 1000), (10, 64), (11, 64),
                                         (12, 64), (13, 64), (14, 64)])
def test_gguf_dequant_matches_cpu(hipmod, qtype, n_sup):
    import torch

    from demodel_amd.engine.formats import gguf

    h = hipmod
    s = h.Stream(0)
    _, be, bb = gguf.GGML_TYPES[qtype]
    n_elems = n_sup * be
    raw = np.random.default_rng(qtype).integers(
        0, 256, size=n_sup * bb, dtype=np.uint8).tobytes()
    want_f32 = gguf.dequant_cpu(qtype, raw, n_elems)
    want = torch.from_numpy(want_f32).to(torch.bfloat16)
    src = _upload(h, raw, s)
    dst = h.DeviceBuffer(n_elems * 2)
    h.gguf_dequant(qtype, src.ptr, dst.ptr, n_sup, s.handle)
    got = torch.frombuffer(bytearray(_download(h, dst, n_elems * 2, s)),
                           dtype=torch.bfloat16)
    gf, wf = got.float(), want.float()
    mask = torch.isfinite(wf)
    assert mask.float().mean() > 0.5  # random f16 scales: mostly finite
    assert torch.allclose(gf[mask], wf[mask], rtol=1 / 64, atol=1e-3), \
        (gf[mask] - wf[mask]).abs().max()


def test_device_buffer_dlpack_roundtrip(hipmod):
    import torch

    h = hipmod
    s = h.Stream(0)
    data = os.urandom(4096)
    buf = _upload(h, data, s)
    t = torch.from_dlpack(buf.to_dlpack())
    assert t.dtype == torch.uint8 and t.is_cuda and t.numel() == 4096
    assert bytes(t.cpu().numpy().tobytes()) == data
    # view as bf16 works (zero-copy)
    v = t.view(torch.bfloat16)
    assert v.numel() == 2048
