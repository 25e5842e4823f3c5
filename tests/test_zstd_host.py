"""Host-compiled zstd decoder (shares zstd_common.h with the CDNA4
kernel) vs pyarrow's reference zstd — runs on CPU every round, so the
format/entropy layer stays verified without GPU time."""

import os
import random

import pytest

pa = pytest.importorskip("pyarrow")


def _native():
    from demodel_amd import _native as n

    return n


def payloads():
    random.seed(7)
    text = (b"zstd frames carry huffman literals and fse sequences; "
            b"this sentence repeats to create matches. " * 300)
    return {
        "text": text,
        "random": os.urandom(100_000),
        "runs": b"\x42" * 50_000,
        "mixed": text + os.urandom(30_000) + b"Z" * 10_000 + text[:5000],
        "tiny": b"hello",
        "empty": b"",
        "big": (text + os.urandom(1000)) * 40,
        "json": (b'{"text": "sample", "meta": {"idx": 12345}}' * 2000),
        "semi": bytes(random.choices(b"abcdefgh \n", k=400_000)),
    }


@pytest.mark.parametrize("level", [1, 3, 9, 19])
def test_zstd_host_matrix(level):
    n = _native()
    for name, data in payloads().items():
        frame = bytes(pa.Codec("zstd", compression_level=level)
                      .compress(data))
        out, status, consumed = n.zstd_decode(frame, len(data) + 16)
        assert status == 0, (name, level, status)
        assert out == data, (name, level)
        assert consumed == len(frame), (name, level)


def test_zstd_host_multiframe_and_skippable():
    n = _native()
    c = pa.Codec("zstd", compression_level=3)
    a, b = b"first frame " * 100, b"second frame " * 200
    skippable = b"\x50\x2a\x4d\x18" + (8).to_bytes(4, "little") + b"x" * 8
    blob = bytes(c.compress(a)) + skippable + bytes(c.compress(b))
    out, status, consumed = n.zstd_decode(blob, len(a) + len(b) + 16)
    assert status == 0
    assert out == a + b
    assert consumed == len(blob)


def test_zstd_host_rejects_garbage():
    n = _native()
    out, status, consumed = n.zstd_decode(os.urandom(100), 1024)
    assert status != 0
