"""Hypothesis fuzzing of the CPU-verifiable decode surfaces.

The host zstd decoder shares zstd_common.h with the CDNA4 kernel, so
property-testing it on CPU hardens the GPU decode logic every round
without GPU time.  The http1 parser fronts every proxied byte.
"""

import pytest
from hypothesis import given, settings, strategies as st

pa = pytest.importorskip("pyarrow")


def _native():
    from demodel_amd import _native as n

    return n


# --- zstd: encoder-roundtrip property ---------------------------------

_chunk = st.one_of(
    st.binary(min_size=0, max_size=2000),                   # random
    st.builds(lambda b, n: b * n,
              st.binary(min_size=1, max_size=64),
              st.integers(min_value=1, max_value=400)),     # repeats
    st.builds(lambda c, n: bytes([c]) * n,
              st.integers(0, 255), st.integers(1, 5000)),   # runs
)
_payload = st.lists(_chunk, min_size=0, max_size=8).map(b"".join)


@settings(max_examples=150, deadline=None)
@given(data=_payload, level=st.sampled_from([1, 3, 9, 19]))
def test_zstd_host_roundtrip_fuzz(data, level):
    n = _native()
    frame = bytes(pa.Codec("zstd", compression_level=level)
                  .compress(data))
    out, status, consumed = n.zstd_decode(frame, len(data) + 16)
    assert status == 0
    assert out == data
    assert consumed == len(frame)


@settings(max_examples=200, deadline=None)
@given(garbage=st.binary(min_size=0, max_size=4000))
def test_zstd_host_rejects_garbage_without_crashing(garbage):
    """Arbitrary bytes must produce an error status (or decode a
    coincidentally-valid prefix), never crash or overrun."""
    n = _native()
    out, status, consumed = n.zstd_decode(garbage, 1 << 16)
    assert consumed <= len(garbage)
    assert len(out) <= (1 << 16)


@settings(max_examples=100, deadline=None)
@given(data=_payload, cut=st.integers(min_value=1, max_value=50))
def test_zstd_host_truncated_frames_error(data, cut):
    """A truncated frame must error, not fabricate output."""
    n = _native()
    frame = bytes(pa.Codec("zstd", compression_level=3).compress(data))
    if cut >= len(frame):
        return
    out, status, consumed = n.zstd_decode(frame[:-cut], len(data) + 16)
    assert status != 0 or out != data  # never "success with full data"


# --- lz4 raw blocks: same roundtrip property via the numpy-side -------
# (the GPU lz4 kernel has no host twin; the format is fuzzed on GPU in
# tests/test_gpu_lz4.py — here we fuzz OUR understanding of pyarrow's
# framing so page-plan bugs surface on CPU)

@settings(max_examples=100, deadline=None)
@given(data=_payload)
def test_lz4_raw_roundtrip_reference(data):
    c = pa.Codec("lz4_raw")
    comp = bytes(c.compress(data))
    assert bytes(c.decompress(comp, len(data))) == data


# --- http1: request-head serialize/parse roundtrip --------------------

_token = st.text(
    alphabet="abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ"
             "0123456789-_", min_size=1, max_size=24)
_value = st.text(
    alphabet=st.characters(min_codepoint=0x20, max_codepoint=0x7E),
    min_size=0, max_size=60).map(str.strip)


@settings(max_examples=150, deadline=None)
@given(method=st.sampled_from(["GET", "HEAD", "POST", "PUT"]),
       path=st.text(alphabet="abcdefghij0123456789/._-", min_size=1,
                    max_size=60).map(lambda p: "/" + p),
       headers=st.lists(st.tuples(_token, _value), max_size=10))
def test_http1_request_head_roundtrip(method, path, headers):
    import asyncio

    from demodel_amd.proxy import http1

    head = http1.RequestHead(method, path, "HTTP/1.1", headers)
    raw = http1.serialize_request(head)

    async def parse():
        reader = asyncio.StreamReader()
        reader.feed_data(raw)
        reader.feed_eof()
        return await http1.read_request_head(reader)

    got = asyncio.new_event_loop().run_until_complete(parse())
    assert got.method == method
    assert got.target == path
    # full ordered list (duplicate names allowed; .get returns first)
    assert got.headers == [(k, v) for k, v in headers]


# --- GGUF header parser: untrusted-registry input ---------------------

@settings(max_examples=200, deadline=None)
@given(garbage=st.binary(min_size=0, max_size=3000))
def test_gguf_parse_garbage_fails_loudly(garbage):
    """Arbitrary bytes (including a valid magic prefix) must raise a
    clean error — never hang, crash, or allocate per declared counts."""
    import struct as _struct

    from demodel_amd.engine.formats import gguf

    for payload in (garbage, b"GGUF" + garbage):
        try:
            gguf.parse_bytes(payload)
        except (ValueError, _struct.error):
            pass


@settings(max_examples=60, deadline=None)
@given(names=st.lists(
    st.text(alphabet="abcdefgh.0123456789_", min_size=1, max_size=30),
    min_size=0, max_size=6, unique=True),
    qtype=st.sampled_from([0, 2, 8, 12, 14]))
def test_gguf_header_roundtrip(names, qtype):
    from demodel_amd.engine.formats import gguf

    blocked = 32 if qtype in (2, 8) else 256
    dims = (blocked * 2, 3) if qtype != 0 else (5, 7)
    tensors = [(n, dims, qtype) for n in names]
    prefix, total = gguf.build_virtual(tensors)
    gg = gguf.parse_bytes(prefix)
    assert [t.name for t in gg.tensors] == names
    for t in gg.tensors:
        assert t.dims == dims
        assert t.type_id == qtype
    assert total >= gg.data_offset


# --- safetensors header parser: untrusted input -----------------------

@settings(max_examples=200, deadline=None)
@given(garbage=st.binary(min_size=0, max_size=2000))
def test_safetensors_parse_garbage_fails_loudly(garbage):
    import json as _json

    from demodel_amd.engine.formats import safetensors as stf

    try:
        stf.parse_header(garbage)
    except (ValueError, _json.JSONDecodeError, KeyError, TypeError,
            AttributeError):
        pass


def test_safetensors_rejects_negative_offsets():
    import json as _json
    import struct as _struct

    from demodel_amd.engine.formats import safetensors as stf

    hdr = _json.dumps({"w": {"dtype": "F32", "shape": [2],
                             "data_offsets": [-8, 0]}}).encode()
    blob = _struct.pack("<Q", len(hdr)) + hdr
    with pytest.raises(ValueError):
        stf.parse_header(blob)


# --- parquet thrift compact walker: untrusted input -------------------

def test_thrift_boolean_list_bomb_raises_fast():
    """Regression: a crafted bool-list header declared 10^15 elements;
    the walker treated struct-bool encoding (zero bytes) as the element
    encoding and would spin for hours.  Compact-protocol list elements
    of bool are ONE byte each, and declared counts are bounded by the
    remaining bytes."""
    import time

    from demodel_amd.engine.formats import parquet as pqf

    # struct field 1, type LIST; list header: n=15 -> varint, elem=TRUE
    bomb = bytes([0x19, 0xF1]) + b"\xff\xff\xff\xff\xff\xff\x7f"
    t0 = time.time()
    with pytest.raises((ValueError, IndexError)):
        pqf.parse_page_header(bomb, 0)
    assert time.time() - t0 < 1.0


@settings(max_examples=300, deadline=None)
@given(garbage=st.binary(min_size=0, max_size=400))
def test_parquet_page_header_garbage_fails_loudly(garbage):
    from demodel_amd.engine.formats import parquet as pqf

    try:
        pqf.parse_page_header(garbage, 0)
    except (ValueError, IndexError, RecursionError):
        pass
