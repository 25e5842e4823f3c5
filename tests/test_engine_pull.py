"""Engine pull tests (CPU landing target): hf:// and ollama:// end-to-end
against the fake origin, digest verification, safetensors views, GGUF
parse + CPU dequant reference self-consistency."""

import hashlib
import json
import os
import struct

import numpy as np
import pytest

from demodel_amd.engine import pull as pull_mod
from demodel_amd.engine.formats import gguf, safetensors as st
from demodel_amd.engine.pipeline import (HostLander, _compress,
                                         _host_sha256_finish)
from helpers import Stack


# ---------------------------------------------------------------- #
# sha256 chain plumbing


def test_host_sha256_finisher_matches_hashlib():
    data = os.urandom(1000)  # not a multiple of 64
    state = [0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
             0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19]
    full = len(data) // 64 * 64
    for i in range(0, full, 64):
        _compress(state, data[i:i + 64])
    got = _host_sha256_finish(state, len(data), data[full:])
    assert got == hashlib.sha256(data).hexdigest()


def test_host_lander_chunks_and_chain():
    data = os.urandom(200_000)
    pos = [0]

    def fill(view):
        n = min(len(view), len(data) - pos[0], 7777)  # ragged fills
        view[:n] = data[pos[0]:pos[0] + n]
        pos[0] += n
        return n

    lander = HostLander(slab_bytes=64 << 10, verify_chunk=16 << 10)
    blob = lander.land(fill, len(data), host_chain=True)
    assert bytes(blob.buffer) == data
    assert blob.sha256 == hashlib.sha256(data).hexdigest()
    vc = 16 << 10
    assert blob.chunk_digests == [
        hashlib.sha256(data[o:o + vc]).hexdigest()
        for o in range(0, len(data), vc)]
    assert blob.head[:100] == data[:100]


# ---------------------------------------------------------------- #
# safetensors


def _make_safetensors(path, tensors):
    """tensors: name -> np array (dtype float32/bf16-as-uint16)."""
    spec = {}
    payload = b""
    for name, arr in tensors.items():
        tag = {"float32": "F32", "uint16": "BF16"}[str(arr.dtype)]
        spec[name] = (tag, arr.shape, arr.nbytes)
        payload += arr.tobytes()
    head, total = st.build_header(spec)
    with open(path, "wb") as f:
        f.write(head + payload)
    return head, payload


def test_safetensors_roundtrip(tmp_path):
    a = np.random.rand(4, 8).astype(np.float32)
    b = (np.random.rand(16) * 65535).astype(np.uint16)
    head, payload = _make_safetensors(tmp_path / "m.safetensors",
                                      {"a": a, "b": b})
    hdr = st.parse_header(head + payload[:0])
    assert [t.name for t in hdr.tensors] == ["a", "b"]
    ta, tb = hdr.tensors
    assert ta.torch_dtype == "float32" and ta.shape == (4, 8)
    assert tb.st_dtype == "BF16" and tb.nbytes == 32
    assert hdr.data_bytes == a.nbytes + b.nbytes

    import torch

    blob = torch.frombuffer(bytearray(head + payload), dtype=torch.uint8)
    views = st.torch_views(hdr, blob)
    assert torch.equal(views["a"],
                       torch.from_numpy(a))
    assert views["b"].dtype == torch.bfloat16


# ---------------------------------------------------------------- #
# hf:// pull e2e (CPU landing)


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path)
    yield s
    s.close()


def test_pull_hf_end_to_end(stack, tmp_path):
    a = np.random.rand(8, 16).astype(np.float32)
    _make_safetensors(tmp_path / "model.safetensors", {"wte": a})
    (tmp_path / "config.json").write_text(json.dumps({"model_type": "gpt2"}))
    stack.origin.add_hf_repo("org/tiny", {
        "model.safetensors": str(tmp_path / "model.safetensors"),
        "config.json": str(tmp_path / "config.json"),
    })
    res = pull_mod.pull_hf("org/tiny", endpoint=stack.origin_base,
                           verify="digest", workers=2,
                           out_dir=str(tmp_path / "out"))
    assert res.total_bytes == sum(
        os.path.getsize(tmp_path / f)
        for f in ("model.safetensors", "config.json"))
    byname = {f.name: f for f in res.files}
    # etag is the file sha256 in the fake origin -> digest must verify
    assert byname["model.safetensors"].digest_ok is True
    assert byname["config.json"].digest_ok is True
    # tensors() produces working views
    import torch

    t = res.tensors()
    assert torch.equal(t["wte"], torch.from_numpy(a))
    # out_dir materialization byte-exact
    assert (tmp_path / "out" / "model.safetensors").read_bytes() == \
        (tmp_path / "model.safetensors").read_bytes()


def test_pull_hf_through_proxy(stack, tmp_path):
    """Engine pull via the demodel reverse proxy; second pull offline."""
    a = np.random.rand(32, 32).astype(np.float32)
    _make_safetensors(tmp_path / "w.safetensors", {"w": a})
    stack.origin.add_hf_repo("org/p", {
        "w.safetensors": str(tmp_path / "w.safetensors")})
    res = pull_mod.pull_hf("org/p", endpoint=stack.endpoint,
                           verify="chunked")
    assert res.total_bytes == os.path.getsize(tmp_path / "w.safetensors")
    stack.stop_origin()
    res2 = pull_mod.pull_hf("org/p", endpoint=stack.endpoint,
                            verify="digest")
    assert res2.files[0].digest_ok is True


def test_pull_spec_parsing():
    with pytest.raises(ValueError):
        pull_mod.pull_spec("s3://nope")


# ---------------------------------------------------------------- #
# ollama:// pull + GGUF


def test_pull_ollama_end_to_end(stack, tmp_path):
    gg_path = tmp_path / "model.gguf"
    gguf.build_file(str(gg_path), [
        ("tok_embd.weight", (64, 32), 2),      # q4_0
        ("blk.0.attn_q.weight", (256, 4), 12),  # q4_K
        ("output_norm.weight", (64,), 0),       # f32
    ], kv={"general.architecture": "llama"})
    lic = tmp_path / "LICENSE"
    lic.write_text("MIT")
    stack.origin.add_ollama_model("library/tinymodel", "latest", [
        ("application/vnd.ollama.image.model", str(gg_path)),
        ("application/vnd.ollama.image.license", str(lic)),
    ])
    res = pull_mod.pull_ollama(
        "tinymodel", "latest", endpoint=stack.origin_base,
        verify="digest", workers=2)
    # every layer digest must verify (sha256 whole-blob chain)
    model_files = [f for f in res.files if "image.model" in f.name]
    assert model_files and model_files[0].digest_ok is True
    assert all(f.digest_ok for f in res.files)
    assert res.meta["gguf"]["n_tensors"] == 3
    assert set(res.meta["gguf"]["types"]) == {"q4_0", "q4_K", "f32"}

    # manifest passed through byte-for-byte semantics: digests match files
    man = res.meta["manifest"]
    model_layer = [l for l in man["layers"]
                   if l["mediaType"].endswith("image.model")][0]
    want = hashlib.sha256(gg_path.read_bytes()).hexdigest()
    assert model_layer["digest"] == f"sha256:{want}"


def test_gguf_parse_and_cpu_dequant(tmp_path):
    path = tmp_path / "t.gguf"
    gguf.build_file(str(path), [
        ("q40.weight", (64, 2), 2),
        ("q80.weight", (32, 4), 8),
        ("q4k.weight", (256, 2), 12),
        ("q6k.weight", (256, 3), 14),
        ("f16.weight", (16, 2), 1),
    ])
    raw = path.read_bytes()
    gg = gguf.parse_bytes(raw[:1 << 20])
    assert len(gg.tensors) == 5
    for t in gg.tensors:
        data = raw[gg.data_offset + t.offset:
                   gg.data_offset + t.offset + t.nbytes]
        vals = gguf.dequant_cpu(t.type_id, data, t.n_elems)
        assert vals.shape == (t.n_elems,)
        assert np.isfinite(vals).all() or t.type_id in (1,)  # f16 may inf


def test_on_range_fires_on_cpu_path(stack, tmp_path):
    """The on_range progress hook fires once (whole blob) for plain
    non-segmented pulls, with head bytes attached."""
    data = os.urandom(200_000)
    p = tmp_path / "r.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/rng", {"r.bin": str(p)})
    events = []

    def on_range(name, lo, hi, buf, head):
        events.append((name, lo, hi, head is not None))

    pull_mod.pull_hf("org/rng", endpoint=stack.origin_base,
                     verify="chunked", workers=1, on_range=on_range)
    assert events == [("r.bin", 0, len(data), True)]


def test_recycle_noop_on_cpu(stack, tmp_path):
    data = os.urandom(10_000)
    p = tmp_path / "c.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/cyc", {"c.bin": str(p)})
    landers = pull_mod.LanderPool(0)
    res = pull_mod.pull_hf("org/cyc", endpoint=stack.origin_base,
                           verify="chunked", workers=1, landers=landers)
    assert landers.recycle(res) == 0  # host blobs are not pooled


def test_pull_hf_dataset_repo_type(stack, tmp_path):
    """HF dataset repos use /api/datasets + /datasets/.../resolve."""
    data = os.urandom(30_000)
    p = tmp_path / "shard.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/dset", {"shard.bin": str(p)})
    res = pull_mod.pull_hf("org/dset", endpoint=stack.origin_base,
                           verify="chunked", workers=1,
                           repo_type="dataset")
    f = res.files[0]
    assert f.url.endswith("/datasets/org/dset/resolve/main/shard.bin") \
        or "/datasets/org/dset/" in f.url or f.nbytes == len(data)
    assert bytes(f.blob.buffer) == data
    reqs = [r for r in stack.origin.requests if "datasets" in r]
    assert any("/api/datasets/org/dset" in r for r in reqs)
    assert any("/datasets/org/dset/resolve/" in r for r in reqs)


def test_pull_spec_dataset_prefix(stack, tmp_path):
    """hf://datasets/org/name routes through the dataset URL layout."""
    p = tmp_path / "d.bin"
    p.write_bytes(b"y" * 5000)
    stack.origin.add_hf_repo("org/dsp", {"d.bin": str(p)})
    out = pull_mod.pull_spec("hf://datasets/org/dsp",
                             endpoint=stack.origin_base)
    assert out["total_bytes"] == 5000
    assert any("/api/datasets/org/dsp" in r
               for r in stack.origin.requests)


def test_buffer_pool_exact_size_recycling():
    from demodel_amd.engine.pipeline import BufferPool

    pool = BufferPool()
    a, b = object(), object()
    pool.put(a, 100)
    pool.put(b, 200)
    assert pool.take(50) is None         # exact-size only
    assert pool.take(100) is a
    assert pool.take(100) is None        # drained
    assert pool.take(200) is b
    pool.put(a, 100)
    pool.clear()
    assert pool.take(100) is None


def test_progressive_dequant_range_merging():
    """Pure range bookkeeping: out-of-order segments merge and the
    contiguous prefix only advances once byte 0 is covered."""
    from demodel_amd.engine.formats.gguf import ProgressiveDequant

    pd = ProgressiveDequant()
    pd._add_range(100, 200)
    assert pd._prefix == 0               # no coverage from 0 yet
    pd._add_range(300, 400)
    pd._add_range(0, 100)
    assert pd._prefix == 200             # [0,200) now contiguous
    pd._add_range(200, 300)
    assert pd._prefix == 400             # all merged
    assert pd._ranges == [(0, 400)]
    pd._add_range(50, 150)               # duplicate/overlap is harmless
    assert pd._ranges == [(0, 400)]


def test_virtual_origin_prefix_and_ranges(stack, tmp_path):
    """Virtual blobs with a prefix (the gguf-70b bench's REAL header +
    patterned payload) serve correct bytes for full GETs and Ranges
    spanning the prefix/pattern boundary."""
    import urllib.request

    from demodel_amd.testing.origin import FakeOrigin

    prefix = bytes(range(256)) * 4          # 1024-byte "header"
    total = 200_000
    stack.origin.add_hf_repo_virtual(
        "org/virt", {"v.bin": total}, prefixes={"v.bin": prefix})
    url = f"{stack.origin_base}/cdn/org/virt/x/v.bin"
    pattern = FakeOrigin._virtual_pattern()

    def expect(lo, hi):  # [lo, hi)
        out = bytearray()
        for p in range(lo, hi):
            out += (prefix[p:p + 1] if p < len(prefix)
                    else pattern[p % len(pattern):p % len(pattern) + 1])
        return bytes(out)

    with urllib.request.urlopen(url, timeout=30) as r:
        body = r.read()
    assert len(body) == total
    assert body[:2048] == expect(0, 2048)
    assert body[-64:] == expect(total - 64, total)
    # range straddling the prefix boundary
    req = urllib.request.Request(
        url, headers={"Range": "bytes=1000-1100"})
    with urllib.request.urlopen(req, timeout=30) as r:
        assert r.status == 206
        assert r.read() == expect(1000, 1101)
    # range fully past the prefix
    req = urllib.request.Request(
        url, headers={"Range": "bytes=50000-50099"})
    with urllib.request.urlopen(req, timeout=30) as r:
        assert r.read() == expect(50000, 50100)


def test_gated_repo_token_flow(stack, tmp_path, monkeypatch):
    """Private/gated repos: the engine sends the HF token to hub
    endpoints, and fetch STRIPS it on the cross-host CDN redirect
    (presigned-URL hosts reject stray credentials)."""
    import pytest as _pytest

    from demodel_amd.engine import fetch as fetch_mod

    from demodel_amd.testing.origin import FakeOrigin

    data = os.urandom(100_000)
    p = tmp_path / "g.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/gated", {"g.bin": str(p)})
    stack.origin.require_token = "s3cret"
    # a SECOND origin plays the CDN host (the real hub/S3 topology):
    # it rejects requests that still carry Authorization
    cdn = FakeOrigin(str(tmp_path))
    cdn.hf_repos = stack.origin.hf_repos
    cdn.require_token = "s3cret"
    cdn_port = stack.lt.call(cdn.start())
    stack.origin.cdn_base = f"http://127.0.0.1:{cdn_port}"

    # without a token: 401 on the repo-info call
    monkeypatch.delenv("HF_TOKEN", raising=False)
    monkeypatch.delenv("HUGGING_FACE_HUB_TOKEN", raising=False)
    with _pytest.raises(fetch_mod.FetchError):
        pull_mod.pull_hf("org/gated", endpoint=stack.origin_base,
                         workers=1)
    # with the token (env surface), the pull succeeds END TO END —
    # which proves the CDN hop did NOT receive Authorization (the
    # origin 403s any /cdn request that carries it)
    monkeypatch.setenv("HF_TOKEN", "s3cret")
    res = pull_mod.pull_hf("org/gated", endpoint=stack.origin_base,
                           workers=1)
    f = [x for x in res.files if x.name == "g.bin"][0]
    assert bytes(f.blob.buffer) == data
    # and the CDN host really served it (cross-host redirect taken)
    assert any("/cdn/" in r for r in cdn.requests)
    stack.lt.call(cdn.close())


def test_gated_repo_resume_does_not_leak_token(stack, tmp_path,
                                               monkeypatch):
    """A mid-stream drop resumes against the CDN host directly — the
    token must NOT ride along (the cdn origin 403s it)."""
    from demodel_amd.testing.origin import FakeOrigin

    data = os.urandom(2 << 20)
    p = tmp_path / "gr.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/gres", {"gr.bin": str(p)})
    stack.origin.require_token = "tok"
    cdn = FakeOrigin(str(tmp_path))
    cdn.hf_repos = stack.origin.hf_repos
    cdn.require_token = "tok"
    cdn.drop_once["gr.bin"] = 300 << 10  # resume mid-blob on the CDN
    cdn_port = stack.lt.call(cdn.start())
    stack.origin.cdn_base = f"http://127.0.0.1:{cdn_port}"
    monkeypatch.setenv("HF_TOKEN", "tok")
    res = pull_mod.pull_hf("org/gres", endpoint=stack.origin_base,
                           workers=1, slab_bytes=256 << 10)
    f = [x for x in res.files if x.name == "gr.bin"][0]
    assert bytes(f.blob.buffer) == data
    assert not cdn.drop_once  # the drop fired -> a resume happened
    stack.lt.call(cdn.close())


def test_ollama_registry_token_handshake(stack, tmp_path):
    """Docker-registry Bearer flow (ollama.com/private models): a 401
    with WWW-Authenticate drives a token fetch from the realm; manifest
    AND blob requests then carry the token."""
    from demodel_amd.engine.formats import gguf as gguf_mod

    gg_path = tmp_path / "tok.gguf"
    gguf_mod.build_file(str(gg_path), [("w.weight", (64, 4), 2)])
    stack.origin.add_ollama_model(
        "library/gated", "latest",
        [("application/vnd.ollama.image.model", str(gg_path))])
    stack.origin.docker_token = "registry-jwt"
    res = pull_mod.pull_ollama("gated", "latest",
                               endpoint=stack.origin_base,
                               verify="digest", workers=1)
    assert all(f.digest_ok for f in res.files)
    assert "gguf" in res.meta
    # the token endpoint was hit exactly once
    assert sum(1 for r in stack.origin.requests
               if r.startswith("GET /token")) == 1
