"""Reverse-mode proxy tests (HF_ENDPOINT-style): client talks origin-form
HTTP to the proxy, proxy forwards to the fake origin, caches, replays.

This is the minimum end-to-end slice of SURVEY.md §7 step 2 /
BASELINE.json config 1 (tiny-random-gpt2, CPU only).
"""

import hashlib
import json
import os
import urllib.request

import pytest

from helpers import Stack


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path)
    yield s
    s.close()


def _get(url, method="GET", headers=None):
    req = urllib.request.Request(url, method=method,
                                 headers=headers or {})
    with urllib.request.urlopen(req, timeout=20) as r:
        return r.status, dict(r.headers), r.read()


def _make_repo(stack, tmp_path, n_files=2, size=100_000):
    files = {}
    for i in range(n_files):
        p = tmp_path / f"model-{i}.bin"
        p.write_bytes(os.urandom(size))
        files[f"model-{i}.bin"] = str(p)
    cfgp = tmp_path / "config.json"
    cfgp.write_text(json.dumps({"model_type": "gpt2"}))
    files["config.json"] = str(cfgp)
    commit = stack.origin.add_hf_repo("test-org/tiny", files)
    return files, commit


def test_api_model_info_proxied(stack, tmp_path):
    _make_repo(stack, tmp_path)
    status, headers, body = _get(
        f"{stack.endpoint}/api/models/test-org/tiny")
    assert status == 200
    info = json.loads(body)
    assert info["id"] == "test-org/tiny"
    assert {s["rfilename"] for s in info["siblings"]} == {
        "model-0.bin", "model-1.bin", "config.json"}


def test_blob_pull_cache_hit_and_offline_replay(stack, tmp_path):
    files, commit = _make_repo(stack, tmp_path)
    url = f"{stack.endpoint}/test-org/tiny/resolve/main/model-0.bin"
    expect = open(files["model-0.bin"], "rb").read()

    s1, h1, b1 = _get(url)
    assert s1 == 200 and b1 == expect
    assert "X-Demodel-Cache" not in h1

    # the origin 302-redirects to /cdn/; proxy must follow internally
    assert any("/cdn/" in r for r in stack.origin.requests)

    n_origin_reqs = len(stack.origin.requests)
    s2, h2, b2 = _get(url)
    assert s2 == 200 and b2 == expect
    assert h2.get("X-Demodel-Cache") == "HIT"
    assert len(stack.origin.requests) == n_origin_reqs  # no upstream traffic

    # offline replay: kill the origin entirely, cache still serves
    stack.stop_origin()
    s3, _, b3 = _get(url)
    assert s3 == 200 and b3 == expect


def test_head_request_not_poisoning_cache(stack, tmp_path):
    files, _ = _make_repo(stack, tmp_path)
    url = f"{stack.endpoint}/test-org/tiny/resolve/main/model-1.bin"
    s, h, b = _get(url, method="HEAD")
    assert s == 200 and b == b""
    expect = open(files["model-1.bin"], "rb").read()
    s2, _, b2 = _get(url)
    assert s2 == 200 and b2 == expect


def test_cache_entry_chunk_digests_match(stack, tmp_path):
    files, commit = _make_repo(stack, tmp_path, n_files=1, size=300_000)
    url = f"{stack.endpoint}/test-org/tiny/resolve/main/model-0.bin"
    _get(url)
    data = open(files["model-0.bin"], "rb").read()
    # the blob was cached under its final (CDN) URI with a whole-body sha256
    # (server-side record lands a beat after the client finishes reading)
    import time
    cached = []
    for _ in range(50):
        cached = [r for r in stack.proxy.transfers.records
                  if r["event"] == "miss" and "/cdn/" in r["uri"]]
        if cached:
            break
        time.sleep(0.05)
    assert cached
    # digests are computed asynchronously after commit
    hit = None
    for _ in range(100):
        hit = stack.proxy.cache.lookup(cached[0]["uri"])
        if hit is not None and hit.sha256:
            break
        time.sleep(0.05)
    assert hit is not None
    assert hit.sha256 == hashlib.sha256(data).hexdigest()
    assert hit.read_body() == data


def test_huggingface_hub_snapshot_download(stack, tmp_path, monkeypatch):
    """The real huggingface_hub client pulls through the proxy (config 1)."""
    hub = pytest.importorskip("huggingface_hub")
    files, commit = _make_repo(stack, tmp_path, n_files=2, size=50_000)

    # the CI image exports HF_HUB_OFFLINE=1 and huggingface_hub bakes it in
    # at import; flip the baked constant for this in-process test
    monkeypatch.delenv("HF_HUB_OFFLINE", raising=False)
    import huggingface_hub.constants as hf_const
    monkeypatch.setattr(hf_const, "HF_HUB_OFFLINE", False)
    monkeypatch.setenv("HF_ENDPOINT", stack.endpoint)
    monkeypatch.setenv("HF_HUB_DISABLE_XET", "1")
    monkeypatch.setenv("HF_HUB_DISABLE_TELEMETRY", "1")
    monkeypatch.setenv("HF_HUB_ETAG_TIMEOUT", "20")
    dst1 = tmp_path / "dl1"
    out = hub.snapshot_download(
        "test-org/tiny", cache_dir=str(tmp_path / "hfcache1"),
        local_dir=str(dst1), endpoint=stack.endpoint)
    for name, src in files.items():
        got = open(os.path.join(out, name), "rb").read()
        assert got == open(src, "rb").read(), name

    # second pull, fresh hub cache: all blobs must come from the proxy cache
    before = len(stack.origin.requests)
    dst2 = tmp_path / "dl2"
    out2 = hub.snapshot_download(
        "test-org/tiny", cache_dir=str(tmp_path / "hfcache2"),
        local_dir=str(dst2), endpoint=stack.endpoint)
    for name, src in files.items():
        got = open(os.path.join(out2, name), "rb").read()
        assert got == open(src, "rb").read(), name
    blob_reqs = [r for r in stack.origin.requests[before:]
                 if "/cdn/" in r or "resolve" in r]
    assert not [r for r in blob_reqs if r.startswith("GET") and "/cdn/" in r]
