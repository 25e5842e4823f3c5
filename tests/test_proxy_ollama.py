"""Ollama registry protocol through the proxy — the reference's worked
example (CONTRIBUTING.md:23-153): manifest fetch cached byte-exact in its
original gzip Content-Encoding, layer blobs by sha256 digest, offline
replay, stats endpoint."""

import gzip
import hashlib
import json
import os
import urllib.request

import pytest

from demodel_amd.cache import cache_key
from demodel_amd.engine import pull as pull_mod
from helpers import Stack


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path)
    yield s
    s.close()


def _get(url, headers=None):
    req = urllib.request.Request(url, headers=headers or {})
    with urllib.request.urlopen(req, timeout=20) as r:
        return r.status, dict(r.headers), r.read()


def test_ollama_manifest_cached_gzip_byte_exact(stack, tmp_path):
    blob = tmp_path / "layer.bin"
    blob.write_bytes(os.urandom(100_000))
    manifest = stack.origin.add_ollama_model(
        "library/nomic-embed-text", "latest",
        [("application/vnd.ollama.image.model", str(blob))])

    url = (f"{stack.endpoint}/v2/library/nomic-embed-text"
           f"/manifests/latest")
    st, h1, body = _get(url)
    assert st == 200
    # urllib does not auto-decode; the proxy forwarded the gzip body
    assert h1.get("Content-Encoding") == "gzip"
    assert body[:2] == b"\x1f\x8b"  # the reference's xxd observation
    decoded = json.loads(gzip.decompress(body))
    assert decoded["schemaVersion"] == 2
    assert decoded["layers"][0]["mediaType"] == \
        "application/vnd.ollama.image.model"

    # cache layout: body stored byte-exact, still gzip
    # (CONTRIBUTING.md:53-121)
    origin_uri = (f"http://127.0.0.1:{stack.origin_port}"
                  f"/v2/library/nomic-embed-text/manifests/latest")
    # cache finalize lands a beat after the client finishes reading
    import time

    entry = None
    for _ in range(50):
        entry = stack.proxy.cache.lookup(origin_uri)
        if entry is not None:
            break
        time.sleep(0.05)
    assert entry is not None
    assert entry.read_body() == body
    assert os.path.basename(entry.body_path) == cache_key(origin_uri)

    # layer blob by digest, then offline replay of everything
    digest = decoded["layers"][0]["digest"]
    bst, _, bbody = _get(f"{stack.endpoint}/v2/library/nomic-embed-text"
                         f"/blobs/{digest}")
    assert bst == 200
    assert hashlib.sha256(bbody).hexdigest() == digest.split(":")[1]

    stack.stop_origin()
    st2, h2, body2 = _get(url)
    assert st2 == 200 and body2 == body
    assert h2.get("X-Demodel-Cache") == "HIT"
    _, _, bbody2 = _get(f"{stack.endpoint}/v2/library/nomic-embed-text"
                        f"/blobs/{digest}")
    assert bbody2 == bbody


def test_engine_ollama_pull_via_proxy(stack, tmp_path):
    blob = tmp_path / "m.bin"
    blob.write_bytes(os.urandom(64_000))
    stack.origin.add_ollama_model("library/m", "latest",
                                  [("application/vnd.ollama.image.model",
                                    str(blob))])
    res = pull_mod.pull_ollama("m", "latest", endpoint=stack.endpoint,
                               verify="digest", workers=1, dequant=False)
    assert all(f.digest_ok for f in res.files)
    # replay offline through the proxy cache
    stack.stop_origin()
    res2 = pull_mod.pull_ollama("m", "latest", endpoint=stack.endpoint,
                                verify="digest", workers=1, dequant=False)
    assert all(f.digest_ok for f in res2.files)


def test_stats_endpoint(stack, tmp_path):
    p = tmp_path / "s.bin"
    p.write_bytes(os.urandom(10_000))
    stack.origin.add_hf_repo("o/s", {"s.bin": str(p)})
    url = f"{stack.endpoint}/o/s/resolve/main/s.bin"
    _get(url)
    _get(url)  # hit
    st, _, body = _get(f"{stack.endpoint}/__demodel/stats")
    assert st == 200
    stats = json.loads(body)
    assert stats["cache_hits"] >= 1
    assert stats["cache_misses"] >= 1
    assert stats["hit_bytes"] >= 10_000
