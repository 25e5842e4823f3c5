"""Cache layout tests — byte-compat with reference CONTRIBUTING.md:53-153."""

import gzip
import hashlib
import json
import os

from demodel_amd.cache import CacheStore, cache_key


def test_key_is_16_hex():
    k = cache_key("https://registry.ollama.ai/v2/library/x/manifests/latest")
    assert len(k) == 16
    int(k, 16)  # parses as hex


def test_body_stored_byte_exact_gzip(tmp_path):
    """Bodies keep their original Content-Encoding (CONTRIBUTING.md:116)."""
    store = CacheStore(str(tmp_path))
    uri = "https://registry.ollama.ai/v2/library/m/manifests/latest"
    raw = json.dumps({"schemaVersion": 2, "layers": []}).encode()
    body = gzip.compress(raw)
    w = store.writer(uri, 200, "OK",
                     [("Content-Encoding", "gzip"),
                      ("Content-Type", "application/json"),
                      ("Transfer-Encoding", "chunked")])  # hop-by-hop dropped
    w.write(body)
    entry = w.finalize()

    # on-disk layout: {root}/{key} + {root}/{key}.meta
    key = cache_key(uri)
    body_path = tmp_path / key
    assert entry.body_path == str(body_path)
    assert body_path.read_bytes() == body
    assert body_path.read_bytes()[:2] == b"\x1f\x8b"  # gzip magic, as cached
    assert os.path.exists(str(body_path) + ".meta")

    hit = store.lookup(uri)
    assert hit is not None
    assert hit.status == 200
    assert hit.read_body() == body
    assert gzip.decompress(hit.read_body()) == raw
    hdrs = dict((k.lower(), v) for k, v in hit.headers)
    assert hdrs["content-encoding"] == "gzip"
    assert "transfer-encoding" not in hdrs  # hop-by-hop never replayed


def test_chunk_digests(tmp_path):
    store = CacheStore(str(tmp_path), chunk_bytes=1024)
    uri = "https://host/x"
    data = os.urandom(3000)
    w = store.writer(uri, 200, "OK", [])
    # write in awkward pieces to exercise chunk-boundary tracking
    w.write(data[:700])
    w.write(data[700:1500])
    w.write(data[1500:])
    entry = w.finalize()
    assert entry.sha256 == hashlib.sha256(data).hexdigest()
    assert entry.chunk_sha256 == [
        hashlib.sha256(data[0:1024]).hexdigest(),
        hashlib.sha256(data[1024:2048]).hexdigest(),
        hashlib.sha256(data[2048:3000]).hexdigest(),
    ]


def test_miss_and_purge(tmp_path):
    store = CacheStore(str(tmp_path))
    assert store.lookup("https://host/none") is None
    w = store.writer("https://host/a", 200, "OK", [])
    w.write(b"hello")
    w.finalize()
    assert store.lookup("https://host/a") is not None
    assert store.purge("https://host/a")
    assert store.lookup("https://host/a") is None


def test_abort_leaves_nothing(tmp_path):
    store = CacheStore(str(tmp_path))
    w = store.writer("https://host/b", 200, "OK", [])
    w.write(b"partial")
    w.abort()
    assert store.lookup("https://host/b") is None
    assert list(p for p in os.listdir(tmp_path) if not p.startswith(".")) == []


def test_torn_entry_is_miss(tmp_path):
    store = CacheStore(str(tmp_path))
    uri = "https://host/c"
    w = store.writer(uri, 200, "OK", [])
    w.write(b"0123456789")
    entry = w.finalize()
    with open(entry.body_path, "wb") as f:
        f.write(b"0123")  # truncate the body behind the meta's back
    assert store.lookup(uri) is None


def test_gc_lru(tmp_path):
    import time as _time

    store = CacheStore(str(tmp_path))
    for i in range(5):
        w = store.writer(f"https://host/gc{i}", 200, "OK", [])
        w.write(b"x" * 1000)
        w.finalize()
        _time.sleep(0.02)
    # touch entry 0 so it becomes most-recent
    e0 = store.lookup("https://host/gc0")
    now = _time.time()
    os.utime(e0.body_path, (now + 10, now + 10))
    res = store.gc(max_bytes=2500)
    assert res["evicted"] == 3
    assert store.lookup("https://host/gc0") is not None  # recently used
    assert store.lookup("https://host/gc4") is not None  # newest
    assert store.lookup("https://host/gc1") is None


def test_verify_cache_cli(tmp_path):
    from demodel_amd.config import Config
    from demodel_amd.engine.pull import verify_cache

    store = CacheStore(str(tmp_path))
    w = store.writer("https://host/v", 200, "OK", [])
    w.write(os.urandom(5000))
    w.finalize()
    cfg = Config(cache_dir=str(tmp_path))
    res = verify_cache(cfg)
    assert res["ok"] and res["checked"] == 1

    # corrupt the body -> verify fails
    e = store.lookup("https://host/v")
    with open(e.body_path, "r+b") as f:
        f.seek(100)
        f.write(b"\x00\x01\x02")
    res2 = verify_cache(cfg)
    assert not res2["ok"] and res2["bad"] == ["https://host/v"]


def test_gc_reaps_orphaned_part_files(tmp_path):
    """A fill temp left by a crashed process is reaped by gc once
    stale; fresh temps (an active fill) are left alone."""
    import os
    import time

    from demodel_amd.cache import CacheStore

    store = CacheStore(str(tmp_path))
    stale = tmp_path / ".deadbeef.abc.part"
    stale.write_bytes(b"x" * 100)
    os.utime(stale, (time.time() - 7200, time.time() - 7200))
    fresh = tmp_path / ".cafef00d.xyz.meta.part"
    fresh.write_bytes(b"y")
    store.gc(10 << 30)
    assert not stale.exists()
    assert fresh.exists()
