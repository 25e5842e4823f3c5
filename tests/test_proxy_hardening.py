"""Round-2 proxy hardening: upstream keep-alive pooling, read timeouts,
bounded transfer log, strict /__demodel routing, streaming PUT/POST
pass-through, off-loop cache fills (VERDICT round-1 items 2 and 7)."""

import hashlib
import json
import os
import urllib.request

import pytest

from demodel_amd.utils.log import TransferLog
from helpers import Stack


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path)
    yield s
    s.close()


def _get(url, timeout=20, method="GET", data=None, headers=None):
    req = urllib.request.Request(url, method=method, data=data,
                                 headers=headers or {})
    return urllib.request.urlopen(req, timeout=timeout)


def test_transfer_log_is_bounded():
    tl = TransferLog(max_records=8)
    for i in range(100):
        tl.record(event="miss", bytes=10)
    for i in range(50):
        tl.record(event="hit", bytes=5)
    assert len(tl.records) == 8          # ring stays bounded
    assert tl.n_requests == 150          # aggregates stay exact
    assert tl.counts == {"miss": 100, "hit": 50}
    assert tl.bytes == {"miss": 1000, "hit": 250}
    assert tl.total_bytes() == 1250


def test_strict_demodel_routing(stack):
    with pytest.raises(urllib.error.HTTPError) as ei:
        _get(f"{stack.endpoint}/__demodel/bogus")
    assert ei.value.code == 404
    with pytest.raises(urllib.error.HTTPError) as ei:
        _get(f"{stack.endpoint}/__demodel/statsfoo")
    assert ei.value.code == 404
    with _get(f"{stack.endpoint}/__demodel/stats") as r:
        obj = json.loads(r.read())
    assert "upstream_pool" in obj


def test_upstream_keepalive_reuse(stack, tmp_path):
    """Sequential proxied requests reuse ONE upstream connection
    (reference/goproxy behavior; round 1 opened one per request)."""
    files = {}
    for i in range(5):
        p = tmp_path / f"f{i}.bin"
        p.write_bytes(os.urandom(10_000))
        files[f"f{i}.bin"] = str(p)
    stack.origin.add_hf_repo("org/ka", files)
    for i in range(5):
        with _get(f"{stack.endpoint}/org/ka/resolve/main/f{i}.bin") as r:
            r.read()
    # each pull is 2 upstream requests (resolve 302 + cdn GET) but they
    # should all ride a handful of pooled connections, not 10
    assert stack.origin.connections <= 2, stack.origin.connections
    with _get(f"{stack.endpoint}/__demodel/stats") as r:
        obj = json.loads(r.read())
    assert obj["upstream_pool"]["reused"] >= 8


def test_upstream_read_timeout_504(stack, tmp_path, monkeypatch):
    import demodel_amd.proxy.server as server_mod

    monkeypatch.setattr(server_mod, "READ_TIMEOUT", 0.5)
    p = tmp_path / "h.bin"
    p.write_bytes(b"x" * 100)
    stack.origin.add_hf_repo("org/hang", {"h.bin": str(p)})
    stack.origin.hang_once.add("h.bin")
    with pytest.raises(urllib.error.HTTPError) as ei:
        _get(f"{stack.endpoint}/org/hang/resolve/main/h.bin", timeout=30)
    assert ei.value.code == 504
    # the fault is consumed; the next request succeeds
    with _get(f"{stack.endpoint}/org/hang/resolve/main/h.bin") as r:
        assert r.read() == b"x" * 100


def test_streaming_post_passthrough(stack):
    """A PUT/POST body larger than the buffer threshold streams through
    the proxy to the origin without truncation."""
    body = os.urandom(3 << 20)  # 3 MiB > _REQ_BUFFER_MAX
    with _get(f"{stack.endpoint}/echo", method="POST", data=body,
              headers={"Content-Type": "application/octet-stream"}) as r:
        obj = json.loads(r.read())
    assert obj["bytes"] == len(body)
    assert obj["sha256"] == hashlib.sha256(body).hexdigest()


def test_small_post_passthrough(stack):
    body = b"hello world" * 10
    with _get(f"{stack.endpoint}/echo", method="POST", data=body) as r:
        obj = json.loads(r.read())
    assert obj["bytes"] == len(body)
    assert obj["sha256"] == hashlib.sha256(body).hexdigest()


@pytest.mark.parametrize("prefetch", [False, True])
def test_concurrent_mixed_load_soak(stack, tmp_path, prefetch):
    """32 threads hammer the proxy with mixed hits/misses/ranges of
    several blobs at once — exercises the upstream pool, the in-flight
    fill registry (thundering herd on a cold URI), and the threaded
    relay under contention.  Everything must come back byte-exact.
    With prefetch=True, auto pull-ahead landings run concurrently with
    the serving load (registry + lander contention)."""
    import concurrent.futures as cf
    import random

    if prefetch:
        from demodel_amd.engine.pull import LanderPool
        from demodel_amd.engine.registry import BlobRegistry

        stack.cfg.gpu_prefetch = "auto"
        stack.proxy.prefetch_landers = LanderPool(0, gpu=False)
        stack.proxy.registry = BlobRegistry()

    blobs = {}
    paths = {}
    for i in range(6):
        data = os.urandom((i + 1) * 300_000)
        p = tmp_path / f"s{i}.bin"
        p.write_bytes(data)
        blobs[f"s{i}.bin"] = data
        paths[f"s{i}.bin"] = str(p)
    stack.origin.add_hf_repo("org/soak", paths)

    rng = random.Random(7)

    def one(k):
        name = rng.choice(list(blobs))
        want = blobs[name]
        url = f"{stack.endpoint}/org/soak/resolve/main/{name}"
        if k % 3 == 0 and len(want) > 10_000:
            lo = rng.randrange(0, len(want) - 5_000)
            hi = lo + 4_999
            req = urllib.request.Request(
                url, headers={"Range": f"bytes={lo}-{hi}"})
            with urllib.request.urlopen(req, timeout=30) as r:
                assert r.read() == want[lo:hi + 1], (name, lo)
        else:
            with urllib.request.urlopen(url, timeout=30) as r:
                assert r.read() == want, name
        return True

    with cf.ThreadPoolExecutor(max_workers=32) as ex:
        assert all(ex.map(one, range(160)))
    # origin traffic stays FAR below one GET per request: full-body
    # misses dedup through the in-flight fill registry, and everything
    # after the first full fill is a cache hit (cold Range requests
    # legitimately pass through until then)
    blob_gets = [r for r in stack.origin.requests if "/cdn/" in r]
    assert len(blob_gets) < 60, len(blob_gets)  # 160 requests issued


def test_relay_upstream_drop_mid_body(stack, tmp_path):
    """Origin dies mid-blob during the threaded relay: the client must
    see a hard failure (truncated/closed), the cache entry must be
    dropped, and the next request must succeed end-to-end."""
    import http.client

    data = os.urandom(4 << 20)
    p = tmp_path / "drop.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/drop", {"drop.bin": str(p)})
    stack.origin.drop_once["drop.bin"] = 1 << 20
    url = f"{stack.endpoint}/org/drop/resolve/main/drop.bin"
    got = None
    try:
        with _get(url) as r:
            got = r.read()
    except (http.client.IncompleteRead, ConnectionResetError,
            urllib.error.HTTPError, OSError):
        got = None
    assert got != data  # truncated or errored, never silently complete
    # the aborted fill must not have poisoned the cache
    with _get(url) as r:
        assert r.read() == data
    stack.stop_origin()  # and the good body IS cached
    with _get(url) as r:
        assert r.read() == data
        assert r.headers["X-Demodel-Cache"] == "HIT"


def test_upstream_pool_unit(stack, tmp_path):
    """Pool mechanics: reuse, idle-TTL expiry, max-idle cap."""
    import asyncio

    from demodel_amd.proxy.server import UpstreamPool

    host, port = "127.0.0.1", stack.origin_port

    async def scenario():
        pool = UpstreamPool(max_idle_per_key=1, idle_ttl=0.3)
        r1, w1, reused = await pool.acquire(host, port, None)
        assert reused is False
        pool.release(host, port, False, r1, w1)
        r2, w2, reused = await pool.acquire(host, port, None)
        assert reused is True and w2 is w1
        # max_idle=1: releasing two keeps only one
        r3, w3, _ = await pool.acquire(host, port, None)
        pool.release(host, port, False, r2, w2)
        pool.release(host, port, False, r3, w3)
        assert len(pool._idle[(host, port, False)]) == 1
        assert w3.is_closing()  # the overflow one was closed
        # TTL expiry: after the idle window the conn is discarded
        await asyncio.sleep(0.9)
        r4, w4, reused = await pool.acquire(host, port, None)
        assert reused is False
        pool.close_all()
        w4.close()

    stack.lt.call(scenario())


def test_cache_fill_still_correct_off_loop(stack, tmp_path):
    """Cache fills now run on a worker thread; the entry must still be
    byte-exact and replayable with the origin down."""
    data = os.urandom(2 << 20)
    p = tmp_path / "c.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/cw", {"c.bin": str(p)})
    url = f"{stack.endpoint}/org/cw/resolve/main/c.bin"
    with _get(url) as r:
        assert r.read() == data
    stack.stop_origin()
    with _get(url) as r:
        assert r.read() == data
        assert r.headers["X-Demodel-Cache"] == "HIT"


def test_proxy_fleet_reuseport(tmp_path):
    """ProxyFleet: 3 acceptor loops on ONE port (SO_REUSEPORT) share
    the cache; concurrent clients get byte-exact bodies and the kernel
    spreads connections across loops."""
    import concurrent.futures as cf

    from demodel_amd.config import Config
    from demodel_amd.proxy.server import ProxyFleet
    from demodel_amd.testing.origin import FakeOrigin
    from helpers import LoopThread

    lt = LoopThread()
    origin = FakeOrigin(str(tmp_path))
    data = os.urandom(2 << 20)
    p = tmp_path / "f.bin"
    p.write_bytes(data)
    origin.add_hf_repo("org/fleet", {"f.bin": str(p)})
    oport = lt.call(origin.start())

    cfg = Config(host="127.0.0.1", port=0,
                 cache_dir=str(tmp_path / "fleetcache"))
    fleet = ProxyFleet(cfg, loops=3)
    for srv in []:
        pass
    port = fleet.start()
    for srv in fleet.servers:
        srv.reverse_routes = [("/", f"http://127.0.0.1:{oport}")]
    try:
        url = f"http://127.0.0.1:{port}/org/fleet/resolve/main/f.bin"

        def one(_):
            with urllib.request.urlopen(url, timeout=30) as r:
                return r.read() == data

        with cf.ThreadPoolExecutor(max_workers=12) as ex:
            assert all(ex.map(one, range(48)))
        st = fleet.stats()
        assert st["requests"] >= 48
        # SO_REUSEPORT balance: with 48 conns, >=2 loops served traffic
        assert sum(1 for n in st["per_loop"] if n > 0) >= 2, st
        # the shared cache bounds origin traffic to the cold herd
        # (in-flight dedup is per-loop and opens at response-head
        # time), far below one fetch per request
        blob_gets = [r for r in origin.requests if "/cdn/" in r]
        assert len(blob_gets) <= 16, len(blob_gets)
    finally:
        fleet.close()
        lt.call(origin.close())
        lt.stop()
