"""Forward-proxy tests: CONNECT + TLS MITM and blind tunnel.

Covers the reference's CONNECT policy & MITM paths (start.go:183-196,
CertStorage start.go:27-165): client sets HTTPS_PROXY at the proxy and
trusts the demodel CA; a MITM'd host is terminated with a minted leaf and
cached; a non-listed host is tunneled blind.
"""

import json
import os
import ssl
import tempfile
import urllib.request

import pytest

from demodel_amd import _native
from demodel_amd.ca import CA
from demodel_amd.certs import LeafStore
from helpers import Stack


def _origin_tls(tmp_path):
    """Fake origin's own (non-demodel) self-signed chain."""
    ca_cert, ca_key = _native.ca_create(ecdsa=True)
    leaf_cert, leaf_key = _native.leaf_create(
        ca_cert, ca_key, "127.0.0.1", ecdsa=True)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    cp = tmp_path / "origin-leaf.crt"
    kp = tmp_path / "origin-leaf.key"
    cp.write_text(leaf_cert + ca_cert)
    kp.write_text(leaf_key)
    ctx.load_cert_chain(str(cp), str(kp))
    cafile = tmp_path / "origin-ca.crt"
    cafile.write_text(ca_cert)
    return ctx, str(cafile)


@pytest.fixture()
def mitm_stack(tmp_path):
    origin_ctx, origin_cafile = _origin_tls(tmp_path)
    demodel_ca_cert, demodel_ca_key = _native.ca_create(ecdsa=True)
    leafs = LeafStore(CA(demodel_ca_cert, demodel_ca_key))
    s = Stack(
        tmp_path,
        origin_tls_ctx=origin_ctx,
        leafs=leafs,
        cfg_kwargs={"upstream_cafile": origin_cafile},
    )
    # MITM exactly the fake origin host:port
    s.cfg.mitm_hosts = [f"127.0.0.1:{s.origin_port}"]
    demodel_cafile = tmp_path / "demodel-ca.crt"
    demodel_cafile.write_text(demodel_ca_cert)
    s.client_cafile = str(demodel_cafile)
    yield s
    s.close()


def _https_via_proxy(url, proxy_port, cafile, method="GET"):
    ctx = ssl.create_default_context(cafile=cafile)
    opener = urllib.request.build_opener(
        urllib.request.ProxyHandler(
            {"https": f"http://127.0.0.1:{proxy_port}"}),
        urllib.request.HTTPSHandler(context=ctx))
    req = urllib.request.Request(url, method=method)
    with opener.open(req, timeout=20) as r:
        return r.status, dict(r.headers), r.read()


def test_mitm_pull_and_cache_hit(mitm_stack, tmp_path):
    s = mitm_stack
    blob = tmp_path / "w.bin"
    blob.write_bytes(os.urandom(200_000))
    s.origin.add_hf_repo("org/m", {"w.bin": str(blob)})
    s.origin.redirect_blobs = False  # serve directly over the MITM'd conn

    url = (f"https://127.0.0.1:{s.origin_port}"
           f"/org/m/resolve/main/w.bin")
    st, h1, b1 = _https_via_proxy(url, s.proxy_port, s.client_cafile)
    assert st == 200 and b1 == blob.read_bytes()
    assert "X-Demodel-Cache" not in h1

    n = len(s.origin.requests)
    st2, h2, b2 = _https_via_proxy(url, s.proxy_port, s.client_cafile)
    assert st2 == 200 and b2 == blob.read_bytes()
    assert h2.get("X-Demodel-Cache") == "HIT"
    assert len(s.origin.requests) == n  # served without touching origin


def test_mitm_leaf_is_demodel_signed(mitm_stack):
    """The client's TLS peer inside the tunnel is the minted leaf."""
    s = mitm_stack
    import socket

    raw = socket.create_connection(("127.0.0.1", s.proxy_port), timeout=10)
    raw.sendall(
        f"CONNECT 127.0.0.1:{s.origin_port} HTTP/1.1\r\n"
        f"Host: 127.0.0.1:{s.origin_port}\r\n\r\n".encode())
    resp = raw.recv(1024)
    assert b"200" in resp.split(b"\r\n")[0]
    ctx = ssl.create_default_context(cafile=s.client_cafile)
    tls = ctx.wrap_socket(raw, server_hostname="127.0.0.1")
    cert = tls.getpeercert()
    issuer = dict(x[0] for x in cert["issuer"])
    assert issuer["commonName"] == "demodel-amd Root CA"
    tls.close()


def test_non_mitm_host_tunnels_blind(mitm_stack, tmp_path):
    """A host not in the MITM list is tunneled byte-for-byte (start.go:195)."""
    s = mitm_stack
    s.cfg.mitm_hosts = ["something-else:443"]  # origin no longer matches
    blob = tmp_path / "t.bin"
    blob.write_bytes(os.urandom(50_000))
    s.origin.add_hf_repo("org/t", {"t.bin": str(blob)})
    s.origin.redirect_blobs = False

    # client must now verify the ORIGIN's cert (no MITM in the middle)
    url = f"https://127.0.0.1:{s.origin_port}/org/t/resolve/main/t.bin"
    st, _, body = _https_via_proxy(
        url, s.proxy_port, cafile=s.cfg.upstream_cafile)
    assert st == 200 and body == blob.read_bytes()
    # and nothing was cached (proxy saw only ciphertext)
    assert s.proxy.cache.lookup(
        f"https://127.0.0.1:{s.origin_port}/org/t/resolve/main/t.bin") is None


def test_mitm_all_and_no_mitm_flags(tmp_path):
    from demodel_amd.config import Config

    cfg = Config(mitm_all=True)
    assert cfg.should_mitm("anything:443")
    cfg = Config(no_mitm=True, mitm_all=True)
    assert not cfg.should_mitm("huggingface.co:443")
    cfg = Config()
    assert cfg.should_mitm("huggingface.co:443")  # bug-fixed default list
    assert not cfg.should_mitm("example.com:443")


def test_mitm_concurrent_soak(mitm_stack, tmp_path):
    """16 threads of MITM'd HTTPS pulls at once: leaf minting memoizes
    under concurrency, the TLS relay (asyncio fallback path — no raw
    socket) stays byte-exact, and upstream TLS conns pool."""
    import concurrent.futures as cf
    import random

    s = mitm_stack
    blobs = {}
    paths = {}
    for i in range(4):
        data = os.urandom((i + 1) * 200_000)
        p = tmp_path / f"m{i}.bin"
        p.write_bytes(data)
        blobs[f"m{i}.bin"] = data
        paths[f"m{i}.bin"] = str(p)
    s.origin.add_hf_repo("org/mitm", paths)
    rng = random.Random(3)

    def one(k):
        name = rng.choice(list(blobs))
        url = (f"https://127.0.0.1:{s.origin_port}"
               f"/org/mitm/resolve/main/{name}")
        status, headers, body = _https_via_proxy(
            url, s.proxy_port, s.client_cafile)
        assert status == 200
        assert body == blobs[name], name
        return True

    with cf.ThreadPoolExecutor(max_workers=16) as ex:
        assert all(ex.map(one, range(48)))
    # one leaf minted total, not one per connection
    assert len(s.proxy.leafs._pems) == 1
