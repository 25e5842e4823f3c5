"""Verified distribution between demodel nodes: a peer's recorded chunk
digests guard pulls from that peer — including detecting a tampered
cache."""

import os
import time
import urllib.request

import pytest

from demodel_amd.engine import pull as pull_mod
from demodel_amd.engine.pipeline import DigestMismatch
from helpers import Stack


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path)
    yield s
    s.close()


def _prime(stack, tmp_path, name="w.bin", size=3 << 20):
    data = os.urandom(size)
    p = tmp_path / name
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/p", {name: str(p)})
    url = f"{stack.endpoint}/org/p/resolve/main/{name}"
    with urllib.request.urlopen(url, timeout=20) as r:
        assert r.read() == data
    return data, url


def _wait_digests(stack, path, timeout=10.0):
    import json

    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            with urllib.request.urlopen(
                    f"{stack.endpoint}/__demodel/digests{path}",
                    timeout=10) as r:
                obj = json.loads(r.read())
                if obj.get("chunk_sha256"):
                    return obj
        except urllib.error.HTTPError:
            pass
        time.sleep(0.1)
    raise AssertionError("peer digests never appeared")


def test_peer_digests_endpoint_and_verified_pull(stack, tmp_path):
    data, url = _prime(stack, tmp_path)
    obj = _wait_digests(stack, "/org/p/resolve/main/w.bin")
    assert obj["body_size"] == len(data)
    assert obj["chunk_bytes"] == 64 << 10
    assert len(obj["chunk_sha256"]) == 48  # 3 MiB / 64 KiB
    import hashlib

    assert obj["sha256"] == hashlib.sha256(data).hexdigest()
    assert obj["chunk_sha256"][0] == \
        hashlib.sha256(data[:64 << 10]).hexdigest()

    # peer-verified pull: engine compares every chunk to the record
    res = pull_mod.pull_hf("org/p", endpoint=stack.endpoint,
                           verify="chunked", workers=1, peer_verify=True)
    f = res.files[0]
    assert f.blob.verify_chunk == 64 << 10
    assert bytes(f.blob.buffer) == data


def test_peer_verify_detects_tamper(stack, tmp_path):
    data, url = _prime(stack, tmp_path, name="t.bin")
    _wait_digests(stack, "/org/p/resolve/main/t.bin")
    # corrupt one byte in the peer's cached body, behind the digests
    cached = [r for r in stack.proxy.transfers.records
              if r["event"] == "miss" and "/cdn/" in r["uri"]]
    entry = stack.proxy.cache.lookup(cached[0]["uri"])
    raw = bytearray(entry.read_body())
    raw[2_000_000] ^= 0xFF
    with open(entry.body_path, "wb") as fh:
        fh.write(raw)

    with pytest.raises(DigestMismatch) as ei:
        pull_mod.pull_hf("org/p", endpoint=stack.endpoint,
                         verify="chunked", workers=1, peer_verify=True)
    assert ei.value.chunk_index == 2_000_000 // (64 << 10)


def test_pull_cli_peer_verify(stack, tmp_path, capsys):
    """`demodel pull --peer-verify` verifies against the peer's digest
    record end-to-end through the CLI."""
    import json as _json
    import time

    from demodel_amd.cli import main as cli_main
    from demodel_amd.engine.pull import fetch_peer_digests

    data = os.urandom(300_000)
    p = tmp_path / "pv.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/pvcli", {"pv.bin": str(p)})
    # prime the peer's cache + wait for its async digest record
    import urllib.request

    urllib.request.urlopen(
        f"{stack.endpoint}/org/pvcli/resolve/main/pv.bin",
        timeout=20).read()
    path = "/org/pvcli/resolve/main/pv.bin"
    t0 = time.time()
    while fetch_peer_digests(stack.endpoint, path) is None:
        assert time.time() - t0 < 30
        time.sleep(0.05)
    rc = cli_main(["pull", "hf://org/pvcli", "--cpu", "--peer-verify",
                   "--endpoint", stack.endpoint])
    assert rc == 0
    out = _json.loads(capsys.readouterr().out)
    f = [x for x in out["files"] if x["name"] == "pv.bin"][0]
    assert f["bytes"] == len(data)
