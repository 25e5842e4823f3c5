"""Proxy -> GPU pipeline unification (VERDICT round-1 item 3).

A client pull through the proxy leaves the node blob-warm: the proxy
lands the cached bytes through the engine's pipeline into the registry
(HBM on GPU boxes; host RAM here), and a subsequent engine pull of the
same repo is served registry-resident with ZERO upstream traffic and
zero landing work in the hot path.
"""

import json
import os
import time
import urllib.request

import pytest

from demodel_amd.engine import pull as pull_mod
from demodel_amd.engine.pull import LanderPool
from helpers import Stack


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path, cfg_kwargs={"gpu_prefetch": "auto"})
    # retrofit prefetch landers (host-RAM landers on this CPU box; the
    # same LanderPool lands into HBM on a GPU box)
    from demodel_amd.engine.registry import BlobRegistry

    s.proxy.prefetch_landers = LanderPool(0, gpu=False)
    s.proxy.registry = BlobRegistry()
    yield s
    s.close()


def _wait_registered(proxy, path, timeout=15.0):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if path in proxy.registry:
            return True
        time.sleep(0.02)
    return False


def _mk_repo(stack, tmp_path, repo, fname, nbytes=1 << 20):
    data = os.urandom(nbytes)
    p = tmp_path / fname
    p.write_bytes(data)
    stack.origin.add_hf_repo(repo, {fname: str(p)})
    return data


def test_auto_prefetch_then_registry_pull(stack, tmp_path):
    data = _mk_repo(stack, tmp_path, "org/pf", "model.safetensors")
    # 1. a client (vLLM-style) pulls api + blob THROUGH the proxy
    urllib.request.urlopen(
        f"{stack.endpoint}/api/models/org/pf/revision/main",
        timeout=20).read()
    got = urllib.request.urlopen(
        f"{stack.endpoint}/org/pf/resolve/main/model.safetensors",
        timeout=20).read()
    assert got == data
    # 2. the proxy lands the blob ahead (auto mode)
    path = "/org/pf/resolve/main/model.safetensors"
    assert _wait_registered(stack.proxy, path), stack.proxy.registry.keys()
    blob = stack.proxy.registry.get(path)
    assert bytes(blob.buffer) == data
    # 3. a subsequent engine pull is served from the registry: zero
    #    upstream traffic, zero fresh landing
    n_origin = len(stack.origin.requests)
    res = pull_mod.pull_hf("org/pf", endpoint=stack.endpoint,
                           registry=stack.proxy.registry,
                           landers=LanderPool(0, gpu=False))
    assert len(stack.origin.requests) == n_origin  # api was cached too
    f = res.files[0]
    assert f.blob is blob          # the very same landed object
    assert f.digest_ok is True
    assert bytes(f.blob.buffer) == data


def test_explicit_prefetch_endpoint(stack, tmp_path):
    data = _mk_repo(stack, tmp_path, "org/pf2", "weights.gguf")
    path = "/org/pf2/resolve/main/weights.gguf"
    # POST a path that is NOT yet cached: the proxy pulls it through its
    # own front door (filling the cache) and lands it
    req = urllib.request.Request(
        f"{stack.endpoint}/__demodel/prefetch",
        method="POST",
        data=json.dumps({"paths": [path]}).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=20) as r:
        assert r.status == 202
        assert json.loads(r.read())["queued"] == [path]
    assert _wait_registered(stack.proxy, path)
    blob = stack.proxy.registry.get(path)
    assert bytes(blob.buffer) == data
    # status document
    with urllib.request.urlopen(
            f"{stack.endpoint}/__demodel/prefetch", timeout=20) as r:
        st = json.loads(r.read())
    assert path in st["registered"]
    assert st["entries"] >= 1


@pytest.mark.gpu
def test_auto_prefetch_lands_in_hbm(stack, tmp_path):
    """On a GPU box the pull-ahead lands in HBM (cuda:0) and the engine
    pull reuses the device-resident blob."""
    stack.proxy.prefetch_landers = LanderPool(0, gpu=True)
    data = _mk_repo(stack, tmp_path, "org/gpupf", "model.safetensors",
                    nbytes=8 << 20)
    urllib.request.urlopen(
        f"{stack.endpoint}/api/models/org/gpupf/revision/main",
        timeout=20).read()
    urllib.request.urlopen(
        f"{stack.endpoint}/org/gpupf/resolve/main/model.safetensors",
        timeout=30).read()
    path = "/org/gpupf/resolve/main/model.safetensors"
    assert _wait_registered(stack.proxy, path, timeout=60)
    blob = stack.proxy.registry.get(path)
    assert blob.device.startswith("cuda")
    n_origin = len(stack.origin.requests)
    res = pull_mod.pull_hf("org/gpupf", endpoint=stack.endpoint,
                           registry=stack.proxy.registry)
    assert len(stack.origin.requests) == n_origin
    f = res.files[0]
    assert f.blob is blob
    got = bytes(f.blob.torch_u8().cpu().numpy().tobytes())
    assert got == data


def test_prefetch_decodes_gzip_bodies(stack, tmp_path):
    """Cached bodies keep their original Content-Encoding (the
    reference's worked example is a gzip Ollama manifest,
    CONTRIBUTING.md:116); the registry copy must be the DECODED
    bytes."""
    import json as _json

    blob = tmp_path / "layer.bin"
    blob.write_bytes(os.urandom(4096))
    manifest = stack.origin.add_ollama_model(
        "library/tiny", "latest",
        [("application/vnd.ollama.image.model", str(blob))])
    # client pulls the manifest through the proxy (cached gzip'd)
    got = urllib.request.urlopen(
        f"{stack.endpoint}/v2/library/tiny/manifests/latest",
        timeout=20).read()
    # urllib doesn't auto-decode; body is gzip per origin config
    import gzip as _gzip

    assert _json.loads(_gzip.decompress(got)) == manifest
    path = "/v2/library/tiny/manifests/latest"
    req = urllib.request.Request(
        f"{stack.endpoint}/__demodel/prefetch", method="POST",
        data=json.dumps({"paths": [path]}).encode())
    with urllib.request.urlopen(req, timeout=20) as r:
        assert json.loads(r.read())["queued"] == [path]
    assert _wait_registered(stack.proxy, path)
    reg = stack.proxy.registry.get(path)
    assert _json.loads(bytes(reg.buffer)) == manifest  # decoded!


def test_prefetch_cli(stack, tmp_path, capsys):
    from demodel_amd.cli import main as cli_main

    data = _mk_repo(stack, tmp_path, "org/cli", "w.safetensors",
                    nbytes=100_000)
    path = "/org/cli/resolve/main/w.safetensors"
    rc = cli_main(["prefetch", path, "--endpoint", stack.endpoint])
    assert rc == 0
    out = json.loads(capsys.readouterr().out)
    assert out["queued"] == [path]
    assert _wait_registered(stack.proxy, path)
    assert bytes(stack.proxy.registry.get(path).buffer) == data
    rc = cli_main(["prefetch", "--endpoint", stack.endpoint])
    assert rc == 0
    st = json.loads(capsys.readouterr().out)
    assert path in st["registered"]


def test_registry_lru_eviction():
    from demodel_amd.engine.pipeline import LandedBlob
    from demodel_amd.engine.registry import BlobRegistry

    reg = BlobRegistry(max_bytes=250)
    for i in range(4):
        reg.put(f"/b{i}", LandedBlob(nbytes=100, device="cpu",
                                     buffer=bytearray(100)))
    assert len(reg.keys()) == 2          # 400 bytes -> evicted to <=250
    assert "/b3" in reg and "/b2" in reg  # LRU kept the newest


def test_recycle_skips_shared_blobs():
    from demodel_amd.engine.pipeline import LandedBlob

    pool = LanderPool(0, gpu=False)

    class _F:
        pass

    f = _F()
    f.blob = LandedBlob(nbytes=10, device="cuda:0", buffer=object(),
                        shared=True)
    r = _F()
    r.files = [f]
    assert pool.recycle(r) == 0
    assert f.blob.buffer is not None


def test_prefetch_cli_wait(stack, tmp_path, capsys):
    from demodel_amd.cli import main as cli_main

    data = _mk_repo(stack, tmp_path, "org/cliw", "w2.safetensors",
                    nbytes=200_000)
    path = "/org/cliw/resolve/main/w2.safetensors"
    rc = cli_main(["prefetch", path, "--wait", "--timeout", "30",
                   "--endpoint", stack.endpoint])
    assert rc == 0
    out = json.loads(capsys.readouterr().out)
    assert out["registered"] == [path]
    assert bytes(stack.proxy.registry.get(path).buffer) == data
    # a bogus path fails within the timeout instead of hanging
    rc = cli_main(["prefetch", "/no/such/thing.safetensors", "--wait",
                   "--timeout", "20", "--endpoint", stack.endpoint])
    assert rc == 1
