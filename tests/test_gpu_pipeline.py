"""GPU landing-pipeline + pull e2e on a real MI355X: fake origin over
loopback -> pinned ring -> HBM -> GPU chunk verify -> zero-copy views."""

import hashlib
import json
import os

import numpy as np
import pytest

from helpers import Stack

pytestmark = pytest.mark.gpu


@pytest.fixture()
def stack(tmp_path):
    s = Stack(tmp_path)
    yield s
    s.close()


def _require_gpu():
    from demodel_amd.gpu import have_gpu

    assert have_gpu()


def test_gpu_land_and_verify(tmp_path):
    _require_gpu()
    from demodel_amd.engine.pipeline import Lander

    data = os.urandom(7 << 20)
    pos = [0]

    def fill(view):
        n = min(len(view), len(data) - pos[0])
        view[:n] = data[pos[0]:pos[0] + n]
        pos[0] += n
        return n

    lander = Lander(slab_bytes=1 << 20, n_slabs=4, verify_chunk=64 << 10)
    blob = lander.land(fill, len(data), gpu_chain=True)
    # exact whole-blob digest from the GPU chain kernel
    assert blob.sha256 == hashlib.sha256(data).hexdigest()
    # chunk digests from the batch kernel
    vc = 64 << 10
    assert blob.chunk_digests == [
        hashlib.sha256(data[o:o + vc]).hexdigest()
        for o in range(0, len(data), vc)]
    # round-trip: the landed bytes really are in HBM
    t = blob.torch_u8()
    assert bytes(t.cpu().numpy().tobytes()) == data


def test_gpu_pull_hf_views(stack, tmp_path):
    _require_gpu()
    import torch

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.formats import safetensors as st

    a = (np.random.rand(256, 128) * 100).astype(np.float32)
    spec = {"weight": ("F32", a.shape, a.nbytes)}
    head, _ = st.build_header(spec)
    p = tmp_path / "m.safetensors"
    p.write_bytes(head + a.tobytes())
    stack.origin.add_hf_repo("org/g", {"m.safetensors": str(p)})

    res = pull_mod.pull_hf("org/g", endpoint=stack.origin_base,
                           verify="digest", workers=1)
    assert res.files[0].digest_ok is True
    assert res.device.startswith("cuda")
    t = res.tensors()["weight"]
    assert t.is_cuda and t.dtype == torch.float32
    assert torch.equal(t.cpu(), torch.from_numpy(a))


def test_gpu_pull_ollama_dequant(stack, tmp_path):
    _require_gpu()
    import torch

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.formats import gguf

    gg_path = tmp_path / "m.gguf"
    gguf.build_file(str(gg_path), [
        ("blk.0.w.weight", (256, 8), 12),     # q4_K
        ("blk.0.n.weight", (64,), 0),         # f32
    ])
    stack.origin.add_ollama_model("library/g", "latest", [
        ("application/vnd.ollama.image.model", str(gg_path)),
    ])
    res = pull_mod.pull_ollama("g", "latest", endpoint=stack.origin_base,
                               verify="digest", workers=1)
    assert all(f.digest_ok for f in res.files)
    gg = res.meta["gguf_model"]
    t = gg.tensor("blk.0.w.weight")
    out = gguf.dequant_tensor_gpu(gg, t)
    assert out.dtype == torch.bfloat16 and out.shape == (8, 256)
    raw = gg_path.read_bytes()[gg.data_offset + t.offset:]
    want = gguf.dequant_cpu(12, raw[:t.nbytes], t.n_elems)
    wf = torch.from_numpy(want).to(torch.bfloat16).float().view(8, 256)
    gf = out.float().cpu()
    mask = torch.isfinite(wf)
    assert torch.allclose(gf[mask], wf[mask], rtol=1 / 64, atol=1e-3)


def test_gpu_peer_verified_pull(stack, tmp_path):
    """Peer-verified pull on the GPU lander: digests recorded by the
    proxy cache (1 MiB chunks) are verified by sha256_batch at landing."""
    _require_gpu()
    import json
    import time
    import urllib.request

    from demodel_amd.engine import pull as pull_mod

    data = os.urandom(5 << 20)
    p = tmp_path / "pv.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/pv", {"pv.bin": str(p)})
    url = f"{stack.endpoint}/org/pv/resolve/main/pv.bin"
    with urllib.request.urlopen(url, timeout=20) as r:
        assert r.read() == data
    # wait for the peer's async digest record
    for _ in range(100):
        try:
            with urllib.request.urlopen(
                    f"{stack.endpoint}/__demodel/digests"
                    f"/org/pv/resolve/main/pv.bin", timeout=10) as r:
                if json.loads(r.read()).get("chunk_sha256"):
                    break
        except urllib.error.HTTPError:
            pass
        time.sleep(0.05)
    res = pull_mod.pull_hf("org/pv", endpoint=stack.endpoint,
                           verify="chunked", workers=1, peer_verify=True)
    f = res.files[0]
    assert f.blob.device.startswith("cuda")
    assert f.blob.verify_chunk == 64 << 10
    assert bytes(f.blob.torch_u8().cpu().numpy().tobytes()) == data


def test_gpu_dequant_all(stack, tmp_path):
    """dequant_all_gpu: whole GGUF -> bf16 state dict in one arena."""
    _require_gpu()
    import numpy as np
    import torch

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.formats import gguf

    gg_path = tmp_path / "all.gguf"
    gguf.build_file(str(gg_path), [
        ("a.weight", (64, 4), 2),      # q4_0
        ("b.weight", (256, 2), 12),    # q4_K
        ("c.weight", (256, 2), 14),    # q6_K
        ("n.weight", (32,), 0),        # f32
    ])
    stack.origin.add_ollama_model("library/da", "latest", [
        ("application/vnd.ollama.image.model", str(gg_path)),
    ])
    res = pull_mod.pull_ollama("da", "latest", endpoint=stack.origin_base,
                               verify="chunked", workers=1)
    gg = res.meta["gguf_model"]
    sd = gguf.dequant_all_gpu(gg)
    assert set(sd) == {"a.weight", "b.weight", "c.weight", "n.weight"}
    raw = gg_path.read_bytes()
    for t in gg.tensors:
        got = sd[t.name].float().cpu().reshape(-1)
        want = torch.from_numpy(gguf.dequant_cpu(
            t.type_id, raw[gg.data_offset + t.offset:
                           gg.data_offset + t.offset + t.nbytes],
            t.n_elems))
        if t.type_id in (2, 3, 6, 7, 8, 10, 11, 12, 13, 14):
            want = want.to(torch.bfloat16).float()
        else:
            want = want.to(torch.bfloat16).float()
        mask = torch.isfinite(want)
        assert torch.allclose(got[mask], want[mask], rtol=1 / 64,
                              atol=1e-3), t.name


def test_gpu_progressive_dequant(stack, tmp_path, monkeypatch):
    """ProgressiveDequant: dequant launches fed by on_range events from
    a segmented pull match the CPU reference."""
    _require_gpu()
    import torch

    import demodel_amd.engine.pull as pm
    from demodel_amd.engine.formats import gguf

    monkeypatch.setattr(pm, "SEGMENT_MIN", 1 << 20)  # force segments
    gg_path = tmp_path / "prog.gguf"
    gguf.build_file(str(gg_path), [
        ("t0.weight", (256, 512), 12),   # q4_K ~1.1 MB
        ("t1.weight", (256, 512), 14),   # q6_K ~1.6 MB
        ("t2.weight", (64, 1024), 2),    # q4_0
        ("n.weight", (128,), 0),         # f32
    ])
    stack.origin.add_hf_repo("org/prog", {"m.gguf": str(gg_path)})
    pd = gguf.ProgressiveDequant()
    res = pm.pull_hf("org/prog", endpoint=stack.endpoint,
                     verify="chunked", workers=2,
                     patterns=["*.gguf"], on_range=pd.on_range)
    sd = pd.finish(res.files[0].blob)
    assert set(sd) == {"t0.weight", "t1.weight", "t2.weight", "n.weight"}
    raw = gg_path.read_bytes()
    gg = gguf.parse_bytes(raw[:1 << 20])
    for t in gg.tensors:
        got = sd[t.name].float().cpu().reshape(-1)
        want = torch.from_numpy(gguf.dequant_cpu(
            t.type_id, raw[gg.data_offset + t.offset:
                           gg.data_offset + t.offset + t.nbytes],
            t.n_elems)).to(torch.bfloat16).float()
        mask = torch.isfinite(want)
        assert torch.allclose(got[mask], want[mask], rtol=1 / 64,
                              atol=1e-3), t.name


def test_gpu_pull_ollama_dequant_tensors(stack, tmp_path):
    """pull_ollama(dequant_tensors=True): bf16 weights arrive with the
    pull (ProgressiveDequant under the hood)."""
    _require_gpu()
    import torch

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.formats import gguf

    gg_path = tmp_path / "dt.gguf"
    gguf.build_file(str(gg_path), [
        ("w.weight", (256, 8), 12),
        ("n.weight", (64,), 0),
    ])
    stack.origin.add_ollama_model("library/dt", "latest", [
        ("application/vnd.ollama.image.model", str(gg_path)),
    ])
    res = pull_mod.pull_ollama("dt", "latest",
                               endpoint=stack.origin_base,
                               verify="chunked", workers=1,
                               dequant_tensors=True)
    sd = res.meta["tensors"]
    assert set(sd) == {"w.weight", "n.weight"}
    raw = gg_path.read_bytes()
    gg = res.meta["gguf_model"]
    t = [x for x in gg.tensors if x.name == "w.weight"][0]
    got = sd["w.weight"].float().cpu().reshape(-1)
    want = torch.from_numpy(gguf.dequant_cpu(
        t.type_id, raw[gg.data_offset + t.offset:
                       gg.data_offset + t.offset + t.nbytes],
        t.n_elems)).to(torch.bfloat16).float()
    mask = torch.isfinite(want)
    assert torch.allclose(got[mask], want[mask], rtol=1 / 64, atol=1e-3)


def test_gpu_buffer_recycle(stack, tmp_path):
    """LanderPool.recycle: a re-pull of the same sizes reuses the HBM
    buffers instead of free+realloc (near-capacity reallocs cost
    seconds of driver page reclaim)."""
    _require_gpu()
    import os as _os

    from demodel_amd.engine import pull as pull_mod

    data = _os.urandom(2 << 20)
    p = tmp_path / "cyc.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/cyc", {"cyc.bin": str(p)})
    landers = pull_mod.LanderPool(0)
    res = pull_mod.pull_hf("org/cyc", endpoint=stack.origin_base,
                           verify="chunked", workers=1, landers=landers)
    f = res.files[0]
    ptr0 = f.blob.buffer.ptr
    body0 = bytes(f.blob.torch_u8().cpu().numpy().tobytes())
    assert body0 == data
    assert landers.recycle(res) == 1
    assert f.blob.buffer is None
    res2 = pull_mod.pull_hf("org/cyc", endpoint=stack.origin_base,
                            verify="chunked", workers=1, landers=landers)
    f2 = res2.files[0]
    assert f2.blob.buffer.ptr == ptr0  # same buffer came back
    assert bytes(f2.blob.torch_u8().cpu().numpy().tobytes()) == data


@pytest.mark.parametrize("inc", ["0", "1"])
def test_segmented_digests_match_host(stack, tmp_path, monkeypatch, inc):
    """Segmented-pull chunk digests equal a host-computed record for
    BOTH verification schedules (tail hash = default; incremental =
    DEMODEL_INC_VERIFY=1, kept as a documented experiment), including
    ragged, chunk-unaligned segment bounds."""
    import hashlib

    import demodel_amd.engine.pull as pm

    monkeypatch.setenv("DEMODEL_INC_VERIFY", inc)
    monkeypatch.setattr(pm, "SEGMENT_MIN", 3 << 20)
    monkeypatch.setattr(pm, "MAX_SEGMENTS", 7)  # unaligned bounds
    data = os.urandom((19 << 20) + 4321)
    p = tmp_path / "rag.bin"
    p.write_bytes(data)
    stack.origin.add_hf_repo("org/rag", {"rag.bin": str(p)})
    res = pm.pull_hf("org/rag", endpoint=stack.origin_base,
                     verify="chunked", workers=4)
    f = [x for x in res.files if x.name == "rag.bin"][0]
    vc = f.blob.verify_chunk
    want = b"".join(hashlib.sha256(data[o:o + vc]).digest()
                    for o in range(0, len(data), vc))
    assert f.blob.digest_blob == want
    assert bytes(f.blob.torch_u8().cpu().numpy().tobytes()) == data
