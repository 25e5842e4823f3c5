"""GPU model-load scatter (K4): landed blob -> existing param tensors."""

import numpy as np
import pytest

from helpers import Stack

pytestmark = pytest.mark.gpu


def test_load_into_params_and_cast(tmp_path):
    import torch

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.formats import safetensors as st
    from demodel_amd.engine.loader import load_into
    from demodel_amd.gpu import have_gpu

    assert have_gpu()
    stack = Stack(tmp_path)
    try:
        a = np.random.rand(128, 64).astype(np.float32)
        b = (np.random.rand(1024) * 60000).astype(np.uint16)  # bf16 bits
        spec = {"w.f32": ("F32", a.shape, a.nbytes),
                "w.bf16": ("BF16", b.shape, b.nbytes)}
        head, _ = st.build_header(spec)
        p = tmp_path / "m.safetensors"
        p.write_bytes(head + a.tobytes() + b.tobytes())
        stack.origin.add_hf_repo("o/l", {"m.safetensors": str(p)})

        res = pull_mod.pull_hf("o/l", endpoint=stack.origin_base,
                               workers=1)
        blob = res.files[0].blob
        hdr = st.parse_header(blob.head)

        # same-dtype scatter + f32->bf16 cast into "model params"
        dst_bf16 = torch.empty(1024, dtype=torch.bfloat16, device="cuda")
        dst_cast = torch.empty(128, 64, dtype=torch.bfloat16,
                               device="cuda")
        loaded = load_into(blob, hdr,
                           {"w.bf16": dst_bf16, "w.f32": dst_cast})
        assert sorted(loaded) == ["w.bf16", "w.f32"]
        want_b = torch.from_numpy(b.copy()).view(torch.bfloat16)
        # compare raw bits: random uint16 payloads include bf16 NaNs
        assert torch.equal(dst_bf16.cpu().view(torch.int16),
                           want_b.view(torch.int16))
        want_a = torch.from_numpy(a).to(torch.bfloat16)
        assert torch.equal(dst_cast.cpu(), want_a)

        # strict mode catches mismatches
        with pytest.raises(ValueError):
            load_into(blob, hdr,
                      {"w.bf16": torch.empty(3, device="cuda",
                                             dtype=torch.bfloat16)})
        with pytest.raises(KeyError):
            load_into(blob, hdr, {"nope": dst_bf16})
    finally:
        stack.close()


def test_load_model_from_pull(tmp_path):
    import torch

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.loader import load_model_from_pull
    from demodel_amd.engine.formats import safetensors as st

    stack = Stack(tmp_path)
    try:
        lin_w = np.random.rand(32, 16).astype(np.float32)
        lin_b = np.random.rand(32).astype(np.float32)
        spec = {"weight": ("F32", lin_w.shape, lin_w.nbytes),
                "bias": ("F32", lin_b.shape, lin_b.nbytes)}
        head, _ = st.build_header(spec)
        p = tmp_path / "lin.safetensors"
        p.write_bytes(head + lin_w.tobytes() + lin_b.tobytes())
        stack.origin.add_hf_repo("o/lin", {"lin.safetensors": str(p)})

        res = pull_mod.pull_hf("o/lin", endpoint=stack.origin_base,
                               workers=1)
        model = torch.nn.Linear(16, 32).cuda()
        n = load_model_from_pull(res, model)
        assert n == 2
        assert torch.allclose(model.weight.detach().cpu(),
                              torch.from_numpy(lin_w))
        assert torch.allclose(model.bias.detach().cpu(),
                              torch.from_numpy(lin_b))
    finally:
        stack.close()
