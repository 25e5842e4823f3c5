"""Config 1 end-to-end (BASELINE.json): a REAL transformers client loads
tiny-random-gpt2 through the demodel proxy via HF_ENDPOINT, twice —
second load offline from the cache."""

import pytest

from helpers import Stack

transformers = pytest.importorskip("transformers")


@pytest.fixture(scope="module")
def tiny_gpt2_dir(tmp_path_factory):
    import torch

    d = tmp_path_factory.mktemp("tiny-gpt2")
    cfg = transformers.GPT2Config(
        n_embd=32, n_layer=2, n_head=2, vocab_size=512,
        n_positions=64)
    torch.manual_seed(0)
    model = transformers.GPT2LMHeadModel(cfg)
    model.save_pretrained(str(d), safe_serialization=True)
    return d


def test_transformers_from_pretrained_via_proxy(tiny_gpt2_dir, tmp_path,
                                                monkeypatch):
    import torch

    files = {p.name: str(p) for p in tiny_gpt2_dir.iterdir()}
    assert "model.safetensors" in files and "config.json" in files

    stack = Stack(tmp_path)
    try:
        stack.origin.add_hf_repo("tiny-random/gpt2", files)

        monkeypatch.delenv("HF_HUB_OFFLINE", raising=False)
        import huggingface_hub.constants as hf_const

        monkeypatch.setattr(hf_const, "HF_HUB_OFFLINE", False)
        # hf_hub bakes the endpoint into TWO import-time constants
        monkeypatch.setattr(hf_const, "ENDPOINT", stack.endpoint)
        monkeypatch.setattr(
            hf_const, "HUGGINGFACE_CO_URL_TEMPLATE",
            stack.endpoint + "/{repo_id}/resolve/{revision}/{filename}")
        monkeypatch.setenv("HF_ENDPOINT", stack.endpoint)
        monkeypatch.setenv("HF_HUB_DISABLE_XET", "1")

        m1 = transformers.GPT2LMHeadModel.from_pretrained(
            "tiny-random/gpt2", cache_dir=str(tmp_path / "hf1"))
        x = torch.randint(0, 512, (1, 8))
        with torch.no_grad():
            out1 = m1(x).logits

        # offline replay: origin gone, proxy cache serves everything
        stack.stop_origin()
        m2 = transformers.GPT2LMHeadModel.from_pretrained(
            "tiny-random/gpt2", cache_dir=str(tmp_path / "hf2"))
        with torch.no_grad():
            out2 = m2(x).logits
        assert torch.equal(out1, out2)
    finally:
        stack.close()


def test_pull_pretrained_cpu(tiny_gpt2_dir, tmp_path):
    """pull_pretrained: pull + build + weight-load in one call (CPU
    landing; the GPU path is the same scatter covered by gpu tests)."""
    import torch

    from demodel_amd.engine.loader import pull_pretrained

    files = {p.name: str(p) for p in tiny_gpt2_dir.iterdir()}
    stack = Stack(tmp_path)
    try:
        stack.origin.add_hf_repo("tiny-random/gpt2", files)
        model, res = pull_pretrained("tiny-random/gpt2",
                                     endpoint=stack.origin_base,
                                     workers=2)
        assert res.total_bytes > 0
        ref = transformers.GPT2LMHeadModel.from_pretrained(
            str(tiny_gpt2_dir))
        x = torch.randint(0, 512, (1, 8))
        with torch.no_grad():
            assert torch.allclose(model(x).logits, ref(x).logits,
                                  atol=1e-5)
    finally:
        stack.close()
