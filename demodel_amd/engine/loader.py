"""Model loading: scatter a landed safetensors blob into existing
tensors (SURVEY.md §2.3 K4's real job).

Zero-copy views (safetensors.torch_views) cover the "give me tensors"
case; this module covers loading into a model whose parameters already
live at allocator-chosen addresses — one scatter_ranges launch moves
every matching tensor's bytes HBM->HBM (3.2 TB/s class), with an
f32->bf16 cast path when the checkpoint dtype is wider than the model's.
"""

from __future__ import annotations

import ctypes
import struct

from ..gpu import hip
from .formats.safetensors import SafetensorsHeader

_PIECE = 1 << 20  # scatter descriptor granularity (see load_into)


def load_into(blob, header: SafetensorsHeader, targets: dict,
              strict: bool = True, stream=None) -> list[str]:
    """Scatter tensors from `blob` into `targets` (name -> torch tensor
    on the same device).  Returns the names loaded.

    Same-dtype tensors batch into one scatter_ranges launch; f32->bf16
    casts launch per tensor.  Raises on shape/dtype mismatches (strict)
    or missing tensors (strict).
    """
    import torch

    if blob.device == "cpu":
        # CPU landing target: plain torch copies from zero-copy views
        from .formats.safetensors import torch_views

        views = torch_views(header, blob.torch_u8())
        loaded = []
        for name, dst in targets.items():
            v = views.get(name)
            if v is None:
                if strict:
                    raise KeyError(f"tensor {name!r} not in checkpoint")
                continue
            if tuple(dst.shape) != tuple(v.shape):
                raise ValueError(f"{name}: shape mismatch")
            with torch.no_grad():
                dst.copy_(v.to(dst.dtype))
            loaded.append(name)
        if strict:
            missing = {t.name for t in header.tensors} - set(targets)
            if missing:
                raise KeyError(f"model is missing tensors: "
                               f"{sorted(missing)[:5]}")
        return loaded

    h = hip()
    own = stream is None
    s = h.Stream(0) if own else stream
    handle = s.handle

    desc = bytearray()
    casts = []
    loaded = []
    names = {t.name: t for t in header.tensors}
    for name, dst in targets.items():
        info = names.get(name)
        if info is None:
            if strict:
                raise KeyError(f"tensor {name!r} not in checkpoint")
            continue
        if tuple(dst.shape) != info.shape:
            raise ValueError(
                f"{name}: shape {tuple(dst.shape)} != {info.shape}")
        if not dst.is_contiguous():
            raise ValueError(f"{name}: target must be contiguous")
        src_off = header.data_offset + info.begin
        if dst.dtype == getattr(torch, info.torch_dtype):
            # one descriptor per <=1 MiB piece: the kernel maps one
            # workgroup per descriptor, so an unsplit 1 GB tensor would
            # crawl on a single WG (~4 GB/s; measured 256 ms for the
            # 16 GB model).  1 MiB pieces give 16k WGs across 256 CUs.
            off = 0
            while off < info.nbytes:
                n = min(_PIECE, info.nbytes - off)
                desc += struct.pack("<4Q", src_off + off,
                                    dst.data_ptr() + off, n, 0)
                off += n
        elif info.torch_dtype == "float32" and dst.dtype == torch.bfloat16:
            casts.append((src_off, dst.data_ptr(), info.nbytes // 4))
        else:
            raise ValueError(
                f"{name}: cannot load {info.torch_dtype} into {dst.dtype}")
        loaded.append(name)
    if strict:
        missing = set(names) - set(targets)
        if missing:
            raise KeyError(f"model is missing tensors: {sorted(missing)[:5]}"
                           f"{'...' if len(missing) > 5 else ''}")

    if desc:
        dbuf = h.DeviceBuffer(len(desc))
        carr = (ctypes.c_char * len(desc)).from_buffer(desc)
        h.h2d_async(dbuf.ptr, ctypes.addressof(carr), len(desc), handle)
        h.scatter_ranges(blob.buffer.ptr, dbuf.ptr, len(desc) // 32,
                         handle)
    for src_off, dst_ptr, n in casts:
        h.cast_f32_to_bf16(blob.buffer.ptr + src_off, dst_ptr, n, handle)
    # synchronous by contract: desc buffer and host staging must outlive
    # the launches
    s.sync()
    return loaded


def pull_pretrained(repo: str, endpoint: str | None = None,
                    device: str | None = None, model_builder=None,
                    **pull_kw):
    """Pull an HF repo and materialize a transformers model with weights
    scattered straight from the landed HBM blobs (no host round-trip on
    GPU machines).

    model_builder(config) -> nn.Module; defaults to
    transformers.AutoModelForCausalLM.from_config.
    Returns (model, PullResult)."""
    import json

    import torch

    from ..gpu import have_gpu
    from .pull import pull_hf

    res = pull_hf(repo, endpoint=endpoint, **pull_kw)
    cfg_file = next((f for f in res.files if f.name == "config.json"),
                    None)
    if cfg_file is None:
        raise FileNotFoundError("repo has no config.json")
    cfg_bytes = (bytes(cfg_file.blob.head[:cfg_file.nbytes])
                 if cfg_file.blob.device != "cpu"
                 else bytes(cfg_file.blob.buffer))
    cfg_dict = json.loads(cfg_bytes)

    import transformers

    config = transformers.AutoConfig.for_model(
        cfg_dict.get("model_type"), **{
            k: v for k, v in cfg_dict.items() if k != "model_type"})
    if device is None:
        device = "cuda" if have_gpu() else "cpu"
    if model_builder is None:
        def model_builder(c):
            return transformers.AutoModelForCausalLM.from_config(c)
    with torch.device(device):
        model = model_builder(config)
    model = model.to(device)
    model.eval()  # match from_pretrained semantics
    n = load_model_from_pull(res, model)
    if n == 0:
        raise RuntimeError("no tensors loaded from pull")
    return model, res


def load_model_from_pull(result, model, strict: bool = False) -> int:
    """Load a pull result's safetensors shards into a torch module's
    named parameters/buffers.  Returns tensors loaded."""
    from .formats import safetensors as st

    targets = dict(model.named_parameters())
    targets.update(dict(model.named_buffers()))
    n = 0
    for f in result.files:
        if not f.name.endswith(".safetensors"):
            continue
        hdr = st.parse_header(f.blob.head)
        present = {t.name for t in hdr.tensors}
        n += len(load_into(f.blob, hdr,
                           {k: v for k, v in targets.items()
                            if k in present},
                           strict=strict))
    return n
