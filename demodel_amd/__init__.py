"""demodel-amd: an MI355X-native model/dataset pull engine.

Caching, syncing, distributing middleware for models and datasets —
the capability contract of moeru-ai/demodel (reference: /root/reference,
see cmd/demodel/main.go:59) rebuilt from scratch MI355X-first:

  * TLS-MITM / HF_ENDPOINT / OLLAMA_HOST-compatible proxy front-end
    (asyncio + per-host leaf certs minted natively via libcrypto),
  * layout-compatible on-disk response cache (.cache/{key} + .meta,
    bodies kept in original Content-Encoding — reference
    CONTRIBUTING.md:53-153),
  * a GPU landing pipeline: chunked blob downloads stream through a
    pinned host ring into HBM3E via hipMemcpyAsync on side streams,
    with hand-written CDNA4 (gfx950) HIP kernels for SHA-256 verify,
    gzip/zstd/snappy/LZ4 decompression, safetensors tensor-scatter and
    GGUF dequantization (q4_0..q8_0 + q2_K..q6_K -> bf16),
  * proxy -> HBM pull-ahead: blobs the proxy caches can land into a
    device-resident registry so later engine pulls are GPU-warm
    (POST /__demodel/prefetch or DEMODEL_GPU_PREFETCH=auto),
  * RCCL-over-xGMI fan-out: broadcast of a pulled model to all GPUs of
    a node, and sharded pulls reassembled with all-gather overlapped
    with the next chunk's download.
"""

__version__ = "0.2.0"

from .config import Config, load_config  # noqa: F401


def __getattr__(name):
    # convenience re-exports without forcing heavy imports at package load
    if name in ("pull_hf", "pull_ollama", "pull_spec"):
        from .engine import pull as _p

        return getattr(_p, name)
    if name == "stream_dataset":
        from .engine.datasets import stream_dataset

        return stream_dataset
    if name in ("pull_pretrained", "load_into", "load_model_from_pull"):
        from .engine import loader as _l

        return getattr(_l, name)
    raise AttributeError(name)
