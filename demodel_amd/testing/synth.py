"""Synthetic model blob generation for tests and benchmarks.

There is no network in the build/bench environments, so BASELINE.json's
configs run on synthetic blobs: random-init Llama-3-shaped safetensors
shards (configs 2-3), GGUF quant files (config 4), and zstd parquet-like
streams (config 5).  Files are written once and served by the in-process
fake origin (testing/origin.py) over loopback via sendfile.
"""

from __future__ import annotations

import json
import os

import numpy as np

from ..engine.formats import safetensors as st

# Llama-3-8B geometry (public config): the bench's model shape.
LLAMA3_8B = {
    "hidden": 4096, "inter": 14336, "layers": 32,
    "heads": 32, "kv_heads": 8, "vocab": 128256,
}
# Llama-3-70B geometry for the sharded 8-rank pull (config 3).
LLAMA3_70B = {
    "hidden": 8192, "inter": 28672, "layers": 80,
    "heads": 64, "kv_heads": 8, "vocab": 128256,
}


def llama_tensor_table(geom: dict) -> list[tuple[str, tuple[int, ...]]]:
    h, inter, L = geom["hidden"], geom["inter"], geom["layers"]
    kvh = geom["kv_heads"] * (h // geom["heads"])
    out = [("model.embed_tokens.weight", (geom["vocab"], h))]
    for i in range(L):
        p = f"model.layers.{i}."
        out += [
            (p + "self_attn.q_proj.weight", (h, h)),
            (p + "self_attn.k_proj.weight", (kvh, h)),
            (p + "self_attn.v_proj.weight", (kvh, h)),
            (p + "self_attn.o_proj.weight", (h, h)),
            (p + "mlp.gate_proj.weight", (inter, h)),
            (p + "mlp.up_proj.weight", (inter, h)),
            (p + "mlp.down_proj.weight", (h, inter)),
            (p + "input_layernorm.weight", (h,)),
            (p + "post_attention_layernorm.weight", (h,)),
        ]
    out += [("model.norm.weight", (h,)),
            ("lm_head.weight", (geom["vocab"], h))]
    return out


def plan_shards(geom: dict, n_shards: int) -> list[dict]:
    """Byte-balanced assignment of llama tensors to shard specs."""
    table = llama_tensor_table(geom)
    shard_specs: list[dict] = [dict() for _ in range(n_shards)]
    shard_bytes = [0] * n_shards
    for name, shape in table:
        nbytes = int(np.prod(shape)) * 2
        i = shard_bytes.index(min(shard_bytes))
        shard_specs[i][name] = ("BF16", shape, nbytes)
        shard_bytes[i] += nbytes
    return shard_specs


def shard_sizes(geom: dict, n_shards: int) -> dict[str, int]:
    """File sizes write_shards would produce, without writing anything
    (virtual-origin benchmarks bigger than the disk)."""
    out = {}
    for i, spec in enumerate(plan_shards(geom, n_shards)):
        head, data_bytes = st.build_header(spec)
        fname = f"model-{i + 1:05d}-of-{n_shards:05d}.safetensors"
        out[fname] = len(head) + data_bytes
    return out


def write_shards(out_dir: str, geom: dict, n_shards: int,
                 seed: int = 0, reuse: bool = True) -> dict[str, str]:
    """Write llama-shaped bf16 safetensors shards; returns
    {rfilename: path}.  Payload is a repeated random block (content is
    irrelevant to the pipeline; generation must not dominate setup)."""
    os.makedirs(out_dir, exist_ok=True)
    shard_specs = plan_shards(geom, n_shards)

    rng = np.random.default_rng(seed)
    rand_block = rng.integers(0, 256, size=64 << 20, dtype=np.uint8)
    rand_block = rand_block.tobytes()

    files = {}
    total = n_shards
    for i, spec in enumerate(shard_specs):
        fname = f"model-{i + 1:05d}-of-{total:05d}.safetensors"
        path = os.path.join(out_dir, fname)
        head, data_bytes = st.build_header(spec)
        want_size = len(head) + data_bytes
        if reuse and os.path.exists(path) and \
                os.path.getsize(path) == want_size:
            files[fname] = path
            continue
        with open(path, "wb") as f:
            f.write(head)
            left = data_bytes
            while left > 0:
                take = min(left, len(rand_block))
                f.write(rand_block[:take])
                left -= take
        files[fname] = path
    cfg_path = os.path.join(out_dir, "config.json")
    with open(cfg_path, "w") as f:
        json.dump({"architectures": ["LlamaForCausalLM"],
                   "hidden_size": geom["hidden"],
                   "num_hidden_layers": geom["layers"],
                   "vocab_size": geom["vocab"],
                   "torch_dtype": "bfloat16"}, f)
    files["config.json"] = cfg_path
    return files


def write_dataset_shards(out_dir: str, n_shards: int = 8,
                         frames_per_shard: int = 256,
                         frame_bytes: int = 2 << 20,
                         level: int = 1, reuse: bool = True
                         ) -> dict[str, str]:
    """Synthetic c4-en-like text dataset: shards of concatenated zstd
    frames (one frame ~= one record batch), plus a .idx.json sidecar per
    shard with frame offsets/sizes (plays the role parquet metadata plays
    for page locations).  Returns {rfilename: path} incl. sidecars."""
    import pyarrow as pa

    os.makedirs(out_dir, exist_ok=True)
    rng = np.random.default_rng(42)
    words = [f"w{i:04d}" for i in range(30000)]
    codec = pa.Codec("zstd", compression_level=level)

    # one representative text block, tiled with per-frame mutation
    idx = rng.integers(0, len(words), size=frame_bytes // 6)
    base_text = " ".join(words[i] for i in idx).encode()[:frame_bytes]

    files: dict[str, str] = {}
    for s in range(n_shards):
        fname = f"c4-train.{s:05d}-of-{n_shards:05d}.zst"
        path = os.path.join(out_dir, fname)
        idx_path = path + ".idx.json"
        if reuse and os.path.exists(path) and os.path.exists(idx_path):
            files[fname] = path
            files[fname + ".idx.json"] = idx_path
            continue
        frames = []
        off = 0
        with open(path, "wb") as f:
            for k in range(frames_per_shard):
                stamp = f"[shard {s} frame {k}] ".encode()
                raw = stamp + base_text[len(stamp):]
                comp = bytes(codec.compress(raw))
                f.write(comp)
                frames.append({"offset": off, "compressed": len(comp),
                               "decompressed": len(raw)})
                off += len(comp)
        with open(idx_path, "w") as f:
            json.dump({"frames": frames}, f)
        files[fname] = path
        files[fname + ".idx.json"] = idx_path
    return files


def write_parquet_shards(out_dir: str, n_shards: int = 8,
                         rows_per_shard: int = 300_000,
                         reuse: bool = True,
                         compression: str = "zstd") -> dict[str, str]:
    """c4-en-like parquet shards (text/url/timestamp columns; ZSTD or
    SNAPPY pages — snappy is parquet's default codec) — BASELINE.json
    config 5's dataset."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    os.makedirs(out_dir, exist_ok=True)
    rng = np.random.default_rng(11)
    vocab = np.array([f"w{i:04d}" for i in range(20000)])

    base_n = 30_000
    docs = [" ".join(vocab[rng.integers(0, len(vocab), size=120)])
            for _ in range(base_n)]
    reps = (rows_per_shard + base_n - 1) // base_n

    files = {}
    for s in range(n_shards):
        fname = f"c4-train.{s:05d}-of-{n_shards:05d}.parquet"
        path = os.path.join(out_dir, fname)
        if reuse and os.path.exists(path) and os.path.getsize(path) > 0:
            files[fname] = path
            continue
        n = rows_per_shard
        text = (docs * reps)[:n]
        table = pa.table({
            "text": text,
            "url": [f"https://example.com/{s}/{i}" for i in range(n)],
            "timestamp": np.arange(n, dtype=np.int64) + s,
        })
        pq.write_table(table, path, compression=compression,
                       data_page_version="1.0")
        files[fname] = path
    return files


def gguf_tensor_specs(geom: dict, qtype: int = 12):
    """(name, ggml_dims, type_id) specs for a llama-shaped GGUF."""
    tensors = []
    for name, shape in llama_tensor_table(geom):
        if len(shape) == 1:
            tensors.append((name, shape, 0))       # f32 norms
        else:
            d0, d1 = shape[1], shape[0]
            tensors.append((name, (d0, d1), qtype))
    return tensors


def gguf_virtual(geom: dict, qtype: int = 12) -> tuple[bytes, int]:
    """(header_prefix, total_size) for a virtual-origin GGUF — the 70B
    nameplate config (41 GB q4_K) without disk backing."""
    from ..engine.formats import gguf

    return gguf.build_virtual(gguf_tensor_specs(geom, qtype))


def write_gguf_model(path: str, geom: dict, qtype: int = 12,
                     reuse: bool = True):
    """Synthetic GGUF (default q4_K) with llama-shaped 2-D tensors."""
    from ..engine.formats import gguf

    if reuse and os.path.exists(path) and os.path.getsize(path) > 0:
        return gguf.parse_bytes(open(path, "rb").read(8 << 20))
    # ggml dims are reversed (fastest first) and dim0 must divide the
    # quant block size — gguf_tensor_specs handles both
    return gguf.build_file(path, gguf_tensor_specs(geom, qtype))
