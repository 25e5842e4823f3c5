#!/usr/bin/env python3
"""demodel-amd flagship benchmark: Llama-3-8B synthetic safetensors pull
-> HBM with GPU chunk-verified landing (BASELINE.json configs 2/3).

Each *step* pulls the whole model (all shards, ~16 GB bf16) from a
per-rank loopback origin through the engine's landing pipeline into this
rank's HBM, with GPU SHA-256 chunk verification and zero-copy tensor
views materialized.  value = whole-job GB/s into HBM (sum over ranks).

Run directly (1 GPU) or under torch.distributed.run with one rank per
GPU; weak scaling (per-rank work fixed).
"""

import argparse
import json
import os
import pathlib
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "tests"))

GEOMS = {
    "llama3-8b": ("LLAMA3_8B", 4),
    "llama3-70b": ("LLAMA3_70B", 16),
    "tiny": (None, 2),
    "gguf-8b": ("LLAMA3_8B", 0),     # Ollama q4_K pull + GPU dequant
    "gguf-70b": ("LLAMA3_70B", 0),   # nameplate llama3:70b (use --virtual)
    "gguf-tiny": (None, 0),
    "dataset": (None, 0),            # zstd frame shards into HBM ring
    "parquet": (None, 0),            # real c4-like parquet, ZSTD pages
}


def log(msg):
    print(f"[bench r{os.environ.get('RANK', '0')}] {msg}", file=sys.stderr,
          flush=True)


class _FanoutResult:
    """Minimal result shim for fan-out steps (log fields only)."""

    def __init__(self, nbytes, seconds=0.0):
        self.total_bytes = nbytes
        self.seconds_to_ready = seconds
        self.gbps = nbytes / max(seconds, 1e-9) / 1e9 if seconds else 0.0


TINY_GEOM = {"hidden": 256, "inter": 688, "layers": 4, "heads": 8,
             "kv_heads": 4, "vocab": 32000}


def make_model_files(model: str, data_dir: str, n_shards_override=None,
                     parquet_codec="zstd"):
    from demodel_amd.testing import synth

    if model == "tiny":
        return synth.write_shards(data_dir, TINY_GEOM, 2)
    if model == "dataset":
        return synth.write_dataset_shards(data_dir)
    if model == "parquet":
        return synth.write_parquet_shards(data_dir,
                                          compression=parquet_codec)
    if model.startswith("gguf"):
        assert model != "gguf-70b", "gguf-70b needs --virtual (41 GB)"
        geom = synth.LLAMA3_8B if model == "gguf-8b" else TINY_GEOM
        path = os.path.join(data_dir, "model.gguf")
        os.makedirs(data_dir, exist_ok=True)
        synth.write_gguf_model(path, geom, qtype=12)
        return {"model.gguf": path}
    geom_name, n_shards = GEOMS[model]
    geom = getattr(synth, geom_name)
    return synth.write_shards(data_dir, geom,
                              n_shards_override or n_shards)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--model", default="llama3-8b",
                    choices=list(GEOMS))
    ap.add_argument("--mode", default="dp",
                    choices=["dp", "shard", "broadcast", "allgather"],
                    help="dp: independent per-rank pulls (weak scaling); "
                         "shard: rank r pulls 1/N of the manifest files, "
                         "file-wise RCCL broadcast reassembly (config 3); "
                         "broadcast: rank 0 pulls, RCCL fan-out (R1); "
                         "allgather: byte-range shards + bucketed "
                         "all-gather overlapped with download (R2)")
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--verify", default="chunked",
                    choices=["chunked", "digest", "gpu-digest", "off"])
    ap.add_argument("--slab-mib", type=int, default=32)
    ap.add_argument("--n-slabs", type=int, default=4)
    ap.add_argument("--data-dir", default=None)
    ap.add_argument("--shards", type=int, default=None,
                    help="override safetensors shard count")
    ap.add_argument("--parquet-codec", default="zstd",
                    choices=["zstd", "snappy", "gzip", "lz4"],
                    help="page codec for --model parquet")
    ap.add_argument("--via", default="direct",
                    choices=["direct", "proxy", "proxy-miss", "peer"],
                    help="direct: engine pulls from the origin (the "
                         "headline path); proxy: pulls go THROUGH the "
                         "demodel proxy with a primed cache (measures "
                         "the cache-HIT data plane a real client sees); "
                         "proxy-miss: caching disabled, measures the "
                         "origin->proxy->client relay; peer: verified "
                         "peer distribution — pull from a primed peer "
                         "node, every chunk GPU-verified against the "
                         "peer's digest record (/__demodel/digests)")
    ap.add_argument("--virtual", action="store_true",
                    help="serve blobs from memory (no disk) — for models "
                         "bigger than the box's disk, e.g. llama3-70b; "
                         "payloads are patterned bytes, so tensor-view "
                         "materialization is skipped")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    import torch

    have_gpu = torch.cuda.is_available()
    if have_gpu:
        torch.cuda.set_device(local_rank)

    dist = None
    if world > 1 or args.mode != "dp":
        # collective modes init the process group even at world 1 so the
        # RCCL backend path (init, dtype, stream interaction with the
        # landing pipeline) is exercised on a single-GPU box
        import datetime

        import torch.distributed as dist_mod

        dist = dist_mod
        # first-try-robust rendezvous on an unfamiliar node: loopback
        # defaults (container hostnames may not resolve), explicit
        # device binding for RCCL, and a bounded timeout so a wedged
        # rank fails the job instead of hanging it
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29513")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", str(world))
        kw = {}
        if have_gpu:
            kw["device_id"] = torch.device("cuda", local_rank)
        dist.init_process_group(
            backend="nccl" if have_gpu else "gloo",
            timeout=datetime.timedelta(seconds=600), **kw)

    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.pull import LanderPool
    from demodel_amd.testing.origin import FakeOrigin
    from helpers import LoopThread

    # ---- setup (untimed): synth model + per-rank loopback origin -------
    data_dir = args.data_dir or os.path.join(
        os.environ.get("TMPDIR", "/tmp"),
        f"demodel_bench_{args.model}"
        + (f"_s{args.shards}" if args.shards else "")
        + (f"_{args.parquet_codec}" if args.model == "parquet"
           and args.parquet_codec != "zstd" else ""))
    t = time.time()
    lt = LoopThread()
    if args.virtual:
        from demodel_amd.testing import synth

        geom_name, n_shards = GEOMS[args.model]
        geom = getattr(synth, geom_name) if geom_name else TINY_GEOM
        n_shards = n_shards or 2
        prefixes = None
        if args.model.startswith("gguf"):
            # REAL GGUF header prefix + patterned quant payload: the
            # 41 GB llama3:70b q4_K config without disk backing
            prefix, total = synth.gguf_virtual(geom)
            sizes = {"model.gguf": total}
            prefixes = {"model.gguf": prefix}
        else:
            sizes = synth.shard_sizes(geom, args.shards or n_shards)
        files = {n: None for n in sizes}
        os.makedirs(data_dir, exist_ok=True)
        origin = FakeOrigin(data_dir, redirect_blobs=True)
        origin.add_hf_repo_virtual("bench/model", sizes,
                                   prefixes=prefixes)
        total_bytes = sum(sizes.values())
    else:
        # rank 0 generates the shared synthetic files; others wait
        if rank == 0:
            files = make_model_files(args.model, data_dir, args.shards,
                                     args.parquet_codec)
        if dist:
            dist.barrier()
        if rank != 0:
            files = make_model_files(args.model, data_dir, args.shards,
                                     args.parquet_codec)
        origin = FakeOrigin(data_dir, redirect_blobs=True)
        origin.add_hf_repo("bench/model", files)
        total_bytes = sum(os.path.getsize(p) for p in files.values())
    log(f"model files ready in {time.time() - t:.1f}s "
        f"({total_bytes / 1e9:.2f} GB{' virtual' if args.virtual else ''})")

    port = lt.call(origin.start())
    endpoint = f"http://127.0.0.1:{port}"

    proxy = None
    if args.via != "direct":
        assert not args.virtual, (
            "--via proxy/peer with --virtual would spool the virtual "
            "blob set to the proxy's DISK cache (e.g. 141 GB)")
        # the client-facing data plane (reference hot loop,
        # start.go:201-204): engine pulls go THROUGH the proxy
        from demodel_amd.config import Config
        from demodel_amd.proxy.server import ProxyServer

        pcfg = Config(host="127.0.0.1", port=0,
                      cache_dir=os.path.join(data_dir,
                                             f"proxycache_r{rank}"))
        proxy = ProxyServer(pcfg)
        proxy.reverse_routes = [("/", endpoint)]
        if args.via == "proxy-miss":
            proxy.cache.cacheable = lambda *a, **k: False
        pport = lt.call(proxy.start())
        endpoint = f"http://127.0.0.1:{pport}"
        if args.via in ("proxy", "peer"):
            # prime the cache (untimed) so timed pulls measure the HIT
            # path: cache file -> sendfile -> engine -> HBM
            import concurrent.futures as cf

            from demodel_amd.engine import fetch

            def prime(name):
                src = fetch.http_get(
                    f"{endpoint}/bench/model/resolve/main/{name}")
                try:
                    assert src.status == 200, (name, src.status)
                    sink = memoryview(bytearray(8 << 20))
                    while src.fill(sink) > 0:
                        pass
                finally:
                    src.close()

            t = time.time()
            with cf.ThreadPoolExecutor(max_workers=8) as ex:
                list(ex.map(prime, files))
            log(f"proxy cache primed in {time.time() - t:.1f}s")
            if args.via == "peer":
                # wait for the peer's async chunk-digest records: the
                # verified-distribution pull compares every chunk
                # against them
                from demodel_amd.engine.pull import fetch_peer_digests

                t = time.time()
                for n in files:
                    while fetch_peer_digests(
                            endpoint,
                            f"/bench/model/resolve/main/{n}") is None:
                        assert time.time() - t < 120, \
                            f"peer digests for {n} never appeared"
                        time.sleep(0.1)
                log(f"peer digest records ready in {time.time() - t:.1f}s")

    landers = LanderPool(local_rank if have_gpu else 0,
                         slab_bytes=args.slab_mib << 20,
                         n_slabs=args.n_slabs)

    def sync():
        if have_gpu:
            torch.cuda.synchronize()
        if dist:
            dist.barrier()

    digest_map: dict = {}
    # pre-allocated scatter targets (name -> tensor), built during the
    # first (untimed) step: the timed step then exercises the K4
    # scatter_ranges kernel exactly as BASELINE config 2 words it
    # ("GPU SHA256 + tensor scatter"), not just zero-copy views
    scatter_targets: dict = {}

    def _build_targets(res):
        from demodel_amd.engine.formats import safetensors as st

        dev = "cuda" if have_gpu else "cpu"
        for f in res.files:
            if not f.name.endswith(".safetensors"):
                continue
            hdr = st.parse_header(f.blob.head)
            tg = {}
            for t in hdr.tensors:
                tg[t.name] = torch.empty(
                    t.shape, dtype=getattr(torch, t.torch_dtype),
                    device=dev)
            scatter_targets[f.name] = (tg)

    def dp_step(record_digests=False):
        from demodel_amd.engine.formats import safetensors as st
        from demodel_amd.engine.loader import load_into

        res = pull_mod.pull_hf(
            "bench/model", endpoint=endpoint, workers=args.workers,
            verify=args.verify, landers=landers,
            digest_map=digest_map or None,
            peer_verify=(args.via == "peer"))
        assert res.total_bytes == total_bytes, res.total_bytes
        if record_digests:
            for f in res.files:
                if f.blob.digest_blob:
                    digest_map[f.name] = f.blob.digest_blob
        # model-ready: tensors scattered into their final (preallocated)
        # HBM addresses + zero-copy views materialized (virtual payloads
        # are patterned bytes, not parseable safetensors)
        n_t = 0
        if not args.virtual:
            if not scatter_targets:
                _build_targets(res)
            t_sc = time.perf_counter()
            for f in res.files:
                tg = scatter_targets.get(f.name)
                if tg:
                    hdr = st.parse_header(f.blob.head)
                    n_t += len(load_into(f.blob, hdr, tg))
            t_sc2 = time.perf_counter()
            n_t += len(res.tensors())
            log(f"scatter {t_sc2 - t_sc:.3f}s views "
                f"{time.perf_counter() - t_sc2:.3f}s")
        if have_gpu:
            torch.cuda.synchronize()
        return res, n_t

    def gguf_step(record_digests=False):
        """Ollama-style pull (config 4): land the q4_K GGUF blob, then
        dequant EVERY quant tensor to bf16 on-GPU; ready = bf16 weights
        resident."""
        from demodel_amd.engine.formats import gguf
        from demodel_amd.gpu import have_gpu as _hg, hip

        pd = (gguf.ProgressiveDequant(
            buffer_pool=landers.buffer_pool) if _hg() else None)
        res = pull_mod.pull_hf(
            "bench/model", endpoint=endpoint, workers=args.workers,
            verify=args.verify, landers=landers,
            digest_map=digest_map or None,
            on_range=pd.on_range if pd else None)
        assert res.total_bytes == total_bytes
        if record_digests:
            for f in res.files:
                digest_map[f.name] = f.blob.digest_blob
        blob = res.files[0].blob
        if pd is not None:
            # per-tensor dequants launched WHILE segments landed; this
            # finishes the stragglers and syncs (one output arena)
            tensors = pd.finish(blob)
            out_bytes = sum(t.numel() * 2 for t in tensors.values())
            n_t = len(tensors)
            log(f"dequant overlap: {pd.launched_early}/{n_t} tensors "
                f"launched before pull completion")
            del tensors
            pd.recycle_arena()  # views dropped; reuse at 70B scale
            del pd
        else:
            gg = gguf.parse(blob)
            out_bytes = sum(t.n_elems * 2 for t in gg.tensors
                            if t.type_id in gguf.ProgressiveDequant.QUANT_IDS)
            n_t = len(gg.tensors)
        if have_gpu:
            torch.cuda.synchronize()
        res.meta["dequant_bf16_bytes"] = out_bytes
        return res, n_t

    def _stream_step(record_digests, patterns):
        """config 5 via the user-facing stream_dataset API: each landed
        shard's GPU decompression overlaps with the remaining pulls;
        counts DECOMPRESSED bytes landed in HBM rings."""
        from demodel_amd.engine.datasets import stream_dataset

        def on_file(f):
            if record_digests and f.blob.digest_blob:
                digest_map[f.name] = f.blob.digest_blob

        t0 = time.perf_counter()
        out_bytes = 0
        n = 0
        for b in stream_dataset("bench/model", endpoint=endpoint,
                                workers=args.workers, verify=args.verify,
                                landers=landers, patterns=patterns,
                                digest_map=digest_map or None,
                                on_file=on_file, inflight=64,
                                eager=False):
            out_bytes += int(b.data.numel())
            n += 1
            del b
        if have_gpu:
            torch.cuda.synchronize()
        return _FanoutResult(out_bytes, time.perf_counter() - t0), n

    def parquet_step(record_digests=False):
        return _stream_step(record_digests, ("*.parquet",))

    def dataset_step(record_digests=False):
        return _stream_step(record_digests, ("*.zst", "*.idx.json"))

    file_sizes = [(n, os.path.getsize(p) if p else sizes[n])
                  for n, p in sorted(files.items())]

    def fanout_step(record_digests=False):
        """shard/broadcast modes: every rank ends with the full model."""
        from demodel_amd.engine.pull import _pull_blob
        from demodel_amd.parallel.fanout import (shard_assignment,
                                                 sharded_pull_fanout)

        if args.mode == "broadcast":
            plan = shard_assignment(file_sizes, 1, 0)  # rank 0 owns all
            plan.my_files = plan.my_files if rank == 0 else []
        else:
            plan = shard_assignment(file_sizes, world, rank)

        def pull_one(name):
            pf = _pull_blob(
                landers, name,
                f"{endpoint}/bench/model/resolve/main/{name}",
                None, args.verify, None, False,
                expected_digests=digest_map.get(name))
            if record_digests:
                digest_map[name] = pf.blob.digest_blob
            return pf.blob.torch_u8()

        def alloc(nb):
            return torch.empty(nb, dtype=torch.uint8,
                               device="cuda" if have_gpu else "cpu")

        out = sharded_pull_fanout(plan, pull_one, alloc)
        got = sum(t.numel() for t in out.values())
        assert got == total_bytes, (got, total_bytes)
        if have_gpu:
            torch.cuda.synchronize()
        return out

    step_bytes = total_bytes  # may be overridden by workloads below

    def allgather_step(record_digests=False):
        """R2: every rank pulls 1/N of the concatenated manifest bytes
        via HTTP Range, reassembly by bucketed all-gather overlapped
        with the remaining download."""
        from demodel_amd.engine import fetch
        from demodel_amd.parallel.fanout import range_sharded_allgather

        names = [n for n, _ in file_sizes]
        sizes = [sz for _, sz in file_sizes]
        prefix = [0]
        for sz in sizes:
            prefix.append(prefix[-1] + sz)

        class _Shim:
            def __init__(self, ptr):
                self.ptr = ptr

        bucket = 256 << 20

        def pull_range(lo, want, dest, bucket_done):
            lander = landers.get()
            buf = (_Shim(dest.data_ptr()) if have_gpu
                   else memoryview(dest.numpy()))
            done = 0
            flushed = 0
            while done < want:
                g = lo + done  # global offset
                fi = next(i for i in range(len(names))
                          if prefix[i + 1] > g)
                f_lo = g - prefix[fi]
                take = min(prefix[fi + 1] - g, want - done,
                           bucket - (done % bucket) or bucket)
                url = (f"{endpoint}/bench/model/resolve/main/"
                       f"{names[fi]}")
                src = fetch.http_get(url, headers={
                    "Range": f"bytes={f_lo}-{f_lo + take - 1}"})
                try:
                    assert src.status == 206, src.status
                    lander.land_into(buf, done, src.fill, take,
                                     file_size=want)
                finally:
                    src.close()
                done += take
                lander.sync()
                while (flushed + 1) * bucket <= done:
                    bucket_done(flushed)
                    flushed += 1
            while flushed * bucket < want:
                bucket_done(flushed)
                flushed += 1

        def alloc(nb):
            return torch.empty(nb, dtype=torch.uint8,
                               device="cuda" if have_gpu else "cpu")

        full = range_sharded_allgather(total_bytes, pull_range, alloc,
                                       bucket_bytes=bucket)
        assert full.numel() >= total_bytes
        if have_gpu:
            torch.cuda.synchronize()
        return _FanoutResult(total_bytes * world), 0

    def one_step(record_digests=False):
        nonlocal step_bytes
        if args.model.startswith("gguf"):
            return gguf_step(record_digests)
        if args.model == "dataset":
            res, n = dataset_step(record_digests)
            step_bytes = res.total_bytes
            return res, n
        if args.model == "parquet":
            res, n = parquet_step(record_digests)
            step_bytes = res.total_bytes
            return res, n
        if args.mode == "dp":
            return dp_step(record_digests)
        if args.mode == "allgather":
            return allgather_step(record_digests)
        res = fanout_step(record_digests)
        return _FanoutResult(total_bytes * world), len(res)

    # warmup; the first pull records chunk digests so every TIMED pull is a
    # fully verified re-pull (GPU sha256_batch compared against the record)
    for i in range(max(args.warmup, 1 if args.verify != "off" else 0)):
        res, _ = one_step(record_digests=(i == 0))
        log(f"warmup {i}: {res.seconds_to_ready:.2f}s "
            f"({res.gbps:.2f} GB/s)")
        # recycle HBM buffers instead of free+realloc: near device
        # capacity (70B = 141 GB twice on 288 GB) a fresh hipMalloc of
        # just-freed pages costs SECONDS of driver page reclaim
        landers.recycle(res)
        del res

    sync()
    t0 = time.perf_counter()
    per_step = []
    for i in range(args.steps):
        res, _ = one_step()
        per_step.append(res.seconds_to_ready)
        landers.recycle(res)
        del res
    sync()
    elapsed = time.perf_counter() - t0

    if dist:
        et = torch.tensor([elapsed])
        if have_gpu:
            et = et.cuda()
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
        elapsed = float(et.item())

    steps_bytes = step_bytes * args.steps * world
    value = steps_bytes / elapsed / 1e9
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        out = {
            "metric": "pull_gbps_into_hbm",
            "value": round(value, 3),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 1),
            "higher_is_better": True,
            "scaling": "weak" if args.mode == "dp" else "strong",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "Llama-3-8B" if args.model == "llama3-8b"
                else args.model,
                "bytes_per_model": total_bytes,
                "files": len(files),
                "verify": args.verify,
                "via": args.via,
                "scatter": bool(scatter_targets),
                "seconds_to_ready": round(ms_per_step / 1000.0, 3),
                "parallelism": (
                    f"independent-pull dp{world}" if args.mode == "dp"
                    else f"{args.mode}-fanout rccl x{world}"),
                "device": "cuda" if have_gpu else "cpu",
            },
        }
        print(json.dumps(out), flush=True)

    if proxy is not None:
        lt.call(proxy.close())
    lt.call(origin.close())
    lt.stop()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
