"""HBM-warm restart probe: a client pulls the 16 GB flagship THROUGH
the proxy (auto pull-ahead lands it in HBM), then an engine pull of the
same repo is served from the registry — measure its seconds-to-ready.

This is the unification headline: a vLLM-style restart against a warm
demodel node skips the network AND the disk.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests"))


def main():
    import concurrent.futures as cf

    from demodel_amd.config import Config
    from demodel_amd.engine import fetch
    from demodel_amd.engine import pull as pull_mod
    from demodel_amd.engine.pull import LanderPool
    from demodel_amd.proxy.server import ProxyServer
    from demodel_amd.testing import synth
    from demodel_amd.testing.origin import FakeOrigin
    from helpers import LoopThread

    data_dir = os.path.join(os.environ.get("TMPDIR", "/tmp"),
                            "demodel_bench_llama3-8b")
    files = synth.write_shards(data_dir, synth.LLAMA3_8B, 4)
    total = sum(os.path.getsize(p) for p in files.values())
    lt = LoopThread()
    origin = FakeOrigin(data_dir, redirect_blobs=True)
    origin.add_hf_repo("bench/model", files)
    oport = lt.call(origin.start())

    cfg = Config(host="127.0.0.1", port=0,
                 cache_dir=os.path.join(data_dir, "pfcache"),
                 gpu_prefetch="auto")
    proxy = ProxyServer(cfg, prefetch_landers=LanderPool(0, gpu=True))
    proxy.reverse_routes = [("/", f"http://127.0.0.1:{oport}")]
    pport = lt.call(proxy.start())
    endpoint = f"http://127.0.0.1:{pport}"

    # 1. client-style pull through the proxy (this is the "first boot")
    def prime(name):
        src = fetch.http_get(f"{endpoint}/bench/model/resolve/main/{name}")
        try:
            assert src.status == 200, (name, src.status)
            sink = memoryview(bytearray(8 << 20))
            while src.fill(sink) > 0:
                pass
        finally:
            src.close()

    t0 = time.perf_counter()
    prime_names = list(files)
    with cf.ThreadPoolExecutor(max_workers=8) as ex:
        list(ex.map(prime, prime_names))
    t_client = time.perf_counter() - t0

    # 2. wait for the auto pull-ahead to land everything in HBM
    want = {f"/bench/model/resolve/main/{n}" for n in files
            if proxy._is_blob_path(f"/bench/model/resolve/main/{n}")}
    t0 = time.perf_counter()
    while not want.issubset(set(proxy.registry.keys())):
        assert time.perf_counter() - t0 < 300, proxy.registry.keys()
        time.sleep(0.1)
    t_land = time.perf_counter() - t0

    # 3. the "restart": engine pull served HBM-resident
    n_origin = len(origin.requests)
    t0 = time.perf_counter()
    res = pull_mod.pull_hf("bench/model", endpoint=endpoint,
                           registry=proxy.registry, verify="off")
    n_t = len(res.tensors())
    t_warm = time.perf_counter() - t0
    blob_bytes = sum(f.nbytes for f in res.files
                     if f.name.endswith(".safetensors"))
    print(json.dumps({
        "op": "hbm_warm_restart",
        "model_gb": round(total / 1e9, 2),
        "client_pull_s": round(t_client, 3),
        "pull_ahead_extra_s": round(t_land, 3),
        "warm_restart_s": round(t_warm, 4),
        "warm_tensors": n_t,
        "warm_blob_gb": round(blob_bytes / 1e9, 2),
        "origin_requests_during_restart": len(origin.requests) - n_origin,
    }), flush=True)
    lt.call(proxy.close())
    lt.call(origin.close())
    lt.stop()


if __name__ == "__main__":
    main()
