"""The ``demodel`` command-line interface.

Parity with the reference CLI (cmd/demodel/main.go:56-81):

* ``demodel`` / ``demodel start`` — run the proxy (start.go:218-230)
* ``demodel init``                — create/load the CA (init.go:156-168)
* ``demodel export-ca [--for X]`` — print or install the CA
  (export_ca.go:22-120; presets python-ssl, python-certifi, plus the
  README-promised ``openssl`` the reference never implemented)

New subcommands beyond the reference:

* ``demodel pull <spec>``  — pull a model/dataset directly (no client
  needed): hf://org/repo[@rev] or ollama://name[:tag], with GPU landing
  auto-selected (force with --gpu / --cpu).
* ``demodel verify <spec>`` — re-verify cached blobs against digests.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import sys

from .config import load_config


def _cmd_start(args) -> int:
    from .proxy.server import run_proxy

    cfg = load_config()
    if args.port is not None:
        cfg.port = args.port
    if getattr(args, "gpu_prefetch", None) is not None:
        cfg.gpu_prefetch = args.gpu_prefetch
    if getattr(args, "loops", None) is not None:
        cfg.loops = args.loops
    if getattr(cfg, "loops", 1) > 1:
        # SO_REUSEPORT fleet: one event loop per core slice, so MITM
        # TLS crypto scales (proxy/server.py ProxyFleet)
        import threading

        from .ca import read_or_new_ca
        from .certs import LeafStore
        from .proxy.server import ProxyFleet

        ca = read_or_new_ca(cfg.ca_use_ecdsa)
        landers = None
        if getattr(cfg, "gpu_prefetch", "off") != "off":
            from .engine.pull import LanderPool
            from .gpu import have_gpu

            landers = LanderPool(0, gpu=True if have_gpu() else False)
        fleet = ProxyFleet(cfg, leafs=LeafStore(ca),
                           prefetch_landers=landers)
        fleet.start()
        try:
            threading.Event().wait()
        except KeyboardInterrupt:
            fleet.close()
        return 0
    try:
        asyncio.run(run_proxy(cfg))
    except KeyboardInterrupt:
        pass
    return 0


def _cmd_init(args) -> int:
    from . import ca as ca_mod

    cfg = load_config()
    ca_mod.read_or_new_ca(cfg.ca_use_ecdsa)
    crt, key = ca_mod.cert_paths()
    print(f"CA certificate: {crt}")
    print(f"CA private key: {key}")
    if args.install:
        ok = ca_mod.install_system()
        print("system trust store: " + ("installed" if ok else "skipped"))
    return 0


def _cmd_export_ca(args) -> int:
    from . import ca as ca_mod

    out = ca_mod.export_ca(args.dest)
    if args.dest is None:
        sys.stdout.write(out)
    else:
        print(f"CA exported to {out}")
    return 0


def _cmd_pull(args) -> int:
    from .engine.pull import pull_spec
    from .gpu import have_gpu

    if args.gpu and not have_gpu():
        print("error: --gpu requested but no AMD GPU is available",
              file=sys.stderr)
        return 1
    gpu = True if args.gpu else (False if args.cpu else None)
    kw = {}
    if getattr(args, "peer_verify", False):
        kw["peer_verify"] = True
    result = pull_spec(args.spec, endpoint=args.endpoint, out_dir=args.out,
                       gpu=gpu, **kw)
    print(json.dumps(result, indent=2, default=str))
    return 0


def _cmd_verify(args) -> int:
    from .engine.pull import verify_cache

    cfg = load_config()
    result = verify_cache(cfg, uri=args.uri)
    print(json.dumps(result, indent=2))
    return 0 if result.get("ok", False) else 1


def _cmd_stats(args) -> int:
    from .engine import fetch

    base = (args.endpoint or "http://127.0.0.1:8080").rstrip("/")
    try:
        print(json.dumps(fetch.get_json(base + "/__demodel/stats"),
                         indent=1))
    except (fetch.FetchError, OSError) as e:
        print(f"error: no demodel proxy reachable at {base} ({e})",
              file=sys.stderr)
        return 1
    return 0


def _cmd_prefetch(args) -> int:
    """Ask a running proxy to land blob paths into its HBM registry
    (POST /__demodel/prefetch); with no paths, print prefetch status."""
    import urllib.request

    base = (args.endpoint or "http://127.0.0.1:8080").rstrip("/")
    url = base + "/__demodel/prefetch"
    try:
        if not args.paths:
            with urllib.request.urlopen(url, timeout=30) as r:
                print(json.dumps(json.loads(r.read()), indent=1))
            return 0
        req = urllib.request.Request(
            url, method="POST",
            data=json.dumps({"paths": args.paths}).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=30) as r:
            out = json.loads(r.read())
    except urllib.error.HTTPError as e:  # before OSError: subclass
        print(f"error: {e} — is the proxy running with prefetch "
              f"landers (--gpu-prefetch)?", file=sys.stderr)
        return 1
    except OSError as e:
        print(f"error: no demodel proxy reachable at {base} ({e})",
              file=sys.stderr)
        return 1
    if getattr(args, "wait", False):
        import time

        deadline = time.time() + args.timeout
        left = set(args.paths)
        while left:
            if time.time() > deadline:
                print(f"error: timed out waiting for {sorted(left)}",
                      file=sys.stderr)
                return 1
            time.sleep(0.2)
            with urllib.request.urlopen(url, timeout=30) as r:
                st = json.loads(r.read())
            left -= set(st.get("registered", []))
            # a path that is neither registered nor in flight failed
            # (lands register BEFORE leaving the in-flight set)
            gone = left - set(st.get("in_flight", []))
            if gone:
                print(f"error: prefetch failed for {sorted(gone)}",
                      file=sys.stderr)
                return 1
        out["registered"] = args.paths
    print(json.dumps(out, indent=1))
    return 0 if out.get("queued") is not None else 1


def _cmd_gc(args) -> int:
    from .cache import CacheStore

    cfg = load_config()
    store = CacheStore(cfg.cache_dir)
    result = store.gc(int(args.max_gb * 1e9))
    print(json.dumps(result))
    return 0


def main(argv: list[str] | None = None) -> int:
    p = argparse.ArgumentParser(
        prog="demodel",
        description="Caching, syncing, distributing middleware for models "
                    "and datasets — MI355X-native.")
    sub = p.add_subparsers(dest="cmd")

    sp = sub.add_parser("start", help="run the caching proxy")
    sp.add_argument("--port", type=int, default=None)
    sp.add_argument("--loops", type=int, default=None,
                    help="acceptor event loops on one port "
                         "(SO_REUSEPORT); >1 scales MITM TLS across "
                         "cores (also DEMODEL_LOOPS)")
    sp.add_argument("--gpu-prefetch", default=None,
                    choices=["off", "auto"],
                    help="land proxy-cached blobs into HBM ahead of "
                         "engine pulls (also DEMODEL_GPU_PREFETCH)")
    sp.set_defaults(fn=_cmd_start)

    ip = sub.add_parser("init", help="create or load the demodel CA")
    ip.add_argument("--install", action="store_true",
                    help="also install into the system trust store")
    ip.set_defaults(fn=_cmd_init)

    ep = sub.add_parser("export-ca", help="export the CA certificate")
    ep.add_argument("--for", dest="dest", default=None,
                    choices=["python-ssl", "python-certifi", "openssl"])
    ep.set_defaults(fn=_cmd_export_ca)

    pp = sub.add_parser("pull", help="pull a model/dataset through the engine")
    pp.add_argument("spec", help="hf://org/repo[@rev] or ollama://name[:tag]")
    pp.add_argument("--endpoint", default=None,
                    help="upstream endpoint override (e.g. a fake origin)")
    pp.add_argument("--gpu", action="store_true",
                    help="land blobs in GPU HBM via the HIP pipeline "
                         "(default: auto)")
    pp.add_argument("--cpu", action="store_true",
                    help="force host-RAM landing even with a GPU present")
    pp.add_argument("--out", default=None, help="materialize files here")
    pp.add_argument("--peer-verify", action="store_true",
                    help="when --endpoint is another demodel node, "
                         "GPU-verify every chunk against the peer's "
                         "digest record (hf:// specs)")
    pp.set_defaults(fn=_cmd_pull)

    vp = sub.add_parser("verify", help="re-verify cached blobs")
    vp.add_argument("--uri", default=None)
    vp.set_defaults(fn=_cmd_verify)

    st = sub.add_parser("stats", help="transfer metrics of a running "
                                      "proxy (GET /__demodel/stats)")
    st.add_argument("--endpoint", default=None,
                    help="proxy base URL (default http://127.0.0.1:8080)")
    st.set_defaults(fn=_cmd_stats)

    gp = sub.add_parser("gc", help="evict LRU cache entries over a size "
                                   "budget")
    gp.add_argument("--max-gb", type=float, required=True)
    gp.set_defaults(fn=_cmd_gc)

    pf = sub.add_parser("prefetch",
                        help="land blob paths into a running proxy's "
                             "HBM registry (POST /__demodel/prefetch); "
                             "no paths = show status")
    pf.add_argument("paths", nargs="*",
                    help="request paths, e.g. "
                         "/org/repo/resolve/main/model.safetensors")
    pf.add_argument("--endpoint", default=None,
                    help="proxy base URL (default http://127.0.0.1:8080)")
    pf.add_argument("--wait", action="store_true",
                    help="block until every path is registered "
                         "(HBM-resident) or --timeout expires")
    pf.add_argument("--timeout", type=float, default=600.0)
    pf.set_defaults(fn=_cmd_prefetch)

    args = p.parse_args(argv)
    if not hasattr(args, "fn"):
        # bare `demodel` runs the proxy, like the reference root command
        # (main.go:68-70)
        return _cmd_start(argparse.Namespace(port=None))
    return args.fn(args)


if __name__ == "__main__":
    raise SystemExit(main())
