"""Test harness: run the asyncio origin + proxy in a background thread so
synchronous clients (urllib, huggingface_hub) can talk to them."""

import asyncio
import threading

from demodel_amd.config import Config
from demodel_amd.proxy.server import ProxyServer
from demodel_amd.testing.origin import FakeOrigin


class LoopThread:
    def __init__(self):
        self.loop = asyncio.new_event_loop()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _run(self):
        asyncio.set_event_loop(self.loop)
        self.loop.run_forever()

    def call(self, coro, timeout=30):
        fut = asyncio.run_coroutine_threadsafe(coro, self.loop)
        return fut.result(timeout)

    def stop(self):
        self.loop.call_soon_threadsafe(self.loop.stop)
        self._thread.join(timeout=5)


class Stack:
    """Origin + proxy running on one background loop."""

    def __init__(self, tmp_path, origin_tls_ctx=None, mitm_hosts=None,
                 leafs=None, origin_kwargs=None, cfg_kwargs=None):
        self.lt = LoopThread()
        self.origin = FakeOrigin(str(tmp_path), tls_ctx=origin_tls_ctx,
                                 **(origin_kwargs or {}))
        self.origin_port = self.lt.call(self.origin.start())
        scheme = "https" if origin_tls_ctx else "http"
        origin_base = f"{scheme}://127.0.0.1:{self.origin_port}"

        cfg = Config(host="127.0.0.1", port=0,
                     cache_dir=str(tmp_path / "proxycache"),
                     **(cfg_kwargs or {}))
        if mitm_hosts is not None:
            cfg.mitm_hosts = mitm_hosts
        self.cfg = cfg
        self.proxy = ProxyServer(cfg, leafs=leafs)
        # route everything at the fake origin
        self.proxy.reverse_routes = [("/", origin_base)]
        self.proxy_port = self.lt.call(self.proxy.start())
        self.origin_base = origin_base

    @property
    def endpoint(self):
        return f"http://127.0.0.1:{self.proxy_port}"

    def stop_origin(self):
        self.lt.call(self.origin.close())

    def close(self):
        try:
            self.lt.call(self.proxy.close())
        except Exception:
            pass
        try:
            self.lt.call(self.origin.close())
        except Exception:
            pass
        self.lt.stop()
