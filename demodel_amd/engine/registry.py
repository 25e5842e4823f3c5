"""HBM-resident blob registry: the bridge between the proxy's cache and
the engine's landing pipeline (VERDICT round-1 item 3; north-star
"single pipeline" — BASELINE.json:5).

The proxy observes blob traffic (or an explicit
``POST /__demodel/prefetch``) and lands the cached bytes into HBM ahead
of time; a subsequent engine pull of the same path is served straight
from the registry — zero upstream traffic, zero disk reads in the hot
path (reference anchor: the request-hook short-circuit contract,
cmd/demodel/start.go:197-200, extended from "serve from disk cache" to
"serve from HBM").

Keys are canonical request paths (e.g.
``/org/repo/resolve/main/model.safetensors`` or
``/v2/library/llama3/blobs/sha256:...``) — exactly what both a proxied
client and the engine-native client request, so the two surfaces meet
on one name.
"""

from __future__ import annotations

import threading
from collections import OrderedDict

from ..utils.log import get_logger

log = get_logger("registry")


class BlobRegistry:
    """Thread-safe LRU of landed blobs keyed by request path.

    Registered blobs are marked ``shared``: LanderPool.recycle refuses
    to steal their HBM buffers, so a pull served from the registry can
    hand out views safely.
    """

    def __init__(self, max_bytes: int | None = None):
        self._lock = threading.Lock()
        self._d: "OrderedDict[str, object]" = OrderedDict()
        self.max_bytes = max_bytes
        self.total_bytes = 0
        self.hits = 0
        self.misses = 0

    def get(self, key: str):
        with self._lock:
            blob = self._d.get(key)
            if blob is None:
                self.misses += 1
                return None
            self._d.move_to_end(key)
            self.hits += 1
            return blob

    def put(self, key: str, blob) -> None:
        blob.shared = True
        with self._lock:
            old = self._d.pop(key, None)
            if old is not None:
                self.total_bytes -= old.nbytes
            self._d[key] = blob
            self.total_bytes += blob.nbytes
            while (self.max_bytes is not None
                   and self.total_bytes > self.max_bytes
                   and len(self._d) > 1):
                k, evicted = self._d.popitem(last=False)
                self.total_bytes -= evicted.nbytes
                log.info("evicted %s (%d bytes) from HBM registry",
                         k, evicted.nbytes)

    def __contains__(self, key: str) -> bool:
        with self._lock:
            return key in self._d

    def keys(self) -> list[str]:
        with self._lock:
            return list(self._d)

    def stats(self) -> dict:
        with self._lock:
            return {
                "entries": len(self._d),
                "bytes": self.total_bytes,
                "hits": self.hits,
                "misses": self.misses,
            }
