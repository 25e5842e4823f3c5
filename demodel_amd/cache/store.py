"""On-disk response cache, layout-compatible with the reference
(bodies byte-exact in original Content-Encoding; 16-hex key + .meta
sidecar — the KEY DERIVATION differs, see below).

Layout contract (reference CONTRIBUTING.md:53-153):

* ``{root}/{key}``       — the response body, byte-exact as received from
  the origin, i.e. still in its original ``Content-Encoding`` (the worked
  example is a gzip body whose file starts with the 1f8b magic —
  CONTRIBUTING.md:116).
* ``{root}/{key}.meta``  — per-URI response metadata (status + headers).
  The reference's meta was an opaque binary (bincode, deleted Rust
  prototype); ours is JSON, which keeps the documented "inspect the
  metadata with cat" workflow honest.

``key`` is 16 lowercase hex chars (64 bits) derived from the canonical
request URI — the same SHAPE as the reference example key
``1b8c2ef6c820e0c0`` (CONTRIBUTING.md:57).  We use the first 8 bytes of
SHA-256(uri); the reference's derivation is undocumented (none of
sha256/md5/sha1/sha512/blake2b/sha3 over the documented URI reproduces
its example), so an existing reference cache directory is NOT hit —
the compatibility claim is layout, not key-for-key.

Extensions beyond the reference (which never shipped cache code in Go —
SURVEY.md §0):

* streaming fill via ``CacheWriter`` (tee while forwarding, atomic rename
  on completion so a crashed fill never leaves a half body);
* per-chunk SHA-256 digests in the meta (``chunks``) so a cache hit can be
  re-verified **in parallel** by the GPU batch-hash kernel instead of one
  sequential whole-file chain;
* whole-body sha256 recorded for end-to-end digest checks.
"""

from __future__ import annotations

import hashlib
import json
import os
import tempfile
import time
from dataclasses import dataclass, field


def cache_key(uri: str) -> str:
    return hashlib.sha256(uri.encode()).hexdigest()[:16]


# Hop-by-hop headers must not be replayed (RFC 9110 §7.6.1).
_HOP_BY_HOP = {
    "connection", "keep-alive", "proxy-authenticate", "proxy-authorization",
    "te", "trailer", "transfer-encoding", "upgrade",
}


@dataclass
class CachedResponse:
    uri: str
    status: int
    reason: str
    headers: list[tuple[str, str]]
    body_path: str
    body_size: int
    sha256: str | None
    chunk_bytes: int
    chunk_sha256: list[str] = field(default_factory=list)
    created: float = 0.0

    def open_body(self):
        return open(self.body_path, "rb")

    def read_body(self) -> bytes:
        with self.open_body() as f:
            return f.read()


class CacheWriter:
    """Streaming cache fill; finalize() commits, abort() discards.

    Digesting modes (store.digest_mode):
      * "sync"  — hash inline while writing (tests, small bodies)
      * "async" — commit immediately, compute whole+chunk sha256 on a
        background worker and patch the meta when done (the proxy's blob
        path: a 10+ GB fill must not run at single-thread hashlib speed)
      * "off"   — no digests
    """

    def __init__(self, store: "CacheStore", uri: str, status: int,
                 reason: str, headers: list[tuple[str, str]]):
        self._store = store
        self.uri = uri
        self.key = cache_key(uri)
        self.status = status
        self.reason = reason
        self.headers = [
            (k, v) for k, v in headers if k.lower() not in _HOP_BY_HOP
        ]
        fd, self._tmp = tempfile.mkstemp(
            prefix=f".{self.key}.", suffix=".part", dir=store.root
        )
        self._f = os.fdopen(fd, "wb")
        self._size = 0
        self._sync = store.digest_mode == "sync"
        self._whole = hashlib.sha256() if self._sync else None
        self._chunk = hashlib.sha256()
        self._chunk_fill = 0
        self._chunk_digests: list[str] = []
        self.chunk_bytes = store.chunk_bytes

    def write(self, data: bytes) -> None:
        self._f.write(data)
        self._size += len(data)
        if not self._sync:
            return
        self._whole.update(data)
        view = memoryview(data)
        while view:
            take = min(len(view), self.chunk_bytes - self._chunk_fill)
            self._chunk.update(view[:take])
            self._chunk_fill += take
            view = view[take:]
            if self._chunk_fill == self.chunk_bytes:
                self._chunk_digests.append(self._chunk.hexdigest())
                self._chunk = hashlib.sha256()
                self._chunk_fill = 0

    def finalize(self) -> CachedResponse:
        if self._sync and self._chunk_fill:
            self._chunk_digests.append(self._chunk.hexdigest())
        self._f.close()
        body_path = os.path.join(self._store.root, self.key)
        os.replace(self._tmp, body_path)
        entry = CachedResponse(
            uri=self.uri, status=self.status, reason=self.reason,
            headers=self.headers, body_path=body_path, body_size=self._size,
            sha256=self._whole.hexdigest() if self._sync else None,
            chunk_bytes=self.chunk_bytes,
            chunk_sha256=self._chunk_digests, created=time.time(),
        )
        self._store._write_meta(entry)
        if self._store.digest_mode == "async":
            self._store._digest_async(entry)
        return entry

    def abort(self) -> None:
        try:
            self._f.close()
        finally:
            if os.path.exists(self._tmp):
                os.unlink(self._tmp)


class CacheStore:
    _digest_pool = None

    def __init__(self, root: str = ".cache", chunk_bytes: int = 1 << 20,
                 digest_mode: str = "sync"):
        self.root = root
        self.chunk_bytes = chunk_bytes
        self.digest_mode = digest_mode
        os.makedirs(root, exist_ok=True)

    def _write_meta(self, entry: CachedResponse) -> None:
        meta = {
            "uri": entry.uri,
            "status": entry.status,
            "reason": entry.reason,
            "headers": entry.headers,
            "body_size": entry.body_size,
            "sha256": entry.sha256,
            "chunk_bytes": entry.chunk_bytes,
            "chunk_sha256": entry.chunk_sha256,
            "created": entry.created or time.time(),
        }
        # UNIQUE temp name: concurrent fills of the same URI (two cold
        # misses racing before either registered in-flight) both commit;
        # a shared ".meta.part" made the loser's os.replace crash with
        # FileNotFoundError (surfaced by the 32-thread soak test)
        fd, tmp_meta = tempfile.mkstemp(
            prefix=f".{os.path.basename(entry.body_path)}.",
            suffix=".meta.part", dir=self.root)
        with os.fdopen(fd, "w") as f:
            json.dump(meta, f, indent=1)
        os.replace(tmp_meta, entry.body_path + ".meta")

    def _digest_async(self, entry: CachedResponse) -> None:
        import concurrent.futures as cf

        if CacheStore._digest_pool is None:
            CacheStore._digest_pool = cf.ThreadPoolExecutor(
                max_workers=2, thread_name_prefix="cache-digest")

        def job():
            try:
                whole = hashlib.sha256()
                chunks = []
                with open(entry.body_path, "rb") as f:
                    while True:
                        data = f.read(entry.chunk_bytes)
                        if not data:
                            break
                        whole.update(data)
                        chunks.append(hashlib.sha256(data).hexdigest())
                entry.sha256 = whole.hexdigest()
                entry.chunk_sha256 = chunks
                self._write_meta(entry)
            except OSError:
                pass  # entry purged/overwritten meanwhile

        CacheStore._digest_pool.submit(job)

    def lookup(self, uri: str) -> CachedResponse | None:
        key = cache_key(uri)
        body_path = os.path.join(self.root, key)
        meta_path = body_path + ".meta"
        if not (os.path.exists(body_path) and os.path.exists(meta_path)):
            return None
        try:
            with open(meta_path) as f:
                meta = json.load(f)
        except (OSError, json.JSONDecodeError):
            return None
        if meta.get("uri") != uri:
            return None  # 64-bit key collision — treat as miss
        if meta.get("body_size") != os.path.getsize(body_path):
            return None  # torn entry
        return CachedResponse(
            uri=uri, status=meta["status"], reason=meta.get("reason", ""),
            headers=[tuple(h) for h in meta["headers"]],
            body_path=body_path, body_size=meta["body_size"],
            sha256=meta.get("sha256"),
            chunk_bytes=meta.get("chunk_bytes", self.chunk_bytes),
            chunk_sha256=meta.get("chunk_sha256", []),
            created=meta.get("created", 0.0),
        )

    def writer(self, uri: str, status: int, reason: str,
               headers: list[tuple[str, str]]) -> CacheWriter:
        return CacheWriter(self, uri, status, reason, headers)

    def cacheable(self, method: str, status: int) -> bool:
        return method == "GET" and status in (200, 301, 302, 307, 308)

    def gc(self, max_bytes: int) -> dict:
        """Evict least-recently-used entries until total body bytes fit
        under max_bytes.  Recency = body file atime (falls back to
        mtime).  Returns {kept, evicted, bytes}."""
        entries = []
        now = time.time()
        for fn in os.listdir(self.root):
            # orphaned fill temps (process died mid-fill): reap after 1 h
            if fn.endswith(".part"):
                p = os.path.join(self.root, fn)
                try:
                    if now - os.path.getmtime(p) > 3600:
                        os.unlink(p)
                except OSError:
                    pass
                continue
            if not fn.endswith(".meta"):
                continue
            body = os.path.join(self.root, fn[:-5])
            if not os.path.exists(body):
                continue
            st_ = os.stat(body)
            entries.append((max(st_.st_atime, st_.st_mtime), body,
                            st_.st_size))
        total = sum(sz for _, _, sz in entries)
        evicted = 0
        entries.sort()  # oldest first
        for _, body, sz in entries:
            if total <= max_bytes:
                break
            for p in (body, body + ".meta"):
                if os.path.exists(p):
                    os.unlink(p)
            total -= sz
            evicted += 1
        return {"kept": len(entries) - evicted, "evicted": evicted,
                "bytes": total}

    def purge(self, uri: str) -> bool:
        key = cache_key(uri)
        body_path = os.path.join(self.root, key)
        hit = False
        for p in (body_path, body_path + ".meta"):
            if os.path.exists(p):
                os.unlink(p)
                hit = True
        return hit
