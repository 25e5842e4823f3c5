"""In-process fake Hugging Face Hub / Ollama registry origin.

Serves the minimal protocol surface the real clients use, so the proxy and
pull engine can be exercised with zero network (SURVEY.md §4's test plan;
the protocol shapes come from the reference's worked example —
CONTRIBUTING.md:127-153 for the Ollama manifest — and the public HF Hub
HTTP API as exercised by huggingface_hub):

* HF:  ``GET /api/{models|datasets}/{repo}[/revision/{rev}]`` (repo
       info + siblings),
       ``GET|HEAD /{repo}/resolve/{rev}/{file}`` with
       ``X-Repo-Commit`` / ``ETag`` / ``Content-Length`` headers and an
       optional 302 hop to a /cdn/ path (mirrors the hub->CDN redirect),
       Range support.
* Ollama: ``GET /v2/{name}/manifests/{tag}`` (Docker-v2 manifest JSON,
       optionally gzip Content-Encoding like the reference capture),
       ``GET|HEAD /v2/{name}/blobs/sha256:{digest}``.

Also doubles as the synthetic origin for bench.py: file-backed blobs are
served via sendfile so the origin is not the bottleneck.
"""

from __future__ import annotations

import asyncio
import gzip
import hashlib
import json
import os
import ssl

from ..proxy import http1
from ..proxy.http1 import RequestHead, ResponseHead


class FakeOrigin:
    def __init__(self, root: str, tls_ctx: ssl.SSLContext | None = None,
                 redirect_blobs: bool = True, gzip_manifests: bool = True):
        self.root = root  # directory holding blob files
        self.tls_ctx = tls_ctx
        self.redirect_blobs = redirect_blobs
        self.gzip_manifests = gzip_manifests
        # repo_id -> {"sha": commit, "files": {rfilename: abspath}}
        self.hf_repos: dict[str, dict] = {}
        # name -> {"tag": ..., "manifest": dict}; blobs by digest
        self.ollama_manifests: dict[tuple[str, str], dict] = {}
        self.blobs: dict[str, str] = {}  # "sha256:<hex>" -> abspath
        self.port: int | None = None
        self._server = None
        self.requests: list[str] = []  # log of "<METHOD> <path>"
        self.connections = 0           # accepted TCP connections
        # fault injection: path-substring -> serve N body bytes then drop
        # the connection (consumed on first match) — exercises the
        # engine's Range-resume path
        self.drop_once: dict[str, int] = {}
        # fault injection: path-substrings that hang (no response bytes)
        # until the client gives up — exercises proxy read timeouts
        self.hang_once: set[str] = set()
        # gated-repo emulation: /api and /resolve require this Bearer
        # token (401 without); /cdn REJECTS requests that still carry
        # Authorization (403 — presigned-URL semantics, like S3)
        self.require_token: str | None = None
        # absolute base for blob redirects (e.g. a second FakeOrigin
        # playing the CDN host — the real hub/S3 topology); None keeps
        # same-host /cdn/ paths
        self.cdn_base: str | None = None
        # docker-registry Bearer gate: /v2/ requests 401 with a
        # WWW-Authenticate pointing at our /token endpoint until the
        # client presents the issued token (the ollama.com/docker flow)
        self.docker_token: str | None = None

    # ------------------------------------------------------------------ #
    # content registration

    def add_hf_repo(self, repo_id: str, files: dict[str, str],
                    commit: str | None = None) -> str:
        commit = commit or hashlib.sha1(repo_id.encode()).hexdigest()
        self.hf_repos[repo_id] = {"sha": commit, "files": dict(files)}
        return commit

    def add_hf_repo_virtual(self, repo_id: str, sizes: dict[str, int],
                            commit: str | None = None,
                            prefixes: dict[str, bytes] | None = None
                            ) -> str:
        """Register a repo whose blobs are served from memory (tiled
        deterministic pattern) — no disk, no page cache; for benchmarks
        bigger than the box's disk (e.g. the 141 GB Llama-3-70B set).

        prefixes: optional per-file leading bytes (e.g. a REAL GGUF
        header) served before the pattern, so format parsers work on
        virtual blobs."""
        commit = commit or hashlib.sha1(repo_id.encode()).hexdigest()
        self.hf_repos[repo_id] = {"sha": commit, "files": {},
                                  "virtual": dict(sizes),
                                  "vprefix": dict(prefixes or {})}
        return commit

    _pattern: bytes | None = None

    @classmethod
    def _virtual_pattern(cls) -> bytes:
        if cls._pattern is None:
            import numpy as np

            cls._pattern = np.random.default_rng(7).integers(
                0, 256, size=64 << 20, dtype=np.uint8).tobytes()
        return cls._pattern

    def add_ollama_model(self, name: str, tag: str,
                         layers: list[tuple[str, str]]) -> dict:
        """layers: list of (media_type, blob_path). Returns the manifest."""
        def register(path: str) -> tuple[str, int]:
            h = hashlib.sha256()
            with open(path, "rb") as f:
                for chunk in iter(lambda: f.read(1 << 20), b""):
                    h.update(chunk)
            digest = "sha256:" + h.hexdigest()
            self.blobs[digest] = path
            return digest, os.path.getsize(path)

        config_path = os.path.join(self.root, f"{name.replace('/', '_')}-config.json")
        with open(config_path, "w") as f:
            json.dump({"model_format": "gguf", "model_family": "llama"}, f)
        cfg_digest, cfg_size = register(config_path)
        manifest = {
            "schemaVersion": 2,
            "mediaType": "application/vnd.docker.distribution.manifest.v2+json",
            "config": {
                "mediaType": "application/vnd.docker.container.image.v1+json",
                "digest": cfg_digest,
                "size": cfg_size,
            },
            "layers": [],
        }
        for media_type, path in layers:
            digest, size = register(path)
            manifest["layers"].append(
                {"mediaType": media_type, "digest": digest, "size": size})
        self.ollama_manifests[(name, tag)] = manifest
        return manifest

    # ------------------------------------------------------------------ #
    # server

    async def start(self, host: str = "127.0.0.1", port: int = 0) -> int:
        self._server = await asyncio.start_server(
            self._handle, host, port, ssl=self.tls_ctx, limit=http1.MAX_HEAD)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def close(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()

    async def _handle(self, reader, writer):
        self.connections += 1
        try:
            while True:
                head = await http1.read_request_head(reader)
                if head is None:
                    return
                self.requests.append(f"{head.method} {head.target}")
                for key in list(self.hang_once):
                    if key in head.target:
                        self.hang_once.discard(key)
                        await asyncio.sleep(3600)  # until client drops
                await self._dispatch(head, writer, reader)
                if head.get("connection", "").lower() == "close":
                    return
        except (http1.ProtocolError, ConnectionResetError,
                asyncio.IncompleteReadError, ssl.SSLError):
            pass
        finally:
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:
                pass

    async def _dispatch(self, head: RequestHead, writer, reader=None):
        path = head.target.split("?")[0]
        parts = [p for p in path.split("/") if p]

        if self.require_token is not None:
            auth = head.get("authorization", "")
            if parts and parts[0] == "cdn":
                if auth:
                    return await self._error(writer, 403)
            elif parts and parts[0] in ("api",) or "resolve" in parts:
                if auth != f"Bearer {self.require_token}":
                    return await self._error(writer, 401)

        # ---- POST/PUT echo (proxy pass-through tests): digest the
        # streamed request body, reply with its size + sha256 ----
        if parts == ["echo"] and head.method in ("POST", "PUT"):
            mode, length = http1.body_mode(head, method=head.method)
            h = hashlib.sha256()
            n = 0
            if reader is not None and mode != "none":
                async for chunk in http1.iter_body(reader, mode, length):
                    h.update(chunk)
                    n += len(chunk)
            body = json.dumps({"bytes": n,
                               "sha256": h.hexdigest()}).encode()
            return await self._reply(
                writer, head, 200,
                [("Content-Type", "application/json")], body)

        # ---- docker token endpoint ----
        if parts and parts[0] == "token" and self.docker_token:
            body = json.dumps({"token": self.docker_token}).encode()
            return await self._reply(
                writer, head, 200,
                [("Content-Type", "application/json")], body)

        # ---- Ollama registry v2 ----
        if parts and parts[0] == "v2":
            if (self.docker_token
                    and head.get("authorization", "")
                    != f"Bearer {self.docker_token}"):
                scheme = "https" if self.tls_ctx else "http"
                hdr = (f'Bearer realm="{scheme}://127.0.0.1:{self.port}'
                       f'/token",service="fake-registry",'
                       f'scope="repository:{"/".join(parts[1:-2])}:pull"')
                body = b'{"errors": [{"code": "UNAUTHORIZED"}]}'
                out = ResponseHead(
                    "HTTP/1.1", 401, "Unauthorized",
                    [("Content-Type", "application/json"),
                     ("Content-Length", str(len(body))),
                     ("WWW-Authenticate", hdr)])
                writer.write(http1.serialize_response(out) + body)
                await writer.drain()
                return
            if len(parts) >= 4 and parts[-2] == "manifests":
                name = "/".join(parts[1:-2])
                tag = parts[-1]
                m = self.ollama_manifests.get((name, tag))
                if m is None:
                    return await self._error(writer, 404)
                body = json.dumps(m).encode()
                hdrs = [("Content-Type",
                         "application/vnd.docker.distribution.manifest.v2+json"),
                        ("Docker-Content-Digest",
                         "sha256:" + hashlib.sha256(body).hexdigest())]
                if self.gzip_manifests:
                    body = gzip.compress(body)
                    hdrs.append(("Content-Encoding", "gzip"))
                return await self._reply(writer, head, 200, hdrs, body)
            if len(parts) >= 4 and parts[-2] == "blobs":
                digest = parts[-1]
                blob = self.blobs.get(digest)
                if blob is None:
                    return await self._error(writer, 404)
                return await self._serve_file(writer, head, blob, etag=digest)
            return await self._error(writer, 404)

        # ---- HF api: /api/models/{repo}/tree/{rev} (paginated file list,
        # what huggingface_hub 1.x snapshot_download walks) ----
        if parts[:2] in (["api", "models"], ["api", "datasets"]) \
                and "tree" in parts:
            i = parts.index("tree")
            repo_id = "/".join(parts[2:i])
            repo = self.hf_repos.get(repo_id)
            if repo is None:
                return await self._error(writer, 404)
            items = []
            for name in sorted(self._repo_names(repo)):
                size, etag, _ = self._file_info(repo, name)
                items.append({
                    "type": "file", "path": name, "size": size,
                    "oid": etag[:40],
                    "lfs": {"oid": etag, "size": size,
                            "pointerSize": 134},
                })
            body = json.dumps(items).encode()
            return await self._reply(
                writer, head, 200,
                [("Content-Type", "application/json")], body)

        # ---- HF api: /api/models/{repo}[/revision/{rev}] ----
        if parts[:2] in (["api", "models"], ["api", "datasets"]):
            rest = parts[2:]
            rev = "main"
            if "revision" in rest:
                i = rest.index("revision")
                rev = rest[i + 1] if i + 1 < len(rest) else "main"
                rest = rest[:i]
            repo_id = "/".join(rest)
            repo = self.hf_repos.get(repo_id)
            if repo is None:
                return await self._error(writer, 404)
            info = {
                "_id": repo["sha"][:24], "id": repo_id,
                "modelId": repo_id, "sha": repo["sha"],
                "private": False, "gated": False, "disabled": False,
                "downloads": 0, "likes": 0, "tags": [],
                "siblings": [{"rfilename": name}
                             for name in sorted(self._repo_names(repo))],
            }
            body = json.dumps(info).encode()
            return await self._reply(
                writer, head, 200,
                [("Content-Type", "application/json")], body)

        # ---- HF resolve: /{repo}/resolve/{rev}/{path...} ----
        if "resolve" in parts:
            i = parts.index("resolve")
            lead = parts[:i]
            # dataset blobs resolve under /datasets/{org}/{name}/...
            if lead and lead[0] == "datasets":
                lead = lead[1:]
            repo_id = "/".join(lead)
            rev = parts[i + 1] if i + 1 < len(parts) else "main"
            fname = "/".join(parts[i + 2:])
            repo = self.hf_repos.get(repo_id)
            if repo is None or fname not in self._repo_names(repo):
                return await self._error(writer, 404)
            size, etag, fpath = self._file_info(repo, fname)
            extra = [("X-Repo-Commit", repo["sha"]),
                     ("X-Linked-Etag", f'"{etag}"'),
                     ("X-Linked-Size", str(size))]
            if self.redirect_blobs:
                loc = ((self.cdn_base or "")
                       + f"/cdn/{repo_id}/{repo['sha']}/{fname}")
                return await self._reply(
                    writer, head, 302,
                    extra + [("Location", loc),
                             ("Content-Type", "text/plain")],
                    b"redirect")
            if fpath is None:
                return await self._serve_virtual(
                    writer, head, size, etag, extra=extra,
                    prefix=repo.get("vprefix", {}).get(fname))
            return await self._serve_file(writer, head, fpath, etag=etag,
                                          extra=extra)

        # ---- CDN: /cdn/{repo}/{commit}/{path...} ----
        if parts and parts[0] == "cdn":
            repo_id = "/".join(parts[1:3])
            fname = "/".join(parts[4:])
            repo = self.hf_repos.get(repo_id)
            if repo is None or fname not in self._repo_names(repo):
                return await self._error(writer, 404)
            size, etag, fpath = self._file_info(repo, fname)
            if fpath is None:
                return await self._serve_virtual(
                    writer, head, size, etag,
                    prefix=repo.get("vprefix", {}).get(fname))
            return await self._serve_file(writer, head, fpath, etag=etag)

        return await self._error(writer, 404)

    def _repo_names(self, repo: dict) -> set[str]:
        return set(repo["files"]) | set(repo.get("virtual", {}))

    def _file_info(self, repo: dict, name: str):
        """-> (size, etag, path_or_None)."""
        if name in repo["files"]:
            p = repo["files"][name]
            return os.path.getsize(p), self._file_etag(p), p
        size = repo["virtual"][name]
        # deliberately 40-hex (sha1-shaped): content is synthetic, so no
        # sha256 etag the engine might try to verify against
        return size, hashlib.sha1(
            f"virtual:{name}:{size}".encode()).hexdigest(), None

    async def _serve_virtual(self, writer, req: RequestHead, size: int,
                             etag: str, extra=None,
                             prefix: bytes | None = None):
        from ..utils.netio import send_pattern_threaded

        start, end = 0, size - 1
        status, reason = 200, "OK"
        rng = req.get("range")
        if rng and rng.startswith("bytes="):
            spec = rng[len("bytes="):].split(",")[0]
            s, _, e = spec.partition("-")
            if s:
                start = int(s)
                end = int(e) if e else size - 1
            else:
                start = max(0, size - int(e))
            status, reason = 206, "Partial Content"
        length = end - start + 1
        headers = [("Content-Type", "application/octet-stream"),
                   ("Content-Length", str(length)),
                   ("Accept-Ranges", "bytes"),
                   ("ETag", f'"{etag}"')] + (extra or [])
        if status == 206:
            headers.append(("Content-Range", f"bytes {start}-{end}/{size}"))
        writer.write(http1.serialize_response(
            ResponseHead("HTTP/1.1", status, reason, headers)))
        await writer.drain()
        if req.method == "HEAD":
            return
        sent = 0
        if prefix and start < len(prefix):
            piece = prefix[start:min(end + 1, len(prefix))]
            writer.write(piece)
            await writer.drain()
            sent = len(piece)
        if length - sent > 0:
            await send_pattern_threaded(writer, self._virtual_pattern(),
                                        start + sent, length - sent)

    _etag_cache: dict[tuple[str, float], str] = {}

    def _file_etag(self, path: str) -> str:
        key = (path, os.path.getmtime(path))
        hit = self._etag_cache.get(key)
        if hit:
            return hit
        h = hashlib.sha256()
        with open(path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                h.update(chunk)
        self._etag_cache[key] = h.hexdigest()
        return self._etag_cache[key]

    async def _sendfile_threaded(self, writer, f, start: int, length: int):
        from ..utils.netio import sendfile_threaded

        await sendfile_threaded(writer, f, start, length)

    async def _error(self, writer, status: int):
        body = json.dumps({"error": status}).encode()
        head = ResponseHead("HTTP/1.1", status, "Error",
                            [("Content-Type", "application/json"),
                             ("Content-Length", str(len(body)))])
        writer.write(http1.serialize_response(head) + body)
        await writer.drain()

    async def _reply(self, writer, req: RequestHead, status: int,
                     headers: list[tuple[str, str]], body: bytes):
        headers = headers + [("Content-Length", str(len(body))),
                             ("Accept-Ranges", "bytes")]
        reason = {200: "OK", 302: "Found"}.get(status, "OK")
        writer.write(http1.serialize_response(
            ResponseHead("HTTP/1.1", status, reason, headers)))
        if req.method != "HEAD":
            writer.write(body)
        await writer.drain()

    async def _serve_file(self, writer, req: RequestHead, path: str,
                          etag: str, extra: list[tuple[str, str]] | None = None):
        size = os.path.getsize(path)
        start, end = 0, size - 1
        status, reason = 200, "OK"
        rng = req.get("range")
        if rng and rng.startswith("bytes="):
            spec = rng[len("bytes="):].split(",")[0]
            s, _, e = spec.partition("-")
            if s:
                start = int(s)
                end = int(e) if e else size - 1
            else:  # suffix range
                start = max(0, size - int(e))
            status, reason = 206, "Partial Content"
        drop_at = None
        for key in list(self.drop_once):
            # key "name" fires on the first request for name; key
            # "name@OFF" only on a request whose Range starts at OFF
            # (targets one segment of a range-parallel pull)
            name, _, want_start = key.partition("@")
            if name in req.target and \
                    (not want_start or int(want_start) == start):
                drop_at = self.drop_once.pop(key)
                break
        length = end - start + 1
        headers = [("Content-Type", "application/octet-stream"),
                   ("Content-Length", str(length)),
                   ("Accept-Ranges", "bytes"),
                   ("ETag", f'"{etag}"')] + (extra or [])
        if status == 206:
            headers.append(("Content-Range", f"bytes {start}-{end}/{size}"))
        writer.write(http1.serialize_response(
            ResponseHead("HTTP/1.1", status, reason, headers)))
        await writer.drain()
        if req.method == "HEAD":
            return
        if drop_at is not None:
            # fault injection: truncate the body and kill the connection
            with open(path, "rb") as f:
                f.seek(start)
                writer.write(f.read(min(drop_at, length)))
                await writer.drain()
            writer.transport.abort()
            return
        loop = asyncio.get_running_loop()
        with open(path, "rb") as f:
            f.seek(start)
            if self.tls_ctx is None:
                try:
                    # blocking sendfile on a worker thread: the asyncio
                    # loop.sendfile path serializes every stream through
                    # the event loop (~3x slower at 4+ parallel blobs —
                    # scripts/net_probe.py)
                    await self._sendfile_threaded(writer, f, start, length)
                    return
                except (NotImplementedError, OSError):
                    f.seek(start)
            remaining = length
            while remaining > 0:
                data = f.read(min(http1.CHUNK, remaining))
                if not data:
                    break
                remaining -= len(data)
                writer.write(data)
                await writer.drain()
