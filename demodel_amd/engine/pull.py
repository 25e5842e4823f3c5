"""High-level pull engine: hf:// and ollama:// specs -> verified blobs in
HBM (or host memory on CPU-only machines), with zero-copy safetensors
tensor views and GGUF dequant.

The reference only relays client traffic; this module is
the engine-native client the north star adds: manifest fetch, concurrent
chunked blob pulls through the landing pipeline, digest verification, and
model-ready tensor materialization.
"""

from __future__ import annotations

import concurrent.futures as cf
import ctypes
import fnmatch
import os
import threading
import time
from dataclasses import dataclass, field

from ..config import Config
from ..utils.log import get_logger
from . import fetch
from .pipeline import HostLander, Lander, make_lander
from .formats import safetensors as st

log = get_logger("pull")

HF_DEFAULT_ENDPOINT = "https://huggingface.co"
OLLAMA_DEFAULT_ENDPOINT = "https://registry.ollama.ai"


@dataclass
class PulledFile:
    name: str
    url: str
    nbytes: int
    blob: object            # LandedBlob
    etag: str | None = None
    digest_ok: bool | None = None
    seconds: float = 0.0


@dataclass
class PullResult:
    spec: str
    files: list[PulledFile] = field(default_factory=list)
    total_bytes: int = 0
    seconds_to_ready: float = 0.0
    device: str = "cpu"
    meta: dict = field(default_factory=dict)

    @property
    def gbps(self) -> float:
        return self.total_bytes / max(self.seconds_to_ready, 1e-9) / 1e9

    def tensors(self):
        """name -> torch tensor views for every safetensors file."""
        out = {}
        for f in self.files:
            if not f.name.endswith(".safetensors"):
                continue
            hdr = st.parse_header(f.blob.head)
            out.update(st.torch_views(hdr, f.blob.torch_u8()))
        return out

    def summary(self) -> dict:
        return {
            "spec": self.spec,
            "device": self.device,
            "files": [
                {"name": f.name, "bytes": f.nbytes,
                 "seconds": round(f.seconds, 4),
                 "gbps": round(f.nbytes / max(f.seconds, 1e-9) / 1e9, 3),
                 "digest_ok": f.digest_ok}
                for f in self.files
            ],
            "total_bytes": self.total_bytes,
            "seconds_to_ready": round(self.seconds_to_ready, 4),
            "gbps": round(self.gbps, 3),
            **self.meta,
        }


class LanderPool:
    """One landing pipeline per worker thread (own pinned ring+streams).

    All landers share one BufferPool: re-allocating a just-freed large
    HBM buffer costs seconds near device capacity (driver page
    reclaim), so steady-state re-pulls recycle via
    `pool.recycle(result)` — caller guarantees no outstanding tensor
    views of the recycled blobs."""

    def __init__(self, device_index: int = 0, **kw):
        self._device_index = device_index
        self._kw = kw
        self._local = threading.local()
        self._all: list = []
        self._lock = threading.Lock()
        from .pipeline import BufferPool

        self.buffer_pool = BufferPool()

    def get(self):
        lander = getattr(self._local, "lander", None)
        if lander is None:
            lander = make_lander(self._device_index,
                                 buffer_pool=self.buffer_pool, **self._kw)
            self._local.lander = lander
            with self._lock:
                self._all.append(lander)
        return lander

    def recycle(self, result) -> int:
        """Hand a PullResult's HBM blob buffers back for reuse; returns
        the number of buffers recycled.  The caller must not touch the
        result's blobs (or views of them) afterwards."""
        n = 0
        files = getattr(result, "files", None)
        if files is None:
            files = [result]
        for f in files:
            blob = getattr(f, "blob", f)
            if getattr(blob, "device", "cpu") == "cpu":
                continue
            if getattr(blob, "shared", False):
                continue  # registry-owned (proxy prefetch): not ours
            buf = blob.buffer
            if buf is not None:
                self.buffer_pool.put(buf, max(blob.nbytes, 1))
                blob.buffer = None
                n += 1
        return n


def blob_to_file(blob, path: str, chunk: int = 32 << 20) -> None:
    """Materialize a landed blob to disk (D2H read-back for GPU blobs)."""
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    if blob.device == "cpu":
        with open(path, "wb") as f:
            f.write(blob.buffer)
        return
    from ..gpu import hip

    h = hip()
    stream = h.Stream(0)
    staging = bytearray(min(chunk, blob.nbytes))
    addr = ctypes.addressof(
        (ctypes.c_char * len(staging)).from_buffer(staging))
    with open(path, "wb") as f:
        off = 0
        while off < blob.nbytes:
            n = min(chunk, blob.nbytes - off)
            h.d2h_async(addr, blob.buffer.ptr + off, n, stream.handle)
            stream.sync()
            f.write(memoryview(staging)[:n])
            off += n


def _etag_sha256(etag: str | None) -> str | None:
    if not etag:
        return None
    tag = etag.strip('"').removeprefix("W/").strip('"')
    if len(tag) == 64 and all(c in "0123456789abcdef" for c in tag.lower()):
        return tag.lower()
    return None


# Segment defaults from the call-10 hardware sweep: a single-file
# 4.5 GB pull runs 23.3 GB/s at 8x512 MiB segments, 25.3 at 16x256 MiB,
# and REGRESSES to 17.1 at 24 segments + 16 workers (thread/conn
# oversubscription); multi-file pulls are insensitive (31.0 vs 31.7).
SEGMENT_MIN = int(os.environ.get("DEMODEL_SEGMENT_MIN_MB", "256")) << 20
MAX_SEGMENTS = int(os.environ.get("DEMODEL_MAX_SEGMENTS", "16"))
RESUME_RETRIES = 4


def _land_with_resume(lander, open_fn, nbytes: int, verify: bool,
                      expected_digests: bytes | None,
                      host_chain: bool = False,
                      retries: int | None = None,
                      verify_chunk: int | None = None):
    """Land a blob with automatic Range-resume on mid-stream failures.

    open_fn(offset) -> fill callable streaming bytes [offset, nbytes).
    Works with both Lander (HBM) and HostLander (RAM).
    """
    import hashlib

    from .pipeline import (LandedBlob, LandingError, check_digests)

    if retries is None:
        retries = RESUME_RETRIES
    buf = lander.alloc(nbytes)
    chain = hashlib.sha256() if host_chain else None
    head = bytearray()
    landed = 0
    attempt = 0
    fill = open_fn(0)
    while landed < nbytes:
        try:
            hd, _ = lander.land_into(
                buf, landed, fill, nbytes - landed, file_size=nbytes,
                chain=chain, keep_head=landed < lander.head_bytes)
            if hd:
                head += hd[:max(0, lander.head_bytes - len(head))]
            landed = nbytes
        except LandingError as e:
            landed += e.landed
            attempt += 1
            if attempt > retries:
                raise IOError(
                    f"pull failed after {attempt} attempts at byte "
                    f"{landed}/{nbytes}") from e
            log.info("resuming pull at byte %d/%d (attempt %d)",
                     landed, nbytes, attempt)
            fill = open_fn(landed)
    lander.sync()
    # A resumed landing captured its head from the resume offset (or not
    # at all — land_into drops its partial head when it raises); rebuild
    # from the landed buffer so header parsing (safetensors/GGUF)
    # survives mid-stream drops.
    if attempt > 0 or len(head) != min(nbytes, lander.head_bytes):
        head = lander.read_head(buf, nbytes)
    vc = verify_chunk or lander.verify_chunk
    device = ("cpu" if isinstance(buf, bytearray)
              else f"cuda:{lander.device_index}")
    blob = LandedBlob(nbytes=nbytes, device=device, buffer=buf,
                      verify_chunk=vc, head=bytes(head))
    if verify or expected_digests is not None:
        blob.digest_blob = lander.finish_verify(buf, nbytes, vc)
        if expected_digests is not None:
            check_digests(blob.digest_blob, expected_digests, vc)
    if chain is not None:
        blob.sha256 = chain.hexdigest()
    return blob


def _pull_blob(landers: LanderPool, name: str, url: str,
               expected_sha: str | None, verify: str,
               cafile, insecure, headers=None,
               expected_digests: bytes | None = None,
               seg_executor: cf.ThreadPoolExecutor | None = None,
               verify_chunk: int | None = None,
               on_range=None) -> PulledFile:
    """on_range(name, lo, hi, buf, head_or_None): progress hook fired
    as byte ranges become resident (per segment for range-parallel
    pulls, once at completion otherwise) — lets consumers start GPU
    work (e.g. GGUF dequant) on the landed prefix while the rest still
    downloads."""
    t0 = time.perf_counter()
    do_verify = verify in ("chunked", "digest", "gpu-digest")
    want_segments = (seg_executor is not None
                     and verify in ("chunked", "off"))
    req_headers = dict(headers or {})
    if want_segments:
        # probe range support by asking for the first segment outright; a
        # 200 means no range support and we just stream the whole body
        req_headers["Range"] = f"bytes=0-{SEGMENT_MIN - 1}"
    src = fetch.http_get(url, cafile=cafile, insecure=insecure,
                         headers=req_headers)
    try:
        if want_segments and src.status == 206:
            cr = src.resp.get("content-range", "")
            total = int(cr.rsplit("/", 1)[1])
            if total > SEGMENT_MIN:
                blob = _pull_segmented(landers, url, total, src,
                                       expected_digests, cafile, insecure,
                                       headers, seg_executor, verify_chunk,
                                       name=name, on_range=on_range)
            else:
                blob = landers.get().land(
                    src.fill, min(total, SEGMENT_MIN),
                    verify=do_verify,
                    expected_digests=expected_digests,
                    verify_chunk=verify_chunk)
        else:
            if src.status != 200:
                raise fetch.FetchError(
                    f"GET {url} -> {src.status} {src.resp.reason}")
            nbytes = src.length
            if nbytes < 0:
                raise fetch.FetchError(f"no content-length for blob {url}")
            lander = landers.get()
            if verify == "gpu-digest":
                # the sequential GPU chain can't resume across re-fetches
                blob = lander.land(
                    src.fill, nbytes, verify=do_verify, gpu_chain=True,
                    expected_digests=expected_digests,
                    verify_chunk=verify_chunk)
            else:
                extra_sources = []
                first = [True]

                def open_fn(offset):
                    if offset == 0 and first[0]:
                        first[0] = False
                        return src.fill
                    rng_headers = dict(headers or {})
                    # the resume hits the FINAL (post-redirect) URL:
                    # drop credentials if that host differs from the
                    # hub (presigned CDN URLs reject them)
                    from urllib.parse import urlsplit as _us

                    if _us(src.resp.url).netloc != _us(url).netloc:
                        for k in list(rng_headers):
                            if k.lower() in ("authorization", "cookie"):
                                del rng_headers[k]
                    if offset:
                        rng_headers["Range"] = f"bytes={offset}-"
                    s2 = fetch.http_get(src.resp.url, cafile=cafile,
                                        insecure=insecure,
                                        headers=rng_headers)
                    want = 206 if offset else 200
                    if s2.status != want:
                        s2.close()
                        raise fetch.FetchError(
                            f"resume GET -> {s2.status} (want {want})")
                    extra_sources.append(s2)
                    return s2.fill

                try:
                    blob = _land_with_resume(
                        lander, open_fn, nbytes, verify=do_verify,
                        expected_digests=expected_digests,
                        host_chain=(verify == "digest"),
                        verify_chunk=verify_chunk)
                finally:
                    for s2 in extra_sources:
                        s2.close()
        nbytes = blob.nbytes
        if on_range is not None and not getattr(blob, "_ranged", False):
            on_range(name, 0, nbytes, blob.buffer, blob.head)
    finally:
        src.close()
    etag = src.resp.get("x-linked-etag") or src.resp.get("etag")
    want = expected_sha or _etag_sha256(etag)
    ok = None
    if blob.sha256 is not None and want is not None:
        ok = blob.sha256 == want
        if not ok:
            raise fetch.FetchError(
                f"digest mismatch for {url}: got {blob.sha256}, "
                f"want {want}")
    pf = PulledFile(name=name, url=url, nbytes=nbytes, blob=blob,
                    etag=etag, digest_ok=ok,
                    seconds=time.perf_counter() - t0)
    t = blob.timings
    log.info("landed %s: %d bytes in %.3fs (%.2f GB/s) on %s%s",
             name, nbytes, pf.seconds, nbytes / max(pf.seconds, 1e-9) / 1e9,
             blob.device,
             (f" [fill {t['fill_s']:.3f}s land {t['land_s']:.3f}s "
              f"verify {t['verify_s']:.3f}s]") if t else "")
    return pf


def _pull_segmented(landers: LanderPool, url: str, total: int, src0,
                    expected_digests, cafile, insecure, headers,
                    seg_executor, verify_chunk=None, name=None,
                    on_range=None) -> "object":
    """Range-parallel landing of one blob: segment 0 comes from the
    already-open 206 stream, the rest are parallel range GETs, all landing
    into disjoint ranges of one HBM buffer through per-thread pinned
    rings.  Chunk verification runs once over the whole buffer."""
    from .pipeline import LandedBlob

    lander0 = landers.get()
    buf = lander0.alloc(total)
    n_segs = min(MAX_SEGMENTS, (total + SEGMENT_MIN - 1) // SEGMENT_MIN)
    # segment 0 covers exactly what the already-open 206 stream serves
    # ([0, SEGMENT_MIN)); the rest of the file splits evenly
    seg0 = min(SEGMENT_MIN, total)
    rest = total - seg0
    bounds = [0, seg0] + [
        seg0 + rest * i // (n_segs - 1) for i in range(1, n_segs)
    ] if n_segs > 1 else [0, total]
    assert bounds[-1] == total, bounds

    from .pipeline import LandingError

    # Incremental verification: each segment hashes its chunk-ALIGNED
    # range the moment it finishes landing (on that lander's verify
    # stream), so the end-of-blob verify work shrinks to the few
    # boundary chunks + one digest D2H — round 1 hashed the whole
    # buffer serially after the last byte.
    # Incremental per-segment verification measured a ~25% REGRESSION
    # on the flagship (same-box A/B, profiles/bench_history.md): the
    # many small hash launches across per-lander verify streams
    # multiplex onto the 4 HW queues and SERIALIZE with the H2D copy
    # streams mid-landing, while the single whole-buffer tail hash is a
    # big efficient launch that overlaps across files anyway.  Kept
    # selectable for re-evaluation (DEMODEL_INC_VERIFY=1), default off.
    inc_verify = os.environ.get("DEMODEL_INC_VERIFY", "0") == "1"
    vc = verify_chunk or lander0.verify_chunk
    n_chunks = (total + vc - 1) // vc
    dig_dev = (lander0._h.DeviceBuffer(n_chunks * 32)
               if inc_verify else None)

    def land_range(i):
        lo, hi = bounds[i], bounds[i + 1]
        if hi <= lo:
            return None
        lander = landers.get()
        at = lo
        attempt = 0
        while at < hi:
            rng = {"Range": f"bytes={at}-{hi - 1}"}
            if headers:
                rng.update(headers)
            s = fetch.http_get(url, cafile=cafile, insecure=insecure,
                               headers=rng)
            try:
                if s.status != 206:
                    raise fetch.FetchError(
                        f"range GET {url} [{at},{hi}) -> {s.status}")
                lander.land_into(buf, at, s.fill, hi - at, file_size=total)
                at = hi
            except LandingError as e:
                at += e.landed
                attempt += 1
                if attempt > RESUME_RETRIES:
                    raise IOError(
                        f"segment [{lo},{hi}) failed after {attempt} "
                        f"attempts at byte {at}") from e
                log.info("resuming segment [%d,%d) at byte %d (attempt "
                         "%d)", lo, hi, at, attempt)
            finally:
                s.close()
        lander.sync()
        hashed = (lander.hash_range_into(buf, lo, hi, vc, dig_dev, total)
                  if inc_verify else None)
        if on_range is not None:
            on_range(name, lo, hi, buf, None)
        return hashed

    futs = [seg_executor.submit(land_range, i) for i in range(1, n_segs)]
    # this thread lands segment 0 from the open stream (it was asked for
    # [0, SEGMENT_MIN) which equals bounds[1] when total >= SEGMENT_MIN)
    try:
        head, _ = lander0.land_into(buf, 0, src0.fill, bounds[1],
                                    file_size=total, keep_head=True)
    except LandingError as e:
        # the probe stream died mid-segment-0: finish it with range GETs
        head = None  # rebuilt from the landed buffer below
        at = e.landed
        attempt = 0
        while at < bounds[1]:
            rng = {"Range": f"bytes={at}-{bounds[1] - 1}"}
            if headers:
                rng.update(headers)
            s = fetch.http_get(url, cafile=cafile, insecure=insecure,
                               headers=rng)
            try:
                if s.status != 206:
                    raise fetch.FetchError(
                        f"range GET {url} [{at},{bounds[1]}) -> "
                        f"{s.status}")
                lander0.land_into(buf, at, s.fill, bounds[1] - at,
                                  file_size=total)
                at = bounds[1]
            except LandingError as e2:
                at += e2.landed
                attempt += 1
                if attempt > RESUME_RETRIES:
                    raise
            finally:
                s.close()
    lander0.sync()
    covered = [lander0.hash_range_into(buf, 0, bounds[1], vc, dig_dev,
                                       total) if inc_verify else None]
    if head is None:
        # seg-0 fallback dropped its in-flight head capture: rebuild the
        # SEGMENT-0 portion now (other segments may still be landing)
        head = lander0.read_head(buf, bounds[1])
    if on_range is not None:
        on_range(name, 0, bounds[1], buf, bytes(head))
    for f in futs:
        covered.append(f.result())
    # head spans min(total, head_bytes), which can extend past segment 0
    # (small files under a small SEGMENT_MIN): complete it now that every
    # segment's lander has synced
    if len(head) < min(total, lander0.head_bytes):
        head = lander0.read_head(buf, total)
    blob = LandedBlob(nbytes=total, device=f"cuda:{lander0.device_index}",
                      buffer=buf, verify_chunk=vc, head=bytes(head))
    blob._ranged = True  # on_range already fired per segment
    if not inc_verify:  # A/B fallback: whole-buffer tail hash
        blob.digest_blob = lander0._gpu_chunk_digests(buf, total, vc)
        if expected_digests is not None:
            from .pipeline import check_digests

            check_digests(blob.digest_blob, expected_digests, vc)
        return blob
    # boundary chunks no segment covered (chunk-unaligned bounds), plus
    # any too-small segment's range: hash the gap runs now — every
    # landing is host-synced at this point
    covered = [c for c in covered if c is not None]
    ivs = sorted((lo_c, hi_c) for _, lo_c, hi_c in covered)
    events = [ev for ev, _, _ in covered]
    h = lander0._h
    at = 0
    gaps = []
    for lo_c, hi_c in ivs:
        if lo_c > at:
            gaps.append((at, lo_c))
        at = max(at, hi_c)
    if at < n_chunks:
        gaps.append((at, n_chunks))
    for a, b in gaps:
        span = min(total, b * vc) - a * vc
        h.sha256_batch(buf.ptr + a * vc, span, vc, dig_dev.ptr + a * 32,
                       b - a, lander0.verify_stream.handle)
    blob.digest_blob = lander0.collect_digests(dig_dev, n_chunks, events)
    if expected_digests is not None:
        from .pipeline import check_digests

        check_digests(blob.digest_blob, expected_digests, vc)
    return blob


def fetch_peer_digests(endpoint: str, path: str, cafile=None,
                       insecure=False) -> tuple[bytes, int] | None:
    """Ask a demodel peer for its recorded chunk digests of `path`
    (GET /__demodel/digests/<path>).  Returns (raw_digests, chunk_bytes)
    or None when the peer has none (not a peer / not cached / digests
    still computing)."""
    try:
        obj = fetch.get_json(f"{endpoint}/__demodel/digests{path}",
                             cafile=cafile, insecure=insecure)
    except (fetch.FetchError, OSError, ValueError):
        return None
    digs = obj.get("chunk_sha256") or []
    if not digs or not obj.get("chunk_bytes"):
        return None
    return bytes.fromhex("".join(digs)), int(obj["chunk_bytes"])


def pull_hf_stream(repo: str, rev: str = "main",
                   endpoint: str | None = None,
                   device_index: int = 0, workers: int = 4,
                   verify: str = "chunked", cafile=None,
                   insecure: bool = False,
                   patterns: list[str] | None = None,
                   landers: LanderPool | None = None,
                   slab_bytes: int = 32 << 20,
                   digest_map: dict[str, bytes] | None = None,
                   peer_verify: bool = False, batched: bool = False,
                   on_range=None, repo_type: str = "model",
                   registry=None, token: str | None = None):
    """Streaming pull: returns (info, names, generator) where the
    generator yields each PulledFile AS IT FINISHES landing, so a
    consumer (e.g. stream_dataset's GPU decompression) overlaps with the
    remaining downloads.  With batched=True it yields LISTS — every file
    that has finished since the last wake-up arrives together, so the
    consumer can coalesce per-burst work (one decode launch per burst).
    The generator owns the worker pools; closing it early cancels
    unstarted pulls."""
    endpoint = (endpoint or os.environ.get("HF_ENDPOINT")
                or HF_DEFAULT_ENDPOINT).rstrip("/")
    # gated/private repos: Bearer token on hub requests; fetch strips
    # it on the cross-host CDN hop (presigned URLs reject credentials)
    if token is None:
        token = (os.environ.get("HF_TOKEN")
                 or os.environ.get("HUGGING_FACE_HUB_TOKEN"))
    auth = {"Authorization": f"Bearer {token}"} if token else None
    # HF dataset repos live under /api/datasets and resolve their blobs
    # at /datasets/{repo}/resolve/... (models have no path prefix)
    api = "datasets" if repo_type == "dataset" else "models"
    prefix = "datasets/" if repo_type == "dataset" else ""
    info = fetch.get_json(f"{endpoint}/api/{api}/{repo}/revision/{rev}",
                          cafile=cafile, insecure=insecure,
                          headers=auth)
    names = [s["rfilename"] for s in info.get("siblings", [])]
    if patterns:
        names = [n for n in names
                 if any(fnmatch.fnmatch(n, p) for p in patterns)]
    landers = landers or LanderPool(device_index, slab_bytes=slab_bytes)
    from ..gpu import have_gpu

    def peer_expected(n):
        if not peer_verify:
            return None
        return fetch_peer_digests(
            endpoint, f"/{prefix}{repo}/resolve/{rev}/{n}",
            cafile=cafile, insecure=insecure)

    def reg_hit(n):
        """HBM-registry short-circuit (proxy pull-ahead): the blob is
        already landed and verified — serve it with zero fetch, zero
        disk (engine/registry.py)."""
        if registry is None:
            return None
        blob = registry.get(f"/{prefix}{repo}/resolve/{rev}/{n}")
        if blob is None:
            return None
        if on_range is not None:
            on_range(n, 0, blob.nbytes, blob.buffer, blob.head)
        return PulledFile(
            name=n, url=f"{endpoint}/{prefix}{repo}/resolve/{rev}/{n}",
            nbytes=blob.nbytes, blob=blob, digest_ok=True, seconds=0.0)

    def gen():
        seg_ex = (cf.ThreadPoolExecutor(max_workers=max(workers, 4))
                  if have_gpu() else None)
        try:
            with cf.ThreadPoolExecutor(max_workers=workers) as ex:
                futs = {}
                hits = []
                for n in names:
                    pf = reg_hit(n)
                    if pf is not None:
                        hits.append(pf)
                        continue
                    pd = peer_expected(n)
                    exp = (digest_map or {}).get(n)
                    vc = None
                    if pd is not None and exp is None:
                        exp, vc = pd
                    futs[ex.submit(
                        _pull_blob, landers, n,
                        f"{endpoint}/{prefix}{repo}/resolve/{rev}/{n}",
                        None, verify, cafile, insecure, auth,
                        exp, seg_ex, vc, on_range)] = n
                try:
                    if batched:
                        if hits:
                            yield hits
                        left = set(futs)
                        while left:
                            finished, left = cf.wait(
                                left, return_when=cf.FIRST_COMPLETED)
                            yield [f.result() for f in finished]
                    else:
                        yield from hits
                        for fut in cf.as_completed(futs):
                            yield fut.result()
                except GeneratorExit:
                    for f in futs:
                        f.cancel()
                    raise
        finally:
            if seg_ex:
                seg_ex.shutdown()

    return info, names, gen()


def pull_hf(repo: str, rev: str = "main", endpoint: str | None = None,
            device_index: int = 0, workers: int = 4,
            verify: str = "chunked", out_dir: str | None = None,
            cafile=None, insecure: bool = False,
            patterns: list[str] | None = None,
            landers: LanderPool | None = None,
            slab_bytes: int = 32 << 20,
            digest_map: dict[str, bytes] | None = None,
            peer_verify: bool = False, on_range=None,
            repo_type: str = "model", registry=None,
            token: str | None = None) -> PullResult:
    """peer_verify: when `endpoint` is another demodel node, fetch its
    recorded chunk digests per blob and GPU-verify the pull against them
    (verified distribution).  on_range: progress hook, see _pull_blob.
    repo_type: "model" or "dataset" (different HF URL layout).
    registry: a BlobRegistry (proxy pull-ahead); files already landed in
    HBM are served from it with zero fetch."""
    t0 = time.perf_counter()
    info, _, gen = pull_hf_stream(
        repo, rev, endpoint=endpoint, device_index=device_index,
        workers=workers, verify=verify, cafile=cafile, insecure=insecure,
        patterns=patterns, landers=landers, slab_bytes=slab_bytes,
        digest_map=digest_map, peer_verify=peer_verify,
        on_range=on_range, repo_type=repo_type, registry=registry,
        token=token)
    result = PullResult(spec=f"hf://{repo}@{rev}")
    result.files = list(gen)
    result.files.sort(key=lambda f: f.name)
    result.total_bytes = sum(f.nbytes for f in result.files)
    result.seconds_to_ready = time.perf_counter() - t0
    result.device = result.files[0].blob.device if result.files else "cpu"
    result.meta["commit"] = info.get("sha")
    if out_dir:
        for f in result.files:
            blob_to_file(f.blob, os.path.join(out_dir, f.name))
    return result


def _registry_token(www_auth: str, cafile=None,
                    insecure: bool = False) -> str | None:
    """Docker-registry Bearer handshake: parse a 401's
    ``WWW-Authenticate: Bearer realm="...",service="...",scope="..."``
    and fetch a token from the realm (how ollama.com / docker
    registries gate private models; registry.ollama.ai's public
    library never 401s)."""
    import re
    from urllib.parse import quote

    if not www_auth or not www_auth.lower().startswith("bearer"):
        return None
    parts = dict(re.findall(r'(\w+)="([^"]*)"', www_auth))
    realm = parts.get("realm")
    if not realm:
        return None
    q = "&".join(f"{k}={quote(parts[k], safe='')}"
                 for k in ("service", "scope") if parts.get(k))
    try:
        obj = fetch.get_json(realm + (f"?{q}" if q else ""),
                             cafile=cafile, insecure=insecure)
    except (fetch.FetchError, OSError, ValueError):
        return None
    return obj.get("token") or obj.get("access_token")


def _registry_get_json(url: str, cafile, insecure,
                       auth: dict | None) -> tuple[dict, dict | None]:
    """GET a registry JSON document, performing the Bearer handshake on
    a 401.  Returns (json, auth_headers_used)."""
    import json as _json

    src = fetch.http_get(url, cafile=cafile, insecure=insecure,
                         headers=auth)
    try:
        if src.status == 401 and auth is None:
            tok = _registry_token(src.resp.get("www-authenticate", ""),
                                  cafile, insecure)
            if tok is not None:
                src.close()
                auth = {"Authorization": f"Bearer {tok}"}
                src = fetch.http_get(url, cafile=cafile,
                                     insecure=insecure, headers=auth)
        if src.status != 200:
            raise fetch.FetchError(
                f"GET {url} -> {src.status} {src.resp.reason}")
        return _json.loads(src.read_all(limit=256 << 20)), auth
    finally:
        src.close()


def pull_ollama(name: str, tag: str = "latest",
                endpoint: str | None = None, device_index: int = 0,
                workers: int = 4, verify: str = "digest",
                out_dir: str | None = None, cafile=None,
                insecure: bool = False,
                landers: LanderPool | None = None,
                dequant: bool = True,
                dequant_tensors: bool = False) -> PullResult:
    """Pull an Ollama model: manifest + layer blobs by sha256 digest
    (protocol shape per reference CONTRIBUTING.md:127-153).

    dequant=True parses the GGUF header into meta["gguf_model"].
    dequant_tensors=True (GPU) additionally dequantizes every tensor to
    bf16 — overlapped with the download via ProgressiveDequant — into
    meta["tensors"] ({name: torch bf16 tensor})."""
    endpoint = (endpoint or os.environ.get("OLLAMA_REGISTRY")
                or OLLAMA_DEFAULT_ENDPOINT).rstrip("/")
    if "/" not in name:
        name = f"library/{name}"
    t0 = time.perf_counter()
    manifest, reg_auth = _registry_get_json(
        f"{endpoint}/v2/{name}/manifests/{tag}", cafile, insecure, None)
    layers = list(manifest.get("layers", []))
    if manifest.get("config"):
        layers.append(manifest["config"])
    landers = landers or LanderPool(device_index)
    result = PullResult(spec=f"ollama://{name}:{tag}")
    from ..gpu import have_gpu

    seg_ex = (cf.ThreadPoolExecutor(max_workers=max(workers, 4))
              if have_gpu() and verify in ("chunked", "off") else None)
    pd = None
    if dequant and dequant_tensors and have_gpu():
        from .formats.gguf import ProgressiveDequant

        pd = ProgressiveDequant(device_index)
    with cf.ThreadPoolExecutor(max_workers=workers) as ex:
        futs = {}
        for layer in layers:
            digest = layer["digest"]
            url = f"{endpoint}/v2/{name}/blobs/{digest}"
            expected = digest.split(":", 1)[1] \
                if digest.startswith("sha256:") else None
            is_model = (layer.get("mediaType")
                        == "application/vnd.ollama.image.model")
            futs[ex.submit(_pull_blob, landers, digest, url, expected,
                           verify, cafile, insecure, reg_auth, None,
                           seg_ex, None,
                           pd.on_range if (pd and is_model) else None)
                 ] = layer
        for fut in cf.as_completed(futs):
            pf = fut.result()
            pf.name = futs[fut].get("mediaType", pf.name)
            result.files.append(pf)
    if seg_ex:
        seg_ex.shutdown()
    result.total_bytes = sum(f.nbytes for f in result.files)
    result.seconds_to_ready = time.perf_counter() - t0
    result.device = result.files[0].blob.device if result.files else "cpu"
    result.meta["manifest"] = manifest
    if dequant:
        model_layers = [
            f for f in result.files
            if f.name == "application/vnd.ollama.image.model"]
        if model_layers:
            from .formats import gguf

            gg = gguf.parse(model_layers[0].blob)
            result.meta["gguf"] = {
                "n_tensors": len(gg.tensors),
                "types": sorted({t.type_name for t in gg.tensors}),
            }
            result.meta["gguf_model"] = gg
            if pd is not None:
                result.meta["tensors"] = pd.finish(model_layers[0].blob)
    if out_dir:
        for f in result.files:
            safe = f.name.replace("/", "_").replace(":", "_")
            blob_to_file(f.blob, os.path.join(out_dir, safe))
    return result


def pull_spec(spec: str, cfg: Config | None = None, endpoint=None,
              gpu: bool | None = None, out_dir=None, **kw) -> dict:
    """CLI entry: parse hf://org/repo[@rev] or ollama://name[:tag].

    gpu=True forces HBM landing, gpu=False host RAM, None auto."""
    if gpu is not None:
        kw.setdefault("landers", LanderPool(0, gpu=gpu))
    if spec.startswith("hf://"):
        body = spec[len("hf://"):]
        # hf://datasets/org/name pulls a DATASET repo (different HF URL
        # layout); anything else is a model repo
        if body.startswith("datasets/"):
            body = body[len("datasets/"):]
            kw.setdefault("repo_type", "dataset")
        repo, _, rev = body.partition("@")
        res = pull_hf(repo, rev or "main", endpoint=endpoint,
                      out_dir=out_dir, **kw)
        return res.summary()
    if spec.startswith("ollama://"):
        kw.pop("peer_verify", None)  # hf-layout only
        body = spec[len("ollama://"):]
        name, _, tag = body.partition(":")
        res = pull_ollama(name, tag or "latest", endpoint=endpoint,
                          out_dir=out_dir, **kw)
        s = res.summary()
        s.pop("manifest", None)
        s.pop("gguf_model", None)
        s.pop("tensors", None)
        return s
    raise ValueError(f"unknown spec {spec!r} (want hf:// or ollama://)")


def verify_cache(cfg: Config, uri: str | None = None) -> dict:
    """Re-verify cached entries' bodies against their recorded chunk
    digests (CPU path; the GPU path re-verifies at landing time)."""
    import hashlib

    from ..cache import CacheStore

    store = CacheStore(cfg.cache_dir)
    checked, bad = 0, []
    metas = []
    if uri is not None:
        e = store.lookup(uri)
        if e is None:
            return {"ok": False, "error": f"no cache entry for {uri}"}
        metas = [e]
    else:
        for fn in os.listdir(cfg.cache_dir):
            if fn.endswith(".meta"):
                import json

                with open(os.path.join(cfg.cache_dir, fn)) as f:
                    meta = json.load(f)
                e = store.lookup(meta["uri"])
                if e is not None:
                    metas.append(e)
    for e in metas:
        checked += 1
        hh = hashlib.sha256()
        ok = True
        with e.open_body() as f:
            for i, want in enumerate(e.chunk_sha256):
                data = f.read(e.chunk_bytes)
                hh.update(data)
                if hashlib.sha256(data).hexdigest() != want:
                    ok = False
                    break
        if ok and e.sha256 and hh.hexdigest() != e.sha256:
            ok = False
        if not ok:
            bad.append(e.uri)
    return {"ok": not bad, "checked": checked, "bad": bad}
