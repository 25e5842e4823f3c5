"""Process configuration from environment variables.

Reimplements the env-var surface of the reference's C2 config loader
(reference cmd/demodel/main.go:15-42) with its two bugs fixed:

* empty ``DEMODEL_PROXY_MITM_HOSTS`` no longer clobbers the default host
  list with ``[""]`` (reference main.go:30-32 splits "" into [""]);
* boolean env vars accept "1"/"true"/"yes"/"on" case-insensitively.

Additional knobs (port, cache dir, chunking, GPU pipeline) are new —
the reference hard-codes :8080 and ``.cache/`` (start.go:206,
CONTRIBUTING.md:62).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field


DEFAULT_MITM_HOSTS = ["huggingface.co:443"]

_TRUTHY = {"1", "true", "yes", "on"}


def _env_bool(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.strip().lower() in _TRUTHY


def _env_list(name: str) -> list[str]:
    """Comma-separated list; empty/unset -> [] (never [""])."""
    v = os.environ.get(name, "")
    return [h.strip() for h in v.split(",") if h.strip()]


def _env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    if v is None or not v.strip():
        return default
    return int(v)


def data_dir() -> str:
    """XDG data dir for persistent state (CA certs).

    Mirrors the reference's use of adrg/xdg (init.go:32-38):
    ``$XDG_DATA_HOME/demodel`` falling back to ``~/.local/share/demodel``.
    """
    base = os.environ.get("XDG_DATA_HOME") or os.path.join(
        os.path.expanduser("~"), ".local", "share"
    )
    return os.path.join(base, "demodel")


@dataclass
class Config:
    # --- proxy / MITM (reference C2 surface) ---
    ca_use_ecdsa: bool = False
    mitm_all: bool = False
    no_mitm: bool = False
    mitm_hosts: list[str] = field(default_factory=lambda: list(DEFAULT_MITM_HOSTS))

    # --- listener ---
    # acceptor event loops on ONE port (SO_REUSEPORT).  >1 scales MITM
    # TLS crypto across cores (proxy/server.py ProxyFleet).
    loops: int = 1
    # 0.0.0.0:8080 mirrors the reference (start.go:206) for drop-in
    # parity, but note what it means: an UNAUTHENTICATED forward proxy
    # with a CONNECT tunnel reachable on every interface.  Set
    # DEMODEL_HOST=127.0.0.1 on any machine with an untrusted network
    # path; ProxyServer.start() logs a warning on wildcard binds.
    host: str = "0.0.0.0"
    port: int = 8080

    # --- cache ---
    cache_dir: str = ".cache"
    cache_max_bytes: int | None = None   # LRU-evict over this (None = off)

    # --- upstream TLS ---
    upstream_cafile: str | None = None  # extra CA for origin verification (tests)
    upstream_insecure: bool = False

    # --- GPU landing pipeline ---
    chunk_bytes: int = 32 << 20          # download/pipeline chunk size
    pinned_slabs: int = 4                # pinned host ring depth per pull
    gpu_verify: str = "chain"            # "chain" | "chunked" | "off"

    # --- proxy -> HBM pull-ahead ---
    # "off": only POST /__demodel/prefetch lands blobs; "auto": every
    # blob-looking response the proxy caches is landed into HBM
    gpu_prefetch: str = "off"
    gpu_cache_max_bytes: int | None = None  # HBM registry LRU budget

    @property
    def mitm_host_set(self) -> set[str]:
        return set(self.mitm_hosts)

    def should_mitm(self, hostport: str) -> bool:
        """CONNECT policy — reference C8 (start.go:183-196)."""
        if self.no_mitm:
            return False
        if self.mitm_all:
            return True
        return hostport in self.mitm_host_set


def load_config(**overrides) -> Config:
    hosts = _env_list("DEMODEL_PROXY_MITM_HOSTS")
    if not hosts:
        hosts = list(DEFAULT_MITM_HOSTS)
    hosts += _env_list("DEMODEL_PROXY_MITM_EXTRA_HOSTS")

    cfg = Config(
        ca_use_ecdsa=_env_bool("DEMODEL_PROXY_CA_USE_ECDSA"),
        mitm_all=_env_bool("DEMODEL_PROXY_MITM_ALL"),
        no_mitm=_env_bool("DEMODEL_PROXY_NO_MITM"),
        mitm_hosts=hosts,
        host=os.environ.get("DEMODEL_HOST", "0.0.0.0"),
        loops=_env_int("DEMODEL_LOOPS", 1),
        port=_env_int("DEMODEL_PORT", 8080),
        cache_dir=os.environ.get("DEMODEL_CACHE_DIR", ".cache"),
        cache_max_bytes=(
            int(float(os.environ["DEMODEL_CACHE_MAX_GB"]) * 1e9)
            if os.environ.get("DEMODEL_CACHE_MAX_GB") else None),
        upstream_cafile=os.environ.get("DEMODEL_UPSTREAM_CAFILE") or None,
        upstream_insecure=_env_bool("DEMODEL_UPSTREAM_INSECURE"),
        chunk_bytes=_env_int("DEMODEL_CHUNK_BYTES", 32 << 20),
        pinned_slabs=_env_int("DEMODEL_PINNED_SLABS", 4),
        gpu_verify=os.environ.get("DEMODEL_GPU_VERIFY", "chain"),
        gpu_prefetch=os.environ.get("DEMODEL_GPU_PREFETCH", "off"),
        gpu_cache_max_bytes=(
            int(float(os.environ["DEMODEL_GPU_CACHE_MAX_GB"]) * 1e9)
            if os.environ.get("DEMODEL_GPU_CACHE_MAX_GB") else None),
    )
    for k, v in overrides.items():
        setattr(cfg, k, v)
    return cfg
