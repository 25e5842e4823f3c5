"""Per-host leaf certificate store for TLS MITM.

Parity with the reference's C7 CertStorage (cmd/demodel/start.go:27-165):
an RWMutex-guarded hostname -> certificate memo; on miss, mint a leaf
signed by the demodel CA.  Here the mint happens in native code
(libcrypto, demodel_amd/csrc/certs.cpp) and the memo additionally caches
the ready-to-use server ``ssl.SSLContext``.
"""

from __future__ import annotations

import os
import ssl
import tempfile
import threading

from .ca import CA


class LeafStore:
    def __init__(self, ca: CA, use_ecdsa: bool = True):
        self._ca = ca
        # ECDSA-P256 leaves by default: ~1 ms mint vs ~1-2 s for RSA-4096,
        # and every modern client accepts them.  (The reference mints leaves
        # with the same algorithm as the CA — start.go:51-55; independent
        # choice here is deliberate.)
        self._use_ecdsa = use_ecdsa
        self._lock = threading.Lock()
        self._contexts: dict[str, ssl.SSLContext] = {}
        self._pems: dict[str, tuple[str, str]] = {}

    def pem_pair(self, hostname: str) -> tuple[str, str]:
        with self._lock:
            hit = self._pems.get(hostname)
        if hit:
            return hit
        from . import _native

        pair = _native.leaf_create(
            self._ca.cert_pem, self._ca.key_pem, hostname,
            ecdsa=self._use_ecdsa,
        )
        with self._lock:
            self._pems.setdefault(hostname, pair)
            return self._pems[hostname]

    def server_context(self, hostname: str) -> ssl.SSLContext:
        """SSLContext presenting a leaf for `hostname`, minting on miss."""
        with self._lock:
            ctx = self._contexts.get(hostname)
        if ctx is not None:
            return ctx
        cert_pem, key_pem = self.pem_pair(hostname)
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        # ssl can only load cert chains from files; use a private tmpdir.
        with tempfile.TemporaryDirectory(prefix="demodel-leaf-") as d:
            cp = os.path.join(d, "leaf.crt")
            kp = os.path.join(d, "leaf.key")
            with open(cp, "w") as f:
                f.write(cert_pem + self._ca.cert_pem)  # serve the chain
            fd = os.open(kp, os.O_WRONLY | os.O_CREAT, 0o600)
            with os.fdopen(fd, "w") as f:
                f.write(key_pem)
            ctx.load_cert_chain(cp, kp)
        with self._lock:
            self._contexts.setdefault(hostname, ctx)
            return self._contexts[hostname]
