"""CA lifecycle: create-or-load the demodel root CA, export it.

Parity with the reference's C5/C6 components (cmd/demodel/init.go:26-168,
cmd/demodel/export_ca.go:18-120), with the reference's bugs fixed:

* CA files live at ``$XDG_DATA_HOME/demodel/certificates/demodel-ca.crt``
  (0644) and ``demodel-ca.pem`` (0600) — same filenames/paths as
  init.go:32-38,135-143.
* ``install()`` installs the file that was actually written (the reference
  passes a never-written cwd path to truststore — init.go:145).
* ``export-ca --for python-ssl`` writes hash-named symlinks too, so OpenSSL
  capath lookup actually finds the CA (reference gap — export_ca.go:78-86).
"""

from __future__ import annotations

import json
import os
import subprocess
import sys

from . import config


class CA:
    def __init__(self, cert_pem: str, key_pem: str):
        self.cert_pem = cert_pem
        self.key_pem = key_pem


def cert_dir() -> str:
    return os.path.join(config.data_dir(), "certificates")


def cert_paths() -> tuple[str, str]:
    d = cert_dir()
    return os.path.join(d, "demodel-ca.crt"), os.path.join(d, "demodel-ca.pem")


def read_or_new_ca(use_ecdsa: bool = False) -> CA:
    """Load the CA PEM pair, or generate + persist a new one.

    Mirrors readOrNewCA (init.go:31-154): steady-state path reads the two
    files; otherwise a fresh CA is generated natively (libcrypto) and
    written crt=0644 / key=0600.
    """
    crt_path, key_path = cert_paths()
    if os.path.exists(crt_path) and os.path.exists(key_path):
        with open(crt_path) as f:
            cert_pem = f.read()
        with open(key_path) as f:
            key_pem = f.read()
        return CA(cert_pem, key_pem)
    if os.path.exists(crt_path) != os.path.exists(key_path):
        # Silently re-minting over a half-present pair would break every
        # client that trusts the surviving cert; make the operator decide.
        have, missing = ((crt_path, key_path)
                         if os.path.exists(crt_path)
                         else (key_path, crt_path))
        raise FileExistsError(
            f"CA pair is half-present: {have} exists but {missing} is "
            f"missing. Restore the missing file, or delete {have} to "
            f"mint a fresh CA (clients trusting the old one will break).")

    from . import _native

    cert_pem, key_pem = _native.ca_create(ecdsa=use_ecdsa)
    os.makedirs(cert_dir(), exist_ok=True)
    with open(crt_path, "w") as f:
        f.write(cert_pem)
    os.chmod(crt_path, 0o644)
    fd = os.open(key_path, os.O_WRONLY | os.O_CREAT | os.O_TRUNC, 0o600)
    with os.fdopen(fd, "w") as f:
        f.write(key_pem)
    return CA(cert_pem, key_pem)


def install_system(cert_path: str | None = None) -> bool:
    """Install the CA into the OS trust store (Linux).

    Equivalent of smallstep/truststore (init.go:145-148) for this target:
    copy into /usr/local/share/ca-certificates and run
    update-ca-certificates.  Returns False (without raising) when not root
    or the tool is missing.
    """
    crt_path, _ = cert_paths()
    cert_path = cert_path or crt_path
    dst_dir = "/usr/local/share/ca-certificates"
    try:
        os.makedirs(dst_dir, exist_ok=True)
        dst = os.path.join(dst_dir, "demodel-ca.crt")
        with open(cert_path) as f, open(dst, "w") as g:
            g.write(f.read())
        r = subprocess.run(
            ["update-ca-certificates"], capture_output=True, text=True
        )
        return r.returncode == 0
    except (OSError, FileNotFoundError):
        return False


def _openssl_subject_hash(cert_path: str) -> str | None:
    try:
        r = subprocess.run(
            ["openssl", "x509", "-subject_hash", "-noout", "-in", cert_path],
            capture_output=True, text=True,
        )
        if r.returncode == 0:
            return r.stdout.strip()
    except FileNotFoundError:
        pass
    return None


def export_ca(dest: str | None, python_exe: str = sys.executable) -> str:
    """Export the CA: to stdout (dest None) or into a client trust store.

    dest presets, parity with export_ca.go:50-105 plus the README-promised
    ``openssl`` preset the reference never implemented (README.md:50):

    * ``python-ssl``    — write into ssl.get_default_verify_paths().capath
      (plus <hash>.0 symlink) or append to cafile.
    * ``python-certifi``— append to certifi.where()'s cacert.pem.
    * ``openssl``       — install into the system store (install_system).
    """
    crt_path, _ = cert_paths()
    if not os.path.exists(crt_path):
        raise FileNotFoundError(
            f"CA certificate not found at {crt_path}; run `demodel init` first"
        )
    with open(crt_path) as f:
        pem = f.read()

    if dest is None:
        return pem

    if dest == "python-ssl":
        probe = (
            "import ssl, json; p = ssl.get_default_verify_paths(); "
            "print(json.dumps({'cafile': p.cafile, 'capath': p.capath, "
            "'openssl_cafile': p.openssl_cafile, "
            "'openssl_capath': p.openssl_capath}))"
        )
        r = subprocess.run([python_exe, "-c", probe],
                           capture_output=True, text=True, check=True)
        paths = json.loads(r.stdout)
        capath = paths.get("capath") or paths.get("openssl_capath")
        if capath and os.path.isdir(capath):
            dst = os.path.join(capath, "demodel-ca.crt")
            with open(dst, "w") as g:
                g.write(pem)
            h = _openssl_subject_hash(dst)
            if h:
                link = os.path.join(capath, f"{h}.0")
                if not os.path.exists(link):
                    os.symlink("demodel-ca.crt", link)
            return dst
        cafile = paths.get("cafile") or paths.get("openssl_cafile")
        if cafile and os.path.exists(cafile):
            _append_pem(cafile, pem)
            return cafile
        raise RuntimeError("no usable python ssl verify path found")

    if dest == "python-certifi":
        probe = "import certifi; print(certifi.where())"
        r = subprocess.run([python_exe, "-c", probe],
                           capture_output=True, text=True, check=True)
        cacert = r.stdout.strip()
        _append_pem(cacert, pem)
        return cacert

    if dest == "openssl":
        if not install_system(crt_path):
            raise RuntimeError("system trust-store install failed")
        return "/usr/local/share/ca-certificates/demodel-ca.crt"

    raise ValueError(f"unknown export destination: {dest!r}")


def _append_pem(path: str, pem: str) -> None:
    with open(path) as f:
        existing = f.read()
    if pem.strip() in existing:
        return  # idempotent — don't grow the bundle on every call
    with open(path, "a") as f:
        if not existing.endswith("\n"):
            f.write("\n")
        f.write(pem)
