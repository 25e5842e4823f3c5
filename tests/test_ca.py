"""Trust layer tests: CA lifecycle + native leaf minting (reference
cmd/demodel/init.go, cmd/demodel/start.go:27-165)."""

import os
import ssl
import subprocess

import pytest

from demodel_amd import ca as ca_mod
from demodel_amd.certs import LeafStore


@pytest.fixture()
def xdg_tmp(tmp_path, monkeypatch):
    monkeypatch.setenv("XDG_DATA_HOME", str(tmp_path / "data"))
    return tmp_path


def test_ca_create_and_reload(xdg_tmp):
    ca1 = ca_mod.read_or_new_ca(use_ecdsa=True)
    assert "BEGIN CERTIFICATE" in ca1.cert_pem
    assert "PRIVATE KEY" in ca1.key_pem
    crt, key = ca_mod.cert_paths()
    assert os.path.exists(crt) and os.path.exists(key)
    # key file must be private (init.go:139-143: pem written 0600)
    assert oct(os.stat(key).st_mode & 0o777) == "0o600"
    # steady-state: second call loads the same CA (init.go:40-62)
    ca2 = ca_mod.read_or_new_ca(use_ecdsa=True)
    assert ca2.cert_pem == ca1.cert_pem


def test_leaf_chain_verifies(xdg_tmp, tmp_path):
    ca = ca_mod.read_or_new_ca(use_ecdsa=True)
    leafs = LeafStore(ca)
    cert_pem, key_pem = leafs.pem_pair("example.test")
    cafile = tmp_path / "ca.crt"
    leaffile = tmp_path / "leaf.crt"
    cafile.write_text(ca.cert_pem)
    leaffile.write_text(cert_pem)
    r = subprocess.run(
        ["openssl", "verify", "-CAfile", str(cafile), str(leaffile)],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr


def test_leaf_memoized(xdg_tmp):
    ca = ca_mod.read_or_new_ca(use_ecdsa=True)
    leafs = LeafStore(ca)
    a = leafs.pem_pair("h1.test")
    b = leafs.pem_pair("h1.test")
    assert a is b or a == b
    ctx = leafs.server_context("h1.test")
    assert isinstance(ctx, ssl.SSLContext)
    assert leafs.server_context("h1.test") is ctx


def test_export_ca_stdout_and_certifi(xdg_tmp, tmp_path, monkeypatch):
    ca_mod.read_or_new_ca(use_ecdsa=True)
    pem = ca_mod.export_ca(None)
    assert "BEGIN CERTIFICATE" in pem

    # python-certifi preset appends (and is idempotent)
    fake_cacert = tmp_path / "cacert.pem"
    fake_cacert.write_text("# existing bundle\n")
    fake_python = tmp_path / "fakepython"
    fake_python.write_text(
        "#!/bin/sh\necho %s\n" % fake_cacert)
    fake_python.chmod(0o755)
    dest = ca_mod.export_ca("python-certifi", python_exe=str(fake_python))
    assert dest == str(fake_cacert)
    content = fake_cacert.read_text()
    assert pem.strip() in content
    ca_mod.export_ca("python-certifi", python_exe=str(fake_python))
    assert fake_cacert.read_text() == content  # no duplicate append


def test_export_unknown_dest_raises(xdg_tmp):
    ca_mod.read_or_new_ca(use_ecdsa=True)
    with pytest.raises(ValueError):
        ca_mod.export_ca("netscape")


def test_export_python_ssl_capath(xdg_tmp, tmp_path):
    """--for python-ssl drops the CA into capath WITH a hash symlink
    (the reference forgot the c_rehash link — SURVEY §2.1)."""
    import json as _json

    ca_mod.read_or_new_ca(use_ecdsa=True)
    capath = tmp_path / "capath"
    capath.mkdir()
    fake_python = tmp_path / "fakepython"
    fake_python.write_text(
        "#!/bin/sh\necho '%s'\n" % _json.dumps(
            {"cafile": None, "capath": str(capath),
             "openssl_cafile": None, "openssl_capath": None}))
    fake_python.chmod(0o755)
    dest = ca_mod.export_ca("python-ssl", python_exe=str(fake_python))
    assert dest == str(capath / "demodel-ca.crt")
    assert (capath / "demodel-ca.crt").exists()
    links = [p for p in capath.iterdir() if p.name.endswith(".0")]
    assert len(links) == 1 and links[0].is_symlink()
