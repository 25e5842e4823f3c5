from .server import ProxyServer  # noqa: F401
