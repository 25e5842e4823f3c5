from .store import CacheStore, CachedResponse, cache_key  # noqa: F401
