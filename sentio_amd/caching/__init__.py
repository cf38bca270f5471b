from sentio_amd.caching.memory import MemoryCache  # noqa: F401
from sentio_amd.caching.manager import CacheManager, get_cache_manager  # noqa: F401
