from sentio_amd.caching.memory import MemoryCache  # noqa: F401
from sentio_amd.caching.manager import CacheManager, get_cache_manager  # noqa: F401
from sentio_amd.caching.disk import DiskCache  # noqa: F401
from sentio_amd.caching.strategies import (  # noqa: F401
    AdaptiveStrategy,
    CacheStrategy,
    LRUStrategy,
    SizeBasedStrategy,
    TTLStrategy,
)
