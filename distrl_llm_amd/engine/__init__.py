from .engine import Engine  # noqa: F401
from .kvcache import KVCachePool, BlockAllocator  # noqa: F401
