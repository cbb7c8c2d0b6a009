from . import hashing  # noqa: F401
from .lru import LRUCache  # noqa: F401
