from .lru_store import LRUStoreConfig, LRUTokenStore  # noqa: F401
from .trie_store import TrieTokenStore  # noqa: F401
