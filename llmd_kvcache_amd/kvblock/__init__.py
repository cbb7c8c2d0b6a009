from .index import Index, IndexConfig, new_index  # noqa: F401
from .keys import DEFAULT_DEVICE_TIER, Key, PodEntry  # noqa: F401
from .token_processor import (  # noqa: F401
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from .in_memory import InMemoryIndex, InMemoryIndexConfig  # noqa: F401
