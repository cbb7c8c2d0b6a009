from .events import (  # noqa: F401
    AllBlocksCleared,
    BlockRemoved,
    BlockStored,
    EventBatch,
    decode_event_batch,
    get_hash_as_uint64,
)
from .pool import EventsConfig, EventsPool, Message  # noqa: F401
