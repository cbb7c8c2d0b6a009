"""vLLM KVEvents wire format (msgpack tagged unions).

Bit-compatible with the reference schema (pkg/kvcache/kvevents/events.go):
 - ``EventBatch`` is a msgpack *array* ``[ts, [raw events...],
   data_parallel_rank?]`` (events.go:38-43);
 - each event is a tagged-union msgpack array whose first element is the tag
   string (events.go:21-28):
     BlockStored  -> ["BlockStored", block_hashes, parent_block_hash,
                      token_ids, block_size, lora_id, medium]
     BlockRemoved -> ["BlockRemoved", block_hashes, medium]
     AllBlocksCleared -> ["AllBlocksCleared"]
 - block hashes may be uint64/int64 ints or byte strings (new vLLM format);
   byte hashes are coerced by taking the last 8 bytes big-endian
   (pool.go:343-367).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, List, Optional

import msgpack

BLOCK_STORED_TAG = "BlockStored"
BLOCK_REMOVED_TAG = "BlockRemoved"
ALL_BLOCKS_CLEARED_TAG = "AllBlocksCleared"


@dataclass
class BlockStored:
    block_hashes: List[Any]
    parent_block_hash: Optional[Any]
    token_ids: List[int]
    block_size: int
    lora_id: Optional[int] = None
    medium: Optional[str] = None

    def to_tagged_union(self) -> List[Any]:
        return [
            BLOCK_STORED_TAG,
            self.block_hashes,
            self.parent_block_hash,
            self.token_ids,
            self.block_size,
            self.lora_id,
            self.medium,
        ]


@dataclass
class BlockRemoved:
    block_hashes: List[Any]
    medium: Optional[str] = None

    def to_tagged_union(self) -> List[Any]:
        return [BLOCK_REMOVED_TAG, self.block_hashes, self.medium]


@dataclass
class AllBlocksCleared:
    def to_tagged_union(self) -> List[Any]:
        return [ALL_BLOCKS_CLEARED_TAG]


@dataclass
class EventBatch:
    ts: float
    events: List[Any]  # event dataclasses (encode) or raw parts (decode)
    data_parallel_rank: Optional[int] = None

    def encode(self) -> bytes:
        """Marshal to the vLLM wire format: batch array of raw-encoded
        tagged-union events."""
        raw_events = [
            msgpack.packb(
                e.to_tagged_union() if hasattr(e, "to_tagged_union") else e,
                use_bin_type=True,
            )
            for e in self.events
        ]
        arr: List[Any] = [self.ts, raw_events]
        if self.data_parallel_rank is not None:
            arr.append(self.data_parallel_rank)
        return msgpack.packb(arr, use_bin_type=True)


class DecodeError(Exception):
    pass


def _decode_event(parts: List[Any]):
    """Decode one tagged-union array into an event dataclass.
    Unknown tags return None (skipped, pool.go:232-234)."""
    if not parts:
        raise DecodeError("malformed tagged union: no tag element")
    tag = parts[0]
    if isinstance(tag, bytes):
        tag = tag.decode("utf-8", "replace")
    body = parts[1:]
    if tag == BLOCK_STORED_TAG:
        if len(body) < 4:
            raise DecodeError(f"BlockStored needs >=4 fields, got {len(body)}")
        return BlockStored(
            block_hashes=list(body[0]) if body[0] is not None else [],
            parent_block_hash=body[1],
            token_ids=[int(t) for t in (body[2] or [])],
            block_size=int(body[3]),
            lora_id=(int(body[4]) if len(body) > 4 and body[4] is not None else None),
            medium=(
                body[5].decode() if len(body) > 5 and isinstance(body[5], bytes)
                else body[5] if len(body) > 5 else None
            ),
        )
    if tag == BLOCK_REMOVED_TAG:
        if len(body) < 1:
            raise DecodeError("BlockRemoved needs >=1 field")
        return BlockRemoved(
            block_hashes=list(body[0]) if body[0] is not None else [],
            medium=(
                body[1].decode() if len(body) > 1 and isinstance(body[1], bytes)
                else body[1] if len(body) > 1 else None
            ),
        )
    if tag == ALL_BLOCKS_CLEARED_TAG:
        return AllBlocksCleared()
    return None  # unknown tag -> skip


def decode_event_batch(payload: bytes) -> EventBatch:
    """Unmarshal a batch; individual malformed events are skipped (poison
    pills must not kill the stream, pool.go:182-187)."""
    try:
        arr = msgpack.unpackb(payload, raw=False, strict_map_key=False)
    except Exception as e:
        raise DecodeError(f"failed to unmarshal event batch: {e}") from e
    if not isinstance(arr, (list, tuple)) or len(arr) < 2:
        raise DecodeError("event batch is not a >=2 element array")

    try:
        ts = float(arr[0])
    except (TypeError, ValueError) as e:
        raise DecodeError(f"bad batch timestamp: {e}") from e
    if not isinstance(arr[1], (list, tuple)):
        raise DecodeError("batch events field is not an array")
    dp_rank = None
    if len(arr) > 2 and arr[2] is not None:
        try:
            dp_rank = int(arr[2])
        except (TypeError, ValueError):
            dp_rank = None

    events = []
    for raw in arr[1]:
        try:
            if isinstance(raw, (bytes, bytearray)):
                parts = msgpack.unpackb(bytes(raw), raw=False, strict_map_key=False)
            else:
                parts = raw  # already-decoded nested array
            ev = _decode_event(list(parts))
            if ev is not None:
                events.append(ev)
        except Exception:
            continue  # skip malformed event, keep the batch
    return EventBatch(ts=ts, events=events, data_parallel_rank=dp_rank)


def get_hash_as_uint64(h: Any) -> int:
    """Hash coercion parity with pool.go:343-367."""
    if isinstance(h, bool):
        raise DecodeError(f"unsupported hash type: {type(h)}")
    if isinstance(h, int):
        return h & 0xFFFFFFFFFFFFFFFF
    if isinstance(h, (bytes, bytearray)):
        b = bytes(h)
        if len(b) == 0:
            raise DecodeError("hash byte slice is empty")
        if len(b) >= 8:
            return int.from_bytes(b[-8:], "big")
        return int.from_bytes(b, "big")
    if isinstance(h, str):
        # msgpack raw=False may give str for bin-less encoders; treat as bytes
        return get_hash_as_uint64(h.encode("utf-8", "surrogateescape"))
    raise DecodeError(f"unsupported hash type: {type(h)}")
