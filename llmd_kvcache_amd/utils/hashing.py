"""Hashing primitives for vLLM-compatible KV-block content addressing.

The reference (llm-d-kv-cache-manager) computes each block key as the FNV-64a
hash of the canonical-CBOR encoding of ``[parent_hash, token_chunk, None]``
(reference: pkg/kvcache/kvblock/token_processor.go:94-112), with the chain
root being ``FNV-64a(hash_seed_bytes)`` (token_processor.go:81-90, aligned
with vLLM's PYTHONHASHSEED-derived NONE_HASH).

This module provides the pure-Python implementation used as the golden
reference for the C++/HIP fast paths (ops/csrc/kvidx_common.h).  The CBOR
subset implemented here is exactly what the fixed payload shape needs:
a 3-element array of [unsigned int, array of unsigned ints, null], encoded
with canonical (shortest-form) integer encoding, matching Go's
fxamacker/cbor CanonicalEncOptions for this payload.
"""

from __future__ import annotations

FNV64_OFFSET_BASIS = 0xCBF29CE484222325
FNV64_PRIME = 0x100000001B3
_MASK64 = 0xFFFFFFFFFFFFFFFF


def fnv1a_64(data: bytes, h: int = FNV64_OFFSET_BASIS) -> int:
    """FNV-1a 64-bit hash (Go hash/fnv New64a semantics)."""
    for b in data:
        h ^= b
        h = (h * FNV64_PRIME) & _MASK64
    return h


def fnv1a_32(data: bytes) -> int:
    """FNV-1a 32-bit hash (Go hash/fnv New32a) - used for event-pool sharding
    (reference: pkg/kvcache/kvevents/pool.go:132-144)."""
    h = 0x811C9DC5
    for b in data:
        h ^= b
        h = (h * 0x01000193) & 0xFFFFFFFF
    return h


def cbor_encode_uint(n: int, major: int = 0) -> bytes:
    """Canonical (shortest-form) CBOR unsigned integer with given major type."""
    mt = major << 5
    if n < 24:
        return bytes([mt | n])
    if n <= 0xFF:
        return bytes([mt | 24, n])
    if n <= 0xFFFF:
        return bytes([mt | 25]) + n.to_bytes(2, "big")
    if n <= 0xFFFFFFFF:
        return bytes([mt | 26]) + n.to_bytes(4, "big")
    return bytes([mt | 27]) + n.to_bytes(8, "big")


def cbor_chunk_payload(parent_hash: int, tokens, extra_null: bool = True) -> bytes:
    """Canonical CBOR of ``[parent, tokens, null]`` - the exact payload the
    reference marshals per chunk (token_processor.go:96-108)."""
    out = bytearray()
    out.append(0x83)  # array(3)
    out += cbor_encode_uint(parent_hash & _MASK64)
    # tokens: array header then each token as canonical unsigned
    n = len(tokens)
    out += cbor_encode_uint(n, major=4)
    for t in tokens:
        out += cbor_encode_uint(int(t) & _MASK64)
    out.append(0xF6)  # null
    return bytes(out)


def chunk_hash(parent_hash: int, tokens) -> int:
    """One link of the block-hash chain: FNV-64a(CBOR([parent, chunk, null]))."""
    return fnv1a_64(cbor_chunk_payload(parent_hash, tokens))


def init_hash(seed: str = "") -> int:
    """Root parent hash: FNV-64a of the seed string bytes
    (token_processor.go:81-90). Empty seed -> FNV offset basis."""
    return fnv1a_64(seed.encode("utf-8"))


# -- pluggable hash algorithms (vLLM changed SHA-256 ->
# FNV-64a once already; keep the chain function swappable) --------------

def sha256_cbor_64(parent_hash: int, tokens) -> int:
    """Alternative chain link: top 8 bytes (big-endian) of SHA-256 over
    the same canonical-CBOR payload - the shape of vLLM's historical
    sha256 content addressing. NOTE: gated to host paths until a vLLM
    golden fixture pins the exact byte convention (ROADMAP #5); the
    HIP/C++ fast paths implement fnv-64a only."""
    import hashlib

    digest = hashlib.sha256(cbor_chunk_payload(parent_hash, tokens)).digest()
    return int.from_bytes(digest[:8], "big")


def sha256_init_hash(seed: str = "") -> int:
    import hashlib

    return int.from_bytes(
        hashlib.sha256(seed.encode("utf-8")).digest()[:8], "big")


CHAIN_ALGOS = {
    "fnv-64a": (chunk_hash, init_hash),
    "sha256-cbor-64": (sha256_cbor_64, sha256_init_hash),
}
