"""llmd_kvcache_amd - MI355X-native KV-cache locality indexer.

A from-scratch rebuild of llm-d/llm-d-kv-cache-manager for AMD MI355X:
the block->pod index lives in HBM3E as an open-addressing hash table probed
by wave-cooperative HIP (gfx950) kernels; scoring is fused into the lookup
kernel; multi-GPU sharding merges per-shard hit masks over RCCL/xGMI.
Wire formats (vLLM KVEvents msgpack over ZMQ, indexer.proto gRPC) are
bit-compatible with the reference.
"""

__version__ = "0.1.0"

from .indexer import Config, Indexer  # noqa: F401
from .scorer import (  # noqa: F401
    KVBlockScorerConfig,
    KVCacheBackendConfig,
    LongestPrefixScorer,
)
