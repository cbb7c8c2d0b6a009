"""JSON config loading for the whole indexer stack.

Parity with the reference's pure-JSON nested config structs with
Default*Config constructors at every level (pkg/kvcache/indexer.go:36-60,
docs/configuration.md): every component config here is a dataclass with
defaults; this module adds dict/JSON (de)serialization so services can be
configured from a file or an env-embedded YAML/JSON blob the way the EPP
embeds the reference's config.
"""

from __future__ import annotations

import dataclasses
import json
from typing import Any, Dict, Optional

from .indexer import Config
from .kvblock.index import IndexConfig
from .scorer import KVBlockScorerConfig, KVCacheBackendConfig


def _to_dict(obj: Any) -> Any:
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        out = {}
        for f in dataclasses.fields(obj):
            if f.name.startswith("_"):
                continue
            out[f.name] = _to_dict(getattr(obj, f.name))
        return out
    if isinstance(obj, (list, tuple)):
        return [_to_dict(x) for x in obj]
    if isinstance(obj, dict):
        return {k: _to_dict(v) for k, v in obj.items()}
    return obj


def config_to_dict(cfg: Config) -> Dict[str, Any]:
    return _to_dict(cfg)


def config_to_json(cfg: Config, **kw) -> str:
    return json.dumps(config_to_dict(cfg), **kw)


def _dataclass_of(field_type: Any) -> Any:
    """Resolve Optional[SomeDataclass] / SomeDataclass annotations."""
    import typing

    for t in typing.get_args(field_type) or (field_type,):
        if dataclasses.is_dataclass(t):
            return t
    return None


def _apply(obj: Any, data: Dict[str, Any]) -> Any:
    import typing

    try:
        hints = typing.get_type_hints(type(obj))
    except Exception:
        hints = {}
    for f in dataclasses.fields(obj):
        if f.name not in data or f.name.startswith("_"):
            continue
        val = data[f.name]
        cur = getattr(obj, f.name)
        if dataclasses.is_dataclass(cur) and isinstance(val, dict):
            _apply(cur, val)
        elif cur is None and isinstance(val, dict):
            # e.g. tokenizers_pool.uds defaults to None: construct the
            # annotated dataclass and overlay into it
            dc = _dataclass_of(hints.get(f.name, f.type))
            if dc is not None:
                setattr(obj, f.name, _apply(dc(), val))
            else:
                setattr(obj, f.name, val)
        else:
            setattr(obj, f.name, val)
    return obj


def config_from_dict(data: Dict[str, Any],
                     base: Optional[Config] = None) -> Config:
    """Overlays a (possibly partial) dict onto defaults - unknown keys are
    ignored, absent keys keep their defaults (reference semantics: first
    non-nil backend config wins, defaults everywhere else)."""
    cfg = base or Config()
    _apply(cfg, {k: v for k, v in data.items()
                 if k not in ("kv_block_index", "backend_configs")})

    if "backend_configs" in data:
        cfg.backend_configs = [
            KVCacheBackendConfig(name=b["name"], weight=float(b["weight"]))
            for b in data["backend_configs"]
        ]
        cfg.scorer = KVBlockScorerConfig(backend_configs=cfg.backend_configs)

    if "kv_block_index" in data:
        cfg.kv_block_index = index_config_from_dict(data["kv_block_index"])
    return cfg


def index_config_from_dict(data: Dict[str, Any]) -> IndexConfig:
    cfg = IndexConfig(
        enable_metrics=bool(data.get("enable_metrics", False)),
        metrics_logging_interval_s=float(
            data.get("metrics_logging_interval_s", 0.0)
        ),
    )
    if "in_memory" in data:
        from .kvblock.in_memory import InMemoryIndexConfig

        cfg.in_memory = _apply(InMemoryIndexConfig(), data["in_memory"] or {})
    if "native" in data:
        from .kvblock.gpu_index import TableIndexConfig

        cfg.native = _apply(TableIndexConfig(), data["native"] or {})
    if "gpu" in data:
        from .kvblock.gpu_index import GpuIndexConfig

        cfg.gpu = _apply(GpuIndexConfig(), data["gpu"] or {})
    if "tiered" in data:
        from .kvblock.gpu_index import GpuIndexConfig, TableIndexConfig
        from .kvblock.tiered import TieredIndexConfig

        sub = data["tiered"] or {}
        hot = _apply(GpuIndexConfig(), sub.get("hot") or {})
        tiered = TieredIndexConfig(hot=hot)
        if sub.get("cold"):
            tiered.cold = _apply(tiered.cold, sub["cold"])
            tiered.cold.device = "cpu"
        cfg.tiered = tiered
    if "cost_aware" in data:
        from .kvblock.cost_aware import CostAwareMemoryIndexConfig

        cfg.cost_aware = _apply(
            CostAwareMemoryIndexConfig(), data["cost_aware"] or {}
        )
    if "redis" in data:
        from .kvblock.redis_index import RedisIndexConfig

        cfg.redis = _apply(RedisIndexConfig(), data["redis"] or {})
    if "valkey" in data:
        from .kvblock.redis_index import RedisIndexConfig

        cfg.valkey = _apply(RedisIndexConfig(), data["valkey"] or {})
    if not any((cfg.in_memory, cfg.native, cfg.gpu, cfg.tiered,
                cfg.cost_aware, cfg.redis, cfg.valkey)):
        from .kvblock.in_memory import InMemoryIndexConfig

        cfg.in_memory = InMemoryIndexConfig()
    return cfg


def config_from_json(blob: str, base: Optional[Config] = None) -> Config:
    return config_from_dict(json.loads(blob), base)
