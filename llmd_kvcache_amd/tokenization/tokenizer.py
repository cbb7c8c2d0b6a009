"""Tokenizer providers: local-dir, HuggingFace-hub, UDS sidecar, composite.

Parity with reference pkg/tokenization/tokenizer.go:
 - ``Tokenizer`` contract {encode, render_chat_template} (:42-47);
 - CachedTokenizer: LRU of 20 loaded tokenizers (:39) with singleflight
   dedup of concurrent loads (:350-371);
 - hf provider loads from the HuggingFace hub/cache (:439-441);
 - local provider loads tokenizer.json from a model->path map (:460-466)
   with auto-discovery of a directory, including HF-cache style
   ``models--org--name`` layouts (:169-263); env overrides
   LOCAL_TOKENIZER_DIR / LOCAL_TOKENIZER_FILENAME (:71-99);
 - CompositeTokenizer: ordered fallback chain local -> uds -> hf
   (:497-553, pool.go:103-127) accumulating errors.

The encoding engine is the HuggingFace ``tokenizers`` package - the same
Rust core the reference links as libtokenizers.a via CGo, here used through
its first-party Python binding inside the PyTorch-ROCm host process (no CGo
bridge needed).
"""

from __future__ import annotations

import os
import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

from ..utils.lru import LRUCache

DEFAULT_TOKENIZER_CACHE_SIZE = 20
DEFAULT_LOCAL_TOKENIZER_FILENAME = "tokenizer.json"

Offset = Tuple[int, int]


class TokenizationError(Exception):
    pass


class Tokenizer:
    """Contract: encode + render_chat_template + type."""

    def encode(
        self, prompt: str, model_name: str, add_special_tokens: bool = True
    ) -> Tuple[List[int], List[Offset]]:
        raise NotImplementedError

    def encode_batch(
        self, prompts: List[str], model_name: str,
        add_special_tokens: bool = True,
    ) -> List[Tuple[List[int], List[Offset]]]:
        """Batch encode; backends with a parallel core (HF tokenizers'
        Rust encode_batch releases the GIL and uses a rayon pool)
        override this - the default is a sequential loop.  The default
        flag value calls the 2-arg form so duck-typed tokenizers that
        omit add_special_tokens keep working."""
        if add_special_tokens:
            return [self.encode(p, model_name) for p in prompts]
        return [self.encode(p, model_name, add_special_tokens)
                for p in prompts]

    def render_chat_template(self, req) -> str:
        raise NotImplementedError

    @property
    def type(self) -> str:
        raise NotImplementedError


class _Call:
    __slots__ = ("ev", "status", "payload")

    def __init__(self) -> None:
        self.ev = threading.Event()
        self.status = "err"
        self.payload: object = TokenizationError("load failed")


class _SingleFlight:
    """Deduplicates concurrent loads of the same key (tokenizer.go:350-371).
    Each in-flight key owns a _Call object that every waiter reads from
    after its event fires - no shared-map pop races."""

    def __init__(self) -> None:
        self._mu = threading.Lock()
        self._calls: Dict[str, _Call] = {}

    def do(self, key: str, fn):
        with self._mu:
            call = self._calls.get(key)
            leader = call is None
            if leader:
                call = _Call()
                self._calls[key] = call
        if leader:
            try:
                call.payload = fn()
                call.status = "ok"
            except Exception as e:  # propagate to all waiters
                call.payload = e
                call.status = "err"
            finally:
                with self._mu:
                    self._calls.pop(key, None)
                call.ev.set()
        else:
            call.ev.wait()
        if call.status == "err":
            e = call.payload
            raise e if isinstance(e, Exception) else TokenizationError(str(e))
        return call.payload


def discover_local_tokenizers(
    directory: str, filename: str = DEFAULT_LOCAL_TOKENIZER_FILENAME
) -> Dict[str, str]:
    """Walk a directory for tokenizer files; understands both plain
    ``<dir>/<model>/tokenizer.json`` layouts and HF-cache
    ``models--org--name/snapshots/<rev>/tokenizer.json`` layouts
    (tokenizer.go:169-263)."""
    mapping: Dict[str, str] = {}
    if not directory or not os.path.isdir(directory):
        return mapping
    for root, _dirs, files in os.walk(directory):
        if filename not in files:
            continue
        path = os.path.join(root, filename)
        rel = os.path.relpath(root, directory)
        parts = rel.split(os.sep)
        model_name = None
        for part in parts:
            if part.startswith("models--"):
                model_name = part[len("models--") :].replace("--", "/")
                break
        if model_name is None:
            # plain layout: the relative dir (minus snapshots/rev) is the name
            clean = [p for p in parts if p not in (".", "snapshots")]
            model_name = "/".join(clean) if clean else os.path.basename(directory)
        # first found wins (stable across walks)
        mapping.setdefault(model_name, path)
    return mapping


@dataclass
class LocalTokenizerConfig:
    tokenizers_map: Dict[str, str] = field(default_factory=dict)
    auto_discover_dir: Optional[str] = None
    filename: str = DEFAULT_LOCAL_TOKENIZER_FILENAME

    @staticmethod
    def from_env() -> "LocalTokenizerConfig":
        return LocalTokenizerConfig(
            auto_discover_dir=os.environ.get("LOCAL_TOKENIZER_DIR"),
            filename=os.environ.get(
                "LOCAL_TOKENIZER_FILENAME", DEFAULT_LOCAL_TOKENIZER_FILENAME
            ),
        )

    def is_enabled(self) -> bool:
        return bool(self.tokenizers_map) or bool(self.auto_discover_dir)


@dataclass
class HFTokenizerConfig:
    tokenizers_cache_dir: Optional[str] = None
    token: Optional[str] = field(
        default_factory=lambda: os.environ.get("HF_TOKEN")
    )

    def is_enabled(self) -> bool:
        return True


class CachedTokenizer(Tokenizer):
    """LRU-cached tokenizer loader over a provider function."""

    def __init__(
        self,
        provider,
        type_name: str,
        cache_size: int = DEFAULT_TOKENIZER_CACHE_SIZE,
    ):
        self._provider = provider
        self._type = type_name
        self._cache = LRUCache(cache_size)
        self._flight = _SingleFlight()

    @property
    def type(self) -> str:
        return self._type

    def _get(self, model_name: str):
        tok, found = self._cache.get(model_name)
        if found:
            return tok
        tok = self._flight.do(model_name, lambda: self._provider(model_name))
        self._cache.add(model_name, tok)
        return tok

    def encode(
        self, prompt: str, model_name: str, add_special_tokens: bool = True
    ) -> Tuple[List[int], List[Offset]]:
        tok = self._get(model_name)
        enc = tok.encode(prompt, add_special_tokens=add_special_tokens)
        return list(enc.ids), [tuple(o) for o in enc.offsets]

    def encode_batch(
        self, prompts: List[str], model_name: str,
        add_special_tokens: bool = True,
    ) -> List[Tuple[List[int], List[Offset]]]:
        tok = self._get(model_name)
        encs = tok.encode_batch(prompts,
                                add_special_tokens=add_special_tokens)
        return [(list(e.ids), [tuple(o) for o in e.offsets])
                for e in encs]

    def render_chat_template(self, req) -> str:
        from ..preprocessing import chat_completions

        return chat_completions.render_chat_template(req)


def _load_tokenizers_lib():
    import tokenizers as hf_tokenizers

    return hf_tokenizers


def new_cached_local_tokenizer(cfg: LocalTokenizerConfig) -> CachedTokenizer:
    mapping = dict(cfg.tokenizers_map)
    if cfg.auto_discover_dir:
        discovered = discover_local_tokenizers(cfg.auto_discover_dir, cfg.filename)
        for k, v in discovered.items():
            mapping.setdefault(k, v)

    def provider(model_name: str):
        path = mapping.get(model_name)
        if path is None:
            # allow a direct filesystem path as the model name
            if os.path.isfile(model_name):
                path = model_name
            elif os.path.isdir(model_name) and os.path.isfile(
                os.path.join(model_name, cfg.filename)
            ):
                path = os.path.join(model_name, cfg.filename)
            else:
                raise TokenizationError(
                    f"no local tokenizer for model {model_name!r}"
                )
        lib = _load_tokenizers_lib()
        return lib.Tokenizer.from_file(path)

    return CachedTokenizer(provider, "local")


def new_cached_hf_tokenizer(cfg: Optional[HFTokenizerConfig] = None) -> CachedTokenizer:
    cfg = cfg or HFTokenizerConfig()

    def provider(model_name: str):
        lib = _load_tokenizers_lib()
        try:
            return lib.Tokenizer.from_pretrained(model_name)
        except Exception as e:
            raise TokenizationError(
                f"failed to load HF tokenizer {model_name!r}: {e}"
            ) from e

    return CachedTokenizer(provider, "hf")


class CompositeTokenizer(Tokenizer):
    """Ordered fallback chain; tries each backend, accumulates errors
    (tokenizer.go:497-553)."""

    def __init__(self, tokenizers_chain: Sequence[Tokenizer]):
        if not tokenizers_chain:
            raise ValueError("composite tokenizer needs at least one backend")
        self.chain = list(tokenizers_chain)

    @property
    def type(self) -> str:
        return "composite(" + ",".join(t.type for t in self.chain) + ")"

    def encode(
        self, prompt: str, model_name: str, add_special_tokens: bool = True
    ) -> Tuple[List[int], List[Offset]]:
        from ..metrics import collector

        errors = []
        for tok in self.chain:
            try:
                t0 = collector.monotonic()
                result = tok.encode(prompt, model_name, add_special_tokens)
                collector.observe_tokenization(tok.type, collector.monotonic() - t0, len(result[0]))
                return result
            except Exception as e:
                errors.append(f"{tok.type}: {e}")
        raise TokenizationError(
            f"all tokenizer backends failed for {model_name!r}: {'; '.join(errors)}"
        )

    def encode_batch(
        self, prompts: List[str], model_name: str,
        add_special_tokens: bool = True,
    ) -> List[Tuple[List[int], List[Offset]]]:
        from ..metrics import collector

        errors = []
        for tok in self.chain:
            try:
                t0 = collector.monotonic()
                result = tok.encode_batch(prompts, model_name,
                                          add_special_tokens)
                collector.observe_tokenization(
                    tok.type, collector.monotonic() - t0,
                    sum(len(r[0]) for r in result))
                return result
            except Exception as e:
                errors.append(f"{tok.type}: {e}")
        raise TokenizationError(
            f"all tokenizer backends failed for {model_name!r}: "
            f"{'; '.join(errors)}"
        )

    def render_chat_template(self, req) -> str:
        errors = []
        for tok in self.chain:
            try:
                return tok.render_chat_template(req)
            except Exception as e:
                errors.append(f"{tok.type}: {e}")
        raise TokenizationError(
            f"all tokenizer backends failed to render template: {'; '.join(errors)}"
        )
