"""Table-backed index: HBM3E-resident on GPU, same layout on CPU.

This is the MI355X-native replacement for the reference's Go-map/LRU
in-memory index (pkg/kvcache/kvblock/in_memory.go): an open-addressing
hash table (ops/csrc/kvidx_common.h) probed by wave-cooperative HIP
kernels on gfx950, or by the bit-identical C++ reference on CPU.

Two concrete backends:
 - ``NativeIndex``  - the table on CPU tensors (fast CPU mode; also the
   golden reference for GPU differential tests);
 - ``GpuIndex``     - the table in HBM on ``cuda:N``; inserts/evicts/
   lookups are batched kernel launches; the read path can bypass the
   generic ``lookup`` entirely via ``fused_scores`` (one kernel does
   probe + longest-prefix scoring).

String identities (pod, model, tier) are interned host-side into dense
ids (Registry); the table stores only ids.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Set, Tuple

import torch

from ..scorer import default_kv_cache_backend_configs
from .index import Index
from .keys import Key, PodEntry

MAX_TIERS = 4
DEFAULT_CAPACITY = 1 << 21  # 2M slots; ~1M keys at 0.5 load factor
DEFAULT_PODS_PER_KEY = 10  # in_memory.go:34
MAX_PODS = 4096


def _to_i64(h: int) -> int:
    """uint64 -> int64 bit reinterpretation for torch tensors."""
    return h - (1 << 64) if h >= (1 << 63) else h


def _to_u64(v: int) -> int:
    return v & 0xFFFFFFFFFFFFFFFF


class Registry:
    """Interns pod/model/tier strings to dense ids (thread-safe)."""

    def __init__(self, tier_names: Optional[Sequence[str]] = None):
        self._lock = threading.Lock()
        self.pod_to_id: Dict[str, int] = {}
        self.id_to_pod: List[str] = []
        self.model_to_id: Dict[str, int] = {}
        self.id_to_model: List[str] = []
        tiers = list(tier_names or [b.name for b in default_kv_cache_backend_configs()])
        self.tier_to_id: Dict[str, int] = {}
        self.id_to_tier: List[str] = []
        for t in tiers:
            self._intern_tier(t)

    def _intern_tier(self, tier: str) -> int:
        tid = self.tier_to_id.get(tier)
        if tid is None:
            if len(self.id_to_tier) >= MAX_TIERS:
                # Silently sharing a slot would mis-weight scores for
                # every tier folded onto it; fail loudly instead (event
                # paths catch this and drop the offending event).
                raise ValueError(
                    f"tier registry full ({MAX_TIERS}): cannot intern "
                    f"{tier!r}; registered tiers: {self.id_to_tier}"
                )
            tid = len(self.id_to_tier)
            self.tier_to_id[tier] = tid
            self.id_to_tier.append(tier)
        return tid

    def pod_id(self, pod: str) -> int:
        with self._lock:
            pid = self.pod_to_id.get(pod)
            if pid is None:
                if len(self.id_to_pod) >= MAX_PODS:
                    raise ValueError(f"pod registry full ({MAX_PODS})")
                pid = len(self.id_to_pod)
                self.pod_to_id[pod] = pid
                self.id_to_pod.append(pod)
            return pid

    def model_id(self, model: str) -> int:
        with self._lock:
            mid = self.model_to_id.get(model)
            if mid is None:
                if len(self.id_to_model) >= 0xFFFF:
                    raise ValueError("model registry full")
                mid = len(self.id_to_model)
                self.model_to_id[model] = mid
                self.id_to_model.append(model)
            return mid

    def tier_id(self, tier: str) -> int:
        with self._lock:
            return self._intern_tier(tier)

    @property
    def num_pods(self) -> int:
        with self._lock:
            return len(self.id_to_pod)


@dataclass
class TableIndexConfig:
    capacity: int = DEFAULT_CAPACITY
    pods_per_key: int = DEFAULT_PODS_PER_KEY
    device: str = "cpu"
    # sharding: the main table stores only keys with
    # chunk_hash % num_shards == shard_id; the engine map is replicated
    # (parallel/sharded.py merges per-shard masks over RCCL)
    shard_id: int = 0
    num_shards: int = 1
    tier_names: List[str] = field(
        default_factory=lambda: [b.name for b in default_kv_cache_backend_configs()]
    )


@dataclass
class GpuIndexConfig(TableIndexConfig):
    device: str = "cuda:0"


class KvTable:
    """Tensor bundle + op dispatch for one table instance."""

    def __init__(self, cfg: TableIndexConfig):
        cap = cfg.capacity
        if cap & (cap - 1):
            raise ValueError("capacity must be a power of two")
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        self.is_cuda = self.device.type == "cuda"
        from ..ops import cpu_ext

        self.ops = cpu_ext.require()
        if self.is_cuda and not self.ops.HAS_HIP:
            raise RuntimeError(
                "native extension built without HIP support but a GPU table "
                "was requested - rebuild with ROCm torch"
            )
        opts = dict(device=self.device)
        self.keys = torch.zeros(cap, dtype=torch.int64, **opts)
        self.meta = torch.zeros(cap, dtype=torch.int32, **opts)
        self.stamp = torch.zeros(cap, dtype=torch.int32, **opts)
        self.pods = torch.zeros(cap * cfg.pods_per_key, dtype=torch.int32, **opts)
        self.e_keys = torch.zeros(cap, dtype=torch.int64, **opts)
        self.e_meta = torch.zeros(cap, dtype=torch.int32, **opts)
        self.e_vals = torch.zeros(cap, dtype=torch.int64, **opts)
        self._epoch = 0
        self._epoch_lock = threading.Lock()

    def next_epoch(self) -> int:
        with self._epoch_lock:
            self._epoch = (self._epoch + 1) & 0x7FFFFFFF
            return self._epoch

    def _t(self):
        return (
            self.keys,
            self.meta,
            self.stamp,
            self.pods,
            self.e_keys,
            self.e_meta,
            self.e_vals,
            self.cfg.pods_per_key,
        )

    # -- raw ops (tensors in the table's device) -----------------------
    def insert(self, engine_hashes, request_hashes, model_id, pod_entries,
               emap_write: bool = True):
        """emap_write=False skips the engine-map write - the tier-
        promotion path reuses request hashes as engine hashes and a
        self-mapping there would corrupt get_request_key()."""
        fn = self.ops.gpu_insert if self.is_cuda else self.ops.cpu_insert
        fn(*self._t(), engine_hashes, request_hashes, model_id, pod_entries,
           self.next_epoch(), self.cfg.shard_id, self.cfg.num_shards,
           1 if emap_write else 0)

    def evict(self, engine_hashes, model_id, pod_entries):
        fn = self.ops.gpu_evict if self.is_cuda else self.ops.cpu_evict
        fn(*self._t(), engine_hashes, model_id, pod_entries)

    def lookup(self, request_hashes, model_id, filter_words, num_pods,
               sharded=True, n_tiers=MAX_TIERS):
        fn = self.ops.gpu_lookup if self.is_cuda else self.ops.cpu_lookup
        cfg = self.cfg
        shard_id = cfg.shard_id if sharded else 0
        num_shards = cfg.num_shards if sharded else 1
        return fn(*self._t(), request_hashes, model_id, filter_words,
                  num_pods, self.next_epoch(), shard_id, num_shards,
                  n_tiers)

    def fused_score(self, hashes, counts_or_offsets, model_id, filter_words,
                    weights, num_pods, max_k=None, n_tiers=MAX_TIERS):
        if self.is_cuda:
            # n_tiers (tiers actually registered) sizes the kernel's LDS:
            # 1-2 real tiers instead of MAX_TIERS=4 keeps 256-pod fleets
            # at 16-32 KB/WG and full occupancy.
            return self.ops.gpu_fused_score(
                *self._t(), hashes, counts_or_offsets, model_id, filter_words,
                weights, num_pods, self.next_epoch(),
                max_k if max_k is not None else 512, n_tiers)
        return self.ops.cpu_fused_score(
            *self._t(), hashes, counts_or_offsets, model_id, filter_words,
            weights, num_pods, self.next_epoch())

    def compact(self, new_capacity: Optional[int] = None) -> None:
        """Rehash every live entry into a fresh tensor bundle, dropping
        tombstones and dead probe windows (ROADMAP #9 - long-running
        tables). Stamps (approximate-LRU) and pod rows carry over;
        new_capacity (power of two) allows growing/shrinking. Costs 2x
        table memory transiently; callers serialize against writes."""
        import copy

        cfg = copy.copy(self.cfg)
        if new_capacity is not None:
            if new_capacity & (new_capacity - 1):
                raise ValueError("capacity must be a power of two")
            cfg.capacity = new_capacity
        fresh = KvTable(cfg)
        fn = self.ops.gpu_compact if self.is_cuda else self.ops.cpu_compact
        fn(*self._t(), *fresh._t()[:-1])
        self.cfg = cfg
        for name in ("keys", "meta", "stamp", "pods",
                     "e_keys", "e_meta", "e_vals"):
            setattr(self, name, getattr(fresh, name))

    def get_request_keys(self, engine_hashes, model_id):
        fn = (self.ops.gpu_get_request_keys if self.is_cuda
              else self.ops.cpu_get_request_keys)
        return fn(*self._t(), engine_hashes, model_id)

    # -- checkpoint/restore --------------------------------------------
    # The reference index is ephemeral by design (rebuilds from the event
    # stream; durability delegated to Redis - docs/architecture.md:129).
    # The table layout makes snapshots nearly free on MI355X, so we offer
    # them as a native extra: one D2H copy of the tensor bundle.
    def state_dict(self):
        return {
            "keys": self.keys.cpu(),
            "meta": self.meta.cpu(),
            "stamp": self.stamp.cpu(),
            "pods": self.pods.cpu(),
            "e_keys": self.e_keys.cpu(),
            "e_meta": self.e_meta.cpu(),
            "e_vals": self.e_vals.cpu(),
            "capacity": self.keys.numel(),
            "pods_per_key": self.cfg.pods_per_key,
            "epoch": self._epoch,
        }

    def load_state_dict(self, state):
        if state["capacity"] != self.keys.numel() or (
            state["pods_per_key"] != self.cfg.pods_per_key
        ):
            raise ValueError("snapshot shape mismatch")
        for name in ("keys", "meta", "stamp", "pods", "e_keys", "e_meta",
                     "e_vals"):
            getattr(self, name).copy_(state[name].to(self.device))
        self._epoch = int(state.get("epoch", 0))


class TableIndex(Index):
    """Index contract over a KvTable (CPU or GPU device)."""

    def __init__(self, cfg: Optional[TableIndexConfig] = None,
                 registry: Optional[Registry] = None):
        self.cfg = cfg or TableIndexConfig()
        self.table = KvTable(self.cfg)
        self.registry = registry or Registry(self.cfg.tier_names)
        self._write_lock = threading.Lock()  # CPU ops are single-writer
        self.device = self.table.device

    # -- helpers -------------------------------------------------------
    def _hashes_tensor(self, keys: Sequence[Key]):
        return torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                            dtype=torch.int64, device=self.device)

    def _entries_tensor(self, entries: Sequence[PodEntry]):
        vals = []
        for e in entries:
            pid = self.registry.pod_id(e.pod_identifier)
            tid = self.registry.tier_id(e.device_tier)
            vals.append((tid << 24) | (pid + 1))
        return torch.tensor(vals, dtype=torch.int32, device=self.device)

    def _filter_tensor(self, pod_identifier_set: Set[str], num_pods: int):
        if not pod_identifier_set:
            return torch.zeros(0, dtype=torch.int64, device=self.device)
        W = (num_pods + 63) // 64
        words = [0] * W
        for pod in pod_identifier_set:
            pid = self.registry.pod_to_id.get(pod)
            if pid is not None and pid < num_pods:
                words[pid // 64] |= 1 << (pid % 64)
        return torch.tensor([_to_i64(w) for w in words], dtype=torch.int64,
                            device=self.device)

    def _num_pods_padded(self) -> int:
        return max(64, (self.registry.num_pods + 63) // 64 * 64)

    # -- Index contract ------------------------------------------------
    def lookup(self, request_keys: Sequence[Key],
               pod_identifier_set: Set[str]) -> Dict[Key, List[PodEntry]]:
        if not request_keys:
            raise ValueError("no request keys provided for lookup")
        model_id = self.registry.model_id(request_keys[0].model_name)
        num_pods = self._num_pods_padded()
        hashes = self._hashes_tensor(request_keys)
        filt = self._filter_tensor(pod_identifier_set, num_pods)
        found, masks = self.table.lookup(
            hashes, model_id, filt, num_pods,
            n_tiers=max(1, len(self.registry.id_to_tier)))
        found = found.cpu().tolist()
        masks_l = masks.cpu().tolist()  # one bulk D2H/convert, no per-item
        W = masks.shape[2]
        n_tiers = masks.shape[1]

        result: Dict[Key, List[PodEntry]] = {}
        for i, key in enumerate(request_keys):
            f = found[i]
            if f == 0:
                continue  # absent: skip, keep walking (in_memory.go:141)
            if f == 2:
                break  # present-but-empty: chain cut (in_memory.go:118-121)
            entries: List[PodEntry] = []
            for t in range(n_tiers):
                tier_name = (self.registry.id_to_tier[t]
                             if t < len(self.registry.id_to_tier) else None)
                if tier_name is None:
                    continue
                for w in range(W):
                    bits = _to_u64(masks_l[i][t][w])
                    while bits:
                        b = (bits & -bits).bit_length() - 1
                        bits &= bits - 1
                        pid = w * 64 + b
                        if pid < len(self.registry.id_to_pod):
                            entries.append(PodEntry(
                                self.registry.id_to_pod[pid], tier_name))
            if entries:
                result[key] = entries
        return result

    def add(self, engine_keys: Sequence[Key], request_keys: Sequence[Key],
            entries: Sequence[PodEntry], write_emap: bool = True) -> None:
        if not engine_keys or not request_keys or not entries:
            raise ValueError("no keys or entries provided for adding to index")
        if len(engine_keys) != len(request_keys):
            raise ValueError("mismatch between engine keys and request keys length")
        model_id = self.registry.model_id(request_keys[0].model_name)
        eh = self._hashes_tensor(engine_keys)
        rh = self._hashes_tensor(request_keys)
        pe = self._entries_tensor(entries)
        with self._write_lock:
            self.table.insert(eh, rh, model_id, pe, emap_write=write_emap)

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        if not entries:
            raise ValueError("no entries provided for eviction from index")
        model_id = self.registry.model_id(engine_key.model_name)
        eh = self._hashes_tensor([engine_key])
        pe = self._entries_tensor(entries)
        with self._write_lock:
            self.table.evict(eh, model_id, pe)

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        model_id = self.registry.model_id(engine_key.model_name)
        eh = self._hashes_tensor([engine_key])
        found, out = self.table.get_request_keys(eh, model_id)
        if int(found[0]) == 0:
            return None
        return Key(engine_key.model_name, _to_u64(int(out[0])))

    # -- fast paths ----------------------------------------------------
    def fused_scores(self, hashes: torch.Tensor, counts_or_offsets: torch.Tensor,
                     model_name: str, pod_identifier_set: Set[str],
                     weights: Optional[torch.Tensor] = None,
                     max_k: Optional[int] = None) -> torch.Tensor:
        """Batched probe + longest-prefix score, one call.

        hashes: int64 flat request hashes; counts (CPU table) or offsets
        [B+1] int32 (GPU table). Returns float32 [B, num_pods] scores in
        registry pod-id order."""
        model_id = self.registry.model_id(model_name)
        num_pods = self._num_pods_padded()
        filt = self._filter_tensor(pod_identifier_set, num_pods)
        if weights is None:
            weights = self.tier_weights()
        if self.table.is_cuda:
            W = (num_pods + 63) // 64
            k = max_k if max_k is not None else 512
            n_tiers = max(1, len(self.registry.id_to_tier))
            if k * n_tiers * W * 8 > 64 * 1024:
                # LDS would overflow (huge fleet x long prompts): fall back
                # to the two-kernel path - global-mask lookup + mask walk.
                found, masks = self.table.lookup(hashes, model_id, filt,
                                                 num_pods, n_tiers=n_tiers)
                del found
                return self.table.ops.gpu_score_from_masks(
                    masks.contiguous(), counts_or_offsets, weights, num_pods)
            return self.table.fused_score(hashes, counts_or_offsets, model_id,
                                          filt, weights, num_pods, max_k,
                                          n_tiers=n_tiers)
        with self._write_lock:
            return self.table.fused_score(hashes, counts_or_offsets, model_id,
                                          filt, weights, num_pods, max_k)

    def tier_weights(self, weight_map: Optional[Dict[str, float]] = None
                     ) -> torch.Tensor:
        """Per-tier-id weight vector; unknown tiers weigh 1.0
        (kvblock_scorer.go:93-99)."""
        if weight_map is None:
            weight_map = {b.name: b.weight
                          for b in default_kv_cache_backend_configs()}
        w = [1.0] * MAX_TIERS
        for i, name in enumerate(self.registry.id_to_tier):
            w[i] = weight_map.get(name, 1.0)
        return torch.tensor(w, dtype=torch.float32, device=self.device)

    def compact(self, new_capacity: Optional[int] = None) -> None:
        """Reclaim tombstoned slots (and optionally resize) - see
        KvTable.compact. Takes the write lock."""
        with self._write_lock:
            self.table.compact(new_capacity)

    def save(self, path: str) -> None:
        """Checkpoint the table + string registries to a file."""
        state = self.table.state_dict()
        state["registry"] = {
            "pods": list(self.registry.id_to_pod),
            "models": list(self.registry.id_to_model),
            "tiers": list(self.registry.id_to_tier),
        }
        torch.save(state, path)

    def load(self, path: str) -> None:
        state = torch.load(path, map_location="cpu", weights_only=False)
        reg = state.pop("registry")
        self.table.load_state_dict(state)
        r = self.registry
        with r._lock:
            r.id_to_pod = list(reg["pods"])
            r.pod_to_id = {p: i for i, p in enumerate(r.id_to_pod)}
            r.id_to_model = list(reg["models"])
            r.model_to_id = {m: i for i, m in enumerate(r.id_to_model)}
            r.id_to_tier = list(reg["tiers"])
            r.tier_to_id = {t: i for i, t in enumerate(r.id_to_tier)}

    def scores_to_map(self, scores: torch.Tensor) -> List[Dict[str, float]]:
        """float [B, num_pods] -> per-prompt {pod: score} (nonzero only).
        One whole-matrix nonzero + one Python pass over the hits (a
        per-row torch.nonzero loop costs ~50us/row at service batch
        sizes)."""
        sc = scores.cpu()
        B = sc.shape[0]
        out: List[Dict[str, float]] = [{} for _ in range(B)]
        nzi = torch.nonzero(sc)
        if nzi.numel() == 0:
            return out
        vals = sc[nzi[:, 0], nzi[:, 1]].tolist()
        rows = nzi[:, 0].tolist()
        cols = nzi[:, 1].tolist()
        names = self.registry.id_to_pod
        n_names = len(names)
        for r, c, v in zip(rows, cols, vals):
            if c < n_names:
                out[r][names[c]] = v
        return out


class _nullcontext:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


class _PinnedUploader:
    """Double-buffered pinned staging for event uploads.

    numpy -> pinned host buffer (CPU memcpy) -> true async H2D DMA,
    instead of pageable `.to(non_blocking=True)` (which bounces through
    the runtime's staging path and serializes).  Two slots rotate; a
    slot's recorded event is synchronized before the slot is rewritten
    so an in-flight DMA never reads a buffer being reused (roadmap #4).
    """

    SLOTS = 2

    def __init__(self):
        self._slots = [dict() for _ in range(self.SLOTS)]
        self._events = [None] * self.SLOTS
        self._i = 0

    def begin(self) -> None:
        import time as _time

        self._i = (self._i + 1) % self.SLOTS
        ev = self._events[self._i]
        if ev is not None:
            # NOT ev.synchronize(): hipEventSynchronize costs ~2.4 ms of
            # interrupt-wakeup latency on this stack even for an already-
            # complete event (measured: scripts/bench_upload.py), which
            # would dominate the ~3 ms batch. query() is ~us and nearly
            # always already true when a slot rotates back around.
            while not ev.query():
                _time.sleep(0)

    def up(self, arr, name: str, device, dtype=None) -> torch.Tensor:
        import numpy as np

        if not isinstance(arr, np.ndarray):
            arr = np.asarray(arr, dtype=dtype)
        src = torch.from_numpy(np.ascontiguousarray(arr)).reshape(-1)
        n = src.numel()
        if n == 0:
            return torch.zeros(0, dtype=src.dtype, device=device)
        slot = self._slots[self._i]
        buf = slot.get(name)
        if buf is None or buf.numel() < n or buf.dtype != src.dtype:
            cap = max(1024, 1 << (n - 1).bit_length())
            buf = torch.empty(cap, dtype=src.dtype, pin_memory=True)
            slot[name] = buf
        buf[:n].copy_(src)
        return buf[:n].to(device, non_blocking=True)

    def end(self) -> None:
        # reuse one event per slot: hipEventCreate/Destroy churn per call
        # is measurable on this stack (scripts/bench_upload.py)
        ev = self._events[self._i]
        if ev is None:
            ev = self._events[self._i] = torch.cuda.Event()
        ev.record()


class NativeIndex(TableIndex):
    """CPU-tensor table (fast CPU backend + GPU-parity reference)."""

    def __init__(self, cfg: Optional[TableIndexConfig] = None, **kw):
        cfg = cfg or TableIndexConfig()
        cfg.device = "cpu"
        super().__init__(cfg, **kw)


class GpuIndex(TableIndex):
    """HBM3E-resident table probed by gfx950 HIP kernels.

    Fails loudly if the HIP extension is missing or no GPU is visible -
    there is no silent CPU fallback on a GPU host."""

    def __init__(self, cfg: Optional[GpuIndexConfig] = None, **kw):
        cfg = cfg or GpuIndexConfig()
        if not torch.cuda.is_available():
            raise RuntimeError(
                "GpuIndex requires a visible AMD GPU (torch.cuda unavailable)"
            )
        super().__init__(cfg, **kw)
        if not self.table.is_cuda:
            raise RuntimeError("GpuIndexConfig.device must be a cuda device")

    def apply_event_batches(self, batches: List[Tuple[str, str, list]],
                            token_processor=None) -> None:
        """Apply decoded KV event batches fully on-device.

        batches: list of (pod_identifier, model_name, [events]) in arrival
        order; per-pod ordering is preserved by grouping (one wave per pod
        group processes its events serially - kvevents/pool.go:132-144).
        """
        from ..kvevents.events import (AllBlocksCleared, BlockRemoved,
                                       BlockStored, get_hash_as_uint64)

        if token_processor is None:
            from .token_processor import ChunkedTokenDatabase

            token_processor = ChunkedTokenDatabase()
        block_size = token_processor.block_size
        init_hash = token_processor.config.init_hash()

        import numpy as np

        def _hashes_np(raw_list):
            """Coerce a block-hash list to u64 bits as an int64 ndarray;
            fast path for plain-int lists/ndarrays, per-element fallback
            for mixed/bytes forms (pool.go:343-367 coercion)."""
            try:
                return np.asarray(raw_list, dtype=np.uint64).view(np.int64)
            except (TypeError, ValueError, OverflowError):
                out = []
                for raw in raw_list:
                    try:
                        out.append(get_hash_as_uint64(raw))
                    except Exception:
                        continue
                return np.asarray(out, dtype=np.uint64).view(np.int64)

        token_arrays: List = []
        n_tokens = 0
        tok_off = [0]
        hash_arrays: List = []
        n_hashes = 0
        eh_off = [0]
        parents: List[int] = []
        has_parent: List[int] = []
        ev_type: List[int] = []
        pod_entry: List[int] = []
        grp_off = [0]
        model_of: List[int] = []  # model id per event (mixed-model
        # batches apply in ONE launch; the kernels read model_of[e])

        by_pod: Dict[Tuple[str, str], list] = {}
        order: List[Tuple[str, str]] = []
        for pod, model, events in batches:
            k = (pod, model)
            if k not in by_pod:
                by_pod[k] = []
                order.append(k)
            by_pod[k].extend(events)

        for pod, model in order:
            model_id = self.registry.model_id(model)
            pod_id = self.registry.pod_id(pod)
            n_events_in_group = 0
            for ev in by_pod[(pod, model)]:
                if isinstance(ev, BlockStored):
                    # parse everything fallible BEFORE appending: a
                    # malformed parent hash or an un-internable tier drops
                    # the whole event, matching _digest_block_stored /
                    # pool.go (CPU and GPU replicas converge on the same
                    # state for malformed streams).
                    if ev.parent_block_hash is not None:
                        try:
                            par = _to_i64(
                                get_hash_as_uint64(ev.parent_block_hash))
                            hp = 1
                        except Exception:
                            continue  # drop event (bad parent hash)
                    else:
                        par, hp = 0, 0
                    try:
                        tier = self.registry.tier_id(
                            ev.medium.lower() if ev.medium else "gpu")
                    except ValueError:
                        continue  # drop event (tier registry full)
                    hs = _hashes_np(ev.block_hashes)
                    if hs.size == 0:
                        continue
                    hash_arrays.append(hs)
                    n_hashes += hs.size
                    eh_off.append(n_hashes)
                    toks = np.asarray(ev.token_ids, dtype=np.int64)
                    token_arrays.append(toks)
                    n_tokens += toks.size
                    tok_off.append(n_tokens)
                    parents.append(par)
                    has_parent.append(hp)
                    ev_type.append(0)
                    model_of.append(model_id)
                    pod_entry.append((tier << 24) | (pod_id + 1))
                    n_events_in_group += 1
                elif isinstance(ev, BlockRemoved):
                    try:
                        tier = self.registry.tier_id(
                            ev.medium.lower() if ev.medium else "gpu")
                    except ValueError:
                        continue  # drop event (tier registry full)
                    hs = _hashes_np(ev.block_hashes)
                    if hs.size == 0:
                        continue
                    hash_arrays.append(hs)
                    n_hashes += hs.size
                    eh_off.append(n_hashes)
                    tok_off.append(n_tokens)
                    parents.append(0)
                    has_parent.append(0)
                    ev_type.append(1)
                    model_of.append(model_id)
                    pod_entry.append((tier << 24) | (pod_id + 1))
                    n_events_in_group += 1
                elif isinstance(ev, AllBlocksCleared):
                    continue
            if n_events_in_group:
                grp_off.append(grp_off[-1] + n_events_in_group)

        if len(grp_off) == 1:
            return

        d = self.device
        tokens_np = (np.concatenate(token_arrays) if token_arrays
                     else np.zeros(0, dtype=np.int64))
        hashes_np = (np.concatenate(hash_arrays) if hash_arrays
                     else np.zeros(0, dtype=np.int64))
        import os as _os

        # Default is PAGEABLE staging: measured on MI355X (ROCm 7.2,
        # scripts/bench_upload.py) the pinned+async path is bimodal -
        # ~0.1 ms typical for the 4 MB token upload but with sporadic
        # 38-96 ms stalls in hipMemcpyAsync-from-pinned that crater
        # ingest 9.9M -> 1.8M blocks/s, while pageable .to() is a steady
        # ~0.1 ms (2.8 ms per 512-event batch end to end).  KVIDX_PINNED=1
        # opts back in for stacks where the async path behaves.
        if _os.environ.get("KVIDX_PINNED", "0") == "1":
            up = getattr(self, "_pinned_up", None)
            if up is None:
                up = self._pinned_up = _PinnedUploader()
        else:
            up = None
        # Kernel selection (host-side data only, BEFORE uploads):
        #  - serial wave-per-group kernel when a batch both stores AND
        #    removes some engine hash (per-pod op order matters);
        #  - lane-per-EVENT transposed-chain kernel when no event's
        #    parent is another event of THIS batch (chains independent
        #    across events; parents resolve via the persistent engine
        #    map): 8x the chain parallelism of lane-per-group, coalesced
        #    int32 token loads, and HALF the token upload bytes;
        #  - lane-per-group phase-split kernel otherwise.
        use_split = True
        if any(ev_type):
            stored, removed = set(), set()
            eh_list = hashes_np.tolist()
            for e, t in enumerate(ev_type):
                span = eh_list[eh_off[e]:eh_off[e + 1]]
                (removed if t else stored).update(span)
            use_split = not (stored & removed)
        use_tr = use_split and block_size <= 64
        if use_tr and any(has_parent):
            par = np.asarray(parents, dtype=np.int64)[
                np.asarray(has_parent, dtype=bool)]
            if np.isin(par, hashes_np).any():
                use_tr = False
        if use_tr:
            # int32 bit pattern of the flat tokens (uint32 token ids;
            # HALF the upload); the [token_pos][event] transpose happens
            # on-device (k_transpose_ev_tokens) - a host-side numpy
            # transpose measured ~2-5 ms/batch and briefly halved ingest
            lens = np.diff(np.asarray(tok_off, dtype=np.int64))
            max_tokens = max(1, int(lens.max()) if len(lens) else 0)
            tok_src = tokens_np.astype(np.uint32).view(np.int32)
        else:
            tok_src = tokens_np

        if up is not None:
            up.begin()
            tok_t = up.up(tok_src, "tok", d)
            common = (
                up.up(tok_off, "tok_off", d, dtype=np.int32),
                up.up(hashes_np, "eh", d),
                up.up(eh_off, "eh_off", d, dtype=np.int32),
                up.up(parents, "par", d, dtype=np.int64),
                up.up(has_parent, "haspar", d, dtype=np.uint8),
                up.up(ev_type, "evt", d, dtype=np.uint8),
                up.up(pod_entry, "pe", d, dtype=np.int32),
                up.up(grp_off, "grp", d, dtype=np.int32),
                up.up(model_of, "mdl", d, dtype=np.int32),
            )
        else:
            i32 = torch.int32
            tok_t = torch.from_numpy(tok_src).to(d, non_blocking=True)
            common = (
                torch.tensor(tok_off, dtype=i32, device=d),
                torch.from_numpy(hashes_np).to(d, non_blocking=True),
                torch.tensor(eh_off, dtype=i32, device=d),
                torch.tensor(parents, dtype=torch.int64, device=d),
                torch.tensor(has_parent, dtype=torch.uint8, device=d),
                torch.tensor(ev_type, dtype=torch.uint8, device=d),
                torch.tensor(pod_entry, dtype=i32, device=d),
                torch.tensor(grp_off, dtype=i32, device=d),
                torch.tensor(model_of, dtype=i32, device=d),
            )
        if use_split:
            counts = np.diff(np.asarray(eh_off, dtype=np.int64))
            ev_of = np.repeat(np.arange(len(counts), dtype=np.int32), counts)
            ev_of_t = (up.up(ev_of, "ev_of", d) if up is not None
                       else torch.from_numpy(ev_of).to(d, non_blocking=True))
            if use_tr:
                self.table.ops.gpu_apply_events_split_tr(
                    *self.table._t(), tok_t, *common[:7],
                    ev_of_t, common[8], _to_i64(init_hash), block_size,
                    self.table.next_epoch(), self.cfg.shard_id,
                    self.cfg.num_shards, max_tokens,
                )
            else:
                self.table.ops.gpu_apply_events_split(
                    *self.table._t(), tok_t, *common[:8], ev_of_t,
                    common[8], _to_i64(init_hash), block_size,
                    self.table.next_epoch(), self.cfg.shard_id,
                    self.cfg.num_shards,
                )
        else:
            self.table.ops.gpu_apply_events(
                *self.table._t(), tok_t, *common,
                _to_i64(init_hash), block_size,
                self.table.next_epoch(), self.cfg.shard_id,
                self.cfg.num_shards,
            )
        if up is not None:
            up.end()
