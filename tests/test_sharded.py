"""Multi-process sharded-index tests over gloo (world_size 2, CPU) -
the distributed path the driver's 8-GPU RCCL bench exercises, minus the
device type.  Spawned with torch.multiprocessing; rendezvous on
127.0.0.1 with a free port."""

import os
import random
import socket

import pytest

torch = pytest.importorskip("torch")
import torch.distributed as dist  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402

from llmd_kvcache_amd.ops import cpu_ext  # noqa: E402

pytestmark = pytest.mark.skipif(
    cpu_ext.maybe_load() is None, reason="native extension not built"
)

MODEL = "m"


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world_size, port, fn_name, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        result = globals()[fn_name](rank, world_size)
        q.put((rank, "ok", result))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "err", f"{e}\n{traceback.format_exc()}"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn_name, world_size=2):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, port, fn_name, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in procs:
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed: {payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    return results


# ---- distributed bodies (run inside workers) -------------------------


def _body_basic_sharded_score(rank, world_size):
    from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig
    from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex

    idx = ShardedIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    # identical replicated writes on every rank
    keys = [Key(MODEL, 100 + i) for i in range(8)]
    idx.add(keys, keys, [PodEntry("pod-a", "gpu")])
    idx.add(keys[:4], keys[:4], [PodEntry("pod-b", "cpu")])

    # ownership actually sharded: local table holds only owned keys
    owned = sum(
        1 for k in keys
        if len(idx.local.lookup([k, Key(MODEL, 999)], set()).get(k, []))
    )
    assert owned < len(keys) or world_size == 1

    scores = idx.score_keys(keys, set())
    return (rank, owned, scores)


def test_sharded_score_merges_across_ranks():
    results = run_distributed("_body_basic_sharded_score")
    # both ranks computed identical merged scores
    s0 = results[0][2]
    s1 = results[1][2]
    assert s0 == s1
    assert s0["pod-a"] == pytest.approx(8.0)
    assert s0["pod-b"] == pytest.approx(4 * 0.8)
    # keys were distributed: neither rank owned everything
    assert results[0][1] + results[1][1] == 8


def _body_matches_single_process(rank, world_size):
    from llmd_kvcache_amd.kvblock.gpu_index import (
        NativeIndex,
        TableIndexConfig,
        _to_i64,
    )
    from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex
    from llmd_kvcache_amd.scorer import new_kv_block_scorer

    rng = random.Random(77)  # same seed on every rank: replicated stream
    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    single = NativeIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    scorer = new_kv_block_scorer()

    for _ in range(100):
        start = rng.randrange(60)
        n = rng.randrange(1, 8)
        keys = [Key(MODEL, 3000 + start + i) for i in range(n)]
        pod = f"pod-{rng.randrange(10)}"
        tier = rng.choice(["gpu", "cpu"])
        sharded.add(keys, keys, [PodEntry(pod, tier)])
        single.add(keys, keys, [PodEntry(pod, tier)])

    mismatches = []
    for _ in range(30):
        start = rng.randrange(60)
        n = rng.randrange(1, 12)
        prompt = [Key(MODEL, 3000 + start + i) for i in range(n)]
        got = sharded.score_keys(prompt, set())
        expected = {
            p: s
            for p, s in scorer.score(prompt, single.lookup(prompt, set())).items()
            if s != 0
        }
        if set(got) != set(expected) or any(
            abs(got[p] - expected[p]) > 1e-4 for p in expected
        ):
            mismatches.append((prompt, got, expected))
    return mismatches


def test_sharded_matches_single_process_scoring():
    results = run_distributed("_body_matches_single_process")
    for rank, mismatches in results.items():
        assert mismatches == [], f"rank {rank}: {mismatches[:3]}"


def _body_batch_scores(rank, world_size):
    from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig, _to_i64
    from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex

    idx = ShardedIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    keys = [Key(MODEL, 500 + i) for i in range(6)]
    idx.add(keys, keys, [PodEntry("pod-a", "gpu")])

    prompts = [keys[:3], keys, [Key(MODEL, 9999)]]
    flat = [_to_i64(k.chunk_hash) for p in prompts for k in p]
    offsets = [0]
    for p in prompts:
        offsets.append(offsets[-1] + len(p))
    scores = idx.sharded_scores(
        torch.tensor(flat, dtype=torch.int64),
        torch.tensor(offsets, dtype=torch.int32),
        MODEL,
        set(),
    )
    return idx.local.scores_to_map(scores)


def test_sharded_batch_scores():
    results = run_distributed("_body_batch_scores")
    for rank, maps in results.items():
        assert maps[0] == {"pod-a": 3.0}
        assert maps[1] == {"pod-a": 6.0}
        assert maps[2] == {}


def _body_service_bridge(rank, world_size):
    import time as _time

    from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig
    from llmd_kvcache_amd.kvblock.token_processor import (
        ChunkedTokenDatabase,
        TokenProcessorConfig,
    )
    from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
    from llmd_kvcache_amd.parallel.service import ShardedIndexService
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex

    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    svc = ShardedIndexService(sharded, tp)

    if rank == 0:
        tokens = list(range(32))
        batch = EventBatch(
            ts=_time.time(),
            events=[BlockStored(list(range(100, 108)), None, tokens, 4)],
        )
        svc.apply_messages([("vllm-pod-1", MODEL, batch.encode())])
        keys = tp.tokens_to_kv_block_keys(None, tokens, MODEL)
        scores = svc.score(keys, set())
        filtered = svc.score(keys, {"nobody"})
        svc.stop()
        return (scores, filtered)
    svc.serve()
    # follower rank must own part of the shard
    from llmd_kvcache_amd.kvblock.keys import Key

    keys = tp.tokens_to_kv_block_keys(None, list(range(32)), MODEL)
    owned = sum(
        1 for k in keys
        if sharded.local.lookup([k, Key(MODEL, 1)], set()).get(k)
    )
    return owned


def test_sharded_service_bridge():
    results = run_distributed("_body_service_bridge")
    scores, filtered = results[0]
    assert scores == {"vllm-pod-1": 8.0}
    assert filtered == {}
    assert results[1] > 0  # rank 1 holds part of the index


def _body_registry_sync(rank, world_size):
    from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig
    from llmd_kvcache_amd.parallel.sharded import (ShardedIndex,
                                                   check_registry_sync,
                                                   registry_fingerprint)

    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 10,
                                            pods_per_key=4))
    # identical interning order on every rank -> fingerprints agree
    for i in range(5):
        sharded.registry.pod_id(f"pod-{i}")
    sharded.registry.model_id("m")
    check_registry_sync(sharded.local)  # must not raise
    fp_before = registry_fingerprint(sharded.registry)

    # rank 1 interns an extra pod (simulating a dropped/extra event):
    # EVERY rank must now raise, not just the divergent one
    if rank == 1:
        sharded.registry.pod_id("pod-ghost")
        assert registry_fingerprint(sharded.registry) != fp_before
    try:
        check_registry_sync(sharded.local)
        return "no-raise"
    except RuntimeError as e:
        assert "registry divergence" in str(e)
        return "raised"


def test_registry_sync_guard():
    results = run_distributed("_body_registry_sync")
    assert set(results.values()) == {"raised"}


def test_sharded_matches_single_process_world4():
    """Same differential as the world-2 test at world_size 4 (VERDICT
    round-1 item 3: extend gloo coverage to 4 ranks)."""
    results = run_distributed("_body_matches_single_process", world_size=4)
    for rank, mismatches in results.items():
        assert mismatches == [], f"rank {rank}: {mismatches[:3]}"


def _body_follower_failure(rank, world_size):
    """A follower whose local probe (or device staging) throws must STILL
    participate in the mask-merge collective - otherwise rank 0 blocks
    there forever.  With the error flag riding in the merge tensor, the
    peers raise instead of returning silently-degraded scores, and the
    follower keeps serving (parallel/service.py serve docstring)."""
    import time as _time

    from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig
    from llmd_kvcache_amd.kvblock.token_processor import (
        ChunkedTokenDatabase,
        TokenProcessorConfig,
    )
    from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
    from llmd_kvcache_amd.parallel.service import ShardedIndexService
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex

    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    svc = ShardedIndexService(sharded, tp)
    tokens = list(range(32))
    keys = tp.tokens_to_kv_block_keys(None, tokens, MODEL)

    if rank == 0:
        batch = EventBatch(
            ts=_time.time(),
            events=[BlockStored(list(range(100, 108)), None, tokens, 4)],
        )
        svc.apply_messages([("vllm-pod-1", MODEL, batch.encode())])
        healthy = svc.score(keys, set())          # both shards contribute
        outcomes = []
        for _ in range(2):  # follower probe fails, then staging fails
            try:
                svc.score(keys, set())
                outcomes.append("no-raise")
            except RuntimeError as e:
                assert "peer rank" in str(e)
                outcomes.append("raised")
        recovered = svc.score(keys, set())        # follower healthy again
        svc.stop()
        return (healthy, outcomes, recovered)

    # Follower: break the local probe for exactly the 2nd score op, and
    # the device move for exactly the 3rd (both asymmetric failures that
    # happen inside sharded_scores' guarded region).
    real_lookup = sharded.local.table.lookup
    real_device = sharded.device
    state = {"scores_seen": 0}

    def flaky_lookup(*a, **kw):
        if state["scores_seen"] == 2:
            raise RuntimeError("injected probe failure")
        return real_lookup(*a, **kw)

    sharded.local.table.lookup = flaky_lookup
    orig_dispatch = svc._dispatch

    def counting_dispatch(op):
        if op[0] == "score":
            state["scores_seen"] += 1
            if state["scores_seen"] == 3:
                sharded.device = "not-a-device"  # .to() raises in-guard
            else:
                sharded.device = real_device
        return orig_dispatch(op)

    svc._dispatch = counting_dispatch
    svc.serve()  # must reach "stop" without hanging despite 2 failures
    return "served"


def test_follower_failure_keeps_collective_alive():
    results = run_distributed("_body_follower_failure")
    healthy, outcomes, recovered = results[0]
    assert results[1] == "served"
    assert outcomes == ["raised", "raised"]  # explicit error, not bad data
    assert recovered == healthy              # follower recovered fully


def _body_reduce_scatter_differential(rank, world_size):
    """Forces the reduce_scatter merge strategy (prompt partitioning,
    error-flag rows, score all_gather reassembly) and differentials it
    against single-process scoring - same checks as the all_reduce path
    test, covering the large-batch RCCL strategy's logic on CPU."""
    from llmd_kvcache_amd.kvblock.gpu_index import (
        NativeIndex,
        TableIndexConfig,
        _to_i64,
    )
    from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex
    from llmd_kvcache_amd.scorer import new_kv_block_scorer

    rng = random.Random(55)
    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    sharded.force_reduce_scatter = True
    single = NativeIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
    scorer = new_kv_block_scorer()

    for _ in range(80):
        start = rng.randrange(50)
        n = rng.randrange(1, 8)
        keys = [Key(MODEL, 7000 + start + i) for i in range(n)]
        pod = f"pod-{rng.randrange(8)}"
        tier = rng.choice(["gpu", "cpu"])
        sharded.add(keys, keys, [PodEntry(pod, tier)])
        single.add(keys, keys, [PodEntry(pod, tier)])

    mismatches = []
    for trial in range(10):
        prompts = []
        for _ in range(rng.randrange(world_size, world_size * 3 + 1)):
            start = rng.randrange(50)
            n = rng.randrange(1, 10)
            prompts.append([Key(MODEL, 7000 + start + i) for i in range(n)])
        flat = [_to_i64(k.chunk_hash) for p in prompts for k in p]
        offsets = [0]
        for p in prompts:
            offsets.append(offsets[-1] + len(p))
        scores = sharded.sharded_scores(
            torch.tensor(flat, dtype=torch.int64),
            torch.tensor(offsets, dtype=torch.int32), MODEL, set())
        maps = sharded.local.scores_to_map(scores)
        for p, got in zip(prompts, maps):
            expected = {
                pod: s for pod, s in
                scorer.score(p, single.lookup(p, set())).items() if s != 0
            }
            if set(got) != set(expected) or any(
                abs(got[k] - expected[k]) > 1e-4 for k in expected
            ):
                mismatches.append((trial, got, expected))
    return mismatches


def test_reduce_scatter_merge_matches_single_process():
    for ws in (2, 4):
        results = run_distributed("_body_reduce_scatter_differential",
                                  world_size=ws)
        for rank, mismatches in results.items():
            assert mismatches == [], f"ws={ws} rank {rank}: {mismatches[:3]}"


def _body_reduce_scatter_follower_failure(rank, world_size):
    """Error-flag propagation through the reduce_scatter strategy."""
    from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig, _to_i64
    from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex

    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 10, pods_per_key=4))
    sharded.force_reduce_scatter = True
    keys = [Key(MODEL, 40 + i) for i in range(8)]
    sharded.add(keys, keys, [PodEntry("pod-a", "gpu")])

    flat = torch.tensor([_to_i64(k.chunk_hash) for k in keys for _ in [0]],
                        dtype=torch.int64)
    offsets = torch.tensor([0, 4, 8], dtype=torch.int32)  # 2 prompts

    if rank == 1:
        real = sharded.local.table.lookup
        sharded.local.table.lookup = lambda *a, **kw: (_ for _ in ()).throw(
            RuntimeError("injected"))
    outcome = None
    try:
        sharded.sharded_scores(flat, offsets, MODEL, set())
        outcome = "no-raise"
    except RuntimeError as e:
        outcome = "peer" if "peer rank" in str(e) else "local"
    if rank == 1:
        sharded.local.table.lookup = real
    # a follow-up call must still work on every rank (no stuck collective)
    scores = sharded.sharded_scores(flat, offsets, MODEL, set())
    ok = abs(float(scores[0].max().item()) - 4.0) < 1e-5
    return (outcome, ok)


def test_reduce_scatter_error_flag():
    results = run_distributed("_body_reduce_scatter_follower_failure")
    assert results[0] == ("peer", True)
    assert results[1] == ("local", True)


def test_partition_prompts_properties():
    """Unit properties of the reduce_scatter prompt partitioner:
    monotone bounds, full coverage, deterministic, reasonable balance."""
    import random

    from llmd_kvcache_amd.parallel.sharded import _partition_prompts

    rng = random.Random(5)
    for _ in range(200):
        world = rng.choice([2, 3, 4, 8])
        B = rng.randrange(world, 200)
        counts = [rng.randrange(0, 64) for _ in range(B)]
        offs = [0]
        for c in counts:
            offs.append(offs[-1] + c)
        offs_t = torch.tensor(offs, dtype=torch.int32)
        bounds = _partition_prompts(offs_t, world)
        assert bounds[0] == 0 and bounds[-1] == B
        assert all(bounds[i] <= bounds[i + 1] for i in range(world))
        assert bounds == _partition_prompts(offs_t, world)  # deterministic
        total = offs[-1]
        if total:
            # no rank should carry more than ~(1/world + one prompt)'s
            # worth of keys beyond the ideal share
            max_keys = max(offs[bounds[r + 1]] - offs[bounds[r]]
                           for r in range(world))
            assert max_keys <= total // world + max(counts)


def _body_rs_ragged_edges(rank, world_size):
    """reduce_scatter merge with deliberately nasty shapes: zero-key
    prompts, heavy skew (one giant prompt), and a prompt count equal to
    world_size (minimum allowed)."""
    from llmd_kvcache_amd.kvblock.gpu_index import (
        NativeIndex,
        TableIndexConfig,
        _to_i64,
    )
    from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
    from llmd_kvcache_amd.parallel.sharded import ShardedIndex
    from llmd_kvcache_amd.scorer import new_kv_block_scorer

    sharded = ShardedIndex(TableIndexConfig(capacity=1 << 10, pods_per_key=4))
    sharded.force_reduce_scatter = True
    single = NativeIndex(TableIndexConfig(capacity=1 << 10, pods_per_key=4))
    scorer = new_kv_block_scorer()
    keys = [Key(MODEL, 800 + i) for i in range(40)]
    for idx in (sharded, single):
        idx.add(keys, keys, [PodEntry("pod-a", "gpu")])
        idx.add(keys[:5], keys[:5], [PodEntry("pod-b", "cpu")])

    cases = [
        # [prompt key-counts]: zero-key prompts + skew
        [0, 40, 0],
        [1, 1, 1],
        [0, 0, 38],
        [12, 0, 12, 0, 12],
    ]
    mismatches = []
    for counts in cases:
        if len(counts) < world_size:
            continue
        prompts, pos = [], 0
        for c in counts:
            prompts.append(keys[pos:pos + c] if c else [])
            pos += c
        flat = [_to_i64(k.chunk_hash) for p in prompts for k in p]
        offsets = [0]
        for p in prompts:
            offsets.append(offsets[-1] + len(p))
        scores = sharded.sharded_scores(
            torch.tensor(flat, dtype=torch.int64),
            torch.tensor(offsets, dtype=torch.int32), MODEL, set())
        maps = sharded.local.scores_to_map(scores)
        for p, got in zip(prompts, maps):
            expected = ({} if not p else {
                pod: s for pod, s in
                scorer.score(p, single.lookup(p, set())).items() if s != 0
            })
            if set(got) != set(expected) or any(
                abs(got[k] - expected[k]) > 1e-4 for k in expected
            ):
                mismatches.append((counts, got, expected))
    return mismatches


def test_reduce_scatter_ragged_edges():
    for ws in (2, 3, 4):
        results = run_distributed("_body_rs_ragged_edges", world_size=ws)
        for rank, mm in results.items():
            assert mm == [], f"ws={ws} rank {rank}: {mm[:2]}"
