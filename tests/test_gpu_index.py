"""GPU (gfx950) kernel tests: the HBM-resident table vs the bit-identical
CPU reference ops, plus the fused lookup+score and hash-chain kernels vs
the pure-Python golden implementations.  All tests require an MI355X."""

import random

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

from llmd_kvcache_amd.kvblock.gpu_index import (  # noqa: E402
    GpuIndex,
    GpuIndexConfig,
    NativeIndex,
    TableIndexConfig,
    _to_i64,
    _to_u64,
)
from llmd_kvcache_amd.kvblock.keys import Key, PodEntry  # noqa: E402
from llmd_kvcache_amd.kvblock.token_processor import (  # noqa: E402
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.kvevents.events import BlockRemoved, BlockStored  # noqa: E402
from llmd_kvcache_amd.scorer import new_kv_block_scorer  # noqa: E402
from llmd_kvcache_amd.utils import hashing  # noqa: E402

MODEL = "m"


@pytest.fixture
def gpu_index():
    return GpuIndex(GpuIndexConfig(capacity=1 << 14, pods_per_key=10))


def test_native_extension_is_loaded():
    """The HIP extension must actually be the in-tree .so (no silent
    eager fallback on a GPU box)."""
    from llmd_kvcache_amd.ops import cpu_ext

    mod = cpu_ext.require()
    assert mod.HAS_HIP
    assert torch.cuda.is_available()


class TestGpuTableOps:
    def test_basic_add_lookup_evict(self, gpu_index):
        keys = [Key(MODEL, h) for h in (1, 2, 3)]
        gpu_index.add(keys, keys, [PodEntry("pod-a", "gpu")])
        result = gpu_index.lookup(keys, set())
        assert set(result.keys()) == set(keys)
        for k in keys:
            assert result[k] == [PodEntry("pod-a", "gpu")]
        assert gpu_index.get_request_key(Key(MODEL, 1)) == Key(MODEL, 1)
        gpu_index.evict(Key(MODEL, 2), [PodEntry("pod-a", "gpu")])
        result = gpu_index.lookup(keys, set())
        assert Key(MODEL, 2) not in result

    def test_dual_keys(self, gpu_index):
        ek = [Key(MODEL, 100)]
        rk = [Key(MODEL, 200)]
        gpu_index.add(ek, rk, [PodEntry("pod-a", "gpu")])
        assert gpu_index.get_request_key(Key(MODEL, 100)) == Key(MODEL, 200)
        assert gpu_index.get_request_key(Key(MODEL, 999)) is None
        gpu_index.evict(Key(MODEL, 100), [PodEntry("pod-a", "gpu")])
        assert Key(MODEL, 200) not in gpu_index.lookup(rk + ek, set())

    @pytest.mark.parametrize("seed", [1, 2])
    def test_differential_vs_cpu_table(self, seed):
        from tests.test_native_index import random_workload, run_op

        rng = random.Random(seed)
        cpu = NativeIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))
        for op in random_workload(rng, n_ops=200):
            r_cpu = run_op(cpu, op)
            r_gpu = run_op(gpu, op)
            if op[0] == "lookup" and r_cpu != "error":
                assert set(r_cpu.keys()) == set(r_gpu.keys()), op
                for k in r_cpu:
                    assert set(r_cpu[k]) == set(r_gpu[k]), (op, k)


class TestFusedScoreKernel:
    @pytest.mark.parametrize("seed", [21, 22])
    def test_matches_python_scorer(self, seed):
        rng = random.Random(seed)
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 14, pods_per_key=10))
        scorer = new_kv_block_scorer()
        for _ in range(200):
            start = rng.randrange(150)
            n = rng.randrange(1, 10)
            keys = [Key(MODEL, 7000 + start + i) for i in range(n)]
            pod = f"pod-{rng.randrange(32)}"
            tier = rng.choice(["gpu", "cpu"])
            gpu.add(keys, keys, [PodEntry(pod, tier)])

        prompts = []
        for _ in range(50):
            start = rng.randrange(150)
            n = rng.randrange(1, 32)
            prompts.append([Key(MODEL, 7000 + start + i) for i in range(n)])

        flat = [_to_i64(k.chunk_hash) for p in prompts for k in p]
        offsets = [0]
        for p in prompts:
            offsets.append(offsets[-1] + len(p))
        hashes = torch.tensor(flat, dtype=torch.int64, device="cuda")
        offs = torch.tensor(offsets, dtype=torch.int32, device="cuda")
        scores = gpu.fused_scores(hashes, offs, MODEL, set(),
                                  max_k=max(len(p) for p in prompts))
        maps = gpu.scores_to_map(scores)

        for i, prompt in enumerate(prompts):
            key_to_pods = gpu.lookup(prompt, set())
            expected = {p: s for p, s in
                        scorer.score(prompt, key_to_pods).items() if s != 0}
            assert maps[i].keys() == expected.keys(), (i, maps[i], expected)
            for p in expected:
                assert maps[i][p] == pytest.approx(expected[p])

    def test_pod_filter(self):
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))
        keys = [Key(MODEL, 1), Key(MODEL, 2)]
        gpu.add(keys, keys, [PodEntry("pod-a", "gpu"), PodEntry("pod-b", "gpu")])
        hashes = torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                              dtype=torch.int64, device="cuda")
        offs = torch.tensor([0, 2], dtype=torch.int32, device="cuda")
        maps = gpu.scores_to_map(
            gpu.fused_scores(hashes, offs, MODEL, {"pod-b"}, max_k=2))
        assert maps[0] == {"pod-b": 2.0}


class TestHashChainKernel:
    @pytest.mark.parametrize("block_size", [4, 16])
    def test_matches_python(self, block_size):
        from llmd_kvcache_amd.ops import cpu_ext

        mod = cpu_ext.require()
        rng = random.Random(5)
        B = 67  # not a multiple of 64: tail-wave coverage
        toks, off = [], [0]
        for _ in range(B):
            n = rng.randrange(1, 12) * block_size
            toks.extend(rng.randrange(0, 2**32) for _ in range(n))
            off.append(len(toks))
        parents = [hashing.init_hash("")] * B
        tokens_t = torch.tensor(toks, dtype=torch.int64, device="cuda")
        off_t = torch.tensor(off, dtype=torch.int64, device="cuda")
        par_t = torch.tensor([_to_i64(p) for p in parents],
                             dtype=torch.int64, device="cuda")
        out, chunk_off = mod.gpu_hash_chain(tokens_t, off_t, par_t, block_size)
        out = out.cpu().tolist()
        chunk_off = chunk_off.cpu().tolist()
        for b in range(B):
            h = parents[b]
            chunk_toks = toks[off[b]: off[b + 1]]
            for c in range((off[b + 1] - off[b]) // block_size):
                h = hashing.chunk_hash(
                    h, chunk_toks[c * block_size:(c + 1) * block_size])
                assert _to_u64(out[chunk_off[b] + c]) == h, (b, c)


class TestHashChainTrKernel:
    @pytest.mark.parametrize("block_size", [4, 16])
    def test_matches_row_major_kernel(self, block_size):
        from llmd_kvcache_amd.ops import cpu_ext

        mod = cpu_ext.require()
        rng = random.Random(13)
        B, max_chunks = 130, 9
        n_chunks = [rng.randrange(1, max_chunks + 1) for _ in range(B)]
        tok = torch.zeros((max_chunks * block_size, B), dtype=torch.int32)
        row_toks, row_off = [], [0]
        for b in range(B):
            n = n_chunks[b] * block_size
            vals = [rng.randrange(0, 1 << 31) for _ in range(n)]
            tok[:n, b] = torch.tensor(vals, dtype=torch.int32)
            row_toks.extend(vals)
            row_off.append(len(row_toks))
        parents = torch.tensor(
            [_to_i64(hashing.init_hash(""))] * B, dtype=torch.int64,
            device="cuda")
        out_t = mod.gpu_hash_chain_tr(
            tok.cuda(), parents,
            torch.tensor(n_chunks, dtype=torch.int32, device="cuda"),
            block_size, max_chunks)
        ref, ref_off = mod.gpu_hash_chain(
            torch.tensor(row_toks, dtype=torch.int64, device="cuda"),
            torch.tensor(row_off, dtype=torch.int64, device="cuda"),
            parents, block_size)
        out_t = out_t.cpu()
        ref = ref.cpu()
        ref_off = ref_off.cpu()
        for b in range(B):
            for c in range(n_chunks[b]):
                assert out_t[c, b] == ref[ref_off[b] + c], (b, c)


class TestApplyEventsKernel:
    def test_stored_then_removed_in_order(self):
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        ev1 = BlockStored([100, 200], None, list(range(8)), 4)
        ev2 = BlockRemoved([100])
        gpu.apply_event_batches([("pod-a", MODEL, [ev1, ev2])], tp)
        torch.cuda.synchronize()
        req = tp.tokens_to_kv_block_keys(None, list(range(8)), MODEL)
        result = gpu.lookup(req, set())
        assert req[0] not in result  # removed
        assert result.get(req[1]) == [PodEntry("pod-a", "gpu")]

    def test_parent_chain_stitching_on_device(self):
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        t1, t2 = [1, 2, 3, 4], [5, 6, 7, 8]
        gpu.apply_event_batches(
            [("pod-a", MODEL, [BlockStored([100], None, t1, 4)])], tp)
        gpu.apply_event_batches(
            [("pod-a", MODEL, [BlockStored([200], 100, t2, 4)])], tp)
        torch.cuda.synchronize()
        full = tp.tokens_to_kv_block_keys(None, t1 + t2, MODEL)
        result = gpu.lookup(full, set())
        assert set(result.keys()) == set(full)

    def test_matches_cpu_pool_digest(self):
        """GPU on-device event application == CPU EventsPool.digest_events."""
        from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool

        rng = random.Random(9)
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        cpu_idx = NativeIndex(TableIndexConfig(capacity=1 << 12, pods_per_key=10))
        gpu_idx = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))
        pool = EventsPool(EventsConfig(concurrency=1), cpu_idx, tp)

        batches = []
        next_hash = 1000
        chains = {}  # pod -> last engine hash
        for _ in range(50):
            pod = f"pod-{rng.randrange(4)}"
            n_blocks = rng.randrange(1, 5)
            toks = [rng.randrange(0, 1 << 31) for _ in range(n_blocks * 4)]
            hs = list(range(next_hash, next_hash + n_blocks))
            next_hash += n_blocks
            parent = chains.get(pod)
            ev = BlockStored(hs, parent, toks, 4)
            chains[pod] = hs[-1]
            batches.append((pod, MODEL, [ev]))

        for pod, model, events in batches:
            pool.digest_events(pod, model, events)
        # apply to GPU in per-pod groups (same order)
        gpu_idx.apply_event_batches(batches, tp)
        torch.cuda.synchronize()

        # compare a sample of request-key lookups
        for pod, model, events in batches:
            ev = events[0]
            req = tp.tokens_to_kv_block_keys(None, ev.token_ids, model)
            # per-event chains may differ (parent stitching), compare via
            # engine->request mapping instead
            for h in ev.block_hashes:
                ck = Key(model, h)
                assert (cpu_idx.get_request_key(ck) ==
                        gpu_idx.get_request_key(ck)), ck
        # spot-check pod visibility through engine-derived request keys
        for h in range(1000, next_hash):
            ck = Key(MODEL, h)
            rk = cpu_idx.get_request_key(ck)
            if rk is None:
                continue
            c = cpu_idx.lookup([rk], set())
            g = gpu_idx.lookup([rk], set())
            assert set(c.get(rk, [])) == set(g.get(rk, [])), rk


    def test_transposed_event_path_ragged_matches_cpu(self):
        """Lane-per-event transposed chain path (parentless batch,
        RAGGED event sizes - exercises the on-device transpose padding)
        produces the same state as the CPU digest."""
        from llmd_kvcache_amd.kvevents.pool import digest_events

        rng = random.Random(21)
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        cpu_idx = NativeIndex(TableIndexConfig(capacity=1 << 12,
                                               pods_per_key=10))
        gpu_idx = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))

        batches = []
        next_hash = 5000
        for i in range(40):
            pod = f"pod-{rng.randrange(6)}"
            n_blocks = rng.randrange(1, 9)  # ragged: 4..32 tokens
            toks = [rng.randrange(0, 1 << 31) for _ in range(n_blocks * 4)]
            hs = list(range(next_hash, next_hash + n_blocks))
            next_hash += n_blocks
            batches.append((pod, MODEL, [BlockStored(hs, None, toks, 4)]))

        for pod, model, events in batches:
            digest_events(cpu_idx, tp, pod, model, events)
        gpu_idx.apply_event_batches(batches, tp)
        torch.cuda.synchronize()

        for pod, model, events in batches:
            ev = events[0]
            req = tp.tokens_to_kv_block_keys(None, ev.token_ids, model)
            c = cpu_idx.lookup(req, set())
            g = gpu_idx.lookup(req, set())
            assert {k: set(v) for k, v in c.items()} == \
                {k: set(v) for k, v in g.items()}, (pod, ev.block_hashes)
            for h in ev.block_hashes:
                ck = Key(model, h)
                assert (cpu_idx.get_request_key(ck) ==
                        gpu_idx.get_request_key(ck)), ck


    def test_mixed_model_single_launch_matches_cpu(self):
        """Two models in ONE apply call (round-2 per-event model ids):
        state must equal the CPU digest, and model scoping must hold
        (same token chains under different models never alias)."""
        from llmd_kvcache_amd.kvevents.pool import digest_events

        rng = random.Random(31)
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        cpu_idx = NativeIndex(TableIndexConfig(capacity=1 << 12,
                                               pods_per_key=10))
        gpu_idx = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))

        batches = []
        next_hash = 9000
        shared_tokens = list(range(16))  # same tokens under BOTH models
        for model in ("model-a", "model-b"):
            hs = list(range(next_hash, next_hash + 4))
            next_hash += 4
            batches.append((f"pod-{model}", model,
                            [BlockStored(hs, None, shared_tokens, 4)]))
        for i in range(20):
            model = rng.choice(["model-a", "model-b"])
            toks = [rng.randrange(1 << 30) for _ in range(rng.randrange(1, 5) * 4)]
            hs = list(range(next_hash, next_hash + len(toks) // 4))
            next_hash += len(toks) // 4
            batches.append((f"pod-{rng.randrange(4)}", model,
                            [BlockStored(hs, None, toks, 4)]))

        for pod, model, events in batches:
            digest_events(cpu_idx, tp, pod, model, events)
        gpu_idx.apply_event_batches(batches, tp)  # ONE call, two models
        torch.cuda.synchronize()

        for pod, model, events in batches:
            ev = events[0]
            req = tp.tokens_to_kv_block_keys(None, ev.token_ids, model)
            c = cpu_idx.lookup(req, set())
            g = gpu_idx.lookup(req, set())
            assert {k: set(v) for k, v in c.items()} == \
                {k: set(v) for k, v in g.items()}, (pod, model)
        # model scoping: identical chains under different models resolve
        # to their own pods only
        ka = tp.tokens_to_kv_block_keys(None, shared_tokens, "model-a")
        kb = tp.tokens_to_kv_block_keys(None, shared_tokens, "model-b")
        ga = gpu_idx.lookup(ka, set())
        gb = gpu_idx.lookup(kb, set())
        assert {e.pod_identifier for v in ga.values() for e in v} == \
            {"pod-model-a"}
        assert {e.pod_identifier for v in gb.values() for e in v} == \
            {"pod-model-b"}


class TestConcurrentGpuInserts:
    def test_many_duplicate_inserts_converge(self):
        """Thousands of threads inserting the same keys concurrently must
        produce one consistent slot per key (lock-free claim correctness)."""
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 14, pods_per_key=10))
        keys = [Key(MODEL, 42)]
        eh = torch.full((5000,), 42, dtype=torch.int64, device="cuda")
        rh = torch.full((5000,), 43, dtype=torch.int64, device="cuda")
        pe = gpu._entries_tensor([PodEntry("pod-a", "gpu")])
        gpu.table.insert(eh, rh, gpu.registry.model_id(MODEL), pe)
        torch.cuda.synchronize()
        result = gpu.lookup([Key(MODEL, 43)], set())
        assert result[Key(MODEL, 43)] == [PodEntry("pod-a", "gpu")]
        assert gpu.get_request_key(Key(MODEL, 42)) == Key(MODEL, 43)

    def test_concurrent_distinct_inserts(self):
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 16, pods_per_key=10))
        n = 20000
        eh = torch.arange(1, n + 1, dtype=torch.int64, device="cuda")
        rh = eh + 1_000_000
        pe = gpu._entries_tensor([PodEntry("pod-a", "gpu")])
        gpu.table.insert(eh, rh, gpu.registry.model_id(MODEL), pe)
        torch.cuda.synchronize()
        sample = [Key(MODEL, 1_000_001 + i) for i in range(0, n, 977)]
        result = gpu.lookup(sample, set())
        assert set(result.keys()) == set(sample)


class TestTieredGpu:
    def test_gpu_hot_cpu_cold(self):
        from llmd_kvcache_amd.kvblock.gpu_index import (
            NativeIndex,
            TableIndexConfig,
        )
        from llmd_kvcache_amd.kvblock.tiered import TieredIndex

        hot = GpuIndex(GpuIndexConfig(capacity=256, pods_per_key=4))
        cold = NativeIndex(
            TableIndexConfig(capacity=1 << 14, pods_per_key=10),
            registry=hot.registry,
        )
        t = TieredIndex(hot=hot, cold=cold)
        all_keys = []
        for h in range(0, 3000, 100):
            ks = [Key(MODEL, 10_000 + h + i) for i in range(100)]
            t.add(ks, ks, [PodEntry("pod-a", "gpu")])
            all_keys.extend(ks)
        torch.cuda.synchronize()
        early = all_keys[:50]
        before = t.hot.lookup(early, set())
        assert len(before) < len(early)  # HBM tier evicted under pressure
        merged = t.lookup(early, set())
        assert len(merged) == len(early)
        # cold hits were promoted back into the HBM tier
        after = t.hot.lookup(early, set())
        assert len(after) == len(early)
        # emap_write=0 on the GPU insert: a key known only to the cold
        # tier gains no engine-map self-mapping when promoted
        solo = Key(MODEL, 99_999)
        cold.add([solo], [solo], [PodEntry("pod-a", "gpu")])
        t.lookup([solo], set())
        assert t.hot.lookup([solo], set())  # promoted
        assert t.hot.get_request_key(solo) is None

    def test_fused_scores_cold_only_on_gpu_hot(self):
        """GPU hot tier: the fused read path must serve keys resident
        only in the CPU capacity tier (and promote them)."""
        from llmd_kvcache_amd.kvblock.gpu_index import (NativeIndex,
                                                        TableIndexConfig,
                                                        _to_i64)
        from llmd_kvcache_amd.kvblock.tiered import TieredIndex

        hot = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=4))
        cold = NativeIndex(TableIndexConfig(capacity=1 << 14,
                                            pods_per_key=10),
                           registry=hot.registry)
        t = TieredIndex(hot=hot, cold=cold)
        keys = [Key(MODEL, 40_000 + i) for i in range(3)]
        cold.add(keys, keys, [PodEntry("pod-cc", "cpu")])
        hashes = torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                              dtype=torch.int64, device="cuda")
        offs = torch.tensor([0, 3], dtype=torch.int32, device="cuda")
        scores = t.fused_scores(hashes, offs, MODEL, set())
        m = t.scores_to_map(scores)[0]
        assert m == {"pod-cc": pytest.approx(3 * 0.8)}
        torch.cuda.synchronize()
        assert len(t.hot.lookup(keys, set())) == 3  # promoted to HBM
        scores2 = t.fused_scores(hashes, offs, MODEL, set())  # pure hot now
        assert t.scores_to_map(scores2)[0] == {"pod-cc":
                                               pytest.approx(3 * 0.8)}


class TestManyPods:
    def test_fused_score_256_pods(self):
        """Multi-word pod masks (W=4): 256-pod fleet through the fused
        kernel (BASELINE config-5 scale)."""
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 14, pods_per_key=10))
        keys = [Key(MODEL, 100 + i) for i in range(4)]
        # register 250 pods; store entries for a spread of pod ids
        for i in range(250):
            gpu.registry.pod_id(f"pod-{i}")
        for pid in (0, 63, 64, 127, 128, 249):
            gpu.add(keys, keys, [PodEntry(f"pod-{pid}", "gpu")])
        gpu.add(keys[:2], keys[:2], [PodEntry("pod-200", "cpu")])
        hashes = torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                              dtype=torch.int64, device="cuda")
        offs = torch.tensor([0, 4], dtype=torch.int32, device="cuda")
        maps = gpu.scores_to_map(
            gpu.fused_scores(hashes, offs, MODEL, set(), max_k=4))
        expected = {f"pod-{p}": 4.0 for p in (0, 63, 64, 127, 128, 249)}
        expected["pod-200"] = 1.6
        assert maps[0].keys() == expected.keys()
        for p, v in expected.items():
            assert maps[0][p] == pytest.approx(v)
        # and agreement with the python scorer on the generic path
        scorer = new_kv_block_scorer()
        ref = {p: s for p, s in
               scorer.score(keys, gpu.lookup(keys, set())).items() if s != 0}
        assert maps[0].keys() == ref.keys()
        for p, v in ref.items():
            assert maps[0][p] == pytest.approx(v)


class TestConcurrentReadWrite:
    def test_reads_race_writes_across_streams(self):
        """Fused scores on the default stream racing k_apply_events on a
        side stream: must not fault, and post-sync lookups must reflect
        all writes (lock-free table invariants under real concurrency)."""
        import numpy as np

        from llmd_kvcache_amd.kvblock.token_processor import (
            ChunkedTokenDatabase,
            TokenProcessorConfig,
        )
        from llmd_kvcache_amd.kvevents.events import BlockStored

        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=16))
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 16, pods_per_key=10))
        rng = np.random.default_rng(3)
        side = torch.cuda.Stream()

        all_tokens = []
        h = 1
        for step in range(10):
            toks = rng.integers(0, 1 << 31, size=16 * 16, dtype=np.int64)
            hs = np.arange(h, h + 16, dtype=np.uint64)
            h += 16
            all_tokens.append(toks)
            with torch.cuda.stream(side):
                gpu.apply_event_batches(
                    [(f"pod-{step % 4}", MODEL,
                      [BlockStored(hs, None, toks, 16)])], tp)
            # concurrent reads on the default stream
            probe = torch.randint(-(2**62), 2**62, (2048,),
                                  dtype=torch.int64, device="cuda")
            offs = torch.arange(0, 2049, 512, dtype=torch.int32,
                                device="cuda")[:5]
            gpu.fused_scores(probe, offs, MODEL, set(), max_k=512)
        torch.cuda.synchronize()

        # every write is visible after sync
        for step, toks in enumerate(all_tokens):
            keys = tp.tokens_to_kv_block_keys(None, toks.tolist(), MODEL)
            result = gpu.lookup(keys, set())
            assert set(result.keys()) == set(keys), step
            assert result[keys[0]] == [PodEntry(f"pod-{step % 4}", "gpu")]


class TestLdsOverflowFallback:
    def test_huge_fleet_long_prompt_falls_back(self):
        """2048 registered pods x 512-key prompt exceeds the fused
        kernel's LDS budget; fused_scores must route through the
        two-kernel global-mask path with identical results."""
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 14, pods_per_key=10))
        for i in range(2048):
            gpu.registry.pod_id(f"pod-{i}")
        K = 512
        keys = [Key(MODEL, 40_000 + i) for i in range(K)]
        gpu.add(keys, keys, [PodEntry("pod-2000", "gpu")])
        hashes = torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                              dtype=torch.int64, device="cuda")
        offs = torch.tensor([0, K], dtype=torch.int32, device="cuda")
        maps = gpu.scores_to_map(
            gpu.fused_scores(hashes, offs, MODEL, set(), max_k=K))
        assert maps[0] == {"pod-2000": float(K)}


class TestHighLoadFactor:
    def test_steal_path_under_pressure(self):
        """GPU twin of the CPU high-load test plus concurrent duplicate
        pressure: fill to 85%, then hammer the hottest window."""
        cap = 1 << 14
        gpu = GpuIndex(GpuIndexConfig(capacity=cap, pods_per_key=4))
        n = int(cap * 0.85)
        eh = torch.arange(1, n + 1, dtype=torch.int64, device="cuda")
        rh = eh + 5_000_000
        pe = gpu._entries_tensor([PodEntry("pod-a", "gpu")])
        gpu.table.insert(eh, rh, gpu.registry.model_id(MODEL), pe)
        torch.cuda.synchronize()
        recent = [Key(MODEL, 5_000_000 + i) for i in range(n - 500, n)]
        found = gpu.lookup(recent, set())
        assert len(found) >= len(recent) * 0.95
        # duplicate hammering on a full table must not corrupt
        dup_eh = torch.full((4096,), n, dtype=torch.int64, device="cuda")
        dup_rh = dup_eh + 5_000_000
        gpu.table.insert(dup_eh, dup_rh, gpu.registry.model_id(MODEL), pe)
        torch.cuda.synchronize()
        assert gpu.lookup([Key(MODEL, 5_000_000 + n)], set())


class TestPinnedUploadOptIn:
    def test_pinned_path_parity(self, monkeypatch):
        """KVIDX_PINNED=1 routes uploads through the double-buffered
        pinned stager (_PinnedUploader); results must match the default
        pageable path exactly (opt-in path kept working; ROADMAP #4)."""
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        results = {}
        for mode in ("0", "1"):
            monkeypatch.setenv("KVIDX_PINNED", mode)
            gpu = GpuIndex(GpuIndexConfig(capacity=1 << 12, pods_per_key=10))
            batches = []
            for e in range(8):
                toks = [e * 100 + i for i in range(8)]
                batches.append((f"pod-{e % 3}", MODEL,
                                [BlockStored([1000 + 2 * e, 1001 + 2 * e],
                                             None, toks, 4)]))
            gpu.apply_event_batches(batches, tp)
            # second call exercises slot rotation + event reuse
            gpu.apply_event_batches(
                [("pod-9", MODEL, [BlockStored([5000], None,
                                               list(range(4)), 4)])], tp)
            torch.cuda.synchronize()
            if mode == "1":
                assert getattr(gpu, "_pinned_up", None) is not None
            snap = {}
            for e in range(8):
                keys = tp.tokens_to_kv_block_keys(
                    None, [e * 100 + i for i in range(8)], MODEL)
                snap[e] = {k: sorted(map(tuple, v)) for k, v in
                           gpu.lookup(keys, set()).items()}
            results[mode] = snap
        assert results["0"] == results["1"]


class TestHashChainTrRowMajor:
    def test_row_major_output_matches_transpose(self):
        """row_major=1 writes [B, maxC] directly - must equal the
        transposed [maxC, B] default (saves the 128 MB/call transpose
        pass the fused-score consumer otherwise needs)."""
        from llmd_kvcache_amd.ops import cpu_ext
        from llmd_kvcache_amd.utils import hashing

        mod = cpu_ext.require()
        rng = random.Random(31)
        B, C, BS = 97, 7, 16
        tok = torch.tensor(
            [[rng.randrange(0, 1 << 31) for _ in range(B)]
             for _ in range(C * BS)], dtype=torch.int32, device="cuda")
        parents = torch.tensor([_to_i64(hashing.init_hash(""))] * B,
                               dtype=torch.int64, device="cuda")
        nch = torch.tensor([rng.randrange(1, C + 1) for _ in range(B)],
                           dtype=torch.int32, device="cuda")
        col = mod.gpu_hash_chain_tr(tok, parents, nch, BS, C, 0, 0)
        row = mod.gpu_hash_chain_tr(tok, parents, nch, BS, C, 0, 1)
        assert col.shape == (C, B) and row.shape == (B, C)
        assert torch.equal(col.t().contiguous(), row)
        # prefetch A/B variant honors the flag too
        row_pf = mod.gpu_hash_chain_tr(tok, parents, nch, BS, C, 8, 1)
        assert torch.equal(row_pf, row)


class TestCompactionGpu:
    def test_gpu_compact_matches_cpu(self):
        """Differential: same churn workload on GPU and CPU tables,
        compact both, lookups must agree (and match pre-compact)."""
        rng = random.Random(77)
        cpu = NativeIndex(TableIndexConfig(capacity=1 << 11, pods_per_key=4))
        gpu = GpuIndex(GpuIndexConfig(capacity=1 << 11, pods_per_key=4))
        keys = [Key(MODEL, 3000 + i) for i in range(300)]
        for idx_, k in enumerate(keys):
            e = [Key(MODEL, 600000 + idx_)]
            pods = [PodEntry(f"pod-{rng.randrange(9)}", "gpu")]
            cpu.add(e, [k], pods)
            gpu.add(e, [k], pods)
        for i in range(0, 300, 4):
            for t in (cpu, gpu):
                t.evict(Key(MODEL, 600000 + i),
                        [PodEntry(f"pod-{i % 9}", "gpu")])
        torch.cuda.synchronize()

        def snap(t):
            return {k: sorted(map(tuple, v))
                    for k, v in t.lookup(keys, set()).items()}

        before = snap(gpu)
        assert before == snap(cpu)
        cpu.compact(new_capacity=1 << 12)
        gpu.compact(new_capacity=1 << 12)
        torch.cuda.synchronize()
        assert snap(gpu) == before
        assert snap(cpu) == before
        tombs = int(((gpu.table.meta.cpu() & 0x40000000) != 0).sum())
        assert tombs == 0
        assert gpu.get_request_key(Key(MODEL, 600001)) == keys[1]
