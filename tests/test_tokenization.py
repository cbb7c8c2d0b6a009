"""Tokenization subsystem tests: local provider + discovery, composite
fallback chain, pool sync/async flow with the 0.8 overlap threshold, and
the UDS sidecar service over a real unix socket
(mirrors pkg/tokenization/pool_test.go and tokenizer.go behaviors)."""

import asyncio
import json
import os
import threading
import time

import pytest

from llmd_kvcache_amd.tokenization.pool import (
    TokenizationConfig,
    TokenizationPool,
)
from llmd_kvcache_amd.tokenization.prefixstore import (
    LRUStoreConfig,
    LRUTokenStore,
)
from llmd_kvcache_amd.tokenization.tokenizer import (
    CompositeTokenizer,
    LocalTokenizerConfig,
    TokenizationError,
    Tokenizer,
    discover_local_tokenizers,
    new_cached_local_tokenizer,
)


@pytest.fixture(scope="module")
def tokenizer_fixture_dir(tmp_path_factory):
    """Builds a real (tiny) HF tokenizers WordLevel tokenizer.json."""
    import tokenizers
    from tokenizers import models, pre_tokenizers

    words = ["hello", "world", "foo", "bar", "baz", "the", "quick", "brown"]
    vocab = {w: i for i, w in enumerate(words)}
    vocab["[UNK]"] = len(vocab)
    tok = tokenizers.Tokenizer(models.WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()

    root = tmp_path_factory.mktemp("tokenizers")
    plain = root / "test-model"
    plain.mkdir()
    tok.save(str(plain / "tokenizer.json"))
    # HF-cache layout for discovery
    hf = root / "models--test-org--cached-model" / "snapshots" / "abc123"
    hf.mkdir(parents=True)
    tok.save(str(hf / "tokenizer.json"))
    return str(root)


class TestLocalProvider:
    def test_discovery_plain_and_hf_layout(self, tokenizer_fixture_dir):
        mapping = discover_local_tokenizers(tokenizer_fixture_dir)
        assert "test-model" in mapping
        assert "test-org/cached-model" in mapping

    def test_encode_with_offsets(self, tokenizer_fixture_dir):
        tok = new_cached_local_tokenizer(
            LocalTokenizerConfig(auto_discover_dir=tokenizer_fixture_dir)
        )
        ids, offsets = tok.encode("hello world foo", "test-model")
        assert len(ids) == 3
        assert offsets == [(0, 5), (6, 11), (12, 15)]

    def test_unknown_model_raises(self, tokenizer_fixture_dir):
        tok = new_cached_local_tokenizer(
            LocalTokenizerConfig(auto_discover_dir=tokenizer_fixture_dir)
        )
        with pytest.raises(TokenizationError):
            tok.encode("x", "nope/never")

    def test_cache_returns_same_instance(self, tokenizer_fixture_dir):
        tok = new_cached_local_tokenizer(
            LocalTokenizerConfig(auto_discover_dir=tokenizer_fixture_dir)
        )
        a = tok._get("test-model")
        b = tok._get("test-model")
        assert a is b


class FailingTokenizer(Tokenizer):
    @property
    def type(self):
        return "failing"

    def encode(self, prompt, model_name, add_special_tokens=True):
        raise TokenizationError("backend down")

    def render_chat_template(self, req):
        raise TokenizationError("backend down")


class StaticTokenizer(Tokenizer):
    @property
    def type(self):
        return "static"

    def encode(self, prompt, model_name, add_special_tokens=True):
        n = len(prompt) // 4
        return [7] * n, [(i * 4, (i + 1) * 4) for i in range(n)]

    def render_chat_template(self, req):
        return "rendered!"


class TestComposite:
    def test_fallback_chain(self):
        comp = CompositeTokenizer([FailingTokenizer(), StaticTokenizer()])
        ids, offsets = comp.encode("x" * 16, "m")
        assert ids == [7, 7, 7, 7]

    def test_all_fail_raises_with_accumulated_errors(self):
        comp = CompositeTokenizer([FailingTokenizer(), FailingTokenizer()])
        with pytest.raises(TokenizationError, match="failing"):
            comp.encode("x", "m")

    def test_type_name(self):
        comp = CompositeTokenizer([FailingTokenizer(), StaticTokenizer()])
        assert comp.type == "composite(failing,static)"


class CountingTokenizer(StaticTokenizer):
    def __init__(self):
        self.encode_calls = 0

    def encode(self, prompt, model_name, add_special_tokens=True):
        self.encode_calls += 1
        return super().encode(prompt, model_name, add_special_tokens)


class TestPool:
    def test_sync_tokenize_and_prefix_cache_skip(self):
        tok = CountingTokenizer()
        store = LRUTokenStore(LRUStoreConfig(cache_size=1000, block_size=16))
        pool = TokenizationPool(
            TokenizationConfig(workers_count=2), indexer=store, tokenizer=tok
        )
        pool.run()
        try:
            prompt = "abcd" * 16  # 64 chars -> 16 tokens
            t1 = pool.tokenize(None, prompt, "m")
            assert len(t1) == 16
            assert tok.encode_calls == 1
            # same prompt again: full coverage -> cached, no re-encode
            t2 = pool.tokenize(None, prompt, "m")
            assert t2 == t1
            assert tok.encode_calls == 1
            # sufficiently-extended prompt: coverage drops below 0.8 ->
            # full re-encode
            pool.tokenize(None, prompt + "zzzz" * 8, "m")
            assert tok.encode_calls == 2
        finally:
            pool.shutdown()

    def test_async_enqueue(self):
        tok = CountingTokenizer()
        store = LRUTokenStore(LRUStoreConfig(cache_size=1000, block_size=16))
        pool = TokenizationPool(
            TokenizationConfig(workers_count=2), indexer=store, tokenizer=tok
        )
        pool.run()
        try:
            pool.enqueue_tokenization(None, "abcd" * 16, "m")
            deadline = time.monotonic() + 5
            while tok.encode_calls == 0 and time.monotonic() < deadline:
                time.sleep(0.01)
            assert tok.encode_calls == 1
        finally:
            pool.shutdown()

    def test_render_req_renders_first(self):
        class RenderCheck(StaticTokenizer):
            def render_chat_template(self, req):
                return "abcd" * 8

        pool = TokenizationPool(
            TokenizationConfig(workers_count=1), tokenizer=RenderCheck()
        )
        tokens = pool.tokenize(object(), "ignored", "m")
        assert len(tokens) == 8

    def test_error_propagates_to_caller(self):
        pool = TokenizationPool(
            TokenizationConfig(workers_count=1), tokenizer=FailingTokenizer()
        )
        pool.run()
        try:
            with pytest.raises(TokenizationError):
                pool.tokenize(None, "x" * 100, "m")
        finally:
            pool.shutdown()


@pytest.fixture
def uds_server(tokenizer_fixture_dir, tmp_path):
    """Runs the real aiohttp UDS sidecar in a background thread."""
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from services.uds_tokenizer import server as uds

    sock_path = str(tmp_path / "tok.socket")
    uds.core.reload({"local_dir": tokenizer_fixture_dir})
    app = uds.make_app()

    loop = asyncio.new_event_loop()
    started = threading.Event()
    runner_box = {}

    def run():
        asyncio.set_event_loop(loop)

        async def start():
            runner = web_runner = None
            from aiohttp import web

            runner = web.AppRunner(app)
            await runner.setup()
            site = web.UnixSite(runner, sock_path)
            await site.start()
            runner_box["runner"] = runner
            started.set()

        loop.run_until_complete(start())
        loop.run_forever()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    assert started.wait(10)
    yield sock_path
    loop.call_soon_threadsafe(loop.stop)
    t.join(timeout=5)


class TestUdsService:
    def test_tokenize_round_trip(self, uds_server):
        from llmd_kvcache_amd.tokenization.uds import (
            UdsTokenizer,
            UdsTokenizerConfig,
        )

        client = UdsTokenizer(UdsTokenizerConfig(socket_path=uds_server))
        ids, offsets = client.encode("hello world", "test-model")
        assert len(ids) == 2
        assert offsets == [(0, 5), (6, 11)]

    def test_health_and_config(self, uds_server):
        import http.client
        import socket as socketlib

        class Conn(http.client.HTTPConnection):
            def __init__(self):
                super().__init__("localhost", timeout=5)

            def connect(self):
                s = socketlib.socket(socketlib.AF_UNIX, socketlib.SOCK_STREAM)
                s.connect(uds_server)
                self.sock = s

        conn = Conn()
        conn.request("GET", "/health")
        assert json.loads(conn.getresponse().read())["status"] == "ok"
        conn = Conn()
        conn.request("GET", "/config")
        cfg = json.loads(conn.getresponse().read())
        assert "local_dir" in cfg
        # hot reload
        conn = Conn()
        body = json.dumps({"add_special_tokens": False}).encode()
        conn.request("POST", "/config", body=body,
                     headers={"Content-Type": "application/json"})
        resp = json.loads(conn.getresponse().read())
        assert resp["config"]["add_special_tokens"] is False

    def test_bad_request_400(self, uds_server):
        from llmd_kvcache_amd.tokenization.uds import (
            UdsTokenizer,
            UdsTokenizerConfig,
        )

        client = UdsTokenizer(
            UdsTokenizerConfig(socket_path=uds_server, max_retries=0)
        )
        with pytest.raises(TokenizationError):
            client._post("/tokenize", {"nope": 1})

    def test_retry_on_dead_socket(self, tmp_path):
        from llmd_kvcache_amd.tokenization.uds import (
            UdsTokenizer,
            UdsTokenizerConfig,
        )

        client = UdsTokenizer(
            UdsTokenizerConfig(
                socket_path=str(tmp_path / "missing.socket"),
                max_retries=1,
                backoff_base_s=0.01,
            )
        )
        t0 = time.monotonic()
        with pytest.raises(TokenizationError):
            client.encode("x", "m")
        assert time.monotonic() - t0 >= 0.01  # backoff happened


class FlakyTokenizer(StaticTokenizer):
    """Fails the first N encodes, then succeeds."""

    def __init__(self, fail_times):
        self.fail_times = fail_times
        self.calls = 0

    def encode(self, prompt, model_name, add_special_tokens=True):
        self.calls += 1
        if self.calls <= self.fail_times:
            raise TokenizationError("transient failure")
        return super().encode(prompt, model_name, add_special_tokens)


class TestAsyncRetry:
    def test_fire_and_forget_retries_with_backoff(self):
        tok = FlakyTokenizer(fail_times=1)
        pool = TokenizationPool(
            TokenizationConfig(workers_count=1), tokenizer=tok
        )
        pool.run()
        try:
            pool.enqueue_tokenization(None, "abcd" * 16, "m")
            deadline = time.monotonic() + 5
            while tok.calls < 2 and time.monotonic() < deadline:
                time.sleep(0.02)
            assert tok.calls >= 2  # retried after the transient failure
        finally:
            pool.shutdown()

    def test_sync_does_not_retry(self):
        tok = FlakyTokenizer(fail_times=1)
        pool = TokenizationPool(
            TokenizationConfig(workers_count=1), tokenizer=tok
        )
        pool.run()
        try:
            with pytest.raises(TokenizationError):
                pool.tokenize(None, "abcd" * 16, "m")
            assert tok.calls == 1
        finally:
            pool.shutdown()


class TestSingleFlight:
    def test_concurrent_loads_deduplicate(self):
        from llmd_kvcache_amd.tokenization.tokenizer import _SingleFlight

        sf = _SingleFlight()
        calls = []
        results = []
        errors = []

        def slow_load():
            calls.append(1)
            time.sleep(0.1)
            return "loaded"

        def worker():
            try:
                results.append(sf.do("k", slow_load))
            except Exception as e:  # pragma: no cover
                errors.append(e)

        threads = [threading.Thread(target=worker) for _ in range(16)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert errors == []
        assert results == ["loaded"] * 16
        assert len(calls) == 1  # exactly one real load

    def test_leader_failure_propagates_then_retries(self):
        from llmd_kvcache_amd.tokenization.tokenizer import _SingleFlight

        sf = _SingleFlight()
        with pytest.raises(TokenizationError):
            sf.do("k", lambda: (_ for _ in ()).throw(
                TokenizationError("boom")))
        # a later call is a fresh flight
        assert sf.do("k", lambda: 42) == 42


class TestInlineModeConcurrency:
    def test_concurrent_inline_tokenize_does_not_cross_tasks(self):
        """ADVICE round-1: two concurrent callers on an UNSTARTED pool
        must each get their own result (the old inline drain could
        process the OTHER caller's task and leave this caller blocked)."""
        import threading

        from llmd_kvcache_amd.tokenization.pool import (TokenizationConfig,
                                                        TokenizationPool)

        class Tok:
            def encode(self, prompt, model):
                return [len(prompt)] * 3, [(0, len(prompt))] * 3

            def render_chat_template(self, req):
                raise AssertionError("not used")

        pool = TokenizationPool(
            TokenizationConfig(workers_count=2), tokenizer=Tok()
        )
        assert not pool._running  # inline mode
        results = {}
        errs = []

        def call(tag, prompt):
            try:
                results[tag] = pool.tokenize(None, prompt, "m")
            except Exception as e:  # pragma: no cover
                errs.append(e)

        threads = [
            threading.Thread(target=call, args=(i, "x" * (i + 1)))
            for i in range(8)
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=10)
        assert not errs
        assert all(not t.is_alive() for t in threads)  # nobody deadlocked
        for i in range(8):
            assert results[i] == [i + 1] * 3


class TestUdsMultiWorker:
    """Prefork multi-worker sidecar (reference runs gunicorn N workers
    with flock-guarded init; here: shared-listener prefork supervisor
    with worker restart)."""

    def _spawn(self, sock_path, tokenizer_dir, workers=2):
        import subprocess
        import sys
        import time as _time

        root = os.path.join(os.path.dirname(__file__), "..")
        env = dict(os.environ, LOCAL_TOKENIZER_DIR=tokenizer_dir,
                   PYTHONPATH=os.path.abspath(root))
        proc = subprocess.Popen(
            [sys.executable,
             os.path.join(root, "services", "uds_tokenizer", "server.py"),
             "--socket", sock_path, "--workers", str(workers)],
            env=env, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        for _ in range(100):
            if os.path.exists(sock_path):
                break
            _time.sleep(0.1)
        else:
            proc.terminate()
            raise RuntimeError("sidecar did not come up")
        return proc

    def test_prefork_serves_and_restarts(self, tokenizer_fixture_dir,
                                         tmp_path):
        import signal
        import subprocess
        import threading
        import time as _time

        from llmd_kvcache_amd.tokenization.uds import (UdsTokenizer,
                                                       UdsTokenizerConfig)

        sock_path = str(tmp_path / "mw.socket")
        proc = self._spawn(sock_path, tokenizer_fixture_dir, workers=2)
        try:
            client = UdsTokenizer(UdsTokenizerConfig(
                socket_path=sock_path, timeout_s=10.0))
            results, errs = [], []

            def call():
                try:
                    ids, offs = client.encode("hello world", "test-model")
                    results.append(ids)
                except Exception as e:  # pragma: no cover
                    errs.append(e)

            threads = [threading.Thread(target=call) for _ in range(8)]
            for t in threads:
                t.start()
            for t in threads:
                t.join(timeout=30)
            assert not errs
            assert len(results) == 8 and all(len(r) == 2 for r in results)

            # kill one worker: the supervisor must restart it and the
            # service keeps answering
            out = subprocess.run(
                ["pgrep", "-P", str(proc.pid)], capture_output=True,
                text=True)
            kids = [int(x) for x in out.stdout.split()]
            assert len(kids) == 2
            os.kill(kids[0], signal.SIGKILL)
            deadline = _time.monotonic() + 15
            while _time.monotonic() < deadline:
                out = subprocess.run(["pgrep", "-P", str(proc.pid)],
                                     capture_output=True, text=True)
                now = [int(x) for x in out.stdout.split()]
                if len(now) == 2 and set(now) != set(kids):
                    break
                _time.sleep(0.2)
            ids, _ = client.encode("hello world", "test-model")
            assert len(ids) == 2
        finally:
            proc.terminate()
            proc.wait(timeout=10)

    def test_flock_preload_serializes(self, tokenizer_fixture_dir,
                                      tmp_path):
        """Two cores preloading under the same lock never overlap."""
        import threading
        import time as _time

        sys_path_root = os.path.join(os.path.dirname(__file__), "..")
        import sys

        sys.path.insert(0, sys_path_root)
        from services.uds_tokenizer.server import TokenizerCore

        lock_path = str(tmp_path / "init.lock")
        active = []
        overlaps = []

        class SlowCore(TokenizerCore):
            def _load(self, model):
                active.append(1)
                if len(active) - len(overlaps) > 1:  # pragma: no cover
                    overlaps.append(1)
                _time.sleep(0.2)
                active.pop()
                return object()

        cores = [SlowCore({"local_dir": tokenizer_fixture_dir})
                 for _ in range(3)]
        threads = [
            threading.Thread(target=c.preload,
                             args=(["test-model"], lock_path))
            for c in cores
        ]
        t0 = _time.monotonic()
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=10)
        assert not overlaps
        assert _time.monotonic() - t0 >= 0.55  # serialized, not parallel


class TestTokenizeBatch:
    def test_batch_matches_sequential_and_uses_store(
            self, tokenizer_fixture_dir):
        from llmd_kvcache_amd.tokenization.pool import (TokenizationConfig,
                                                        TokenizationPool)
        from llmd_kvcache_amd.tokenization.tokenizer import (
            LocalTokenizerConfig, new_cached_local_tokenizer)

        tok = new_cached_local_tokenizer(
            LocalTokenizerConfig(auto_discover_dir=tokenizer_fixture_dir))
        pool = TokenizationPool(TokenizationConfig(), tokenizer=tok)
        # prompts must span full 256-char prefix-store blocks to be
        # cacheable (lru_store semantics: partial tail chunks drop)
        prompts = ["hello world " * 300, "hello there " * 280,
                   "hello world " * 300]
        cold = pool.tokenize_batch(prompts, "test-model")
        # sequential warm calls (store populated by the batch above);
        # warm results are token-PREFIX subsets of the cold encodes -
        # reference pool.go semantics (coverage>=0.8 returns the cached
        # prefix, which excludes tokens in the partial tail block)
        seq_warm = [pool.tokenize(None, p, "test-model") for p in prompts]
        calls = {"n": 0}
        orig = tok.encode_batch

        def counting(*a, **kw):
            calls["n"] += 1
            return orig(*a, **kw)

        tok.encode_batch = counting
        warm = pool.tokenize_batch(prompts, "test-model")
        assert calls["n"] == 0          # all from the prefix store
        assert warm == seq_warm          # batch == sequential, same state
        for c, w in zip(cold, warm):
            assert len(w) <= len(c) and c[:len(w)] == w

    def test_composite_encode_batch_falls_back(self):
        from llmd_kvcache_amd.tokenization.tokenizer import (
            CompositeTokenizer, Tokenizer)

        class Bad(Tokenizer):
            @property
            def type(self):
                return "bad"

            def encode(self, *a, **kw):
                raise RuntimeError("nope")

        class Good(Tokenizer):
            @property
            def type(self):
                return "good"

            def encode(self, prompt, model, add_special_tokens=True):
                return [len(prompt)], [(0, len(prompt))]

        comp = CompositeTokenizer([Bad(), Good()])
        out = comp.encode_batch(["ab", "abc"], "m")
        assert out == [([2], [(0, 2)]), ([3], [(0, 3)])]
