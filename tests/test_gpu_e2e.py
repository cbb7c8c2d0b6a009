"""Full-stack GPU integration: online service shape (HTTP scoring +
Prometheus + ZMTP events pool) over the HBM-resident index on a real
MI355X - vLLM-sim publisher in, scores out."""

import json
import struct
import time
import urllib.request

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

MODEL = "test-model"
BLOCK_SIZE = 16


@pytest.fixture
def tokenizer_dir(tmp_path):
    import tokenizers
    from tokenizers import models, pre_tokenizers

    words = [f"w{i}" for i in range(100)]
    vocab = {w: i for i, w in enumerate(words)}
    vocab["[UNK]"] = len(vocab)
    tok = tokenizers.Tokenizer(models.WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    d = tmp_path / MODEL
    d.mkdir()
    tok.save(str(d / "tokenizer.json"))
    return str(tmp_path)


def test_full_service_stack_on_gpu(tokenizer_dir):
    from llmd_kvcache_amd.indexer import Config, Indexer
    from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
    from llmd_kvcache_amd.kvblock.token_processor import TokenProcessorConfig
    from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
    from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
    from llmd_kvcache_amd.kvevents.zmtp import PubSocket
    from llmd_kvcache_amd.service.http_server import HttpService
    from llmd_kvcache_amd.tokenization.pool import TokenizationPool
    from llmd_kvcache_amd.tokenization.prefixstore import LRUTokenStore
    from llmd_kvcache_amd.tokenization.tokenizer import (
        LocalTokenizerConfig,
        new_cached_local_tokenizer,
    )

    cfg = Config()
    cfg.token_processor = TokenProcessorConfig(block_size=BLOCK_SIZE)
    index = GpuIndex(GpuIndexConfig(capacity=1 << 16, pods_per_key=10))
    tokenizer = new_cached_local_tokenizer(
        LocalTokenizerConfig(auto_discover_dir=tokenizer_dir)
    )
    pool = TokenizationPool(cfg.tokenizers_pool, indexer=LRUTokenStore(),
                            tokenizer=tokenizer)
    pool.run()
    indexer = Indexer(cfg, tokenization_pool=pool, kv_block_index=index)

    events = EventsPool(
        EventsConfig(zmq_endpoint="tcp://127.0.0.1:0", concurrency=2),
        index,
        indexer.tokens_processor,
    )
    events.start(with_subscriber=True)
    http = HttpService(indexer, host="127.0.0.1", port=0)
    http.start()
    pub = PubSocket()
    try:
        deadline = time.monotonic() + 10
        while events._subscriber.port is None and time.monotonic() < deadline:
            time.sleep(0.05)
        pub.connect(f"tcp://127.0.0.1:{events._subscriber.port}")
        assert pub.wait_for_subscriber(10.0)

        # a vLLM pod reports caching a 64-token prompt (4 blocks of 16)
        prompt = " ".join(f"w{i % 100}" for i in range(64))
        tokens, _ = tokenizer.encode(prompt, MODEL)
        assert len(tokens) == 64
        batch = EventBatch(
            ts=time.time(),
            events=[BlockStored(list(range(500, 504)), None, tokens,
                                BLOCK_SIZE)],
        )
        pub.send_multipart(
            [f"kv@vllm-gpu-pod@{MODEL}".encode(), struct.pack(">Q", 1),
             batch.encode()]
        )

        # score over HTTP until the event lands in HBM
        body = json.dumps({"prompt": prompt, "model": MODEL}).encode()
        scores = {}
        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            req = urllib.request.Request(
                f"http://127.0.0.1:{http.port}/score_completions",
                data=body, headers={"Content-Type": "application/json"},
            )
            with urllib.request.urlopen(req, timeout=5) as resp:
                scores = json.loads(resp.read())
            if scores:
                break
            time.sleep(0.1)
        assert scores == {"vllm-gpu-pod": 4.0}, scores

        # metrics endpoint alive
        with urllib.request.urlopen(
            f"http://127.0.0.1:{http.port}/metrics", timeout=5
        ) as resp:
            assert resp.status == 200

        # batched scoring surface (one fused kernel for the batch)
        body = json.dumps({"prompts": [prompt, "w0 w1 w2"],
                           "model": MODEL}).encode()
        req = urllib.request.Request(
            f"http://127.0.0.1:{http.port}/score_batch",
            data=body, headers={"Content-Type": "application/json"},
        )
        with urllib.request.urlopen(req, timeout=5) as resp:
            out = json.loads(resp.read())
        assert out["scores"][0] == {"vllm-gpu-pod": 4.0}
        assert out["scores"][1] == {}
    finally:
        pub.close()
        http.stop()
        events.shutdown()
        indexer.shutdown()


def test_wirefront_on_gpu(tokenizer_dir):
    """Native wire front over the HBM table: text request (tokenization
    included), token request, prompt-cache warm repeat."""
    import socket

    from llmd_kvcache_amd.indexer import Config, Indexer
    from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
    from llmd_kvcache_amd.kvblock.keys import PodEntry
    from llmd_kvcache_amd.kvblock.token_processor import (
        ChunkedTokenDatabase, TokenProcessorConfig)
    from llmd_kvcache_amd.service.wirefront import WireIndexerService
    from llmd_kvcache_amd.tokenization.pool import TokenizationPool
    from llmd_kvcache_amd.tokenization.prefixstore import LRUTokenStore
    from llmd_kvcache_amd.tokenization.tokenizer import (
        LocalTokenizerConfig,
        new_cached_local_tokenizer,
    )

    cfg = Config()
    cfg.token_processor = TokenProcessorConfig(block_size=BLOCK_SIZE)
    index = GpuIndex(GpuIndexConfig(capacity=1 << 16, pods_per_key=10))
    tokenizer = new_cached_local_tokenizer(
        LocalTokenizerConfig(auto_discover_dir=tokenizer_dir))
    pool = TokenizationPool(cfg.tokenizers_pool, indexer=LRUTokenStore(),
                            tokenizer=tokenizer)
    pool.run()
    indexer = Indexer(cfg, tokenization_pool=pool, kv_block_index=index)

    prompt = " ".join(f"w{i % 100}" for i in range(64))
    tokens, _ = tokenizer.encode(prompt, MODEL)
    tp = ChunkedTokenDatabase(cfg.token_processor)
    keys = tp.tokens_to_kv_block_keys(None, tokens, MODEL)
    index.add(keys, keys, [PodEntry("gpu-pod", "gpu")])

    svc = WireIndexerService(indexer)
    port = svc.start(port=0, n_io=2)
    try:
        def post(obj):
            body = json.dumps(obj).encode()
            s = socket.create_connection(("127.0.0.1", port), timeout=10)
            s.sendall((f"POST /score HTTP/1.1\r\nhost: x\r\n"
                       f"content-length: {len(body)}\r\n\r\n").encode()
                      + body)
            buf = b""
            while b"\r\n\r\n" not in buf:
                buf += s.recv(65536)
            head, rest = buf.split(b"\r\n\r\n", 1)
            clen = int([ln.split(b":")[1] for ln in head.split(b"\r\n")
                        if ln.lower().startswith(b"content-length")][0])
            while len(rest) < clen:
                rest += s.recv(65536)
            s.close()
            return int(head.split(b" ")[1]), json.loads(rest[:clen])

        status, out = post({"model": MODEL, "prompt": prompt})
        assert status == 200 and out["scores"] == {"gpu-pod": 4.0}
        status, out = post({"model": MODEL, "tokens": tokens})
        assert status == 200 and out["scores"] == {"gpu-pod": 4.0}
        status, out = post({"model": MODEL, "prompt": prompt})
        assert status == 200 and out["scores"] == {"gpu-pod": 4.0}
        assert svc._front.prompt_cache_hits() >= 1
    finally:
        svc.stop()
        pool.shutdown()
