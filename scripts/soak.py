"""Service soak: full stack (ZMTP events in over TCP, HTTP scores out)
against the GPU index under sustained load, with HBM/host memory tracked.

    python scripts/soak.py --seconds 180
"""
import argparse
import json
import random
import statistics
import struct
import sys
import threading
import time
import urllib.request

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
from llmd_kvcache_amd.kvblock.token_processor import TokenProcessorConfig
from llmd_kvcache_amd.kvevents.events import BlockRemoved, BlockStored, EventBatch
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
from llmd_kvcache_amd.kvevents.zmtp import PubSocket
from llmd_kvcache_amd.service.http_server import HttpService
from llmd_kvcache_amd.tokenization.pool import TokenizationPool
from llmd_kvcache_amd.tokenization.tokenizer import Tokenizer

BS = 16
MODEL = "soak-model"


class SyntheticTokenizer(Tokenizer):
    """Deterministic prompt->tokens (1 token per 4 chars)."""

    @property
    def type(self):
        return "synthetic"

    def encode(self, prompt, model_name, add_special_tokens=True):
        n = len(prompt) // 4
        toks = [(hash(prompt[i * 4:(i + 1) * 4]) & 0x7FFFFFFF)
                for i in range(n)]
        return toks, [(i * 4, (i + 1) * 4) for i in range(n)]

    def render_chat_template(self, req):
        raise NotImplementedError


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=180)
    ap.add_argument("--publishers", type=int, default=4)
    ap.add_argument("--front", choices=["http", "wire"], default="http",
                    help="scoring surface: stdlib HTTP service or the "
                         "native C++ wirefront")
    args = ap.parse_args()
    assert torch.cuda.is_available()

    cfg = Config()
    cfg.token_processor = TokenProcessorConfig(block_size=BS)
    index = GpuIndex(GpuIndexConfig(capacity=1 << 20, pods_per_key=10))
    pool = TokenizationPool(cfg.tokenizers_pool,
                            tokenizer=SyntheticTokenizer())
    pool.run()
    indexer = Indexer(cfg, tokenization_pool=pool, kv_block_index=index)
    events = EventsPool(EventsConfig(zmq_endpoint="tcp://127.0.0.1:0",
                                     concurrency=4),
                        index, indexer.tokens_processor)
    events.start(with_subscriber=True)
    if args.front == "wire":
        from llmd_kvcache_amd.service.wirefront import WireIndexerService

        front = WireIndexerService(indexer)
        front_port = front.start(port=0, n_io=4)
        score_path = "/score"
    else:
        front = HttpService(indexer, host="127.0.0.1", port=0)
        front.start()
        front_port = front.port
        score_path = "/score_completions"
    while events._subscriber.port is None:
        time.sleep(0.05)

    stop = threading.Event()
    stats = {"events": 0, "scores": 0, "errors": 0, "hits": 0}
    lock = threading.Lock()
    prompts = []  # (prompt_string, pod) pairs known to be stored

    def publisher(pid):
        rng = random.Random(pid)
        pub = PubSocket()
        pub.connect(f"tcp://127.0.0.1:{events._subscriber.port}")
        pub.wait_for_subscriber(10)
        seq = 0
        next_hash = pid * 10_000_000 + 1
        stored = []
        while not stop.is_set():
            # store a new 4-block prompt (64 tokens -> 256-char string)
            s = "".join(rng.choice("abcdefgh") for _ in range(256))
            toks, _ = SyntheticTokenizer().encode(s, MODEL)
            hs = list(range(next_hash, next_hash + 4))
            next_hash += 4
            batch = EventBatch(ts=time.time(),
                               events=[BlockStored(hs, None, toks, BS)])
            pub.send_multipart([f"kv@soak-pod-{pid}@{MODEL}".encode(),
                                struct.pack(">Q", seq), batch.encode()])
            seq += 1
            stored.append(hs)
            with lock:
                stats["events"] += 1
                prompts.append((s, f"soak-pod-{pid}"))
                if len(prompts) > 2000:
                    prompts.pop(0)
            if stored and rng.random() < 0.1:  # occasional removal
                victim = stored.pop(rng.randrange(len(stored)))
                batch = EventBatch(ts=time.time(),
                                   events=[BlockRemoved(victim)])
                pub.send_multipart([f"kv@soak-pod-{pid}@{MODEL}".encode(),
                                    struct.pack(">Q", seq), batch.encode()])
                seq += 1
                with lock:
                    stats["events"] += 1
            time.sleep(0.002)
        pub.close()

    def scorer():
        rng = random.Random(99)
        lat = []
        while not stop.is_set():
            with lock:
                if not prompts:
                    time.sleep(0.05)
                    continue
                prompt, pod = prompts[rng.randrange(len(prompts))]
            body = json.dumps({"prompt": prompt, "model": MODEL}).encode()
            t0 = time.monotonic()
            try:
                req = urllib.request.Request(
                    f"http://127.0.0.1:{front_port}{score_path}",
                    data=body,
                    headers={"Content-Type": "application/json"})
                with urllib.request.urlopen(req, timeout=5) as resp:
                    scores = json.loads(resp.read())
                if "scores" in scores:  # wirefront response envelope
                    scores = scores["scores"]
                with lock:
                    stats["scores"] += 1
                    if scores.get(pod, 0) > 0:
                        stats["hits"] += 1
                lat.append(time.monotonic() - t0)
            except Exception:
                with lock:
                    stats["errors"] += 1
        if lat:
            with lock:
                stats["score_p50_ms"] = round(
                    statistics.median(lat) * 1000, 2)

    threads = ([threading.Thread(target=publisher, args=(i,), daemon=True)
                for i in range(args.publishers)]
               + [threading.Thread(target=scorer, daemon=True)
                  for _ in range(4)])
    for t in threads:
        t.start()

    t_end = time.monotonic() + args.seconds
    mem0 = torch.cuda.memory_allocated()
    while time.monotonic() < t_end:
        time.sleep(10)
        with lock:
            snap = dict(stats)
        print(json.dumps({
            **snap,
            "hbm_alloc_mb": round(torch.cuda.memory_allocated() / 2**20, 1),
        }), flush=True)
    stop.set()
    time.sleep(1)
    events.shutdown()
    front.stop()
    indexer.shutdown()
    mem1 = torch.cuda.memory_allocated()
    with lock:
        final = dict(stats)
    final["hbm_growth_mb"] = round((mem1 - mem0) / 2**20, 1)
    hit_rate = final["hits"] / max(final["scores"], 1)
    final["hit_rate"] = round(hit_rate, 3)
    print("FINAL", json.dumps(final), flush=True)
    assert final["errors"] == 0, "scoring errors during soak"
    assert hit_rate > 0.8, f"hit rate too low: {hit_rate}"
    print("SOAK OK")


if __name__ == "__main__":
    main()
