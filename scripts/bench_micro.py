"""Micro-benchmarks (parity with the reference's Go benchmarks:
tokenization stress pool_test.go:211-281, chat-template cold/warm
cgo_functions_test.go:450-545, plus the hashing hot loop).

    python scripts/bench_micro.py
"""
import random
import statistics
import string
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])


def bench(name, fn, iters=5):
    ts = []
    for _ in range(iters):
        t0 = time.perf_counter()
        n = fn()
        ts.append((time.perf_counter() - t0) / max(n, 1))
    best = min(ts)
    print(f"{name:<46} {best*1e6:10.2f} us/op  "
          f"(median {statistics.median(ts)*1e6:.2f})", flush=True)


def main():
    rng = random.Random(42)
    words = ["".join(rng.choices(string.ascii_lowercase, k=rng.randint(2, 9)))
             for _ in range(2000)]

    # --- hashing hot loop (C++ vs Python) -----------------------------
    from llmd_kvcache_amd.ops import cpu_ext
    from llmd_kvcache_amd.utils import hashing

    tokens = [rng.randrange(0, 1 << 31) for _ in range(8192)]
    mod = cpu_ext.maybe_load()

    def py_chain():
        h = hashing.init_hash("")
        for c in range(len(tokens) // 16):
            h = hashing.chunk_hash(h, tokens[c * 16:(c + 1) * 16])
        return len(tokens) // 16

    bench("hash chain (python, per chunk)", py_chain, 3)
    if mod:
        bench("hash chain (C++ native, per chunk)",
              lambda: len(mod.tokens_to_chunk_hashes(
                  tokens, hashing.init_hash(""), 16)), 10)

    # --- tokenization stress (sync pool) ------------------------------
    import tokenizers as hf_tok
    from tokenizers import models, pre_tokenizers

    vocab = {w: i for i, w in enumerate(dict.fromkeys(words))}
    vocab["[UNK]"] = len(vocab)
    tk = hf_tok.Tokenizer(models.WordLevel(vocab, unk_token="[UNK]"))
    tk.pre_tokenizer = pre_tokenizers.Whitespace()

    from llmd_kvcache_amd.tokenization.pool import (
        TokenizationConfig,
        TokenizationPool,
    )
    from llmd_kvcache_amd.tokenization.tokenizer import Tokenizer

    class Local(Tokenizer):
        @property
        def type(self):
            return "local"

        def encode(self, prompt, model_name, add_special_tokens=True):
            enc = tk.encode(prompt)
            return list(enc.ids), [tuple(o) for o in enc.offsets]

    prompts = [" ".join(rng.choices(words, k=rng.randint(50, 400)))
               for _ in range(500)]
    pool = TokenizationPool(TokenizationConfig(workers_count=5),
                            tokenizer=Local())
    pool.run()

    def sync_stress():
        for p in prompts:
            pool.tokenize(None, p, "m")
        return len(prompts)

    bench("sync tokenization (cold+warm mix, per prompt)", sync_stress, 3)
    pool.shutdown()

    # --- chat template render cold/warm -------------------------------
    from llmd_kvcache_amd.preprocessing import chat_completions as cc

    template = ("{% for m in messages %}<|{{ m['role'] }}|>{{ m['content'] }}"
                "{% endfor %}")
    req = cc.RenderJinjaTemplateRequest(
        conversations=[[{"role": "user", "content": "hello " * 50}]],
        chat_template=template,
    )
    t0 = time.perf_counter()
    cc.render_jinja_template(req)
    print(f"{'chat template render (cold)':<46} "
          f"{(time.perf_counter()-t0)*1e6:10.2f} us/op", flush=True)
    bench("chat template render (warm)",
          lambda: (cc.render_jinja_template(req), 1)[1], 5)

    # --- prefix store ---------------------------------------------------
    from llmd_kvcache_amd.tokenization.prefixstore import LRUTokenStore

    store = LRUTokenStore()
    big = " ".join(rng.choices(words, k=3000))
    enc = tk.encode(big)
    store.add_tokenization(big, list(enc.ids), [tuple(o) for o in enc.offsets])
    bench("prefix store lookup (full hit, per call)",
          lambda: (store.find_longest_contained_tokens(big), 1)[1], 10)


if __name__ == "__main__":
    main()
