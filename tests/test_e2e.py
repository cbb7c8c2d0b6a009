"""End-to-end suite: full Indexer wiring with a real (tiny) tokenizer,
events in over ZMTP, scores out over gRPC - the shape of the reference's
e2e_redis_mock suite (tests/e2e/e2e_test.go; block size 4 for small
tests like e2e_suite_test.go:72-73)."""

import struct
import time

import pytest

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.keys import PodEntry
from llmd_kvcache_amd.kvblock.token_processor import TokenProcessorConfig
from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
from llmd_kvcache_amd.tokenization.pool import TokenizationPool
from llmd_kvcache_amd.tokenization.prefixstore import (
    LRUStoreConfig,
    LRUTokenStore,
)
from llmd_kvcache_amd.tokenization.tokenizer import (
    LocalTokenizerConfig,
    new_cached_local_tokenizer,
)

BLOCK_SIZE = 4
MODEL = "test-model"


@pytest.fixture(scope="module")
def fixture_dir(tmp_path_factory):
    import tokenizers
    from tokenizers import models, pre_tokenizers

    words = [f"w{i}" for i in range(50)]
    vocab = {w: i for i, w in enumerate(words)}
    vocab["[UNK]"] = len(vocab)
    tok = tokenizers.Tokenizer(models.WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    root = tmp_path_factory.mktemp("e2e-tok")
    d = root / MODEL
    d.mkdir()
    tok.save(str(d / "tokenizer.json"))
    return str(root)


@pytest.fixture
def stack(fixture_dir):
    cfg = Config()
    cfg.token_processor = TokenProcessorConfig(block_size=BLOCK_SIZE)
    index = InMemoryIndex(InMemoryIndexConfig(size=100_000, pod_cache_size=10))
    store = LRUTokenStore(LRUStoreConfig(cache_size=10_000, block_size=16))
    tokenizer = new_cached_local_tokenizer(
        LocalTokenizerConfig(auto_discover_dir=fixture_dir)
    )
    pool = TokenizationPool(cfg.tokenizers_pool, indexer=store,
                            tokenizer=tokenizer)
    indexer = Indexer(cfg, tokenization_pool=pool, kv_block_index=index)
    indexer.tokens_indexer = store
    events = EventsPool(EventsConfig(concurrency=2), index,
                        indexer.tokens_processor)
    events.start(with_subscriber=False)
    yield indexer, index, events, tokenizer
    events.shutdown()
    indexer.shutdown()


def words(n, start=0):
    return " ".join(f"w{(start + i) % 50}" for i in range(n))


def store_prompt(indexer, index, pod, prompt, tokenizer):
    """Simulates a vLLM pod reporting it cached this prompt."""
    tokens, _ = tokenizer.encode(prompt, MODEL)
    keys = indexer.tokens_processor.tokens_to_kv_block_keys(None, tokens, MODEL)
    index.add(keys, keys, [PodEntry(pod, "gpu")])
    return len(keys)


class TestEndToEnd:
    def test_cache_miss_scores_empty(self, stack):
        indexer, *_ = stack
        scores = indexer.get_pod_scores(None, words(40), MODEL, [])
        assert scores == {}

    def test_cache_hit_scores_pod(self, stack):
        indexer, index, _, tokenizer = stack
        prompt = words(40)
        n = store_prompt(indexer, index, "pod-a", prompt, tokenizer)
        scores = indexer.get_pod_scores(None, prompt, MODEL, [])
        assert scores == {"pod-a": float(n)}

    def test_prefix_reduction(self, stack):
        """A shorter prefix of a stored prompt still hits its blocks."""
        indexer, index, _, tokenizer = stack
        prompt = words(48)
        store_prompt(indexer, index, "pod-a", prompt, tokenizer)
        short = words(24)
        scores = indexer.get_pod_scores(None, short, MODEL, [])
        assert scores.get("pod-a", 0) == 24 // BLOCK_SIZE

    def test_prefix_expansion(self, stack):
        """A longer prompt scores only the stored prefix."""
        indexer, index, _, tokenizer = stack
        prompt = words(24)
        n = store_prompt(indexer, index, "pod-a", prompt, tokenizer)
        longer = prompt + " " + words(24, start=24)
        scores = indexer.get_pod_scores(None, longer, MODEL, [])
        assert scores.get("pod-a", 0) == n

    def test_pod_filter(self, stack):
        indexer, index, _, tokenizer = stack
        prompt = words(20)
        store_prompt(indexer, index, "pod-a", prompt, tokenizer)
        store_prompt(indexer, index, "pod-b", prompt, tokenizer)
        scores = indexer.get_pod_scores(None, prompt, MODEL, ["pod-a"])
        assert "pod-b" not in scores and scores.get("pod-a", 0) > 0

    def test_long_prompt(self, stack):
        """~4500-token prompt (reference e2e_test.go:207-244)."""
        indexer, index, _, tokenizer = stack
        prompt = words(4500)
        n = store_prompt(indexer, index, "pod-long", prompt, tokenizer)
        assert n == 4500 // BLOCK_SIZE
        scores = indexer.get_pod_scores(None, prompt, MODEL, [])
        assert scores == {"pod-long": float(n)}

    def test_events_to_scores_loop(self, stack):
        """BlockStored events through the pool -> scoring sees the pod."""
        indexer, index, events, tokenizer = stack
        prompt = words(32)
        tokens, _ = tokenizer.encode(prompt, MODEL)
        from llmd_kvcache_amd.kvevents.pool import Message

        batch = EventBatch(
            ts=time.time(),
            events=[BlockStored(list(range(100, 100 + len(tokens) // BLOCK_SIZE)),
                                None, tokens, BLOCK_SIZE)],
        )
        events.add_task(Message("kv@vllm-1@" + MODEL, batch.encode(), 1,
                                "vllm-1", MODEL))
        events.drain()
        scores = indexer.get_pod_scores(None, prompt, MODEL, [])
        assert scores.get("vllm-1", 0) == len(tokens) // BLOCK_SIZE

    def test_score_tokens_fast_path(self, stack):
        indexer, index, _, tokenizer = stack
        prompt = words(16)
        store_prompt(indexer, index, "pod-t", prompt, tokenizer)
        tokens, _ = tokenizer.encode(prompt, MODEL)
        scores = indexer.score_tokens(tokens, MODEL, [])
        assert scores.get("pod-t", 0) == len(tokens) // BLOCK_SIZE


class TestChatTemplating:
    @pytest.fixture(scope="class")
    def chat_model_dir(self, tmp_path_factory):
        """Local AutoTokenizer-loadable dir with a chat template."""
        import json

        d = tmp_path_factory.mktemp("chat-model")
        vocab = {chr(97 + i): i for i in range(26)}
        vocab["[UNK]"] = 26
        import tokenizers
        from tokenizers import models

        tok = tokenizers.Tokenizer(models.WordLevel(vocab, unk_token="[UNK]"))
        tok.save(str(d / "tokenizer.json"))
        (d / "tokenizer_config.json").write_text(json.dumps({
            "tokenizer_class": "PreTrainedTokenizerFast",
            "chat_template": "{% for m in messages %}<|{{ m['role'] }}|>"
                             "{{ m['content'] }}{% endfor %}"
                             "{% if add_generation_prompt %}<|assistant|>"
                             "{% endif %}",
        }))
        return str(d)

    def test_render_jinja_template(self):
        from llmd_kvcache_amd.preprocessing import chat_completions as cc

        req = cc.RenderJinjaTemplateRequest(
            conversations=[[{"role": "user", "content": "hi"}]],
            chat_template="{% for m in messages %}[{{ m['content'] }}]"
                          "{% endfor %}",
        )
        resp = cc.render_jinja_template(req)
        assert resp.rendered_chats == ["[hi]"]

    def test_fetch_template_from_local_model(self, chat_model_dir):
        from llmd_kvcache_amd.preprocessing import chat_completions as cc

        template, tvars = cc.get_model_chat_template(
            cc.FetchChatTemplateRequest(model=chat_model_dir)
        )
        assert "<|" in template
        # cached on second fetch
        t2, _ = cc.get_model_chat_template(
            cc.FetchChatTemplateRequest(model=chat_model_dir)
        )
        assert t2 == template

    def test_full_chat_render(self, chat_model_dir):
        from llmd_kvcache_amd.preprocessing import chat_completions as cc

        template, tvars = cc.get_model_chat_template(
            cc.FetchChatTemplateRequest(model=chat_model_dir)
        )
        req = cc.RenderJinjaTemplateRequest(
            conversations=[[
                {"role": "user", "content": "hello"},
                {"role": "assistant", "content": "hey"},
            ]],
            chat_template=template,
            add_generation_prompt=True,
        )
        out = cc.render_chat_template(req)
        assert out == "<|user|>hello<|assistant|>hey<|assistant|>"

    def test_clear_caches(self):
        from llmd_kvcache_amd.preprocessing import chat_completions as cc

        cc.clear_caches()
        assert cc._template_cache == {}
