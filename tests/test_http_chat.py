"""HTTP /score_chat_completions endpoint (explicit template; no model
fetch needed) + error paths - completes the online-binary surface tests."""

import json
import urllib.error
import urllib.request

import pytest

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.keys import PodEntry
from llmd_kvcache_amd.service.http_server import HttpService
from llmd_kvcache_amd.tokenization.pool import TokenizationPool
from llmd_kvcache_amd.tokenization.tokenizer import Tokenizer


class CharTokenizer(Tokenizer):
    @property
    def type(self):
        return "char"

    def encode(self, prompt, model_name, add_special_tokens=True):
        n = len(prompt) // 2
        return ([ord(prompt[i * 2]) for i in range(n)],
                [(i * 2, (i + 1) * 2) for i in range(n)])

    def render_chat_template(self, req):
        raise NotImplementedError


@pytest.fixture
def service():
    cfg = Config()
    cfg.token_processor.block_size = 4
    index = InMemoryIndex(InMemoryIndexConfig(size=10_000, pod_cache_size=10))
    pool = TokenizationPool(cfg.tokenizers_pool, tokenizer=CharTokenizer())
    idx = Indexer(cfg, tokenization_pool=pool, kv_block_index=index)
    svc = HttpService(idx, host="127.0.0.1", port=0)
    svc.start()
    yield svc, idx, index
    svc.stop()


def post(port, path, payload):
    req = urllib.request.Request(
        f"http://127.0.0.1:{port}{path}",
        data=json.dumps(payload).encode(),
        headers={"Content-Type": "application/json"},
    )
    with urllib.request.urlopen(req, timeout=5) as resp:
        return json.loads(resp.read())


TEMPLATE = ("{% for m in messages %}<{{ m['role'] }}>{{ m['content'] }}"
            "{% endfor %}")


class TestScoreChatCompletions:
    def test_renders_and_scores(self, service):
        svc, idx, index = service
        messages = [{"role": "user", "content": "hello there friend"}]
        # precompute the rendered prompt and store its blocks for pod-c
        from llmd_kvcache_amd.preprocessing import chat_completions as cc

        rendered = cc.render_chat_template(
            cc.RenderJinjaTemplateRequest(conversations=[messages],
                                          chat_template=TEMPLATE))
        tokens, _ = CharTokenizer().encode(rendered, "m")
        keys = idx.tokens_processor.tokens_to_kv_block_keys(None, tokens, "m")
        index.add(keys, keys, [PodEntry("pod-c", "gpu")])

        out = post(svc.port, "/score_chat_completions", {
            "model": "m",
            "messages": messages,
            "chat_template": TEMPLATE,
        })
        assert out["templated_messages"] == rendered
        assert out["podScores"].get("pod-c", 0) > 0

    def test_bad_json_400(self, service):
        svc, *_ = service
        req = urllib.request.Request(
            f"http://127.0.0.1:{svc.port}/score_chat_completions",
            data=b"{nope",
            headers={"Content-Type": "application/json"},
        )
        with pytest.raises(urllib.error.HTTPError) as ei:
            urllib.request.urlopen(req, timeout=5)
        assert ei.value.code == 400

    def test_unknown_path_404(self, service):
        svc, *_ = service
        req = urllib.request.Request(
            f"http://127.0.0.1:{svc.port}/nope", data=b"{}",
            headers={"Content-Type": "application/json"})
        with pytest.raises(urllib.error.HTTPError) as ei:
            urllib.request.urlopen(req, timeout=5)
        assert ei.value.code == 404
