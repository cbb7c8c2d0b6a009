"""ASGI service front - a dependency-free ASGI 3.0 application with the
same endpoint surface as the reference's online binary (main.go:269-365),
deployable under uvicorn/gunicorn:

    uvicorn --factory llmd_kvcache_amd.service.asgi:create_default_app

(Written as a raw ASGI callable rather than on a web framework: the
payloads mirror the reference's loosely-typed JSON exactly and the
module has zero extra dependencies.)
"""

from __future__ import annotations

import json
from typing import Any, Dict

from ..indexer import Indexer
from ..preprocessing import chat_completions as cc


class AsgiIndexerApp:
    def __init__(self, indexer: Indexer, coalesce: bool = True):
        """coalesce=True (default) batches concurrent /score_completions
        requests into shared fused kernel launches (service/coalesce.py);
        the blocking tokenize+score runs on the default executor so the
        event loop keeps accepting requests while a batch is in flight."""
        self.indexer = indexer
        self.coalescer = None
        if (coalesce and hasattr(indexer, "tokenizers_pool")
                and hasattr(indexer, "score_tokens_batch")):
            from .coalesce import CoalescingScorer

            self.coalescer = CoalescingScorer(indexer)
            self.coalescer.start()

    def _score_one_blocking(self, prompt: str, model: str, pods):
        tokens = self.indexer.tokenizers_pool.tokenize(None, prompt, model)
        return self.coalescer.score(tokens, model, pods)

    async def __call__(self, scope, receive, send):
        if scope["type"] == "lifespan":
            await self._lifespan(receive, send)
            return
        if scope["type"] != "http":
            return
        method = scope["method"]
        path = scope["path"]
        try:
            if method == "POST" and path == "/score_completions":
                body = await self._read_json(receive)
                await self._score_completions(send, body)
            elif method == "POST" and path == "/score_batch":
                body = await self._read_json(receive)
                await self._score_batch(send, body)
            elif method == "POST" and path == "/score_chat_completions":
                body = await self._read_json(receive)
                await self._score_chat(send, body)
            elif method == "GET" and path == "/health":
                await self._json(send, 200, {"status": "ok"})
            elif method == "GET" and path == "/metrics":
                await self._metrics(send)
            else:
                await self._text(send, 404, "not found")
        except _BadRequest as e:
            await self._text(send, 400, str(e))
        except Exception as e:  # pragma: no cover - defensive
            await self._text(send, 500, f"error: {e}")

    # -- handlers ------------------------------------------------------
    async def _score_completions(self, send, body: Dict[str, Any]):
        prompt = body.get("prompt", "")
        if not prompt:
            raise _BadRequest("field 'prompt' required")
        model = body.get("model", "")
        if self.coalescer is not None:
            import asyncio

            loop = asyncio.get_running_loop()
            scores = await loop.run_in_executor(
                None, self._score_one_blocking, prompt, model,
                body.get("pods", []))
        else:
            scores = self.indexer.get_pod_scores(None, prompt, model, [])
        await self._json(send, 200, scores or {})

    async def _score_batch(self, send, body: Dict[str, Any]):
        prompts = body.get("prompts")
        if not isinstance(prompts, list) or not prompts:
            raise _BadRequest("field 'prompts' (non-empty list) required")
        model = body.get("model", "")
        pool = self.indexer.tokenizers_pool
        if hasattr(pool, "tokenize_batch"):
            token_lists = pool.tokenize_batch(list(prompts), model)
        else:
            token_lists = [pool.tokenize(None, p, model) for p in prompts]
        scores = self.indexer.score_tokens_batch(
            token_lists, model, body.get("pods", []))
        await self._json(send, 200, {"scores": scores})

    async def _score_chat(self, send, body: Dict[str, Any]):
        model = body.get("model", "")
        template = body.get("chat_template")
        kwargs = dict(body.get("chat_template_kwargs") or {})
        if template is None:
            fetched, tvars = cc.get_model_chat_template(
                cc.FetchChatTemplateRequest(model=model)
            )
            template = fetched
            merged = dict(tvars)
            merged.update(kwargs)
            kwargs = merged
        rendered = cc.render_chat_template(
            cc.RenderJinjaTemplateRequest(
                conversations=[body.get("messages", [])],
                chat_template=template,
                tools=body.get("tools"),
                documents=body.get("documents"),
                add_generation_prompt=body.get("add_generation_prompt", False),
                continue_final_message=body.get("continue_final_message",
                                                False),
                chat_template_kwargs=kwargs,
            )
        )
        pods = self.indexer.get_pod_scores(None, rendered, model, [])
        await self._json(send, 200, {"podScores": pods or {},
                                     "templated_messages": rendered})

    async def _metrics(self, send):
        from prometheus_client import generate_latest

        await self._raw(send, 200, generate_latest(),
                        b"text/plain; version=0.0.4")

    # -- plumbing ------------------------------------------------------
    @staticmethod
    async def _lifespan(receive, send):
        while True:
            message = await receive()
            if message["type"] == "lifespan.startup":
                await send({"type": "lifespan.startup.complete"})
            elif message["type"] == "lifespan.shutdown":
                await send({"type": "lifespan.shutdown.complete"})
                return

    @staticmethod
    async def _read_json(receive) -> Dict[str, Any]:
        chunks = []
        while True:
            message = await receive()
            chunks.append(message.get("body", b""))
            if not message.get("more_body"):
                break
        try:
            body = json.loads(b"".join(chunks))
        except Exception:
            raise _BadRequest("invalid JSON body")
        if not isinstance(body, dict):
            raise _BadRequest("JSON object expected")
        return body

    @staticmethod
    async def _raw(send, status: int, payload: bytes, ctype: bytes):
        await send({
            "type": "http.response.start",
            "status": status,
            "headers": [(b"content-type", ctype),
                        (b"content-length", str(len(payload)).encode())],
        })
        await send({"type": "http.response.body", "body": payload})

    async def _json(self, send, status: int, obj) -> None:
        await self._raw(send, status, json.dumps(obj).encode(),
                        b"application/json")

    async def _text(self, send, status: int, msg: str) -> None:
        await self._raw(send, status, msg.encode(), b"text/plain")


class _BadRequest(Exception):
    pass


def build_app(indexer: Indexer) -> AsgiIndexerApp:
    return AsgiIndexerApp(indexer)


def create_default_app() -> AsgiIndexerApp:
    """uvicorn --factory entry point: builds an indexer from env (same
    surface as examples/online_service.py)."""
    import os

    from ..indexer import Config
    from ..kvblock.index import new_index
    from ..kvblock.token_processor import TokenProcessorConfig

    config = Config()
    config.token_processor = TokenProcessorConfig(
        block_size=int(os.environ.get("BLOCK_SIZE", "16")),
        hash_seed=os.environ.get("PYTHONHASHSEED", ""),
    )
    indexer = Indexer(config, kv_block_index=new_index(config.kv_block_index))
    indexer.run()
    return build_app(indexer)
