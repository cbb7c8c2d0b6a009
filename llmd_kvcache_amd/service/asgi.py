"""ASGI (FastAPI) service front - the production-grade alternative to the
stdlib HTTP server, same endpoint surface as the reference's online
binary (main.go:269-365):

    uvicorn --factory llmd_kvcache_amd.service.asgi:create_default_app

or embed:  app = build_app(indexer)

Scoring runs in the ASGI thread pool (the indexer is thread-safe and the
GPU fast path releases the GIL inside kernels/tokenizers).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..indexer import Indexer
from ..preprocessing import chat_completions as cc


def build_app(indexer: Indexer):
    # Raw-body handlers (no pydantic request models): keeps the endpoint
    # payloads byte-identical to the reference's loosely-typed JSON and
    # independent of the installed fastapi/pydantic major versions.
    from fastapi import FastAPI, HTTPException, Request
    from fastapi.responses import PlainTextResponse

    app = FastAPI(title="llmd_kvcache_amd indexer")

    async def _json(request: Request) -> Dict[str, Any]:
        try:
            body = await request.json()
        except Exception:
            raise HTTPException(400, "invalid JSON body")
        if not isinstance(body, dict):
            raise HTTPException(400, "JSON object expected")
        return body

    @app.post("/score_completions")
    async def score_completions(request: Request):
        body = await _json(request)
        prompt = body.get("prompt", "")
        if not prompt:
            raise HTTPException(400, "field 'prompt' required")
        model = body.get("model", "")
        return indexer.get_pod_scores(None, prompt, model, []) or {}

    @app.post("/score_chat_completions")
    async def score_chat_completions(request: Request):
        body = await _json(request)
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
                add_generation_prompt=body.get("add_generation_prompt",
                                               False),
                continue_final_message=body.get("continue_final_message",
                                                False),
                chat_template_kwargs=kwargs,
            )
        )
        pods = indexer.get_pod_scores(None, rendered, model, [])
        return {"podScores": pods or {}, "templated_messages": rendered}

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.get("/metrics")
    def metrics():
        from prometheus_client import generate_latest

        return PlainTextResponse(generate_latest(),
                                 media_type="text/plain; version=0.0.4")

    return app


def create_default_app():
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
