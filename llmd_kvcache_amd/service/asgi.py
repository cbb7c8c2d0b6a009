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
    from fastapi import FastAPI, HTTPException
    from fastapi.responses import PlainTextResponse
    from pydantic import BaseModel

    class ScoreRequest(BaseModel):
        prompt: str
        model: str = ""

    class ChatScoreRequest(BaseModel):
        model: str = ""
        messages: List[Dict[str, Any]] = []
        chat_template: Optional[str] = None
        tools: Optional[List[Dict[str, Any]]] = None
        documents: Optional[List[Dict[str, Any]]] = None
        add_generation_prompt: bool = False
        continue_final_message: bool = False
        chat_template_kwargs: Dict[str, Any] = {}

    app = FastAPI(title="llmd_kvcache_amd indexer")

    @app.post("/score_completions")
    def score_completions(req: ScoreRequest):
        if not req.prompt:
            raise HTTPException(400, "field 'prompt' required")
        return indexer.get_pod_scores(None, req.prompt, req.model, []) or {}

    @app.post("/score_chat_completions")
    def score_chat_completions(req: ChatScoreRequest):
        template = req.chat_template
        kwargs = dict(req.chat_template_kwargs)
        if template is None:
            fetched, tvars = cc.get_model_chat_template(
                cc.FetchChatTemplateRequest(model=req.model)
            )
            template = fetched
            merged = dict(tvars)
            merged.update(kwargs)
            kwargs = merged
        rendered = cc.render_chat_template(
            cc.RenderJinjaTemplateRequest(
                conversations=[req.messages],
                chat_template=template,
                tools=req.tools,
                documents=req.documents,
                add_generation_prompt=req.add_generation_prompt,
                continue_final_message=req.continue_final_message,
                chat_template_kwargs=kwargs,
            )
        )
        pods = indexer.get_pod_scores(None, rendered, req.model, [])
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
