"""Chat-completions preprocessing: template fetch + Jinja rendering.

The reference runs this inside an embedded CPython interpreter bridged over
a C ABI (pkg/preprocessing/chat_completions/cgo_functions.c:40-301 and
render_jinja_template_wrapper.py:81-207).  The MI355X-native host layer is
already Python, so the whole CGo/embed bridge disappears: we call
``transformers.utils.chat_template_utils.render_jinja_template`` directly,
keeping the same request/response shapes and the same process-level template
cache semantics (wrapper :50,163-173).
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple


@dataclass
class RenderJinjaTemplateRequest:
    """Mirrors the Go request struct (cgo_functions.go / wrapper alignment):
    ``messages`` is one conversation; wrapped into ``conversations`` for
    transformers (wrapper :113-115)."""

    conversations: List[List[Dict[str, Any]]]
    chat_template: Optional[str] = None
    tools: Optional[List[Dict[str, Any]]] = None
    documents: Optional[List[Dict[str, Any]]] = None
    return_assistant_tokens_mask: bool = False
    continue_final_message: bool = False
    add_generation_prompt: bool = False
    chat_template_kwargs: Dict[str, Any] = field(default_factory=dict)

    @staticmethod
    def from_messages(messages: List[Dict[str, Any]], **kwargs) -> "RenderJinjaTemplateRequest":
        return RenderJinjaTemplateRequest(conversations=[messages], **kwargs)


@dataclass
class RenderJinjaTemplateResponse:
    rendered_chats: List[str]
    generation_indices: List[Any]


@dataclass
class FetchChatTemplateRequest:
    model: str
    chat_template: Optional[str] = None
    revision: Optional[str] = None
    token: Optional[str] = None
    is_local_path: bool = False


# Process-level template cache: model -> (template, template_vars)
_template_cache: Dict[str, Tuple[Optional[str], Dict[str, Any]]] = {}
_cache_lock = threading.Lock()

_SPECIAL_TOKEN_ATTRS = (
    "bos_token",
    "eos_token",
    "eot_token",
    "pad_token",
    "unk_token",
    "sep_token",
    "additional_special_tokens",
)


def clear_caches() -> None:
    with _cache_lock:
        _template_cache.clear()


def get_model_chat_template(
    req: FetchChatTemplateRequest,
) -> Tuple[Optional[str], Dict[str, Any]]:
    """Fetch a model's chat template string + rendering variables, with a
    process-level cache (wrapper :130-207)."""
    cache_key = f"{req.model}@{req.revision or 'main'}@{req.chat_template or ''}"
    with _cache_lock:
        if cache_key in _template_cache:
            return _template_cache[cache_key]

    from transformers import AutoTokenizer

    kwargs: Dict[str, Any] = {}
    if req.revision:
        kwargs["revision"] = req.revision
    if req.token:
        kwargs["token"] = req.token
    tokenizer = AutoTokenizer.from_pretrained(req.model, **kwargs)

    template = req.chat_template
    if template is None:
        template = getattr(tokenizer, "chat_template", None)
        if isinstance(template, dict):
            template = template.get("default")

    template_vars: Dict[str, Any] = {}
    for attr in _SPECIAL_TOKEN_ATTRS:
        v = getattr(tokenizer, attr, None)
        if v is not None:
            template_vars[attr] = v

    with _cache_lock:
        _template_cache[cache_key] = (template, template_vars)
    return template, template_vars


def render_jinja_template(
    req: RenderJinjaTemplateRequest,
) -> RenderJinjaTemplateResponse:
    """Render conversations through the transformers Jinja engine
    (wrapper :81-127)."""
    from transformers.utils.chat_template_utils import (
        render_jinja_template as _render,
    )

    from ..metrics import collector

    t0 = time.monotonic()
    kwargs: Dict[str, Any] = dict(req.chat_template_kwargs)
    rendered_chats, generation_indices = _render(
        conversations=req.conversations,
        tools=req.tools,
        documents=req.documents,
        chat_template=req.chat_template,
        return_assistant_tokens_mask=req.return_assistant_tokens_mask,
        continue_final_message=req.continue_final_message,
        add_generation_prompt=req.add_generation_prompt,
        **kwargs,
    )
    collector.observe_render_latency(time.monotonic() - t0)
    return RenderJinjaTemplateResponse(
        rendered_chats=list(rendered_chats),
        generation_indices=list(generation_indices),
    )


def render_chat_template(req: RenderJinjaTemplateRequest) -> str:
    """Convenience wrapper returning the first rendered conversation - the
    shape the tokenization pool consumes (pkg/tokenization/pool.go:198-215)."""
    resp = render_jinja_template(req)
    if not resp.rendered_chats:
        raise ValueError("chat template rendering produced no output")
    return resp.rendered_chats[0]
