"""HTTP service surface - parity with the shipped online binary
(examples/kv_events/online/main.go:260-389):

 - POST /score_completions        {"prompt", "model"} -> {pod: score}
 - POST /score_batch              {"prompts": [...], "model"} ->
                                  {"scores": [{pod: score}, ...]}  (one fused kernel)
 - POST /score_chat_completions   chat-completions request -> {"podScores",
   "templated_messages"}  (template fetched for the model when absent)
 - GET  /metrics                  Prometheus exposition

Implemented on the stdlib http.server (threaded) to stay dependency-light
and startable from tests; the env-config surface matches the reference
(HTTP_PORT, online/main.go:41-58).
"""

from __future__ import annotations

import json
import logging
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

from ..indexer import Indexer
from ..preprocessing import chat_completions as cc

logger = logging.getLogger("llmd_kvcache_amd.http")


def make_handler(indexer: Indexer, coalescer=None):
    class Handler(BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, fmt, *args):  # route through logging
            logger.debug("http: " + fmt, *args)

        def _send_json(self, code: int, payload) -> None:
            body = json.dumps(payload).encode("utf-8")
            self.send_response(code)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def _send_error(self, code: int, msg: str) -> None:
            body = msg.encode("utf-8")
            self.send_response(code)
            self.send_header("Content-Type", "text/plain")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def _read_json(self):
            length = int(self.headers.get("Content-Length", 0))
            return json.loads(self.rfile.read(length))

        def do_GET(self):
            if self.path == "/metrics":
                try:
                    from prometheus_client import generate_latest

                    body = generate_latest()
                except Exception as e:
                    self._send_error(500, f"metrics error: {e}")
                    return
                self.send_response(200)
                self.send_header("Content-Type", "text/plain; version=0.0.4")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)
            elif self.path == "/health":
                self._send_json(200, {"status": "ok"})
            else:
                self._send_error(404, "not found")

        def do_POST(self):
            try:
                if self.path == "/score_completions":
                    self._score_completions()
                elif self.path == "/score_batch":
                    self._score_batch()
                elif self.path == "/score_chat_completions":
                    self._score_chat_completions()
                else:
                    self._send_error(404, "not found")
            except json.JSONDecodeError:
                self._send_error(400, "invalid JSON body")
            except Exception as e:
                logger.exception("request failed")
                self._send_error(500, f"error: {e}")

        def _score_completions(self):
            req = self._read_json()
            prompt = req.get("prompt", "")
            model = req.get("model", "")
            if not prompt:
                self._send_error(400, "field 'prompt' required")
                return
            if coalescer is not None:
                tokens = indexer.tokenizers_pool.tokenize(
                    None, prompt, model)
                pods = coalescer.score(tokens, model, req.get("pods", []))
            else:
                pods = indexer.get_pod_scores(None, prompt, model, [])
            self._send_json(200, pods or {})

        def _score_batch(self):
            # MI355X-native addition: N prompts -> ONE fused scoring
            # kernel (the batched hot interface the bench measures)
            req = self._read_json()
            prompts = req.get("prompts")
            model = req.get("model", "")
            if not isinstance(prompts, list) or not prompts:
                self._send_error(400, "field 'prompts' (non-empty list) "
                                      "required")
                return
            pool = indexer.tokenizers_pool
            if hasattr(pool, "tokenize_batch"):
                token_lists = pool.tokenize_batch(list(prompts), model)
            else:
                token_lists = [pool.tokenize(None, p, model)
                               for p in prompts]
            scores = indexer.score_tokens_batch(
                token_lists, model, req.get("pods", []))
            self._send_json(200, {"scores": scores})

        def _score_chat_completions(self):
            req = self._read_json()
            model = req.get("model", "")
            messages = req.get("messages", [])
            chat_template = req.get("chat_template") or None
            kwargs = req.get("chat_template_kwargs") or {}
            if chat_template is None:
                template, tvars = cc.get_model_chat_template(
                    cc.FetchChatTemplateRequest(model=model)
                )
                chat_template = template
                merged = dict(tvars)
                merged.update(kwargs)
                kwargs = merged
            render_req = cc.RenderJinjaTemplateRequest(
                conversations=[messages],
                chat_template=chat_template,
                tools=req.get("tools"),
                documents=req.get("documents"),
                add_generation_prompt=req.get("add_generation_prompt", False),
                continue_final_message=req.get("continue_final_message", False),
                chat_template_kwargs=kwargs,
            )
            rendered = cc.render_chat_template(render_req)
            pods = indexer.get_pod_scores(None, rendered, model, [])
            self._send_json(
                200,
                {"podScores": pods or {}, "templated_messages": rendered},
            )

    return Handler


class HttpService:
    def __init__(self, indexer: Indexer, host: str = "0.0.0.0",
                 port: int = 8080, coalesce: bool = True):
        """coalesce=True batches concurrent /score_completions requests
        from the threaded handlers into shared fused kernel launches
        (service/coalesce.py) - same default as the gRPC front."""
        self._coalescer = None
        if (coalesce and hasattr(indexer, "tokenizers_pool")
                and hasattr(indexer, "score_tokens_batch")):
            from .coalesce import CoalescingScorer

            self._coalescer = CoalescingScorer(indexer)
            self._coalescer.start()
        self.server = ThreadingHTTPServer(
            (host, port), make_handler(indexer, self._coalescer))
        self.port = self.server.server_address[1]
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self.server.serve_forever, name="http-service", daemon=True
        )
        self._thread.start()
        logger.info("HTTP service listening on :%d", self.port)

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
        if self._coalescer is not None:
            self._coalescer.stop()
        if self._thread:
            self._thread.join(timeout=2.0)
