"""UDS tokenizer sidecar service.

Parity with reference services/uds_tokenizer/server.py: an aiohttp server
on a Unix domain socket (default /tmp/tokenizer/tokenizer-uds.socket,
server.py:43,286) with endpoints:

  POST /tokenize       {"prompt", "model", "add_special_tokens"?}
                       -> {"input_ids", "offset_mapping"}
  POST /chat-template  chat request -> {"rendered"}
  GET  /health         liveness
  GET  /config         current config
  POST /config         hot reload (server.py:169-209)

plus an optional TCP health-probe port (server.py:293).  Tokenizers load
through HF AutoTokenizer with an in-process cache and BOS dedup logic
mirroring tokenizer_service/tokenizer.py:225-270.

Run:  python services/uds_tokenizer/server.py [--socket PATH] [--probe-port N]
"""

from __future__ import annotations

import argparse
import asyncio
import json
import logging
import os
import threading
from typing import Any, Dict, Optional, Tuple

from aiohttp import web

logger = logging.getLogger("uds_tokenizer")

DEFAULT_SOCKET = "/tmp/tokenizer/tokenizer-uds.socket"


class TokenizerCore:
    """Loads and caches tokenizers; encodes with offsets; dedups BOS."""

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        self.config: Dict[str, Any] = {
            "local_dir": os.environ.get("LOCAL_TOKENIZER_DIR"),
            "add_special_tokens": True,
            "cache_size": 8,
        }
        if config:
            self.config.update(config)
        self._cache: Dict[str, Any] = {}
        self._lock = threading.Lock()

    def reload(self, new_config: Dict[str, Any]) -> None:
        with self._lock:
            self.config.update(new_config)
            self._cache.clear()

    def _load(self, model: str):
        with self._lock:
            tok = self._cache.get(model)
            if tok is not None:
                return tok
        import tokenizers as hf_tokenizers

        local_dir = self.config.get("local_dir")
        tok = None
        if local_dir:
            from llmd_kvcache_amd.tokenization.tokenizer import (
                discover_local_tokenizers,
            )

            path = discover_local_tokenizers(local_dir).get(model)
            if path:
                tok = hf_tokenizers.Tokenizer.from_file(path)
        if tok is None:
            try:
                tok = hf_tokenizers.Tokenizer.from_pretrained(model)
            except Exception as hf_err:
                # ModelScope fallback, parity with the reference sidecar
                # (tokenizer_service/tokenizer.py:47-207): HF-unreachable
                # deployments can still pull the tokenizer.json.
                tok = self._load_modelscope(model, hf_tokenizers, hf_err)
        with self._lock:
            if len(self._cache) >= int(self.config.get("cache_size", 8)):
                self._cache.pop(next(iter(self._cache)))
            self._cache[model] = tok
        return tok

    @staticmethod
    def _load_modelscope(model: str, hf_tokenizers, hf_err: Exception):
        try:
            from modelscope.hub.snapshot_download import snapshot_download
        except ImportError:
            raise RuntimeError(
                f"HF load failed ({hf_err}) and modelscope is not "
                "installed for fallback"
            ) from hf_err
        path = snapshot_download(model, allow_patterns=["tokenizer.json"])
        return hf_tokenizers.Tokenizer.from_file(
            os.path.join(path, "tokenizer.json"))

    def preload(self, models: list, lock_path: str) -> None:
        """Flock-guarded warmup: with N prefork workers only ONE at a
        time loads/downloads each model into the shared HF cache - the
        same serialization the reference's gunicorn.conf.py:20-21 does
        around its pre-download step."""
        import fcntl

        with open(lock_path, "a+") as lf:
            fcntl.flock(lf, fcntl.LOCK_EX)
            try:
                for m in models:
                    try:
                        self._load(m)
                        logger.info("preloaded tokenizer %s", m)
                    except Exception as e:
                        logger.error("preload of %s failed: %s", m, e)
            finally:
                fcntl.flock(lf, fcntl.LOCK_UN)

    def tokenize(
        self, prompt: str, model: str, add_special_tokens: Optional[bool]
    ) -> Tuple[list, list]:
        tok = self._load(model)
        if add_special_tokens is None:
            add_special_tokens = bool(self.config.get("add_special_tokens", True))
        enc = tok.encode(prompt, add_special_tokens=add_special_tokens)
        ids = list(enc.ids)
        offsets = [list(o) for o in enc.offsets]
        # BOS dedup: if the prompt itself starts with the BOS literal AND
        # the tokenizer prepended one, drop the duplicate
        # (tokenizer_service/tokenizer.py:225-270).
        if len(ids) >= 2 and ids[0] == ids[1] and offsets[0] == offsets[1]:
            ids, offsets = ids[1:], offsets[1:]
        return ids, offsets


core = TokenizerCore()


async def handle_tokenize(request: web.Request) -> web.Response:
    try:
        body = await request.json()
        prompt = body["prompt"]
        model = body.get("model") or body.get("model_name")
        if not model:
            raise KeyError("model")
    except (json.JSONDecodeError, KeyError) as e:
        return web.json_response({"error": f"bad request: {e}"}, status=400)
    try:
        ids, offsets = await asyncio.get_event_loop().run_in_executor(
            None, core.tokenize, prompt, model, body.get("add_special_tokens")
        )
    except Exception as e:
        logger.exception("tokenize failed")
        return web.json_response({"error": str(e)}, status=500)
    return web.json_response({"input_ids": ids, "offset_mapping": offsets})


async def handle_chat_template(request: web.Request) -> web.Response:
    try:
        body = await request.json()
    except json.JSONDecodeError as e:
        return web.json_response({"error": f"bad request: {e}"}, status=400)

    def render():
        from llmd_kvcache_amd.preprocessing import chat_completions as cc

        req = cc.RenderJinjaTemplateRequest(
            conversations=[body.get("messages", [])],
            chat_template=body.get("chat_template"),
            tools=body.get("tools"),
            documents=body.get("documents"),
            add_generation_prompt=body.get("add_generation_prompt", False),
            continue_final_message=body.get("continue_final_message", False),
            chat_template_kwargs=body.get("chat_template_kwargs") or {},
        )
        if req.chat_template is None and body.get("model"):
            template, tvars = cc.get_model_chat_template(
                cc.FetchChatTemplateRequest(model=body["model"])
            )
            req.chat_template = template
            merged = dict(tvars)
            merged.update(req.chat_template_kwargs)
            req.chat_template_kwargs = merged
        return cc.render_chat_template(req)

    try:
        rendered = await asyncio.get_event_loop().run_in_executor(None, render)
    except Exception as e:
        logger.exception("chat-template failed")
        return web.json_response({"error": str(e)}, status=500)
    return web.json_response({"rendered": rendered})


async def handle_health(request: web.Request) -> web.Response:
    return web.json_response({"status": "ok"})


async def handle_get_config(request: web.Request) -> web.Response:
    return web.json_response(core.config)


async def handle_post_config(request: web.Request) -> web.Response:
    try:
        body = await request.json()
    except json.JSONDecodeError as e:
        return web.json_response({"error": f"bad request: {e}"}, status=400)
    core.reload(body)
    return web.json_response({"status": "reloaded", "config": core.config})


def make_app() -> web.Application:
    app = web.Application()
    app.router.add_post("/tokenize", handle_tokenize)
    app.router.add_post("/chat-template", handle_chat_template)
    app.router.add_get("/health", handle_health)
    app.router.add_get("/config", handle_get_config)
    app.router.add_post("/config", handle_post_config)
    return app


def _serve_on_socket(sock, probe_port: int, preload: list,
                     lock_path: str) -> None:
    """One worker: aiohttp on an inherited, already-bound UDS listener.
    Accept races across workers are resolved by the kernel."""
    app = make_app()
    if preload:
        core.preload(preload, lock_path)
    if probe_port:
        # TCP liveness probe alongside the UDS endpoint (server.py:293);
        # port 0 + SO_REUSEPORT keeps N workers from colliding
        probe = web.Application()
        probe.router.add_get("/health", handle_health)

        async def start_probe(_app):
            runner = web.AppRunner(probe)
            await runner.setup()
            site = web.TCPSite(runner, "0.0.0.0", probe_port,
                               reuse_port=True)
            await site.start()

        app.on_startup.append(start_probe)

    async def run():
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.SockSite(runner, sock)
        await site.start()
        while True:
            await asyncio.sleep(3600)

    try:
        asyncio.run(run())
    except KeyboardInterrupt:  # pragma: no cover
        pass


def main() -> None:
    import signal
    import socket as socket_mod
    import sys

    ap = argparse.ArgumentParser()
    ap.add_argument("--socket", default=os.environ.get("UDS_SOCKET", DEFAULT_SOCKET))
    ap.add_argument("--probe-port", type=int,
                    default=int(os.environ.get("PROBE_PORT", "0")))
    ap.add_argument("--workers", type=int,
                    default=int(os.environ.get("UDS_WORKERS", "1")),
                    help="prefork worker processes sharing the socket "
                         "(reference runs gunicorn multi-worker, "
                         "gunicorn.conf.py)")
    ap.add_argument("--preload-model", action="append", default=None,
                    help="tokenizer(s) to load before serving; guarded "
                         "by an flock so N workers initialize serially")
    args = ap.parse_args()

    logging.basicConfig(level=os.environ.get("LOG_LEVEL", "INFO"))
    os.makedirs(os.path.dirname(args.socket) or ".", exist_ok=True)
    if os.path.exists(args.socket):
        os.unlink(args.socket)

    sock = socket_mod.socket(socket_mod.AF_UNIX, socket_mod.SOCK_STREAM)
    sock.bind(args.socket)
    sock.listen(1024)
    preload = args.preload_model or []
    lock_path = args.socket + ".init.lock"

    if args.workers <= 1:
        _serve_on_socket(sock, args.probe_port, preload, lock_path)
        return

    # prefork supervisor: N workers accept on the shared listener; a
    # crashed worker is restarted (failure-recovery parity with a
    # gunicorn master), SIGTERM/SIGINT tears the set down.
    children: Dict[int, bool] = {}
    shutting_down = {"v": False}

    def spawn() -> int:
        pid = os.fork()
        if pid == 0:  # child
            # drop the supervisor's handlers: a worker must die on
            # SIGTERM, not try to signal its siblings
            signal.signal(signal.SIGTERM, signal.SIG_DFL)
            signal.signal(signal.SIGINT, signal.SIG_DFL)
            try:
                _serve_on_socket(sock, args.probe_port, preload, lock_path)
            finally:
                os._exit(0)
        return pid

    def on_term(signum, frame):  # pragma: no cover - signal path
        shutting_down["v"] = True
        for pid in list(children):
            try:
                os.kill(pid, signal.SIGTERM)
            except ProcessLookupError:
                pass

    signal.signal(signal.SIGTERM, on_term)
    signal.signal(signal.SIGINT, on_term)

    for _ in range(args.workers):
        children[spawn()] = True
    logger.info("uds_tokenizer: %d workers on %s", args.workers,
                args.socket)
    while children:
        try:
            pid, status = os.wait()
        except ChildProcessError:  # pragma: no cover
            break
        except InterruptedError:
            continue
        children.pop(pid, None)
        if not shutting_down["v"]:
            logger.warning("worker %d exited (status %d); restarting",
                           pid, status)
            children[spawn()] = True
    sys.exit(0)


if __name__ == "__main__":
    main()
