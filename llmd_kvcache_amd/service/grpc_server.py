"""gRPC IndexerService server + client.

Parity with the reference's gRPC service
(examples/kv_cache_index_service/server/server.go:70-96): GetPodScores
wraps Indexer.get_pod_scores; empty pod_identifiers means "all pods".
Stubs are wired through grpc generic handlers with the hand-written
proto3 codec (service/proto.py) - wire-compatible with any client built
from the reference's api/indexer.proto.
"""

from __future__ import annotations

import logging
from concurrent import futures
from typing import Dict, Optional

import grpc

from ..indexer import Indexer
from .proto import (
    GET_POD_SCORES_METHOD,
    SERVICE_NAME,
    GetPodScoresRequest,
    GetPodScoresResponse,
    PodScore,
)

logger = logging.getLogger("llmd_kvcache_amd.grpc")


class IndexerServicer:
    def __init__(self, indexer: Indexer, coalescer=None):
        """coalescer: optional service.coalesce.CoalescingScorer - when
        set (the default in serve()), concurrent RPCs share one fused
        kernel launch per micro-batch instead of one launch each."""
        self.indexer = indexer
        self.coalescer = coalescer

    def GetPodScores(self, request: GetPodScoresRequest, context):
        try:
            if self.coalescer is not None:
                tokens = self.indexer.tokenizers_pool.tokenize(
                    None, request.prompt, request.model_name)
                scores: Dict[str, float] = self.coalescer.score(
                    tokens, request.model_name, request.pod_identifiers)
            else:
                scores = self.indexer.get_pod_scores(
                    None, request.prompt, request.model_name,
                    request.pod_identifiers,
                )
        except Exception as e:
            logger.exception("GetPodScores failed")
            context.abort(grpc.StatusCode.INTERNAL, str(e))
            return GetPodScoresResponse()
        return GetPodScoresResponse(
            scores=[PodScore(pod=p, score=s) for p, s in (scores or {}).items()]
        )


def serve(
    indexer: Indexer,
    address: str = "0.0.0.0:50051",
    max_workers: int = 16,
    coalesce: bool = True,
) -> grpc.Server:
    """Starts a non-blocking gRPC server; returns the server handle.
    coalesce=True (default) batches concurrent RPCs into shared fused
    kernel launches (service/coalesce.py)."""
    coalescer = None
    if (coalesce and hasattr(indexer, "tokenizers_pool")
            and hasattr(indexer, "score_tokens_batch")):
        from .coalesce import CoalescingScorer

        coalescer = CoalescingScorer(indexer)
        coalescer.start()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    servicer = IndexerServicer(indexer, coalescer)
    server._kvidx_coalescer = coalescer  # stopped with the server by callers
    handlers = {
        "GetPodScores": grpc.unary_unary_rpc_method_handler(
            servicer.GetPodScores,
            request_deserializer=GetPodScoresRequest.decode,
            response_serializer=GetPodScoresResponse.encode,
        )
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(SERVICE_NAME, handlers),)
    )
    port = server.add_insecure_port(address)
    server.start()
    logger.info("IndexerService listening on %s (port %d)", address, port)
    server._kvidx_port = port  # convenience for port-0 binds
    return server


class IndexerClient:
    """Minimal client for IndexerService.GetPodScores."""

    def __init__(self, target: str, timeout_s: float = 5.0):
        self.channel = grpc.insecure_channel(target)
        self.timeout_s = timeout_s
        self._call = self.channel.unary_unary(
            GET_POD_SCORES_METHOD,
            request_serializer=GetPodScoresRequest.encode,
            response_deserializer=GetPodScoresResponse.decode,
        )

    def get_pod_scores(
        self,
        prompt: str,
        model_name: str,
        pod_identifiers: Optional[list] = None,
    ) -> Dict[str, float]:
        resp = self._call(
            GetPodScoresRequest(
                prompt=prompt,
                model_name=model_name,
                pod_identifiers=list(pod_identifiers or []),
            ),
            timeout=self.timeout_s,
        )
        return {s.pod: s.score for s in resp.scores}

    def close(self) -> None:
        self.channel.close()
