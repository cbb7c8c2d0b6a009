"""Multi-rank sharded indexer service entry point (BASELINE config 3).

Launch one rank per GPU (docs/deployment.md):

    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
        examples/sharded_service_main.py

Rank 0 owns the external surfaces (ZMQ KVEvents SUB that vLLM
publishers connect to, HTTP scoring + /metrics) and replicates raw
event payloads to the other ranks over torch.distributed object
broadcasts; every rank holds one `chunk_hash % world` shard and
participates in the RCCL mask-merge all_reduce
(llmd_kvcache_amd/parallel/service.py).

Environment (same surface as examples/online_service.py where it
applies): HTTP_PORT, ZMQ_ENDPOINT, ZMQ_TOPIC, BLOCK_SIZE,
PYTHONHASHSEED, TABLE_CAPACITY (power of two; default 2^21 slots).
"""

import logging
import os
import queue
import signal
import sys
import threading

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch
import torch.distributed as dist

from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.parallel.service import ShardedIndexService
from llmd_kvcache_amd.parallel.sharded import ShardedIndex

logging.basicConfig(level=os.environ.get("LOG_LEVEL", "INFO"))
logger = logging.getLogger("sharded_service")


def main():
    use_gpu = torch.cuda.is_available()
    dist.init_process_group("nccl" if use_gpu else "gloo")
    rank = dist.get_rank()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if use_gpu:
        torch.cuda.set_device(local_rank)
    device = f"cuda:{local_rank}" if use_gpu else "cpu"

    capacity = int(os.environ.get("TABLE_CAPACITY", str(1 << 21)))
    block_size = int(os.environ.get("BLOCK_SIZE", "16"))
    sharded = ShardedIndex(TableIndexConfig(capacity=capacity, device=device))
    tp = ChunkedTokenDatabase(TokenProcessorConfig(
        block_size=block_size,
        hash_seed=os.environ.get("PYTHONHASHSEED", ""),
    ))
    service = ShardedIndexService(sharded, tp)
    logger.info("rank %d/%d: shard ready on %s (capacity %d)",
                rank, dist.get_world_size(), device, capacity)

    if rank != 0:
        service.serve()  # follower loop until rank 0 broadcasts stop
        dist.destroy_process_group()
        return

    # ---- rank 0: sockets + dispatch -----------------------------------
    from llmd_kvcache_amd.kvevents.pool import Message
    from llmd_kvcache_amd.kvevents.zmq_subscriber import ZmqSubscriber

    inbox: "queue.Queue[Message]" = queue.Queue(maxsize=4096)

    class _QueuePool:
        @staticmethod
        def add_task(msg: Message) -> None:
            try:
                inbox.put_nowait(msg)
            except queue.Full:
                logger.warning("event inbox full; dropping message")

    subscriber = ZmqSubscriber(
        _QueuePool,
        os.environ.get("ZMQ_ENDPOINT", "tcp://*:5557"),
        os.environ.get("ZMQ_TOPIC", "kv@"),
    )
    subscriber.start()
    logger.info("KVEvents SUB bound on %s", subscriber.endpoint)

    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *a: stop.set())
    signal.signal(signal.SIGINT, lambda *a: stop.set())

    # Scoring surface: a router embedding this process calls
    # service.score(request_keys, pod_filter) -> {pod: score}; every
    # call participates in the RCCL mask merge across ranks. (An HTTP
    # front composes HttpService with a tokenizing adapter on top of
    # service.score - see examples/online_service.py for the single-GPU
    # version of that surface.)
    try:
        while not stop.is_set():
            # drain a burst (mirrors the events pool's GPU burst size)
            msgs = []
            try:
                msgs.append(inbox.get(timeout=0.25))
                while len(msgs) < 64:
                    msgs.append(inbox.get_nowait())
            except queue.Empty:
                pass
            if msgs:
                service.apply_messages([
                    (m.pod_identifier, m.model_name, m.payload)
                    for m in msgs
                ])
    finally:
        logger.info("shutting down")
        subscriber.stop()
        service.stop()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
