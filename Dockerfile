# Container image for the MI355X-native KV-cache indexer service.
#
# Reference analog: /root/reference/Dockerfile (Go builder + distroless
# runner shipping the kv_events/online binary).  Here the runtime is
# PyTorch-ROCm and the service binary is examples/online_service.py
# (HTTP + ZMQ KVEvents + metrics) with the native extension built for
# gfx950 at image build time.
#
# Build (on a machine with the ROCm PyTorch base available):
#   docker build -t llmd-kvcache-amd --build-arg BASE=<rocm-torch-image> .
# Run (GPU node):
#   docker run --device=/dev/kfd --device=/dev/dri --network=host \
#     -e ZMQ_ENDPOINT=tcp://*:5557 -e HTTP_PORT=8080 llmd-kvcache-amd
#
# The base must provide: ROCm 7.x with hipcc, PyTorch-ROCm, Python 3.10+.
# (This repo's CI environment has no registry access; the image is
# validated by scripts/container_smoke.sh against a local build.)
ARG BASE=rocm/pytorch:latest
FROM ${BASE} AS build

WORKDIR /opt/llmd-kvcache-amd
COPY setup.py pyproject.toml ./
COPY llmd_kvcache_amd ./llmd_kvcache_amd
# build the gfx950 extension in-tree (no GPU needed; hipcc cross-compiles)
RUN PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace && \
    python -c "from llmd_kvcache_amd.ops import cpu_ext; \
m = cpu_ext.require(); assert m.HAS_HIP"

FROM ${BASE}
WORKDIR /opt/llmd-kvcache-amd
COPY --from=build /opt/llmd-kvcache-amd/llmd_kvcache_amd ./llmd_kvcache_amd
COPY examples ./examples
COPY services ./services

ENV PYTHONPATH=/opt/llmd-kvcache-amd \
    HSA_ENABLE_IPC_MODE_LEGACY=0 \
    HTTP_PORT=8080 \
    ZMQ_ENDPOINT=tcp://*:5557 \
    POOL_CONCURRENCY=4 \
    BLOCK_SIZE=16

EXPOSE 8080 5557
# same surface as the reference's shipped binary: HTTP scoring endpoints
# + /metrics + ZMQ KVEvents subscriber (examples/kv_events/online)
ENTRYPOINT ["python", "examples/online_service.py"]
