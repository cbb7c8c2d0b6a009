#!/usr/bin/env bash
# Container smoke test (docs/deployment.md): builds the image and checks
# the service comes up and scores.  Requires docker + a ROCm torch base.
set -euo pipefail
BASE=${BASE:-rocm/pytorch:latest}
IMG=${IMG:-llmd-kvcache-amd:smoke}

docker build -t "$IMG" --build-arg BASE="$BASE" .
CID=$(docker run -d --device=/dev/kfd --device=/dev/dri --network=host \
      -e HTTP_PORT=18080 "$IMG")
trap 'docker rm -f "$CID" >/dev/null' EXIT
for i in $(seq 1 60); do
  if curl -fsS http://127.0.0.1:18080/health >/dev/null 2>&1; then
    break
  fi
  sleep 2
done
curl -fsS http://127.0.0.1:18080/health
curl -fsS -X POST http://127.0.0.1:18080/score_completions \
  -H 'content-type: application/json' \
  -d '{"prompt": "hello world", "model": "test-model"}'
echo "container smoke OK"
