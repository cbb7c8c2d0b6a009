"""Capture vLLM golden fixtures for wire-compatibility tests.

Run this wherever a pinned vLLM is installed (CPU-only is enough - only
the KV-event schema and block-hash code are used):

    python scripts/capture_vllm_fixtures.py \
        --out tests/testdata/vllm_golden.json

It records, for a set of fixed token sequences:
  - vLLM's own block hashes (the "engine hashes" BlockStored reports),
  - a msgpack-encoded EventBatch exactly as vLLM's ZMQ publisher frames
    it (tagged-union arrays),
  - the hash seed configuration used.

tests/test_vllm_golden.py then asserts that this repo's event decoder
and token processor reproduce those bytes/hashes bit-for-bit.  The
reference's own integration fixture went stale when vLLM moved from
SHA-256 to FNV-64a (reference tests/integration/prompt_to_block_test.go
is skipped upstream); regenerating against a PINNED version keeps the
fixture trustworthy - record the version, and re-run on upgrades.

This environment has no package index, so the fixture cannot be
generated in-repo; the test skips with a clear message until someone
runs this script on a vLLM-equipped host and commits the JSON.
"""

from __future__ import annotations

import argparse
import base64
import json
import sys


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="tests/testdata/vllm_golden.json")
    ap.add_argument("--seed", default="", help="PYTHONHASHSEED the fleet "
                    "would run with (must match indexer config)")
    ap.add_argument("--block-size", type=int, default=16)
    args = ap.parse_args()

    try:
        import vllm  # noqa: F401
    except ImportError:
        print("vLLM is not importable here - run on a vLLM-equipped host",
              file=sys.stderr)
        return 2

    # Exact module paths move between vLLM versions; try the known homes
    # and record which one produced the fixture.
    hash_impl = None
    for path in ("vllm.v1.core.kv_cache_utils",
                 "vllm.core.block.prefix_caching_block"):
        try:
            mod = __import__(path, fromlist=["*"])
            if hasattr(mod, "hash_block_tokens"):
                hash_impl = (path, mod)
                break
        except ImportError:
            continue
    if hash_impl is None:
        print("could not locate vLLM's hash_block_tokens - inspect the "
              "installed version and extend this script", file=sys.stderr)
        return 3

    try:
        from vllm.distributed.kv_events import BlockStored, KVEventBatch
        import msgspec

        enc = msgspec.msgpack.Encoder()
    except ImportError as e:
        print(f"KV-event schema import failed: {e}", file=sys.stderr)
        return 3

    path, mod = hash_impl
    cases = []
    sequences = [
        list(range(args.block_size * 4)),
        list(range(1000, 1000 + args.block_size * 8)),
        [7] * (args.block_size * 2),
    ]
    if hasattr(mod, "init_none_hash"):
        mod.init_none_hash(hash)  # builtin-hash mode (PYTHONHASHSEED)
    for tokens in sequences:
        hashes = []
        parent = None
        for i in range(0, len(tokens), args.block_size):
            chunk = tokens[i:i + args.block_size]
            h = mod.hash_block_tokens(hash, parent, chunk, None)
            parent = h
            hashes.append(int(getattr(h, "hash_value", h)))
        ev = BlockStored(block_hashes=hashes, parent_block_hash=None,
                         token_ids=tokens, block_size=args.block_size,
                         lora_id=None)
        batch = KVEventBatch(ts=0.0, events=[ev])
        cases.append({
            "tokens": tokens,
            "block_hashes": hashes,
            "event_batch_msgpack_b64":
                base64.b64encode(enc.encode(batch)).decode(),
        })

    out = {
        "vllm_version": vllm.__version__,
        "hash_module": path,
        "hash_seed": args.seed,
        "block_size": args.block_size,
        "cases": cases,
    }
    with open(args.out, "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {args.out} (vLLM {vllm.__version__}, {len(cases)} cases)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
