"""Driver-contract smoke for bench.py: tiny CPU configurations run as
subprocesses; the final stdout line must be ONE JSON object carrying the
fields the round driver parses (BASELINE.json metric shape)."""

import json
import os
import subprocess
import sys

import pytest


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TINY = ["--device", "cpu", "--batch", "32", "--blocks", "2048",
        "--steps", "2", "--warmup", "1", "--calls-per-step", "1"]

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}
REQUIRED_CONFIG = {"model", "global_batch", "seq_len", "parallelism"}


def run_bench(extra):
    env = dict(os.environ, MASTER_PORT=str(_free_port()))
    proc = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py")] + TINY + extra,
        capture_output=True, text=True, timeout=300, cwd=ROOT, env=env)
    assert proc.returncode == 0, proc.stderr[-800:]
    line = proc.stdout.strip().splitlines()[-1]
    return json.loads(line)


@pytest.mark.timeout(600)
class TestBenchContract:
    def test_default_mode_json_shape(self):
        d = run_bench([])
        assert REQUIRED <= set(d)
        assert REQUIRED_CONFIG <= set(d["config"])
        assert d["metric"].startswith("Score() QPS")
        assert isinstance(d["value"], (int, float)) and d["value"] > 0
        assert d["higher_is_better"] is True
        assert d["scaling"] == "weak"
        assert d["data"] == "synthetic"
        assert d["config"]["seq_len"] == 8192
        assert d["config"]["block_size"] == 16

    def test_sharded_mode(self):
        d = run_bench(["--force-sharded"])
        assert d["scaling"] == "strong"
        assert d["config"]["parallelism"] == "shard1"
        assert d["value"] > 0

    def test_prefix_frac_and_pods_flags(self):
        d = run_bench(["--prefix-frac", "1.0", "--pods", "8"])
        assert d["config"]["prefix_frac"] == 1.0
        assert d["config"]["num_pods"] == 8


@pytest.mark.timeout(600)
class TestBenchTorchrun:
    """The driver launches N>1 exactly like this (one rank per GPU over
    RCCL; gloo here): shape must keep working."""

    def run_torchrun(self, extra, port):
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run",
             "--nnodes=1", "--nproc-per-node", "2",
             "--master-addr", "127.0.0.1", "--master-port", str(port),
             os.path.join(ROOT, "bench.py")] + TINY +
            ["--batch", "16", "--blocks", "1024"] + extra,
            capture_output=True, text=True, timeout=300, cwd=ROOT)
        assert proc.returncode == 0, proc.stderr[-800:]
        json_lines = [l for l in proc.stdout.splitlines()
                      if l.startswith("{")]
        assert len(json_lines) == 1, "exactly one rank prints the result"
        return json.loads(json_lines[0])

    def test_replicated_world2(self):
        d = self.run_torchrun([], _free_port())
        assert d["config"]["parallelism"] == "replicated2"
        assert d["scaling"] == "weak"
        assert d["config"]["global_batch"] == 32  # whole-job aggregate

    def test_sharded_world2(self):
        d = self.run_torchrun(["--sharded"], _free_port())
        assert d["config"]["parallelism"] == "shard2"
        assert d["scaling"] == "strong"
