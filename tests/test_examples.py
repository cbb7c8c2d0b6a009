"""Examples must keep working: run each self-contained example as a
subprocess (they print OK / demo output on success)."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXAMPLES = os.path.join(ROOT, "examples")


def run_example(name, timeout=120, expect=None):
    proc = subprocess.run(
        [sys.executable, os.path.join(EXAMPLES, name)],
        capture_output=True,
        text=True,
        timeout=timeout,
        cwd=EXAMPLES,
    )
    assert proc.returncode == 0, f"{name} failed:\n{proc.stdout}\n{proc.stderr}"
    if expect:
        assert expect in proc.stdout, (name, proc.stdout[-500:])
    return proc.stdout


@pytest.mark.timeout(180)
class TestExamples:
    def test_offline_events(self):
        run_example("offline_events.py", expect="OK")

    def test_valkey_example(self):
        run_example("valkey_example.py", expect="OK")

    def test_kv_cache_index(self):
        run_example("kv_cache_index.py", expect="OK")

    def test_grpc_service_demo(self):
        out = run_example("grpc_service.py")
        assert "library scores" in out

    def test_scorer_sketch(self):
        run_example("kv_cache_aware_scorer.py", expect="ready")
