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


class TestOnlineServiceConfig:
    """build_index_config honors every KVCACHE_INDEX_BACKEND value the
    online binary documents (main.go env surface)."""

    def _build(self, monkeypatch, backend, **env):
        sys.path.insert(0, EXAMPLES)
        try:
            import online_service
        finally:
            sys.path.pop(0)
        monkeypatch.setenv("KVCACHE_INDEX_BACKEND", backend)
        for k, v in env.items():
            monkeypatch.setenv(k, v)
        return online_service.build_index_config()

    def test_in_memory(self, monkeypatch):
        cfg = self._build(monkeypatch, "in_memory")
        assert cfg.in_memory is not None and cfg.enable_metrics

    def test_native(self, monkeypatch):
        cfg = self._build(monkeypatch, "native")
        assert cfg.native is not None

    def test_cost_aware(self, monkeypatch):
        cfg = self._build(monkeypatch, "cost_aware")
        assert cfg.cost_aware is not None

    def test_valkey_addr(self, monkeypatch):
        cfg = self._build(monkeypatch, "valkey",
                          REDIS_ADDR="valkey://10.0.0.9:6380")
        assert cfg.valkey is not None
        assert cfg.valkey.address == "valkey://10.0.0.9:6380"

    def test_unknown_backend_exits(self, monkeypatch):
        with pytest.raises(SystemExit):
            self._build(monkeypatch, "bogus")


def test_sharded_service_main_importable():
    """The torchrun entry point from docs/deployment.md must stay
    importable (its collective wiring is exercised by test_sharded's
    service-bridge test)."""
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "sharded_service_main",
        os.path.join(EXAMPLES, "sharded_service_main.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    assert callable(mod.main)


class TestPackaging:
    """Deployment packaging parity (reference Dockerfile + deploy/)."""

    ROOT = __import__("os").path.dirname(__import__("os").path.dirname(
        __import__("os").path.abspath(__file__)))

    def test_dockerfile_entrypoint_exists(self):
        import os

        df = os.path.join(self.ROOT, "Dockerfile")
        assert os.path.exists(df)
        content = open(df).read()
        assert "examples/online_service.py" in content
        assert os.path.exists(os.path.join(
            self.ROOT, "examples", "online_service.py"))
        assert "gfx950" in content  # builds the native extension

    def test_deploy_manifests_parse(self):
        import os

        import yaml

        for name in ("kustomization.yaml", "deployment.yaml",
                     "service.yaml"):
            path = os.path.join(self.ROOT, "deploy", name)
            with open(path) as f:
                doc = yaml.safe_load(f)
            assert isinstance(doc, dict), name
        dep = yaml.safe_load(open(os.path.join(self.ROOT, "deploy",
                                               "deployment.yaml")))
        container = dep["spec"]["template"]["spec"]["containers"][0]
        ports = {p["name"]: p["containerPort"] for p in container["ports"]}
        assert ports == {"http": 8080, "kvevents": 5557}
        assert container["livenessProbe"]["httpGet"]["path"] == "/health"
