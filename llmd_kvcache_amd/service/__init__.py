"""Service fronts: native wirefront (C++ epoll), gRPC, HTTP, raw ASGI,
and the shared request-coalescing scorer."""

from .coalesce import CoalescingScorer  # noqa: F401

__all__ = ["CoalescingScorer", "WireIndexerService"]


def __getattr__(name):
    # WireIndexerService pulls in the native extension; import lazily so
    # `import llmd_kvcache_amd.service` works before the first build.
    if name == "WireIndexerService":
        from .wirefront import WireIndexerService

        return WireIndexerService
    raise AttributeError(name)
