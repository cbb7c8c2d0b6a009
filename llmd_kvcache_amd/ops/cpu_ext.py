"""Loader for the native extension (C++ CPU ops; HIP ops on ROCm).

The extension is built in-tree (setup.py build_ext --inplace) so the .so
travels with the repo snapshot.  maybe_load() returns the module or None;
GPU code paths must use require() which fails loudly when the native
extension is missing on a GPU host.
"""
from __future__ import annotations

_mod = None
_tried = False


def maybe_load():
    global _mod, _tried
    if _tried:
        return _mod
    _tried = True
    try:
        import torch  # noqa: F401  (loads libc10/libtorch for the ext)

        from . import _kvidx_C  # type: ignore

        _mod = _kvidx_C
    except ImportError:
        try:
            import _kvidx_C  # type: ignore  # in-tree build fallback

            _mod = _kvidx_C
        except ImportError:
            _mod = None
    return _mod


def require():
    mod = maybe_load()
    if mod is None:
        raise RuntimeError(
            "native extension _kvidx_C is not built; run "
            "`python setup.py build_ext --inplace` (required on GPU hosts)"
        )
    return mod
