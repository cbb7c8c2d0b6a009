"""Leveled trace logging.

Parity with reference pkg/utils/logging/levels.go:17-20 (DEBUG=4,
TRACE=5 as logr V-levels): Python logging gains a TRACE level below
DEBUG, and every pipeline stage logs at TRACE so an operator can follow
a request end to end (indexer.go:135-163 style)."""

from __future__ import annotations

import logging

TRACE = 5
logging.addLevelName(TRACE, "TRACE")


def get_logger(name: str) -> logging.Logger:
    return logging.getLogger(f"llmd_kvcache_amd.{name}")


def trace(logger: logging.Logger, msg: str, *args) -> None:
    if logger.isEnabledFor(TRACE):
        logger.log(TRACE, msg, *args)


def enable_trace() -> None:
    """Convenience: turn on TRACE for the whole package (the reference's
    TRACE=5 verbosity flag)."""
    logging.getLogger("llmd_kvcache_amd").setLevel(TRACE)
