"""Logging + lightweight latency/throughput tracking.

Mirrors the reference's observability shape (torchstore ``logging.py:13-66``):
a root-logger setup gated on ``TORCHSTORE_AMD_LOG_LEVEL``, and a
:class:`LatencyTracker` that records per-step wall time plus optional GB/s.
On ROCm builds we additionally emit roctx ranges (when the rocprofiler sdk
is loadable) so rocprofv3 timelines show store phases next to kernels.
"""

from __future__ import annotations

import logging
import os
import time
from contextlib import contextmanager
from typing import Dict, List, Optional, Tuple

_INITIALIZED = False


def init_logging() -> None:
    global _INITIALIZED
    if _INITIALIZED:
        return
    _INITIALIZED = True
    level = os.environ.get("TORCHSTORE_AMD_LOG_LEVEL", "WARNING").upper()
    logging.basicConfig(
        level=getattr(logging, level, logging.WARNING),
        format="[%(asctime)s %(process)d %(name)s %(levelname)s] %(message)s",
    )


def get_logger(name: str) -> logging.Logger:
    init_logging()
    return logging.getLogger(name)


_roctx = None


def _load_roctx():
    global _roctx
    if _roctx is not None:
        return _roctx
    try:
        import ctypes

        lib = ctypes.CDLL("libroctx64.so")
        lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
        _roctx = lib
    except OSError:
        _roctx = False
    return _roctx


@contextmanager
def roctx_range(name: str):
    """rocprof-visible named range; no-op when roctx is unavailable."""
    lib = _load_roctx()
    if lib:
        lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        if lib:
            lib.roctxRangePop()


class LatencyTracker:
    """Wall-clock per named step + end-to-end, with optional GB/s."""

    def __init__(self, name: str, logger: Optional[logging.Logger] = None):
        self.name = name
        self.logger = logger or get_logger("torchstore_amd.latency")
        self._t0 = time.perf_counter()
        self._last = self._t0
        self.steps: List[Tuple[str, float, Optional[int]]] = []

    def step(self, label: str, nbytes: Optional[int] = None) -> float:
        now = time.perf_counter()
        dt = now - self._last
        self._last = now
        self.steps.append((label, dt, nbytes))
        if nbytes:
            self.logger.debug(
                "%s/%s: %.3f ms, %.2f GB/s",
                self.name, label, dt * 1e3, nbytes / dt / 1e9,
            )
        else:
            self.logger.debug("%s/%s: %.3f ms", self.name, label, dt * 1e3)
        return dt

    def e2e(self, nbytes: Optional[int] = None) -> float:
        dt = time.perf_counter() - self._t0
        if nbytes:
            self.logger.info(
                "%s e2e: %.3f ms, %.2f GB/s", self.name, dt * 1e3, nbytes / dt / 1e9
            )
        else:
            self.logger.info("%s e2e: %.3f ms", self.name, dt * 1e3)
        return dt

    def summary(self) -> Dict[str, float]:
        return {label: dt for label, dt, _ in self.steps}
