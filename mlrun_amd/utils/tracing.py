# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Per-step tracing: rocTX ranges + wall-time span collection.

Parity target: the reference's only tracing is the opt-in Horovod
Timeline env (mpijob/abstract.py:110 with_tracing).  Here tracing is a
first-class opt-in: when enabled, every serving-graph step and engine
phase is wrapped in a rocTX range (visible in rocprofv3 --sys-trace /
--marker-trace timelines; torch.cuda.nvtx maps to rocTX on ROCm) and
recorded as a wall-time span retrievable per event.
"""

import contextlib
import os
import threading
import time
import typing

_enabled = os.environ.get("MLRUN_TRACE", "") not in ("", "0", "false")
_local = threading.local()


def enable(on: bool = True):
    global _enabled

    _enabled = on


def is_enabled() -> bool:
    return _enabled


def _roctx():
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.nvtx  # rocTX on ROCm builds
    except Exception:
        pass
    return None


@contextlib.contextmanager
def span(name: str):
    """Record a named span (and a rocTX range when on GPU)."""
    if not _enabled:
        yield
        return
    roctx = _roctx()
    if roctx is not None:
        roctx.range_push(name)
    start = time.perf_counter()
    try:
        yield
    finally:
        elapsed_ms = (time.perf_counter() - start) * 1000.0
        if roctx is not None:
            roctx.range_pop()
        spans = getattr(_local, "spans", None)
        if spans is not None:
            spans.append({"name": name, "ms": round(elapsed_ms, 3)})


@contextlib.contextmanager
def collect() -> typing.Iterator[list]:
    """Collect spans recorded on this thread inside the block."""
    previous = getattr(_local, "spans", None)
    _local.spans = []
    try:
        yield _local.spans
    finally:
        _local.spans = previous


def trace_step(step_name: str):
    return span(f"step:{step_name}")
