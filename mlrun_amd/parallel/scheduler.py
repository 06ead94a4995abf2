# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Node-local GPU allocator.

The MI355X node (8 GPUs) is the "cluster": runs that request GPUs get
device leases (exposed via HIP_VISIBLE_DEVICES) from this allocator
instead of k8s GPU limits (reference: runtimes/pod.py:1125
with_limits(gpus=...)).  File-lock based so multiple processes on the
node cooperate.
"""

import os
import threading
import time
import typing

from ..config import config
from ..errors import MLRunRuntimeError, MLRunTimeoutError


def detect_gpu_count() -> int:
    """Number of visible HIP devices (0 when no GPU / no torch-rocm)."""
    visible = os.environ.get("HIP_VISIBLE_DEVICES",
                             os.environ.get("CUDA_VISIBLE_DEVICES"))
    if visible is not None and visible != "":
        return len([d for d in visible.split(",") if d != ""])
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return 0


class GpuLease:
    def __init__(self, allocator: "GpuAllocator", devices: typing.List[int],
                 owner: str):
        self.allocator = allocator
        self.devices = devices
        self.owner = owner
        self._released = False

    def release(self):
        if not self._released:
            self.allocator.release(self)
            self._released = True

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.release()


class GpuAllocator:
    """In-process GPU device allocator with optional on-disk state for
    cross-process coordination."""

    def __init__(self, total: int = None, state_path: str = None):
        if total is None:
            total = detect_gpu_count() or int(config.gpu.devices_per_node)
        self.total = total
        self._lock = threading.Lock()
        self._in_use: dict = {}  # device -> owner
        self.state_path = state_path

    def available(self) -> typing.List[int]:
        with self._lock:
            return [d for d in range(self.total) if d not in self._in_use]

    def acquire(self, count: int, owner: str = "", timeout: float = 300,
                block: bool = True) -> GpuLease:
        if count > self.total:
            raise MLRunRuntimeError(
                f"requested {count} GPUs but node has {self.total}")
        deadline = time.monotonic() + timeout
        while True:
            with self._lock:
                free = [d for d in range(self.total) if d not in self._in_use]
                if len(free) >= count:
                    devices = free[:count]
                    for dev in devices:
                        self._in_use[dev] = owner
                    return GpuLease(self, devices, owner)
            if not block:
                raise MLRunRuntimeError("not enough free GPUs")
            if time.monotonic() > deadline:
                raise MLRunTimeoutError(
                    f"timed out waiting for {count} free GPUs")
            time.sleep(0.1)

    def release(self, lease: GpuLease):
        with self._lock:
            for dev in lease.devices:
                if self._in_use.get(dev) == lease.owner or \
                        dev in self._in_use:
                    self._in_use.pop(dev, None)

    def usage(self) -> dict:
        with self._lock:
            return dict(self._in_use)


_allocator = None
_allocator_lock = threading.Lock()


def get_gpu_allocator() -> GpuAllocator:
    global _allocator

    with _allocator_lock:
        if _allocator is None:
            _allocator = GpuAllocator()
        return _allocator


def set_gpu_allocator(allocator: GpuAllocator):
    """Install a fake allocator (test seam — mirrors the reference's
    mocked-k8s unit tier, tests/api/runtime_handlers/base.py:102)."""
    global _allocator

    with _allocator_lock:
        _allocator = allocator
