# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Distributed / parallel runtime: node-local GPU scheduler, rank
launcher, RCCL-over-xGMI collectives, DDP-style gradient bucketing,
tensor parallelism."""

from .scheduler import (  # noqa: F401
    GpuAllocator,
    GpuLease,
    detect_gpu_count,
    get_gpu_allocator,
    set_gpu_allocator,
)
