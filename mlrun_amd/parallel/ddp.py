# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Data-parallel training over RCCL/xGMI: bucketed gradient all-reduce
overlapped with backward.

This is the Horovod-DistributedOptimizer equivalent of the reference
(frameworks/pytorch/mlrun_interface.py:638-642: broadcast params ->
wrap optimizer -> per-step bucketed allreduce).  MI355X-first design:
- one process per GPU; backend "nccl" IS RCCL on ROCm
- xGMI is point-to-point (7 links x ~153 GB/s): ring all-reduce is
  per-link bound, so buckets default to 64 MB (config
  distributed.bucket_cap_mb) — large enough to amortize ring latency,
  small enough that several buckets overlap with backward
- buckets are filled in reverse parameter order (= backward order) and
  reduced on a dedicated comm stream as soon as they are full
"""

import os
import typing

import torch
import torch.distributed as dist

from ..config import config
from ..utils import logger


def init_process_group(backend: str = None, timeout_seconds: int = 600):
    """Initialize torch.distributed from the launcher env (idempotent)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    backend = backend or (str(config.distributed.backend)
                          if torch.cuda.is_available() else "gloo")
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 1
    os.environ.setdefault("MASTER_ADDR",
                          str(config.distributed.master_addr))
    os.environ.setdefault("MASTER_PORT",
                          str(config.distributed.master_port))
    for key, value in config.distributed.rccl_env.to_dict().items():
        os.environ.setdefault(key, str(value))
    import datetime

    dist.init_process_group(
        backend=backend, rank=rank, world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_seconds))
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local_rank)
    return rank, world_size


def broadcast_module(module: torch.nn.Module, src: int = 0):
    """Broadcast parameters + buffers from rank src (Horovod
    broadcast_parameters analog)."""
    if not dist.is_initialized() or dist.get_world_size() <= 1:
        return
    for tensor in list(module.parameters()) + list(module.buffers()):
        dist.broadcast(tensor.data, src=src)


class _Bucket:
    __slots__ = ["params", "numel", "ready", "flat", "work"]

    def __init__(self):
        self.params: list = []
        self.numel = 0
        self.ready = 0
        self.flat = None
        self.work = None


class DistributedModel(torch.nn.Module):
    """Module wrapper: forwards to the inner module; all-reduces
    gradients in buckets during backward.  Call finalize_backward()
    after loss.backward() and before optimizer.step()."""

    def __init__(self, module: torch.nn.Module, bucket_cap_mb: int = None,
                 process_group=None):
        super().__init__()
        self.module = module
        self.process_group = process_group
        self.world_size = dist.get_world_size(process_group) \
            if dist.is_initialized() else 1
        cap_mb = bucket_cap_mb or int(config.distributed.bucket_cap_mb)
        self._bucket_cap = cap_mb * 1024 * 1024
        self._buckets: typing.List[_Bucket] = []
        self._param_bucket: dict = {}
        self._hooks = []
        self._comm_stream = None
        self._use_comm_stream = torch.cuda.is_available() and \
            next(module.parameters()).is_cuda
        if self._use_comm_stream:
            self._comm_stream = torch.cuda.Stream()
        # grad-accumulation contract (torch-DDP no_sync analog):
        # reduction fires only when require_backward_grad_sync is True
        # — set it False for non-boundary micro-steps, True for the
        # final micro-step of the accumulation window
        self.require_backward_grad_sync = True
        if self.world_size > 1:
            broadcast_module(module)
            self._build_buckets()
            self._register_hooks()

    def _build_buckets(self):
        """Buckets in reverse parameter order = backward completion
        order, so the first-finished grads reduce first."""
        params = [p for p in self.module.parameters() if p.requires_grad]
        bucket = _Bucket()
        for p in reversed(params):
            bytes_ = p.numel() * p.element_size()
            if bucket.numel > 0 and \
                    (bucket.numel * p.element_size() + bytes_ >
                     self._bucket_cap):
                self._buckets.append(bucket)
                bucket = _Bucket()
            bucket.params.append(p)
            bucket.numel += p.numel()
            self._param_bucket[p] = bucket
        if bucket.params:
            self._buckets.append(bucket)
        logger.debug("ddp buckets built", buckets=len(self._buckets))

    def _register_hooks(self):
        for p in self._param_bucket:
            hook = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hooks.append(hook)

    def _on_grad_ready(self, param):
        if not self.require_backward_grad_sync:
            return  # accumulating: reduce on the boundary step only
        bucket = self._param_bucket[param]
        bucket.ready += 1
        if bucket.ready == len(bucket.params):
            self._reduce_bucket(bucket)

    def _reduce_bucket(self, bucket: _Bucket):
        grads = [p.grad for p in bucket.params]
        if self._use_comm_stream:
            # launch on the comm stream so the ring overlaps with the
            # rest of backward
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                bucket.flat = torch._utils._flatten_dense_tensors(grads)
                bucket.work = dist.all_reduce(
                    bucket.flat, op=dist.ReduceOp.SUM,
                    group=self.process_group, async_op=True)
        else:
            bucket.flat = torch._utils._flatten_dense_tensors(grads)
            bucket.work = dist.all_reduce(bucket.flat,
                                          op=dist.ReduceOp.SUM,
                                          group=self.process_group,
                                          async_op=True)

    def finalize_backward(self):
        """Wait for all bucket reductions, scatter averaged grads back."""
        if self.world_size <= 1:
            return
        inv = 1.0 / self.world_size
        for bucket in self._buckets:
            if bucket.work is None:
                # params with no grad this step (e.g. frozen branch)
                bucket.ready = 0
                continue
            bucket.work.wait()
            if self._use_comm_stream:
                torch.cuda.current_stream().wait_stream(self._comm_stream)
            grads = [p.grad for p in bucket.params]
            synced = torch._utils._unflatten_dense_tensors(bucket.flat,
                                                           grads)
            torch._foreach_mul_(list(synced), inv)
            for grad, avg in zip(grads, synced):
                grad.copy_(avg)
            bucket.work = None
            bucket.flat = None
            bucket.ready = 0

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)


class DistributedOptimizer:
    """Optimizer wrapper that finalizes gradient reduction on step()
    (direct analog of Horovod's DistributedOptimizer for code that
    can't call finalize_backward itself)."""

    def __init__(self, optimizer: torch.optim.Optimizer,
                 ddp_model: DistributedModel):
        self._optimizer = optimizer
        self._ddp = ddp_model

    def step(self, closure=None):
        self._ddp.finalize_backward()
        return self._optimizer.step(closure)

    def zero_grad(self, set_to_none: bool = True):
        return self._optimizer.zero_grad(set_to_none=set_to_none)

    def __getattr__(self, name):
        return getattr(self._optimizer, name)
