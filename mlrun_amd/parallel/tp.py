# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Tensor parallelism over xGMI.

Column-parallel projections (qkv, gate|up) shard output rows across
ranks; row-parallel projections (attn-out, mlp-down) shard input
columns and all-reduce the partial outputs over RCCL — one collective
per transformer block half, riding the 7-link xGMI mesh.  Used by the
Llama-70B TP=8 serving config (BASELINE.json config 5).

The reference has NO tensor parallelism anywhere —
this is a new MI355X-native capability.
"""


import torch
import torch.distributed as dist



def init_tp_group(tp_size: int = None, backend: str = None):
    """Create (or reuse) the process group used for tensor parallelism.
    Returns (group, tp_rank, tp_size).  With world_size == tp_size the
    default group is used."""
    if not dist.is_initialized():
        from .ddp import init_process_group

        init_process_group(backend=backend)
    if not dist.is_initialized():
        return None, 0, 1
    world = dist.get_world_size()
    rank = dist.get_rank()
    tp_size = tp_size or world
    if tp_size == world:
        return None, rank, world  # default group
    assert world % tp_size == 0, "world size must be divisible by tp size"
    group = None
    my_group_start = (rank // tp_size) * tp_size
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        new_group = dist.new_group(ranks)
        if start == my_group_start:
            group = new_group
    return group, rank % tp_size, tp_size


def all_reduce_tensor(t: torch.Tensor, group=None) -> torch.Tensor:
    """All-reduce that tolerates bf16-on-CPU backends (gloo lacks bf16
    reduction): round-trips through f32 when needed."""
    if not dist.is_initialized():
        return t
    if t.is_cuda or t.dtype not in (torch.bfloat16, torch.float16):
        dist.all_reduce(t, group=group)
        return t
    f32 = t.float()
    dist.all_reduce(f32, group=group)
    t.copy_(f32.to(t.dtype))
    return t


def shard_llama_state(full_state: dict, cfg, tp_rank: int,
                      tp_size: int) -> dict:
    """Slice a full Llama checkpoint into this rank's TP shard.

    Row (column-parallel) weights: wqkv (per q/k/v head groups),
    wgu (gate half + up half separately).  Column (row-parallel):
    wo, wdown.  embed/lm_head/norms replicated.
    """
    d = cfg.head_dim
    hq_full, hkv_full = cfg.num_heads, cfg.num_kv_heads
    hq = hq_full // tp_size
    hkv = max(hkv_full // tp_size, 1)
    inter_full = cfg.intermediate_size
    inter = inter_full // tp_size
    out = {}
    for key, value in full_state.items():
        if key.endswith(".wqkv"):
            q_full = value[:hq_full * d]
            k_full = value[hq_full * d:(hq_full + hkv_full) * d]
            v_full = value[(hq_full + hkv_full) * d:]
            q = q_full[tp_rank * hq * d:(tp_rank + 1) * hq * d]
            if hkv_full >= tp_size:
                k = k_full[tp_rank * hkv * d:(tp_rank + 1) * hkv * d]
                v = v_full[tp_rank * hkv * d:(tp_rank + 1) * hkv * d]
            else:  # replicate kv heads when tp > kv heads
                k, v = k_full, v_full
            out[key] = torch.cat([q, k, v], dim=0).contiguous()
        elif key.endswith(".wo"):
            out[key] = value[:, tp_rank * hq * d:(tp_rank + 1) * hq * d] \
                .contiguous()
        elif key.endswith(".wgu"):
            gate = value[:inter_full]
            up = value[inter_full:]
            out[key] = torch.cat([
                gate[tp_rank * inter:(tp_rank + 1) * inter],
                up[tp_rank * inter:(tp_rank + 1) * inter]],
                dim=0).contiguous()
        elif key.endswith(".wdown"):
            out[key] = value[:, tp_rank * inter:(tp_rank + 1) * inter] \
                .contiguous()
        else:
            out[key] = value
    return out
