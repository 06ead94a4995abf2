#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Baseline config 5: Llama-3-70B serving with tensor parallelism over
xGMI (TP = world size).  Launch on an 8-GPU node:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 scripts/bench_serving_tp.py \
        --steps 5 --warmup 2

Every rank holds 1/TP of the 70B weights (column/row sharded); one
RCCL all-reduce after attn-out and one after mlp-down per layer.
Rank 0 prints a bench-style JSON line.
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--steps", type=int, default=5)
    parser.add_argument("--warmup", type=int, default=2)
    parser.add_argument("--model", default="llama-3-70b")
    parser.add_argument("--batch", type=int, default=16)
    parser.add_argument("--prompt-len", type=int, default=128)
    parser.add_argument("--gen-tokens", type=int, default=32)
    args = parser.parse_args()

    import torch

    from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine
    from mlrun_amd.parallel.tp import init_tp_group

    on_gpu = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if on_gpu:
        torch.cuda.set_device(local_rank)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    group, tp_rank, tp_size = init_tp_group(
        backend="nccl" if on_gpu else "gloo")
    device = f"cuda:{local_rank}" if on_gpu else "cpu"

    if args.model in ("llama-3-70b", "70b"):
        cfg = LlamaConfig.llama3_70b()
    elif args.model == "tiny":
        cfg = LlamaConfig.tiny(num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
    else:
        cfg = LlamaConfig.llama3_8b()
    batch = args.batch if on_gpu else 2
    prompt_len = args.prompt_len if on_gpu else 8
    # TP decode is hipGraph-captured: the per-layer RCCL all-reduces
    # are recorded into the graph and pair up across ranks at replay
    # (capture_error_mode=thread_local dodges the NCCL watchdog).
    # MLRUN_TP_GRAPH=0 falls back to eager for A/B.
    use_graph = on_gpu and os.environ.get("MLRUN_TP_GRAPH", "1") != "0"
    engine = LlamaDecodeEngine(cfg, batch, device=device, tp_group=group,
                               tp_rank=tp_rank, tp_size=tp_size,
                               use_graph=use_graph, seed=7)

    gen = torch.Generator().manual_seed(77)

    def make_prompts():
        return torch.randint(1, cfg.vocab_size - 1, (batch, prompt_len),
                             generator=gen)

    import torch.distributed as dist

    for _ in range(args.warmup):
        engine.reset()
        engine.generate(make_prompts(), max_new_tokens=args.gen_tokens)
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.reset()
        engine.generate(make_prompts(), max_new_tokens=args.gen_tokens)
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t0

    if rank == 0:
        reqs = batch * args.steps / elapsed
        print(json.dumps({
            "metric": "serving req/sec (Llama-3-70B, TP over xGMI)",
            "value": round(reqs, 3),
            "unit": "req/s",
            "n_gpus": tp_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {"model": cfg.name, "global_batch": batch,
                       "seq_len": prompt_len + args.gen_tokens,
                       "parallelism": f"tp{tp_size}",
                       "hipgraph": use_graph},
        }))
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
