#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""BASELINE config 4: MpiRuntime-style distributed fine-tune of
Llama-3-8B bf16 over RCCL/xGMI (tokens/s + weak scaling).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 scripts/bench_train.py \
        --steps 10 --warmup 3

One rank per GPU; gradients ride the bucketed bf16 all-reduce engine
(mlrun_amd/parallel/ddp.py) overlapped with backward.  Rank 0 prints a
bench-style JSON line (value = WHOLE-JOB tokens/s).
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
    parser.add_argument("--steps", type=int, default=10)
    parser.add_argument("--warmup", type=int, default=3)
    parser.add_argument("--model", default="llama-3-8b")
    parser.add_argument("--batch", type=int, default=4,
                        help="per-rank micro-batch (sequences)")
    parser.add_argument("--seq-len", type=int, default=2048)
    parser.add_argument("--grad-accum", type=int, default=1)
    args = parser.parse_args()

    import torch
    import torch.distributed as dist

    from mlrun_amd.models.llama import LlamaConfig
    from mlrun_amd.models.llama_train import LlamaTrainer

    on_gpu = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if on_gpu:
        torch.cuda.set_device(local_rank)
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("nccl" if on_gpu else "gloo",
                                rank=rank, world_size=world)

    if args.model in ("llama-3-8b", "8b") and on_gpu:
        cfg = LlamaConfig.llama3_8b(max_seq_len=args.seq_len)
    else:
        cfg = LlamaConfig.tiny()
    batch = args.batch if on_gpu else 2
    seq_len = args.seq_len if on_gpu else 64
    trainer = LlamaTrainer(cfg, device=f"cuda:{local_rank}"
                           if on_gpu else "cpu",
                           grad_accum_steps=args.grad_accum)
    gen = torch.Generator().manual_seed(123 + rank)

    def make_batch():
        return torch.randint(1, cfg.vocab_size - 1, (batch, seq_len),
                             generator=gen)

    for _ in range(args.warmup):
        trainer.train_step(make_batch())
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    t0 = time.perf_counter()
    loss = 0.0
    for _ in range(args.steps):
        loss = trainer.train_step(make_batch())
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64)
        if on_gpu:
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens = world * batch * seq_len * args.steps
    if rank == 0:
        print(json.dumps({
            "metric": "fine-tune tokens/sec (Llama-3-8B bf16, "
                      "RCCL over xGMI)",
            "value": round(tokens / elapsed, 1),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if on_gpu else "fp32(cpu)",
            "data": "synthetic (random tokens, random-init weights)",
            "config": {"model": cfg.name,
                       "global_batch": world * batch,
                       "seq_len": seq_len,
                       "grad_accum": args.grad_accum,
                       "parallelism": f"dp{world} (bucketed bf16 "
                                      f"all-reduce overlap)",
                       "final_loss": round(loss, 4)},
        }))
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
