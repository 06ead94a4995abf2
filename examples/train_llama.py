#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Distributed Llama fine-tune (baseline config 4).

Run via the mpijob runtime (one rank per MI355X, RCCL over xGMI):

    fn = mlrun_amd.new_function(kind="mpijob",
                                command="examples/train_llama.py")
    fn.with_replicas(8)
    fn.run(params={"model": "llama-3-8b", "steps": 20})

or standalone / under torchrun.  Synthetic token data, random init
(no network).  Prints per-rank tokens/s; rank 0 logs aggregate results
and checkpoints the model artifact.
"""

import argparse
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default=None)
    parser.add_argument("--steps", type=int, default=None)
    parser.add_argument("--warmup", type=int, default=2)
    parser.add_argument("--batch", type=int, default=None)
    parser.add_argument("--seq-len", type=int, default=None)
    parser.add_argument("--checkpoint", action="store_true")
    args = parser.parse_args()

    import torch

    import mlrun_amd
    from mlrun_amd.models.llama import LlamaConfig
    from mlrun_amd.models.llama_train import LlamaTrainer
    from mlrun_amd.parallel.ddp import init_process_group

    ctx = mlrun_amd.get_or_create_ctx("train-llama")
    model_name = args.model or ctx.get_param("model", None) or (
        "llama-3-8b" if torch.cuda.is_available() else "tiny")
    steps = args.steps or int(ctx.get_param("steps", 10))
    batch = args.batch or int(ctx.get_param("batch",
                                            4 if torch.cuda.is_available()
                                            else 2))
    seq_len = args.seq_len or int(ctx.get_param(
        "seq_len", 2048 if torch.cuda.is_available() else 64))

    backend = os.environ.get("MLRUN_DIST_BACKEND") or (
        "nccl" if torch.cuda.is_available() else "gloo")
    rank, world = init_process_group(backend=backend)
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"

    if model_name in ("llama-3-8b", "8b"):
        cfg = LlamaConfig.llama3_8b(max_seq_len=seq_len)
    elif model_name == "tiny":
        cfg = LlamaConfig.tiny(max_seq_len=seq_len)
    else:
        cfg = LlamaConfig.llama3_70b(max_seq_len=seq_len)

    torch.manual_seed(1234)  # same init across ranks (then broadcast)
    trainer = LlamaTrainer(cfg, device=device, context=ctx)

    gen = torch.Generator().manual_seed(4321 + rank)
    def make_batch():
        return torch.randint(0, cfg.vocab_size, (batch, seq_len),
                             generator=gen)

    for _ in range(args.warmup):
        trainer.train_step(make_batch())
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if world > 1:
        import torch.distributed as dist

        dist.barrier()
    start = time.perf_counter()
    losses = []
    for _ in range(steps):
        losses.append(trainer.train_step(make_batch()))
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if world > 1:
        import torch.distributed as dist

        dist.barrier()
    elapsed = time.perf_counter() - start
    tokens_per_sec = world * batch * seq_len * steps / elapsed
    print(f"rank {rank}: {steps} steps in {elapsed:.2f}s "
          f"loss {losses[0]:.3f} -> {losses[-1]:.3f}")
    if ctx.is_logging_worker():
        ctx.log_result("tokens_per_sec", round(tokens_per_sec, 1))
        ctx.log_result("ms_per_step", round(elapsed / steps * 1000, 2))
        ctx.log_result("final_loss", losses[-1])
        ctx.log_result("first_loss", losses[0])
        ctx.log_result("world_size", world)
        if args.checkpoint or ctx.get_param("checkpoint", False):
            trainer.save_checkpoint("llama-finetuned")
        ctx.commit(completed=True)
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
