#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Flagship benchmark: serving-graph req/sec, Llama-3-8B V2ModelServer.

Measures the BASELINE.json north-star metric: requests/second (+ p50)
through the serving graph (router -> LlamaServer.do_event -> predict ->
postprocess) on N GPUs of one node, one serving replica per GPU
(weak scaling — the reference scales nuclio replicas the same way).

One step = one serving-graph event carrying BATCH requests, each a
PROMPT_LEN-token synthetic prompt generating GEN_TOKENS tokens
(random-init weights, synthetic data — no network).

Usage:  python bench.py --gpus N --steps K --warmup W
(for N>1 the driver launches via torch.distributed.run, one rank/GPU)
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=10)
    parser.add_argument("--warmup", type=int, default=3)
    parser.add_argument("--model", default="llama-3-8b",
                        help="llama-3-8b | llama-3-70b | tiny")
    parser.add_argument("--batch", type=int, default=32,
                        help="requests per serving event")
    parser.add_argument("--prompt-len", type=int, default=128)
    parser.add_argument("--gen-tokens", type=int, default=32)
    parser.add_argument("--no-graph", action="store_true")
    parser.add_argument("--kv", default="bf16", choices=["bf16", "fp8"],
                        help="KV-cache dtype (fp8 = e4m3 + row scales)")
    parser.add_argument("--weights", default="bf16",
                        help="bf16 (headline) | fp8 (opt-in fp8-weight "
                             "decode; activations/KV stay bf16)")
    parser.add_argument("--replicas", type=int, default=2,
                        help="engine replicas per GPU, each on its own "
                             "HIP stream with private KV/buffers and "
                             "shared weights (default 2: decode is "
                             "latency-bound, 2 overlapped streams are "
                             "+16%% req/s — measured; p50 rises "
                             "accordingly and is reported)")
    parser.add_argument("--inflight", type=int, default=0,
                        help="concurrent events (default = replicas)")
    args = parser.parse_args()
    inflight = args.inflight or args.replicas

    import torch

    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    n_gpus = max(args.gpus, world_size)
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)
    device = f"cuda:{local_rank}" if on_gpu else "cpu"

    dist = None
    if world_size > 1:
        import torch.distributed as tdist

        dist = tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend="nccl" if on_gpu else "gloo",
                                rank=rank, world_size=world_size)

    import mlrun_amd
    from mlrun_amd.models.llama import LlamaServer

    model_cfg = args.model if on_gpu else "tiny"
    batch = args.batch if on_gpu else 2
    fn = mlrun_amd.new_function(name="bench-serving", kind="serving")
    fn.add_model("llama", class_name=LlamaServer, config=model_cfg,
                 batch_size=batch, max_new_tokens=args.gen_tokens,
                 device=device, use_graph=not args.no_graph,
                 replicas=args.replicas,
                 weight_dtype="fp8w" if args.weights == "fp8" else "bf16",
                 kv_dtype=args.kv)
    server = fn.to_mock_server()

    import random

    random.seed(1234 + rank)
    vocab = 1000 if model_cfg == "tiny" else 120000
    prompt_len = args.prompt_len if on_gpu else 8

    def make_body():
        return {
            "inputs": [[random.randrange(1, vocab)
                        for _ in range(prompt_len)] for _ in range(batch)],
            "max_tokens": args.gen_tokens,
        }

    path = "/v2/models/llama/infer"

    import concurrent.futures

    pool = concurrent.futures.ThreadPoolExecutor(max_workers=inflight)

    def one_step():
        s0 = time.perf_counter()
        server.test(path, body=make_body())
        return (time.perf_counter() - s0) * 1000.0

    # warmup (includes weight init, first prefill, hipGraph capture)
    list(pool.map(lambda _: one_step(), range(max(args.warmup, inflight))))
    if on_gpu:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()

    t0 = time.perf_counter()
    latencies = list(pool.map(lambda _: one_step(), range(args.steps)))
    if on_gpu:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max elapsed over ranks = whole-job time
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if on_gpu:
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_requests = n_gpus * batch * args.steps
    reqs_per_sec = total_requests / elapsed
    latencies.sort()
    p50 = latencies[len(latencies) // 2]

    headline = None
    if rank == 0:
        result = {
            "metric": "serving-graph req/sec (Llama-3-8B V2ModelServer)",
            "value": round(reqs_per_sec, 3),
            "unit": "req/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("bf16" if args.weights != "fp8" else
                      "bf16-act/fp8-weight (opt-in; headline is bf16)") +
                     ("" if args.kv != "fp8" else " + fp8-kv (opt-in)"),
            "data": "synthetic (random prompts, random-init weights)",
            "config": {
                "model": model_cfg,
                "global_batch": n_gpus * batch,
                "seq_len": prompt_len + args.gen_tokens,
                "prompt_len": prompt_len,
                "gen_tokens": args.gen_tokens,
                "parallelism": f"dp{n_gpus} ({args.replicas} engine "
                               f"replica(s)/GPU, {inflight} in flight)",
                "p50_ms": round(p50, 3),
                "tokens_per_sec": round(
                    reqs_per_sec * args.gen_tokens, 1),
                "hipgraph": not args.no_graph,
            },
        }
        print(json.dumps(result), flush=True)
        headline = result

    # BASELINE config 5 rider: with >1 rank on GPUs, also measure the
    # Llama-3-70B TP=world hipGraph-captured decode point (RCCL
    # all-reduces in-graph).  The headline line above is already on
    # stdout — this prints to STDERR and a JSON file only, and a
    # watchdog hard-exits if a collective wedges so the DP record is
    # never lost.
    # only at the full-node run (world 8): at N=2/4 the rider would
    # eat the driver's scaling-sweep budget for a non-headline point
    if dist and on_gpu and world_size >= 8 and \
            os.environ.get("MLRUN_BENCH_TP", "1") != "0":
        import threading

        watchdog = threading.Timer(420.0, lambda: os._exit(0))
        watchdog.daemon = True
        watchdog.start()
        try:
            tp_result = _bench_tp70b(dist, rank, world_size, local_rank)
            if rank == 0 and tp_result:
                print("TP70B " + json.dumps(tp_result), file=sys.stderr,
                      flush=True)
                with open(os.path.join(REPO, "gpurun_out",
                                       "tp70b_bench.json")
                          if os.path.isdir(os.path.join(REPO,
                                                        "gpurun_out"))
                          else "tp70b_bench.json", "w") as f:
                    json.dump({"headline": headline,
                               "tp70b": tp_result}, f, indent=1)
        except Exception as exc:  # never endanger the DP record
            print(f"TP70B bench failed: {exc}", file=sys.stderr)
        finally:
            watchdog.cancel()

    if dist:
        dist.destroy_process_group()


def _bench_tp70b(dist, rank, world_size, local_rank, steps=3, warmup=2,
                 batch=16, prompt_len=128, gen_tokens=32, cfg=None,
                 device=None, use_graph=True):
    """Llama-3-70B decode at TP=world_size over xGMI, hipGraph-captured
    (BASELINE config 5).  Reuses the already-initialized process
    group.  cfg/device overrides exist so the exact code path also
    runs as a 2-rank gloo CPU test (tests/test_tensor_parallel.py)."""
    import torch

    from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

    cfg = cfg or LlamaConfig.llama3_70b()
    device = device or f"cuda:{local_rank}"
    on_gpu = device.startswith("cuda")
    engine = LlamaDecodeEngine(cfg, batch, device=device,
                               tp_group=None, tp_rank=rank,
                               tp_size=world_size,
                               use_graph=use_graph and on_gpu,
                               seed=7)
    gen = torch.Generator().manual_seed(77)

    def make_prompts():
        return torch.randint(1, cfg.vocab_size - 1, (batch, prompt_len),
                             generator=gen)

    for _ in range(warmup):
        engine.reset()
        engine.generate(make_prompts(), max_new_tokens=gen_tokens)
    if on_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(steps):
        engine.reset()
        engine.generate(make_prompts(), max_new_tokens=gen_tokens)
    if on_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    elapsed = time.perf_counter() - t0
    t = torch.tensor([elapsed], dtype=torch.float64)
    if on_gpu:
        t = t.cuda()
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())
    if rank != 0:
        return None
    return {
        "metric": "serving req/sec (Llama-3-70B, TP over xGMI)",
        "value": round(batch * steps / elapsed, 3),
        "unit": "req/s",
        "n_gpus": world_size,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(elapsed / steps * 1000.0, 3),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic (random prompts, random-init weights)",
        "config": {"model": cfg.name, "global_batch": batch,
                   "seq_len": prompt_len + gen_tokens,
                   "parallelism": f"tp{world_size}", "hipgraph": True},
    }


if __name__ == "__main__":
    main()
