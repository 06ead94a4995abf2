#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""HTTP-level serving load benchmark.

Measures the FULL stack — uvicorn HTTP host -> serving graph ->
V2ModelServer -> dynamic batcher -> hipGraph decode — under concurrent
clients, the shape of the reference's serving benchmark harness
(N concurrent clients, single-prompt requests, req/s + latency
percentiles).  bench.py times the engine/graph path; this script adds
the network + batching layer on top.

  python scripts/bench_serving_http.py --clients 16 --requests 512 \
      --gen-tokens 32 [--model llama-3-8b] [--weights fp8]
"""

import argparse
import json
import multiprocessing
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import torch  # noqa: E402


def _client_proc(url, payload, gen_tokens, remaining, results):
    import requests as http

    session = http.Session()
    headers = {"content-type": "application/json"}
    while True:
        with remaining.get_lock():
            if remaining.value <= 0:
                return
            remaining.value -= 1
        start = time.perf_counter()
        resp = session.post(url, data=payload, headers=headers,
                            timeout=300)
        elapsed = (time.perf_counter() - start) * 1000
        resp.raise_for_status()
        out = resp.json()["outputs"]
        assert len(out[0]) == gen_tokens, out
        results.put(elapsed)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--clients", type=int, default=16)
    parser.add_argument("--requests", type=int, default=256)
    parser.add_argument("--model", default=None)
    parser.add_argument("--prompt-len", type=int, default=128)
    parser.add_argument("--gen-tokens", type=int, default=32)
    parser.add_argument("--batch-window-ms", type=float, default=8.0)
    parser.add_argument("--batch", type=int, default=32)
    parser.add_argument("--replicas", type=int, default=1)
    parser.add_argument("--weights", default="bf16",
                        choices=["bf16", "fp8"])
    parser.add_argument("--warmup", type=int, default=32)
    parser.add_argument("--scheduling", default="batch",
                        choices=["batch", "continuous"])
    parser.add_argument("--workers", type=int, default=0,
                        help="serving worker processes (L4 round-robin)")
    parser.add_argument("--max-workers", type=int, default=0,
                        help="autoscale ceiling (with_replicas max)")
    args = parser.parse_args()

    on_gpu = torch.cuda.is_available()
    model = args.model or ("llama-3-8b" if on_gpu else "tiny")

    import mlrun_amd
    from mlrun_amd.models.llama import LlamaServer

    fn = mlrun_amd.new_function(name="http-bench", kind="serving")
    fn.add_model("llm", class_name=LlamaServer, config=model,
                 batch_size=args.batch,
                 max_new_tokens=args.gen_tokens,
                 batch_window_ms=args.batch_window_ms,
                 replicas=args.replicas,
                 weight_dtype="fp8w" if args.weights == "fp8" else "bf16",
                 scheduling=args.scheduling)
    if args.max_workers:
        fn.with_replicas(max(args.workers, 1), args.max_workers)
        address = fn.deploy()
    else:
        address = fn.deploy(workers=args.workers)
    print(f"serving at {address}", file=sys.stderr)

    vocab = 1000
    rng = torch.Generator().manual_seed(0)
    prompt = torch.randint(0, vocab, (args.prompt_len,),
                           generator=rng).tolist()
    url = f"{address}/v2/models/llm/infer"
    payload = json.dumps({"inputs": [prompt],
                          "max_tokens": args.gen_tokens})

    # clients are separate PROCESSES — one client process running
    # dozens of threads is GIL-bound and under-drives the server
    ctx = multiprocessing.get_context("fork")
    remaining = ctx.Value("i", 0)
    results: multiprocessing.Queue = ctx.Queue()

    def run_wave(n_requests):
        with remaining.get_lock():
            remaining.value = n_requests
        procs = [ctx.Process(target=_client_proc,
                             args=(url, payload, args.gen_tokens,
                                   remaining, results))
                 for _ in range(args.clients)]
        start = time.perf_counter()
        [p.start() for p in procs]
        [p.join() for p in procs]
        elapsed = time.perf_counter() - start
        lat = []
        while not results.empty():
            lat.append(results.get())
        return elapsed, lat

    run_wave(args.warmup)  # captures hipGraphs, fills caches
    wall, latencies = run_wave(args.requests)
    fn.stop()

    latencies.sort()
    result = {
        "metric": "HTTP serving req/sec (Llama V2ModelServer, "
                  "concurrent clients)",
        "value": round(args.requests / wall, 2),
        "unit": "req/s",
        "p50_ms": round(statistics.median(latencies), 2),
        "p99_ms": round(latencies[int(len(latencies) * 0.99) - 1], 2),
        "clients": args.clients,
        "requests": args.requests,
        "gen_tokens": args.gen_tokens,
        "tokens_per_sec": round(args.requests * args.gen_tokens / wall, 1),
        "batch_window_ms": args.batch_window_ms,
        "scheduling": args.scheduling,
        "workers": args.workers,
        "max_workers": args.max_workers,
        "workers_final": len(fn._worker_pool.ports)
        if getattr(fn, "_worker_pool", None) else args.workers or 1,
        "model": model,
        "weights": args.weights,
        "replicas": args.replicas,
    }
    print(json.dumps(result))


if __name__ == "__main__":
    main()
