#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Long-context decode-attention nsplit sweep (NOTES.md lead #3).

Times attn_decode at llama-3-8b GQA shapes (Hq=32, Hkv=8, D=128) for
context lengths 512..8192 and nsplit 1..16, printing us/call and the
best split per (B, S).  Run on a GPU box:

  python scripts/bench_attn_longctx.py [--batch 32]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import torch  # noqa: E402

from mlrun_amd import ops  # noqa: E402


def time_call(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    start = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - start) / iters * 1e6


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--batch", type=int, default=32)
    parser.add_argument("--iters", type=int, default=50)
    parser.add_argument("--kv", default="bf16", choices=["bf16", "fp8"])
    args = parser.parse_args()
    assert torch.cuda.is_available()
    device = "cuda:0"
    B, Hq, Hkv, D = args.batch, 32, 8, 128

    print(f"# B={B} Hq={Hq} Hkv={Hkv} D={D} "
          f"(grid base = B*Hkv = {B * Hkv} workgroups)")
    print(f"{'S':>6} {'bytes_kv':>9} " +
          " ".join(f"ns{n:>2}" for n in [1, 2, 4, 8, 16]) +
          "   best  heuristic  TB/s@best")
    for S in [512, 1024, 2048, 4096, 8192]:
        Smax = S
        q = torch.randn(B, Hq, D, device=device, dtype=torch.bfloat16)
        k = torch.randn(B, Hkv, Smax, D, device=device,
                        dtype=torch.bfloat16)
        v = torch.randn_like(k)
        ks = vs = None
        if args.kv == "fp8":
            k, ks = ops.quantize_kv_rows(k)
            v, vs = ops.quantize_kv_rows(v)
            k, v = k.contiguous(), v.contiguous()
            ks, vs = ks.contiguous(), vs.contiguous()
        seq_lens = torch.full((B,), S, device=device, dtype=torch.int32)
        out = torch.empty(B, Hq, D, device=device, dtype=torch.bfloat16)
        bytes_per = 1 if args.kv == "fp8" else 2
        kv_bytes = 2 * B * Hkv * S * D * bytes_per
        times = {}
        for nsplit in [1, 2, 4, 8, 16]:
            ws = torch.empty(B, Hq, nsplit, D + 2, device=device,
                             dtype=torch.float32) if nsplit > 1 else None
            times[nsplit] = time_call(
                lambda: ops.attn_decode(q, k, v, seq_lens, out=out,
                                        partial_ws=ws, nsplit=nsplit,
                                        k_scale=ks, v_scale=vs),
                args.iters)
        best = min(times, key=times.get)
        heur = ops.pick_attn_nsplit(B, Hkv)
        tbs = kv_bytes / (times[best] * 1e-6) / 1e12
        print(f"{S:>6} {kv_bytes >> 20:>8}M " +
              " ".join(f"{times[n]:>4.0f}" for n in [1, 2, 4, 8, 16]) +
              f"   ns{best:<3} ns{heur:<3}    {tbs:.2f}")


if __name__ == "__main__":
    main()
