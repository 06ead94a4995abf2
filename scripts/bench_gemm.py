#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Skinny-GEMM variant sweep on MI355X: times each (shape, variant,
ksplit) cell and reports effective weight-read bandwidth.  Shapes are
the Llama-3-8B decode projections."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import torch  # noqa: E402

from mlrun_amd import ops  # noqa: E402

SHAPES = [
    ("qkv", 16, 6144, 4096),
    ("wo", 16, 4096, 4096),
    ("wgu", 16, 28672, 4096),
    ("wdown", 16, 4096, 14336),
    ("lm_head", 16, 128256, 4096),
]


def time_gemm(a, weights, out, scratch, ksplit, variant, iters=32):
    """weights: LIST of weight copies rotated per iteration so the
    working set exceeds the 256 MiB Infinity Cache — timing a single
    reused W measures L3 bandwidth, not HBM (the in-graph reality;
    see profiles/README.md L3-trap note)."""
    torch.cuda.synchronize()
    for i in range(4):
        ops.skinny_gemm(a, weights[i % len(weights)], out=out,
                        c_f32=scratch, ksplit=ksplit, variant=variant)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(iters):
        ops.skinny_gemm(a, weights[i % len(weights)], out=out,
                        c_f32=scratch, ksplit=ksplit, variant=variant)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    results = []
    for name, m, n, k in SHAPES:
        torch.manual_seed(1)
        a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda") * 0.1
        bytes_w = n * k * 2
        # enough rotating copies that the set exceeds 2x L3 (512 MB)
        n_copies = max(2, (512 * 2 ** 20) // bytes_w + 1)
        n_copies = min(n_copies, 12)
        weights = [torch.randn(n, k, dtype=torch.bfloat16,
                               device="cuda") * 0.1
                   for _ in range(n_copies)]
        w = weights[0]
        out = torch.empty(m, n, dtype=torch.bfloat16, device="cuda")
        ref = (a.float() @ w.float().t())
        best = None
        for variant in (0, 1, 2):
            for ksplit in (1, 2, 4, 8, 16):
                scratch = torch.empty(max(ksplit * m * n, 1),
                                      dtype=torch.float32, device="cuda")
                ops.skinny_gemm(a, w, out=out, c_f32=scratch, ksplit=ksplit,
                                variant=variant)
                err = (out.float() - ref).abs().max().item()
                if err > 0.02 * (k ** 0.5):
                    print(f"{name} v{variant} ks{ksplit}: WRONG "
                          f"(err {err:.3f})")
                    continue
                dt = time_gemm(a, weights, out, scratch, ksplit, variant)
                bw = bytes_w / dt / 1e12
                line = (f"{name:8s} v{variant} ks{ksplit:2d}: "
                        f"{dt * 1e6:8.1f}us  {bw:5.2f} TB/s")
                print(line, flush=True)
                results.append((name, variant, ksplit, dt, bw))
                if best is None or dt < best[3]:
                    best = (name, variant, ksplit, dt, bw)
        print(f"BEST {name}: v{best[1]} ks{best[2]} {best[4]:.2f} TB/s\n",
              flush=True)
    # default-plan numbers
    print("--- default plan (pick_gemm_plan):")
    for name, m, n, k in SHAPES:
        ksplit, variant = ops.pick_gemm_plan(m, n, k)
        print(f"{name:8s} -> v{variant} ks{ksplit}")


if __name__ == "__main__":
    main()
