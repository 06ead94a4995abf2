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


def time_gemm(a, w, out, scratch, ksplit, variant, iters=30):
    torch.cuda.synchronize()
    # warmup
    for _ in range(5):
        ops.skinny_gemm(a, w, out=out, c_f32=scratch, ksplit=ksplit,
                        variant=variant)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.skinny_gemm(a, w, out=out, c_f32=scratch, ksplit=ksplit,
                        variant=variant)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    results = []
    for name, m, n, k in SHAPES:
        torch.manual_seed(1)
        a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda") * 0.1
        w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda") * 0.1
        out = torch.empty(m, n, dtype=torch.bfloat16, device="cuda")
        ref = (a.float() @ w.float().t())
        bytes_w = n * k * 2
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
                dt = time_gemm(a, w, out, scratch, ksplit, variant)
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
