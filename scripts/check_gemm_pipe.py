#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Correctness check: pipelined-ring GEMM vs the burst kernel
(MLRUN_GEMM_PIPE must be set BEFORE import; run once per depth)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from mlrun_amd import ops

torch.manual_seed(3)
dev = "cuda:0"
ok = True
for (m, n, k, ks) in [(32, 6144, 4096, 1), (32, 4096, 4096, 2),
                      (32, 28672, 4096, 1), (32, 4096, 14336, 4),
                      (32, 128256, 4096, 1), (17, 4096, 4096, 2)]:
    a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.5)
    w = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.02)
    c32 = torch.empty(ks * m * n, dtype=torch.float32, device=dev)
    out = torch.empty(m, n, dtype=torch.bfloat16, device=dev)
    ops.skinny_gemm(a, w, out=out, c_f32=c32, ksplit=ks, variant=1)
    ref = (a.float() @ w.float().t())
    err = (out.float() - ref).abs().max().item()
    rel = err / ref.abs().max().item()
    status = "OK" if rel < 2e-2 else "FAIL"
    ok &= rel < 2e-2
    print(f"pipe={os.environ.get('MLRUN_GEMM_PIPE','0')} "
          f"{m}x{n}x{k} ks{ks}: rel {rel:.2e} {status}", flush=True)
print("ALL_OK" if ok else "SOME_FAILED")
sys.exit(0 if ok else 1)
