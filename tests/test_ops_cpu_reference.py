# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Independent validation of the CPU reference ops.

The GPU numerics tests compare HIP kernels against these CPU paths —
a bug present in BOTH would slip through, so here each CPU reference
is checked against a formula written independently from torch
primitives (or, for rope, against the published rotation identity)."""

import math

import pytest
import torch

from mlrun_amd import ops


class TestAgainstIndependentFormulas:
    def test_rmsnorm_formula(self):
        torch.manual_seed(0)
        x = torch.randn(4, 64, dtype=torch.bfloat16)
        w = torch.randn(64, dtype=torch.bfloat16)
        got = ops.rmsnorm(x, w, eps=1e-5).float()
        xf = x.float()
        expect = xf / torch.sqrt((xf * xf).mean(-1, keepdim=True) + 1e-5) \
            * w.float()
        assert torch.allclose(got, expect, atol=2e-2, rtol=2e-2)

    def test_fused_add_rmsnorm_updates_residual(self):
        torch.manual_seed(1)
        x = torch.randn(3, 32, dtype=torch.bfloat16)
        res = torch.randn(3, 32, dtype=torch.bfloat16)
        w = torch.ones(32, dtype=torch.bfloat16)
        res_before = res.clone()
        out = ops.fused_add_rmsnorm(x, w, residual=res, eps=1e-5).float()
        summed = (x.float() + res_before.float())
        assert torch.allclose(res.float(), summed, atol=2e-2, rtol=2e-2)
        expect = summed / torch.sqrt(
            (summed * summed).mean(-1, keepdim=True) + 1e-5)
        assert torch.allclose(out, expect, atol=3e-2, rtol=3e-2)

    def test_rope_rotation_identity(self):
        """Rope preserves pairwise norms and rotates by pos*theta_i:
        check |(x_i, x_{i+d/2})| invariance AND the explicit angle."""
        d = 32
        table = ops.build_rope_cos_sin(16, d, theta=10000.0)
        x = torch.randn(1, 1, d, dtype=torch.bfloat16)
        rotated = x.clone()
        positions = torch.tensor([3], dtype=torch.int32)
        ops.rope_inplace(rotated, positions, table)
        xf, rf = x.float()[0, 0], rotated.float()[0, 0]
        half = d // 2
        for i in range(half):
            n_before = math.hypot(xf[i], xf[i + half])
            n_after = math.hypot(rf[i], rf[i + half])
            assert n_after == pytest.approx(n_before, rel=2e-2, abs=2e-2)
            angle = 3 * (10000.0 ** (-i / half))
            expect_lo = xf[i] * math.cos(angle) - xf[i + half] * \
                math.sin(angle)
            assert rf[i] == pytest.approx(float(expect_lo), rel=3e-2,
                                          abs=3e-2)

    def test_swiglu_formula(self):
        torch.manual_seed(2)
        gu = torch.randn(4, 64, dtype=torch.bfloat16)  # gate|up fused
        got = ops.swiglu_fused(gu).float()
        g, u = gu.float().chunk(2, dim=-1)
        expect = torch.nn.functional.silu(g) * u
        assert torch.allclose(got, expect, atol=2e-2, rtol=2e-2)

    def test_softmax_matches_torch(self):
        torch.manual_seed(3)
        x = torch.randn(5, 40, dtype=torch.bfloat16)
        got = ops.softmax(x).float()
        expect = torch.softmax(x.float(), dim=-1)
        assert torch.allclose(got, expect, atol=1e-2, rtol=1e-2)

    def test_attn_decode_matches_sdpa(self):
        """CPU reference attention vs torch SDPA (GQA expanded)."""
        B, Hq, Hkv, S, D = 2, 8, 2, 24, 128
        torch.manual_seed(4)
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16)
        kc = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16)
        vc = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16)
        lens = torch.full((B,), S, dtype=torch.int32)
        got = ops.attn_decode(q, kc, vc, lens).float()
        expect = torch.nn.functional.scaled_dot_product_attention(
            q.float().unsqueeze(2),  # [B,Hq,1,D]
            kc.float().repeat_interleave(Hq // Hkv, dim=1),
            vc.float().repeat_interleave(Hq // Hkv, dim=1),
        ).squeeze(2)
        assert torch.allclose(got, expect, atol=2e-2, rtol=2e-2)

    def test_skinny_gemm_matches_matmul(self):
        torch.manual_seed(5)
        a = torch.randn(8, 64, dtype=torch.bfloat16)
        w = torch.randn(32, 64, dtype=torch.bfloat16)
        got = ops.skinny_gemm(a, w).float()
        expect = a.float() @ w.float().t()
        assert torch.allclose(got, expect, atol=5e-2, rtol=5e-2)

    def test_tree_ensemble_single_tree_by_hand(self):
        # one tree: f0 < 0 -> leaf 1.0 else leaf 2.0
        nodes = {
            "feature_idx": torch.tensor([0, -1, -1], dtype=torch.int32),
            "threshold": torch.tensor([0.0, 0.0, 0.0]),
            "left": torch.tensor([1, 0, 0], dtype=torch.int32),
            "right": torch.tensor([2, 0, 0], dtype=torch.int32),
            "leaf_value": torch.tensor([0.0, 1.0, 2.0]),
            "tree_offsets": torch.tensor([0, 3], dtype=torch.int32),
        }
        feats = torch.tensor([[-1.0], [1.0], [0.0]])
        out = ops.tree_ensemble_predict(feats, nodes, base_score=0.5)
        assert out.tolist() == [1.5, 2.5, 2.5]  # 0.0 goes right (<)
