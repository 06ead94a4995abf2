# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""GPU numerics tests: every HIP kernel vs. the plain-torch fp32
reference of the same op (asymmetric random inputs — transposes and
layout bugs must not survive)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


def _rand_bf16(*shape, scale=1.0, seed=None):
    if seed is not None:
        torch.manual_seed(seed)
    return (torch.randn(*shape, dtype=torch.float32) * scale).to(
        torch.bfloat16)


@requires_gpu
class TestHipOpsLoaded:
    def test_extension_is_native(self):
        """The HIP extension must actually be loaded on a GPU box."""
        from mlrun_amd.ops import HAVE_HIP_OPS

        assert HAVE_HIP_OPS, "_hip_ops extension missing on GPU box"
        import mlrun_amd._hip_ops as ext

        assert ext.__file__.endswith(".so")


@requires_gpu
class TestRMSNorm:
    def test_rmsnorm(self):
        from mlrun_amd import ops

        x = _rand_bf16(9, 4096, seed=0)
        w = _rand_bf16(4096, seed=1)
        ref = ops.rmsnorm(x, w)
        got = ops.rmsnorm(x.cuda(), w.cuda()).cpu()
        assert torch.allclose(ref.float(), got.float(), atol=2e-2, rtol=2e-2)

    def test_fused_residual(self):
        from mlrun_amd import ops

        x = _rand_bf16(4, 2048, seed=2)
        res = _rand_bf16(4, 2048, seed=3)
        w = _rand_bf16(2048, seed=4)
        res_ref = res.clone()
        ref = ops.fused_add_rmsnorm(x, w, residual=res_ref)
        res_gpu = res.clone().cuda()
        got = ops.fused_add_rmsnorm(x.cuda(), w.cuda(), residual=res_gpu)
        assert torch.allclose(ref.float(), got.cpu().float(), atol=2e-2,
                              rtol=2e-2)
        assert torch.allclose(res_ref.float(), res_gpu.cpu().float(),
                              atol=2e-2, rtol=2e-2)


@requires_gpu
class TestRoPE:
    def test_rope_matches_reference(self):
        from mlrun_amd import ops

        T, H, D = 5, 8, 128
        q = _rand_bf16(T, H, D, seed=5)
        positions = torch.tensor([0, 3, 17, 100, 511], dtype=torch.int32)
        table = ops.build_rope_cos_sin(1024, D)
        q_ref = q.clone()
        ops.rope_inplace(q_ref, positions, table)
        q_gpu = q.clone().cuda()
        ops.rope_inplace(q_gpu, positions.cuda(), table.cuda())
        assert torch.allclose(q_ref.float(), q_gpu.cpu().float(), atol=2e-2,
                              rtol=2e-2)


@requires_gpu
class TestSiluMul:
    def test_silu_mul(self):
        from mlrun_amd import ops

        g = _rand_bf16(16, 1024, seed=6)
        u = _rand_bf16(16, 1024, seed=7)
        ref = ops.silu_mul(g, u)
        got = ops.silu_mul(g.cuda(), u.cuda()).cpu()
        assert torch.allclose(ref.float(), got.float(), atol=2e-2, rtol=2e-2)


@requires_gpu
class TestSkinnyGemm:
    @pytest.mark.parametrize("m,n,k", [(1, 128, 256), (16, 4096, 4096),
                                       (13, 1000, 512), (8, 6144, 4096),
                                       (32, 4096, 4096), (24, 1024, 512)])
    def test_vs_fp32(self, m, n, k):
        from mlrun_amd import ops

        a = _rand_bf16(m, k, scale=0.5, seed=m)
        w = _rand_bf16(n, k, scale=0.5, seed=n)
        ref = a.float() @ w.float().t()
        got = ops.skinny_gemm(a.cuda(), w.cuda()).cpu().float()
        # bf16 inputs: tolerance scales with sqrt(K)
        tol = 0.02 * math.sqrt(k)
        assert (ref - got).abs().max().item() < tol, \
            f"max err {(ref - got).abs().max().item()} at m={m} n={n} k={k}"

    def test_ksplit_paths(self):
        from mlrun_amd import ops

        a = _rand_bf16(16, 2048, scale=0.5, seed=42)
        w = _rand_bf16(256, 2048, scale=0.5, seed=43)
        ref = a.float() @ w.float().t()
        for ksplit in (1, 4, 8):
            got = ops.skinny_gemm(a.cuda(), w.cuda(),
                                  ksplit=ksplit).cpu().float()
            assert (ref - got).abs().max().item() < 1.0, f"ksplit={ksplit}"


@requires_gpu
class TestAttnDecode:
    @pytest.mark.parametrize("b,hq,hkv,s", [(2, 8, 2, 64), (4, 32, 8, 300),
                                            (16, 32, 8, 1024)])
    def test_vs_fp32(self, b, hq, hkv, s):
        from mlrun_amd import ops

        D = 128
        torch.manual_seed(b * 100 + s)
        q = _rand_bf16(b, hq, D)
        smax = s + 17
        kc = _rand_bf16(b, hkv, smax, D)
        vc = _rand_bf16(b, hkv, smax, D)
        seq_lens = torch.randint(1, s + 1, (b,), dtype=torch.int32)
        ref = ops.attn_decode(q, kc, vc, seq_lens)
        got = ops.attn_decode(q.cuda(), kc.cuda(), vc.cuda(),
                              seq_lens.cuda()).cpu()
        assert torch.allclose(ref.float(), got.float(), atol=3e-2, rtol=3e-2)

    def test_kv_append(self):
        from mlrun_amd import ops

        B, Hkv, Smax, D = 3, 4, 32, 128
        kc = torch.zeros(B, Hkv, Smax, D, dtype=torch.bfloat16)
        vc = torch.zeros_like(kc)
        knew = _rand_bf16(B, Hkv, D, seed=9)
        vnew = _rand_bf16(B, Hkv, D, seed=10)
        positions = torch.tensor([0, 5, 31], dtype=torch.int32)
        kc_ref, vc_ref = kc.clone(), vc.clone()
        ops.kv_append(kc_ref, vc_ref, knew, vnew, positions)
        kc_gpu, vc_gpu = kc.cuda(), vc.cuda()
        ops.kv_append(kc_gpu, vc_gpu, knew.cuda(), vnew.cuda(),
                      positions.cuda())
        assert torch.equal(kc_ref, kc_gpu.cpu())
        assert torch.equal(vc_ref, vc_gpu.cpu())


@requires_gpu
class TestSoftmax:
    def test_vs_fp32(self):
        from mlrun_amd import ops

        x = _rand_bf16(32, 1000, scale=3.0, seed=11)
        ref = ops.softmax(x)
        got = ops.softmax(x.cuda()).cpu()
        assert torch.allclose(ref.float(), got.float(), atol=1e-2, rtol=1e-2)


@requires_gpu
class TestTreeEnsemble:
    def test_vs_reference(self):
        from mlrun_amd import ops
        from mlrun_amd.frameworks.tree import random_forest_nodes

        torch.manual_seed(12)
        nodes = random_forest_nodes(n_trees=20, depth=5, n_features=16,
                                    seed=12)
        feats = torch.randn(512, 16)
        ref = ops.tree_ensemble_predict(feats, nodes, base_score=0.5)
        nodes_gpu = {k: v.cuda() for k, v in nodes.items()}
        got = ops.tree_ensemble_predict(feats.cuda(), nodes_gpu,
                                        base_score=0.5).cpu()
        assert torch.allclose(ref, got, atol=1e-4)


@requires_gpu
class TestWindowAgg:
    def test_ingest_reduce(self):
        from mlrun_amd import ops

        torch.manual_seed(13)
        n_keys, n_periods = 64, 8
        n_events = 5000
        keys = torch.randint(0, n_keys, (n_events,), dtype=torch.int32)
        values = torch.randn(n_events)
        period_idx = torch.randint(0, n_periods, (n_events,),
                                   dtype=torch.int32)
        ring_ref = torch.zeros(n_keys, n_periods, 4)
        ops.window_ingest(ring_ref, keys, values, period_idx)
        ref = ops.window_reduce(ring_ref, window_periods=4, current_period=6)

        ring_gpu = torch.zeros(n_keys, n_periods, 4, device="cuda")
        ops.window_ingest(ring_gpu, keys.cuda(), values.cuda(),
                          period_idx.cuda())
        got = ops.window_reduce(ring_gpu, window_periods=4,
                                current_period=6).cpu()
        assert torch.allclose(ref[:, :2], got[:, :2], atol=1e-2, rtol=1e-3)
        assert torch.allclose(ref[:, 2], got[:, 2], atol=1e-3, rtol=1e-3)

    def test_ingest_reduce_f64_sq(self):
        """f64 sumsq/count/sum ring (stdvar precision path)."""
        from mlrun_amd import ops

        torch.manual_seed(14)
        n_keys, n_periods = 32, 6
        n_events = 3000
        keys = torch.randint(0, n_keys, (n_events,), dtype=torch.int32)
        values = torch.randn(n_events) * 50 + 60  # large mean, small var
        period_idx = torch.randint(0, n_periods, (n_events,),
                                   dtype=torch.int32)
        ring_ref = torch.zeros(n_keys, n_periods, 4, dtype=torch.float64)
        ops.window_ingest(ring_ref, keys, values, period_idx)
        ref = ops.window_reduce(ring_ref, window_periods=3,
                                current_period=4)
        ring_gpu = torch.zeros(n_keys, n_periods, 4, dtype=torch.float64,
                               device="cuda")
        ops.window_ingest(ring_gpu, keys.cuda(), values.cuda(),
                          period_idx.cuda())
        got = ops.window_reduce(ring_gpu, window_periods=3,
                                current_period=4).cpu()
        assert torch.allclose(ref, got, atol=1e-6, rtol=1e-9)

    def test_min_max_first_last_gpu_matches_cpu(self):
        """GPU per-period min/max + packed first/last cells vs the
        numpy fallback — all four windowed ops bit-match."""
        from mlrun_amd import ops

        torch.manual_seed(15)
        n_keys, n_periods = 48, 8
        n_events = 4000
        keys = torch.randint(0, n_keys, (n_events,), dtype=torch.int32)
        values = torch.randn(n_events) * 10
        ts = torch.randint(1_000_000, 2_000_000, (n_events,),
                           dtype=torch.int32)
        period_idx = torch.randint(0, n_periods, (n_events,),
                                   dtype=torch.int32)

        mm_ref = torch.empty(n_keys, n_periods, 2, dtype=torch.int32)
        mm_ref[:, :, 0] = ops.MM_MIN_EMPTY
        mm_ref[:, :, 1] = ops.MM_MAX_EMPTY
        fl_ref = torch.empty(n_keys, n_periods, 2, dtype=torch.int64)
        fl_ref[:, :, 0] = ops.FL_FIRST_EMPTY
        fl_ref[:, :, 1] = ops.FL_LAST_EMPTY
        ops.window_ingest_mm(mm_ref, keys, values, period_idx)
        ops.window_ingest_fl(fl_ref, keys, values, ts, period_idx)
        ref = ops.window_reduce_mmfl(mm_ref, fl_ref, 5, 6)

        mm_gpu = mm_ref.clone().zero_().cuda()
        mm_gpu[:, :, 0] = ops.MM_MIN_EMPTY
        mm_gpu[:, :, 1] = ops.MM_MAX_EMPTY
        fl_gpu = fl_ref.clone().zero_().cuda()
        fl_gpu[:, :, 0] = ops.FL_FIRST_EMPTY
        fl_gpu[:, :, 1] = ops.FL_LAST_EMPTY
        ops.window_ingest_mm(mm_gpu, keys.cuda(), values.cuda(),
                             period_idx.cuda())
        ops.window_ingest_fl(fl_gpu, keys.cuda(), values.cuda(),
                             ts.cuda(), period_idx.cuda())
        got = ops.window_reduce_mmfl(mm_gpu, fl_gpu, 5, 6).cpu()
        assert torch.equal(ref, got)

    def test_window_ring_all_ops_gpu_matches_cpu(self):
        """End-to-end WindowRing on GPU vs CPU: all 10 ops agree."""
        import math

        from mlrun_amd.feature_store.online import WindowRing

        torch.manual_seed(16)
        n = 2000
        keys = torch.randint(0, 20, (n,))
        values = torch.randn(n) * 30 + 100
        ts = torch.sort(torch.randint(0, 3600, (n,)).float()).values
        cpu_ring = WindowRing(60, 61, device="cpu", capacity=32)
        gpu_ring = WindowRing(60, 61, device="cuda:0", capacity=32)
        cpu_ring.ingest(keys, values, ts)
        gpu_ring.ingest(keys, values, ts)
        now = float(ts.max())
        got_cpu = cpu_ring.window_values(1800, now)
        got_gpu = gpu_ring.window_values(1800, now)
        for op in ("count", "sum", "avg", "sqr", "stdvar", "stddev",
                   "min", "max", "first", "last"):
            a, b = got_cpu[op], got_gpu[op]
            mask = ~torch.isnan(a)
            assert torch.equal(torch.isnan(a), torch.isnan(b)), op
            assert torch.allclose(a[mask], b[mask], atol=1e-2,
                                  rtol=1e-4), (op, a[mask], b[mask])


@requires_gpu
class TestFusedDecodeOps:
    def test_swiglu_fused(self):
        from mlrun_amd import ops

        gu = _rand_bf16(16, 2048, seed=20)
        ref = ops.swiglu_fused(gu)
        got = ops.swiglu_fused(gu.cuda()).cpu()
        assert torch.allclose(ref.float(), got.float(), atol=2e-2, rtol=2e-2)

    def test_rope_kv_fused(self):
        from mlrun_amd import ops

        B, Hq, Hkv, D, Smax = 4, 8, 2, 128, 64
        qkv = _rand_bf16(B, (Hq + 2 * Hkv) * D, seed=21)
        kc = torch.zeros(B, Hkv, Smax, D, dtype=torch.bfloat16)
        vc = torch.zeros_like(kc)
        positions = torch.tensor([0, 3, 10, 63], dtype=torch.int32)
        table = ops.build_rope_cos_sin(Smax, D)

        qkv_ref, kc_ref, vc_ref = qkv.clone(), kc.clone(), vc.clone()
        ops.rope_kv_fused(qkv_ref, kc_ref, vc_ref, positions, table, Hq)
        qkv_gpu, kc_gpu, vc_gpu = qkv.cuda(), kc.cuda(), vc.cuda()
        ops.rope_kv_fused(qkv_gpu, kc_gpu, vc_gpu, positions.cuda(),
                          table.cuda(), Hq)
        assert torch.allclose(qkv_ref.float(), qkv_gpu.cpu().float(),
                              atol=2e-2, rtol=2e-2)
        assert torch.allclose(kc_ref.float(), kc_gpu.cpu().float(),
                              atol=2e-2, rtol=2e-2)
        assert torch.equal(vc_ref, vc_gpu.cpu())


@requires_gpu
class TestFP8Gemm:
    @pytest.mark.parametrize("m,n,k,ks", [(16, 4096, 4096, 1),
                                          (32, 6144, 4096, 2),
                                          (16, 1024, 2048, 4)])
    def test_vs_fp32(self, m, n, k, ks):
        from mlrun_amd import ops

        torch.manual_seed(m + n)
        a = (torch.randn(m, k) * 0.5).to(torch.bfloat16)
        w = (torch.randn(n, k) * 0.02).to(torch.bfloat16)
        ref = a.float() @ w.float().t()
        w8, w_scale = ops.quantize_fp8_weight(w)
        a8, a_scale = ops.quant_fp8_rows(a.cuda())
        got = ops.skinny_gemm_fp8(a8, a_scale, w8.cuda(), w_scale.cuda(),
                                  ksplit=ks).cpu().float()
        # fp8 e4m3 ~2-3 relative digits; dot over K averages error
        rel = (ref - got).abs() / ref.abs().clamp(min=0.5)
        assert rel.median().item() < 0.05, rel.median().item()
        assert (ref - got).abs().max().item() < 0.2 * (k ** 0.5)

    def test_quant_matches_torch(self):
        from mlrun_amd import ops

        a = (torch.randn(8, 512) * 3).to(torch.bfloat16)
        ref8, ref_scale = ops.quant_fp8_rows(a)  # cpu reference
        got8, got_scale = ops.quant_fp8_rows(a.cuda())
        assert torch.allclose(ref_scale, got_scale.cpu(), rtol=1e-3)
        ref_v = ref8.view(torch.float8_e4m3fn).float()
        got_v = got8.cpu().view(torch.float8_e4m3fn).float()
        assert (ref_v - got_v).abs().max().item() <= 8.0  # 1-ulp at amax
        assert (ref_v != got_v).float().mean().item() < 0.02


@requires_gpu
class TestFP8KV:
    """fp8 KV cache: e4m3 bytes + per-row scales (quantized append +
    dequant-while-staging attention)."""

    def test_rope_kv_fused_q8_matches_python_quant(self):
        from mlrun_amd import ops

        B, Hq, Hkv, D, Smax = 4, 8, 2, 128, 64
        qkv = _rand_bf16(B, (Hq + 2 * Hkv) * D, seed=33)
        positions = torch.tensor([0, 3, 10, 63], dtype=torch.int32)
        table = ops.build_rope_cos_sin(Smax, D)

        # CPU reference path (python rope + quantize_kv_rows)
        kc_ref = torch.zeros(B, Hkv, Smax, D, dtype=torch.uint8)
        vc_ref = torch.zeros_like(kc_ref)
        ks_ref = torch.ones(B, Hkv, Smax)
        vs_ref = torch.ones(B, Hkv, Smax)
        qkv_ref = qkv.clone()
        ops.rope_kv_fused(qkv_ref, kc_ref, vc_ref, positions, table, Hq,
                          k_scale=ks_ref, v_scale=vs_ref)

        kc = torch.zeros_like(kc_ref).cuda()
        vc = torch.zeros_like(vc_ref).cuda()
        ks = torch.ones(B, Hkv, Smax).cuda()
        vs = torch.ones(B, Hkv, Smax).cuda()
        qkv_gpu = qkv.cuda()
        ops.rope_kv_fused(qkv_gpu, kc, vc, positions.cuda(), table.cuda(),
                          Hq, k_scale=ks, v_scale=vs)
        # compare DEQUANTIZED rows.  The device quantizes the rotated
        # values BEFORE bf16 rounding, the python path after, so the
        # row scales can differ by ~bf16 eps and individual codes by
        # one e4m3 step — bound the diff by the per-row quant step
        # (scale * 2^(4-3) covers the largest binade's step twice).
        for deq_a, deq_b, sc in (
                (ops.dequantize_kv_rows(kc_ref, ks_ref),
                 ops.dequantize_kv_rows(kc.cpu(), ks.cpu()), ks_ref),
                (ops.dequantize_kv_rows(vc_ref, vs_ref),
                 ops.dequantize_kv_rows(vc.cpu(), vs.cpu()), vs_ref)):
            step = sc.unsqueeze(-1) * 32.0  # e4m3 step in [256, 448)
            assert ((deq_a - deq_b).abs() <= 2 * step + 1e-3).all()

    @pytest.mark.parametrize("b,hq,hkv,s", [(2, 8, 2, 64),
                                            (4, 32, 8, 300),
                                            (16, 32, 8, 1024)])
    def test_attn_decode_q8_vs_fp32(self, b, hq, hkv, s):
        from mlrun_amd import ops

        D = 128
        torch.manual_seed(b * 7 + s)
        q = _rand_bf16(b, hq, D)
        smax = s + 17
        kc = _rand_bf16(b, hkv, smax, D)
        vc = _rand_bf16(b, hkv, smax, D)
        k8, ks = ops.quantize_kv_rows(kc)
        v8, vs = ops.quantize_kv_rows(vc)
        seq_lens = torch.randint(1, s + 1, (b,), dtype=torch.int32)
        # fp32 reference over the DEQUANTIZED caches (isolates the
        # kernel's staging/dequant from the quantization error)
        ref = ops.attn_decode(q, k8, v8, seq_lens,
                              k_scale=ks, v_scale=vs)
        got = ops.attn_decode(q.cuda(), k8.contiguous().cuda(),
                              v8.contiguous().cuda(), seq_lens.cuda(),
                              k_scale=ks.contiguous().cuda(),
                              v_scale=vs.contiguous().cuda()).cpu()
        assert torch.allclose(ref.float(), got.float(), atol=4e-2,
                              rtol=4e-2)


@requires_gpu
class TestVectorIndexGPU:
    def test_search_uses_gemm_and_matches_cpu(self):
        import torch

        from mlrun_amd.serving import VectorIndex

        gen = torch.Generator().manual_seed(4)
        emb = torch.randn(512, 256, generator=gen)
        payloads = [{"id": i} for i in range(512)]
        gpu = VectorIndex(dim=256, device="cuda:0")
        gpu.add(emb, payloads)
        cpu = VectorIndex(dim=256, device="cpu")
        cpu.add(emb, payloads)
        q = torch.randn(4, 256, generator=gen)
        got = gpu.search(q.cuda(), k=5)
        ref = cpu.search(q, k=5)
        for row_got, row_ref in zip(got, ref):
            assert [h["payload"]["id"] for h in row_got] == \
                   [h["payload"]["id"] for h in row_ref]


@requires_gpu
class TestOnlineFeatureServiceGPU:
    def test_end_to_end_online_service_on_gpu(self):
        """Config-3 e2e on hardware: ingest -> GPU rings -> online
        vector service get/as_list, results matching the CPU table."""
        import time

        import numpy as np
        import pandas as pd

        from mlrun_amd import feature_store as fstore
        from mlrun_amd.feature_store.online import OnlineTable

        fstore.reset_online_tables()
        rng = np.random.default_rng(5)
        now = time.time()
        df = pd.DataFrame({
            "key": rng.integers(0, 500, 20000).astype(str),
            "amount": rng.normal(10, 3, 20000),
            "ts": pd.to_datetime(now - rng.uniform(0, 3000, 20000),
                                 unit="s"),
        })
        fset = fstore.FeatureSet("gpu-svc", entities=["key"],
                                 timestamp_key="ts")
        fset.add_aggregation(
            "amount",
            ["sum", "count", "avg", "min", "max", "first", "last",
             "stddev"], ["1h"], "5m")
        gpu_table = OnlineTable(fset, device="cuda:0")
        cpu_table = OnlineTable(fset, device="cpu")
        gpu_table.ingest_batch(df)
        cpu_table.ingest_batch(df)
        keys = [{"key": str(k)} for k in rng.integers(0, 500, 256)]
        got_gpu = gpu_table.get(keys, now_ts=now)
        got_cpu = cpu_table.get(keys, now_ts=now)
        for g, c in zip(got_gpu, got_cpu):
            for op in ("sum", "count", "avg", "min", "max", "first",
                       "last", "stddev"):
                name = f"amount_{op}_1h"
                if c[name] is None:
                    assert g[name] is None, (name, g[name])
                else:
                    assert abs(g[name] - c[name]) < 1e-2 + \
                        1e-4 * abs(c[name]), (name, g[name], c[name])
        # columnar fast path agrees too
        names = ["amount_sum_1h", "amount_min_1h", "amount_last_1h"]
        m_gpu = gpu_table.get_agg_matrix(keys, names, now_ts=now)
        m_cpu = cpu_table.get_agg_matrix(keys, names, now_ts=now)
        import numpy.testing as npt

        npt.assert_allclose(m_gpu, m_cpu, rtol=1e-4, atol=1e-2)
