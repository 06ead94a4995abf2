# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Tensor-parallel decode tests: TP=2 over gloo (CPU) must match the
single-rank engine (new capability — the reference has no TP)."""

import textwrap

import pytest
import torch

import mlrun_amd
from mlrun_amd.model import RunStates

TP_SCRIPT = textwrap.dedent("""
    import os
    import torch
    import torch.distributed as dist
    import mlrun_amd
    from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine
    from mlrun_amd.parallel.tp import init_tp_group, shard_llama_state

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    group, tp_rank, tp_size = init_tp_group(backend="gloo")
    cfg = LlamaConfig.tiny(num_heads=4, num_kv_heads=2, hidden_size=512,
                           intermediate_size=1024, vocab_size=2048)

    # every rank builds the same FULL model (same seed), then shards
    full = LlamaDecodeEngine(cfg, batch_size=2, device="cpu", seed=77)
    full_state = full.weights.state_dict()

    tp_engine = LlamaDecodeEngine(cfg, batch_size=2, device="cpu",
                                  tp_group=group, tp_rank=tp_rank,
                                  tp_size=tp_size, seed=1)
    tp_engine.weights.load_state_dict(
        shard_llama_state(full_state, cfg, tp_rank, tp_size))

    prompt = torch.randint(0, 2000, (2, 6),
                           generator=torch.Generator().manual_seed(9))

    ref_logits = full.prefill(prompt.clone())
    tp_logits = tp_engine.prefill(prompt.clone())
    prefill_err = (ref_logits - tp_logits).abs().max().item()

    full.reset(); tp_engine.reset()
    ref_tokens = full.generate(prompt.clone(), max_new_tokens=3)
    tp_tokens = tp_engine.generate(prompt.clone(), max_new_tokens=3)
    match = (ref_tokens == tp_tokens).float().mean().item()

    ctx = mlrun_amd.get_or_create_ctx("tp-check")
    if ctx.is_logging_worker():
        ctx.log_result("prefill_max_err", prefill_err)
        ctx.log_result("token_match", match)
        ctx.commit(completed=True)
    dist.destroy_process_group()
""")


class TestTensorParallel:
    def test_shard_shapes(self):
        from mlrun_amd.models.llama import LlamaConfig, LlamaWeights
        from mlrun_amd.parallel.tp import shard_llama_state

        cfg = LlamaConfig.tiny(num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024)
        full = LlamaWeights(cfg, "cpu", seed=3)
        state = full.state_dict()
        for rank in (0, 1):
            shard = shard_llama_state(state, cfg, rank, 2)
            tp_w = LlamaWeights(cfg, "cpu", tp_rank=rank, tp_size=2, seed=9)
            for key, value in tp_w.state_dict().items():
                assert shard[key].shape == value.shape, \
                    f"{key}: {shard[key].shape} != {value.shape}"

    def test_shard_reconstruction(self):
        """Concatenating both ranks' shards must reproduce the full
        weights (no rows lost/duplicated)."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaWeights
        from mlrun_amd.parallel.tp import shard_llama_state

        cfg = LlamaConfig.tiny(num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024)
        state = LlamaWeights(cfg, "cpu", seed=4).state_dict()
        s0 = shard_llama_state(state, cfg, 0, 2)
        s1 = shard_llama_state(state, cfg, 1, 2)
        d = cfg.head_dim
        wqkv = state["layers.0.wqkv"]
        # q rows: rank0 holds heads 0-1, rank1 heads 2-3
        assert torch.equal(s0["layers.0.wqkv"][:2 * d], wqkv[:2 * d])
        assert torch.equal(s1["layers.0.wqkv"][:2 * d],
                           wqkv[2 * d:4 * d])
        # wo columns split
        wo = state["layers.0.wo"]
        assert torch.equal(
            torch.cat([s0["layers.0.wo"], s1["layers.0.wo"]], dim=1), wo)
        # gate|up interleaved halves reconstruct
        wgu = state["layers.0.wgu"]
        inter = cfg.intermediate_size
        assert torch.equal(s0["layers.0.wgu"][:inter // 2],
                           wgu[:inter // 2])
        assert torch.equal(s1["layers.0.wgu"][inter // 2:],
                           wgu[inter + inter // 2:])

    def test_tp2_matches_single_rank(self, tmp_path):
        script = tmp_path / "tp.py"
        script.write_text(TP_SCRIPT)
        fn = mlrun_amd.new_function(name="tp", kind="mpijob",
                                    command=str(script))
        fn.with_replicas(2)
        run = fn.run(name="tp-check")
        assert run.status.state == RunStates.completed, run.status.error
        assert run.status.results["prefill_max_err"] < 0.25, \
            run.status.results
        assert run.status.results["token_match"] >= 0.5


RIDER_SCRIPT = textwrap.dedent("""
    import os, sys
    import torch
    import torch.distributed as dist

    sys.path.insert(0, os.getcwd())
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)

    from bench import _bench_tp70b
    from mlrun_amd.models.llama import LlamaConfig

    cfg = LlamaConfig.tiny(num_heads=4, num_kv_heads=2, hidden_size=512,
                           intermediate_size=1024, vocab_size=2048)
    result = _bench_tp70b(dist, rank, world, 0, steps=1, warmup=1,
                          batch=2, prompt_len=6, gen_tokens=3,
                          cfg=cfg, device="cpu")
    if rank == 0:
        assert result["value"] > 0, result
        assert result["config"]["parallelism"] == f"tp{world}"
        print("RIDER_OK", result["value"])
    dist.destroy_process_group()
""")


class TestTpRiderPath:
    def test_bench_tp70b_code_path_gloo(self, tmp_path):
        """The TP rider in bench.py runs exactly once, unattended, on
        the driver's 8-GPU box — exercise the EXACT function over
        2-rank gloo with a tiny config so API breaks are caught
        here."""
        import subprocess
        import sys

        script = tmp_path / "rider.py"
        script.write_text(RIDER_SCRIPT)
        procs = []
        import os as _os

        for rank in range(2):
            env = dict(_os.environ, RANK=str(rank), WORLD_SIZE="2",
                       MASTER_ADDR="127.0.0.1", MASTER_PORT="29433",
                       LOCAL_RANK=str(rank))
            procs.append(subprocess.Popen(
                [sys.executable, str(script)], env=env,
                cwd=_os.getcwd(), stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT))
        outs = [p.communicate(timeout=180)[0].decode() for p in procs]
        assert all(p.returncode == 0 for p in procs), outs
        assert "RIDER_OK" in outs[0] + outs[1], outs
