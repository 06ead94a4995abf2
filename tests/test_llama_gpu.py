# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""GPU Llama tests: the full custom-kernel decode chain vs. the
prefill (hipBLASLt/SDPA) path, and hipGraph capture."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@requires_gpu
class TestLlamaGPU:
    def test_decode_matches_prefill_gpu(self):
        """End-to-end kernel-chain validation: greedy decode tokens on
        the HIP kernel path must match next-token argmax of the
        library-GEMM prefill path."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

        cfg = LlamaConfig.tiny(num_layers=4, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        engine_a = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                     use_graph=False, seed=21)
        prompt = torch.randint(0, 2000, (4, 10),
                               generator=torch.Generator().manual_seed(5))
        generated = engine_a.generate(prompt, max_new_tokens=4)

        engine_b = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                     use_graph=False, seed=21)
        extended = torch.cat([prompt, generated[:, :3].cpu()], dim=1)
        logits = engine_b.prefill(extended)
        predicted = logits.argmax(dim=-1).cpu()
        # bf16 rounding differences between the two paths can flip an
        # argmax on random weights; demand >= 3/4 agreement
        agree = (predicted == generated[:, 3].cpu()).sum().item()
        assert agree >= 3, f"decode/prefill disagree: {agree}/4"

    def test_hipgraph_capture_replay(self):
        """Graph-captured decode must equal eager decode."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        prompt = torch.randint(0, 2000, (4, 8),
                               generator=torch.Generator().manual_seed(6))
        eager = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                  use_graph=False, seed=31)
        out_eager = eager.generate(prompt, max_new_tokens=6).cpu()
        graphed = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                    use_graph=True, seed=31)
        out_graph = graphed.generate(prompt, max_new_tokens=6).cpu()
        assert graphed._graph is not None, "hipGraph was not captured"
        assert torch.equal(out_eager, out_graph)

    def test_serving_graph_on_gpu(self):
        import mlrun_amd
        from mlrun_amd.models.llama import LlamaServer

        fn = mlrun_amd.new_function(name="llmgpu", kind="serving")
        fn.add_model("gen", class_name=LlamaServer, config="tiny",
                     batch_size=2, max_new_tokens=4, device="cuda:0")
        server = fn.to_mock_server()
        resp = server.test("/v2/models/gen/infer",
                           body={"inputs": [[1, 2, 3], [4, 5]],
                                 "max_tokens": 4})
        assert len(resp["outputs"]) == 2
        assert len(resp["outputs"][0]) == 4


@requires_gpu
class TestTrainGPU:
    def test_train_step_8b_single_gpu(self):
        """Config-4 smoke on 1 GPU: a few 8B train steps fit in 288 GB
        HBM and the loss is finite."""
        import torch
        from mlrun_amd.models.llama import LlamaConfig
        from mlrun_amd.models.llama_train import LlamaTrainer

        cfg = LlamaConfig.llama3_8b(max_seq_len=512)
        trainer = LlamaTrainer(cfg, device="cuda:0")
        batch = torch.randint(0, cfg.vocab_size, (2, 512))
        losses = [trainer.train_step(batch) for _ in range(2)]
        assert all(torch.isfinite(torch.tensor(losses))), losses
        del trainer
        torch.cuda.empty_cache()


@requires_gpu
class TestFP8Serving:
    def test_fp8_decode_close_to_bf16(self):
        """fp8-weight decode must track the bf16 engine closely on the
        same weights (opt-in mode; headline stays bf16)."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        prompt = torch.randint(0, 2000, (4, 8),
                               generator=torch.Generator().manual_seed(3))
        bf16 = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                 use_graph=False, seed=41)
        out_bf16 = bf16.generate(prompt, max_new_tokens=4).cpu()
        fp8 = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                use_graph=False, seed=1,
                                weight_dtype="fp8w")
        fp8.weights.load_state_dict(bf16.weights.state_dict())
        fp8.rebuild_fp8_packs()  # re-quantize from the loaded weights
        out_fp8 = fp8.generate(prompt, max_new_tokens=4).cpu()
        match = (out_bf16 == out_fp8).float().mean().item()
        assert match >= 0.5, f"fp8 decode diverged: match={match}"


@requires_gpu
class TestFP8KVServing:
    def test_fp8_kv_decode_close_to_bf16(self):
        """fp8 KV cache must track the bf16-cache engine closely on
        identical weights (opt-in mode; headline stays bf16)."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        prompt = torch.randint(0, 2000, (4, 8),
                               generator=torch.Generator().manual_seed(5))
        bf16 = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                 use_graph=False, seed=41)
        out_bf16 = bf16.generate(prompt, max_new_tokens=4).cpu()
        q8 = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                               use_graph=False, seed=41, kv_dtype="fp8")
        out_q8 = q8.generate(prompt, max_new_tokens=4).cpu()
        match = (out_bf16 == out_q8).float().mean().item()
        assert match >= 0.5, f"fp8 KV decode diverged: match={match}"

    def test_fp8_kv_under_hipgraph(self):
        from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        prompt = torch.randint(0, 2000, (4, 8),
                               generator=torch.Generator().manual_seed(9))
        eager = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                  use_graph=False, seed=17,
                                  kv_dtype="fp8")
        graphed = LlamaDecodeEngine(cfg, batch_size=4, device="cuda:0",
                                    use_graph=True, seed=17,
                                    kv_dtype="fp8")
        out_e = eager.generate(prompt, max_new_tokens=4).cpu()
        out_g = graphed.generate(prompt, max_new_tokens=4).cpu()
        assert torch.equal(out_e, out_g)


@requires_gpu
class TestFP8Replicas:
    def test_fp8_server_with_replicas(self):
        """Replica engines must get their own quant-sidecar buffers
        (regression: shared-weight replicas crashed in fp8w mode)."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaServer

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        server = LlamaServer(name="m", config=cfg, batch_size=4,
                             max_new_tokens=4, weight_dtype="fp8w",
                             replicas=2, use_graph=False)
        server.load()

        class _Ev:
            body = {"inputs": [[1, 2, 3]] * 6, "max_tokens": 4}
            path = "/infer"
            id = "t"

        out = server.do_event(_Ev()).body["outputs"]
        assert len(out) == 6 and all(len(o) == 4 for o in out)


@requires_gpu
class TestContinuousBatchingGPU:
    def test_continuous_matches_batch_under_hipgraph(self):
        from mlrun_amd.models.llama import LlamaConfig, LlamaServer

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        cont = LlamaServer(name="c", config=cfg, batch_size=4,
                           max_new_tokens=6, scheduling="continuous",
                           use_graph=True, device="cuda:0")
        cont.load()
        batch = LlamaServer(name="b", config=cfg, batch_size=4,
                            max_new_tokens=6, use_graph=True,
                            device="cuda:0")
        batch.load()
        batch.engines[0].weights.load_state_dict(
            cont.engines[0].weights.state_dict())

        class _Ev:
            body = {"inputs": [[1, 2, 3], [9, 8, 7, 6]],
                    "max_tokens": 5}
            path = "/infer"
            id = "t"

        out_c = cont.do_event(_Ev()).body["outputs"]
        out_b = batch.do_event(_Ev()).body["outputs"]
        assert out_c == out_b

    def test_slot_reuse_on_gpu(self):
        import threading

        from mlrun_amd.models.llama import LlamaConfig, LlamaServer

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        cont = LlamaServer(name="c2", config=cfg, batch_size=2,
                           max_new_tokens=4, scheduling="continuous",
                           use_graph=True, device="cuda:0")
        cont.load()
        results = {}

        def one(i):
            class _Ev:
                body = {"inputs": [[i + 1, i + 2]], "max_tokens": 3}
                path = "/infer"
                id = "t"

            results[i] = cont.do_event(_Ev()).body["outputs"][0]

        threads = [threading.Thread(target=one, args=(i,))
                   for i in range(6)]
        [t.start() for t in threads]
        [t.join(timeout=120) for t in threads]
        assert len(results) == 6
        assert all(len(r) == 3 for r in results.values())


@requires_gpu
class TestStreamingGPU:
    def test_stream_matches_nonstream_under_hipgraph(self):
        import json

        from mlrun_amd.models.llama import LlamaConfig, LlamaServer

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        srv = LlamaServer(name="sg", config=cfg, batch_size=2,
                          max_new_tokens=6, scheduling="continuous",
                          use_graph=True, device="cuda:0")
        srv.load()

        class _Ev:
            path = "/infer"
            id = "t"

        ev = _Ev()
        ev.body = {"inputs": [[3, 1, 4]], "max_tokens": 5,
                   "stream": True}
        per = []
        for line in srv.do_event(ev).body:
            per.append(json.loads(line)["token"])
        ev2 = _Ev()
        ev2.body = {"inputs": [[3, 1, 4]], "max_tokens": 5}
        out = srv.do_event(ev2).body["outputs"][0]
        assert per == out


@requires_gpu
class TestGoldenPathGPU:
    def test_train_then_serve_from_artifact_on_gpu(self, tmp_path):
        """Tiny llama fine-tune on cuda:0 -> model artifact -> decode
        engine serves from the artifact bit-exactly (the train->serve
        bridge on real hardware)."""
        import mlrun_amd
        from mlrun_amd.models.llama import LlamaConfig
        from mlrun_amd.models.llama_train import (LlamaTrainer,
                                                  export_decode_state)

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        fn = mlrun_amd.new_function(name="gtrain", kind="local")

        state = {}

        def train_handler(context):
            trainer = LlamaTrainer(cfg, device="cuda:0", lr=1e-3,
                                   context=context)
            tokens = torch.randint(
                0, cfg.vocab_size, (2, 32),
                generator=torch.Generator().manual_seed(0)).cuda()
            for _ in range(3):
                loss = trainer.train_step(tokens)
            context.log_result("final_loss", loss)
            state["decode"] = export_decode_state(trainer.model)
            trainer.save_checkpoint("model")

        run = fn.run(handler=train_handler, local=True)
        model_uri = run.outputs["model"]

        from mlrun_amd.models.llama import LlamaServer

        server = LlamaServer(name="g", config=cfg, batch_size=2,
                             max_new_tokens=4, model_path=model_uri,
                             device="cuda:0", use_graph=True)
        server.load()

        # the served engine must match a direct export of the trained
        # weights
        from mlrun_amd.models.llama import LlamaDecodeEngine

        direct = LlamaDecodeEngine(cfg, 2, device="cuda:0",
                                   use_graph=True, seed=99)
        direct.weights.load_state_dict(state["decode"])
        prompt = torch.randint(0, cfg.vocab_size, (2, 8),
                               generator=torch.Generator().manual_seed(4))
        out_direct = direct.generate(prompt, max_new_tokens=4).cpu()
        out_served = torch.tensor(server.do_event(type(
            "E", (), {"body": {"inputs": prompt.tolist(),
                               "max_tokens": 4},
                      "path": "/infer", "id": "t"})()).body["outputs"])
        assert torch.equal(out_direct, out_served)


@requires_gpu
class TestSamplingGPU:
    def test_sampling_under_hipgraph(self):
        """RNG ops captured in the hipGraph must draw fresh noise per
        replay (token diversity) while T=0 stays greedy-exact."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaDecodeEngine

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        prompt = torch.randint(0, 2000, (2, 8),
                               generator=torch.Generator().manual_seed(1))
        sampler = LlamaDecodeEngine(cfg, 2, device="cuda:0",
                                    use_graph=True, seed=31,
                                    temperature=1.0)
        out = sampler.generate(prompt, max_new_tokens=8).cpu()
        # fresh noise per replay: a high-T sample should not be one
        # token repeated 8 times for both rows
        assert not all(row.eq(row[0]).all() for row in out)
        greedy_g = LlamaDecodeEngine(cfg, 2, device="cuda:0",
                                     use_graph=True, seed=31)
        greedy_e = LlamaDecodeEngine(cfg, 2, device="cuda:0",
                                     use_graph=False, seed=31)
        assert torch.equal(greedy_g.generate(prompt, 6).cpu(),
                           greedy_e.generate(prompt, 6).cpu())

    def test_per_request_temperature_under_hipgraph(self):
        from mlrun_amd.models.llama import LlamaConfig, LlamaServer

        cfg = LlamaConfig.tiny(num_layers=2, num_heads=4, num_kv_heads=2,
                               hidden_size=512, intermediate_size=1024,
                               vocab_size=2048)
        greedy = LlamaServer(name="pg", config=cfg, batch_size=2,
                             max_new_tokens=6, use_graph=True,
                             device="cuda:0")
        greedy.load()
        mixed = LlamaServer(name="pm", config=cfg, batch_size=2,
                            max_new_tokens=6, scheduling="continuous",
                            use_graph=True, device="cuda:0",
                            temperature=-1)
        mixed.load()
        mixed.engines[0].weights.load_state_dict(
            greedy.engines[0].weights.state_dict())

        class _Ev:
            path = "/infer"
            id = "t"

        base = _Ev()
        base.body = {"inputs": [[7, 8, 9]], "max_tokens": 5}
        expect = greedy.do_event(base).body["outputs"][0]
        cold = _Ev()
        cold.body = {"inputs": [[7, 8, 9]], "max_tokens": 5,
                     "temperature": 0}
        assert mixed.do_event(cold).body["outputs"][0] == expect
        hot = _Ev()
        hot.body = {"inputs": [[7, 8, 9]], "max_tokens": 5,
                    "temperature": 100.0}
        assert mixed.do_event(hot).body["outputs"][0] != expect


@requires_gpu
class TestRcclInGraphCapture:
    """RCCL collectives recorded inside hipGraph capture (the TP decode
    mechanism, VERDICT round-1 item 1).  Runs in a subprocess with a
    world-size-1 RCCL group on one GPU: a single-rank all-reduce is an
    identity collective, so the captured-TP engine must emit exactly
    the eager single-rank tokens — this validates capture of the
    collective kernel itself (pairing across ranks is exercised by
    scripts/bench_serving_tp.py on an 8-GPU node)."""

    def test_allreduce_inside_capture_single_rank(self, tmp_path):
        import subprocess
        import sys
        import textwrap

        script = textwrap.dedent("""
            import os
            import torch
            import torch.distributed as dist

            from mlrun_amd.models.llama import LlamaConfig, \\
                LlamaDecodeEngine

            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29471")
            dist.init_process_group("nccl", rank=0, world_size=1)
            cfg = LlamaConfig.tiny(num_layers=2, num_heads=4,
                                   num_kv_heads=2, hidden_size=512,
                                   intermediate_size=1024,
                                   vocab_size=2048)
            prompt = torch.randint(
                0, 2000, (4, 8),
                generator=torch.Generator().manual_seed(6))

            eager = LlamaDecodeEngine(cfg, 4, device="cuda:0",
                                      use_graph=False, seed=31)
            eager.tp_size = 2  # same (allreduce) code path as graphed
            eager.tp_group = None
            out_eager = eager.generate(prompt, max_new_tokens=6).cpu()

            # tp_size=2 forces _maybe_allreduce to issue
            # dist.all_reduce on the world-1 group (identity);
            # weights were built before the override so shapes are
            # the full single-rank shapes
            graphed = LlamaDecodeEngine(cfg, 4, device="cuda:0",
                                        use_graph=True, seed=31)
            graphed.tp_size = 2
            graphed.tp_group = None
            out_graph = graphed.generate(prompt,
                                         max_new_tokens=6).cpu()
            assert graphed._graph is not None, "capture failed"
            assert torch.equal(out_eager, out_graph), (
                out_eager, out_graph)
            dist.destroy_process_group()
            print("RCCL_IN_GRAPH_OK")
        """)
        proc = subprocess.run([sys.executable, "-c", script],
                              capture_output=True, text=True,
                              timeout=600)
        assert proc.returncode == 0, proc.stderr[-3000:]
        assert "RCCL_IN_GRAPH_OK" in proc.stdout


@requires_gpu
class TestMonitoringRingsOnGPU:
    def test_stream_stats_on_gpu_rings(self):
        """model_endpoint_monitoring.device=auto places the 5m/1h
        serving-stat rings in HBM and serves identical stats."""
        import time

        from mlrun_amd.model_monitoring.stream import (
            EventStreamProcessor,
            ModelMonitoringEvent,
        )

        gpu = EventStreamProcessor("gpu-mon", device="cuda:0")
        cpu = EventStreamProcessor("cpu-mon", device="cpu")
        now = time.time()
        for i in range(64):
            event = dict(endpoint_id="ep", model="m",
                         latency_ms=float(i), timestamp=now - i,
                         error="boom" if i % 8 == 0 else None)
            gpu.push(ModelMonitoringEvent(**event))
            cpu.push(ModelMonitoringEvent(**event))
        a = gpu.endpoint_stats("ep", now_ts=now)
        b = cpu.endpoint_stats("ep", now_ts=now)
        assert gpu._latency_ring.ring.is_cuda
        for window in ("300", "3600"):
            for key in ("count", "error_count", "avg_latency_ms",
                        "max_latency_ms", "min_latency_ms"):
                assert abs(a[window][key] - b[window][key]) < 1e-3, (
                    window, key, a[window], b[window])


@requires_gpu
class TestAsyncEngineWithGpuModel:
    def test_async_flow_serves_gpu_model(self):
        """engine="async" with a CUDA model step: step bodies run on
        per-step executor threads — decode must still work and the
        responder contract must hold."""
        import mlrun_amd
        from mlrun_amd.models.llama import LlamaServer

        fn = mlrun_amd.new_function(name="async-gpu", kind="serving")
        graph = fn.set_topology("flow", engine="async")

        class Tag:
            def __init__(self, context=None, name=None):
                pass

            def do(self, event):
                event.body = dict(event.body)
                event.body.setdefault("tags", []).append("pre")
                return event

        graph.to(Tag, name="pre").to(
            LlamaServer, name="gen", config="tiny", batch_size=2,
            max_new_tokens=4, device="cuda:0").respond()
        server = fn.to_mock_server(namespace={"Tag": Tag})
        try:
            resp = server.test("/infer",
                               body={"inputs": [[1, 2, 3]],
                                     "max_tokens": 4})
            assert len(resp["outputs"]) == 1
            assert len(resp["outputs"][0]) == 4
            # a second event reuses the loaded engine
            resp2 = server.test("/infer",
                                body={"inputs": [[1, 2, 3]],
                                      "max_tokens": 4})
            assert resp2["outputs"] == resp["outputs"]  # greedy
        finally:
            server.graph.shutdown()
