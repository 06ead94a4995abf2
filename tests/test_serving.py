# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Serving-graph engine tests (mirror the reference's
tests/serving/test_serving.py / test_flow.py / test_parallel.py tiers,
all in-process via mock servers)."""

import json

import pytest

import mlrun_amd
from mlrun_amd.serving import (
    Event,
    GraphServer,
    RouterStep,
    TaskStep,
    V2ModelServer,
    create_graph_server,
)


class EchoModel(V2ModelServer):
    def load(self):
        self.model = "loaded"

    def predict(self, request):
        return [x * 2 for x in request["inputs"]]


class ConstModel(V2ModelServer):
    def load(self):
        self.value = self.get_param("value", 1)

    def predict(self, request):
        return [self.value] * len(request["inputs"])


class FailModel(V2ModelServer):
    def load(self):
        pass

    def predict(self, request):
        raise RuntimeError("predict exploded")


def double_handler(body):
    body["x"] = body.get("x", 0) * 2
    return body


def add_one(body):
    body["x"] = body.get("x", 0) + 1
    return body


def _serving_fn():
    return mlrun_amd.new_function(name="srv", kind="serving")


class TestRouterTopology:
    def test_infer(self):
        fn = _serving_fn()
        fn.add_model("m1", class_name=EchoModel)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/m1/infer", body={"inputs": [1, 2, 3]})
        assert resp["outputs"] == [2, 4, 6]
        assert resp["model_name"] == "m1"

    def test_multiple_models_and_list(self):
        fn = _serving_fn()
        fn.add_model("a", class_name=EchoModel)
        fn.add_model("b", class_name=ConstModel, value=7)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/b/infer", body={"inputs": [0, 0]})
        assert resp["outputs"] == [7, 7]
        listing = server.test("/v2/models/", method="GET")
        assert set(listing["models"]) == {"a", "b"}

    def test_body_routing(self):
        fn = _serving_fn()
        fn.add_model("a", class_name=EchoModel)
        fn.add_model("b", class_name=ConstModel, value=5)
        server = fn.to_mock_server()
        resp = server.test("/", body={"model": "b", "inputs": [1]})
        assert resp["outputs"] == [5]

    def test_json_string_body(self):
        fn = _serving_fn()
        fn.add_model("m", class_name=EchoModel)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/m/infer",
                           body=json.dumps({"inputs": [4]}))
        assert resp["outputs"] == [8]

    def test_model_error(self):
        fn = _serving_fn()
        fn.add_model("bad", class_name=FailModel)
        server = fn.to_mock_server()
        with pytest.raises(RuntimeError):
            server.test("/v2/models/bad/infer", body={"inputs": [1]})
        resp = server.test("/v2/models/bad/infer", body={"inputs": [1]},
                           silent=True)
        assert resp.error

    def test_ready_and_metadata(self):
        fn = _serving_fn()
        fn.add_model("m", class_name=EchoModel)
        server = fn.to_mock_server()
        assert server.test("/v2/models/m/ready")["ready"] is True
        meta = server.test("/v2/models/m", method="GET")
        assert meta["name"] == "m"

    def test_unknown_model(self):
        fn = _serving_fn()
        fn.add_model("m", class_name=EchoModel)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/nope/infer", body={"inputs": [1]},
                           silent=True)
        assert resp.error


class TestFlowTopology:
    def test_handler_chain(self):
        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=double_handler, name="d1") \
            .to(handler=add_one, name="a1")
        server = fn.to_mock_server()
        resp = server.test("/", body={"x": 5})
        assert resp["x"] == 11

    def test_flow_with_router(self):
        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=double_handler, name="pre") \
            .to(RouterStep(name="router")) \
            .to(handler=add_one, name="post")
        fn.add_model("m", class_name=EchoModel)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/m/infer", body={"x": 1, "inputs": [3]})
        # pre doubles x (ignored by model), model doubles inputs,
        # post runs on the response body
        assert resp["outputs"] == [6]
        assert resp["x"] == 1

    def test_error_handler(self):
        def boom(body):
            raise ValueError("nope")

        def catcher(event):
            event.body = {"caught": event.error}
            event.error = None
            event.terminated = True
            return event

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        step = graph.to(handler=boom, name="boom")
        graph.add_step(handler=catcher, name="catcher", after=[],
                       full_event=True)
        # detach catcher from the implicit chain
        graph["catcher"].after = []
        graph._build_links()
        step.error_handler(name="catcher")
        server = fn.to_mock_server()
        resp = server.test("/", body={"x": 1})
        assert "nope" in resp["caught"]

    def test_respond_semantics(self):
        """A responder step's output is the response even when the chain
        continues."""
        seen = []

        def tail(body):
            seen.append(dict(body))
            body["tail"] = True
            return body

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=double_handler, name="first").respond() \
            .to(handler=tail, name="tail")
        server = fn.to_mock_server()
        resp = server.test("/", body={"x": 2}, get_body=False)
        assert resp.responded

    def test_queue_step(self):
        import time

        results = []

        def sink(body):
            results.append(body)
            return body

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=double_handler, name="head") \
            .to("$queue", name="q1") \
            .to(handler=sink, name="sink")
        server = fn.to_mock_server()
        server.test("/", body={"x": 2})
        deadline = time.monotonic() + 5
        while not results and time.monotonic() < deadline:
            time.sleep(0.01)
        assert results and results[0]["x"] == 4


class TestEnsembles:
    def test_voting_classification(self):
        fn = _serving_fn()
        fn.set_topology("router", class_name="VotingEnsemble")
        fn.add_model("m1", class_name=ConstModel, value=1)
        fn.add_model("m2", class_name=ConstModel, value=1)
        fn.add_model("m3", class_name=ConstModel, value=0)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/infer", body={"inputs": [0, 0]})
        assert resp["outputs"] == [1, 1]

    def test_voting_regression_mean(self):
        class Half(V2ModelServer):
            def load(self):
                pass

            def predict(self, request):
                return [0.5 for _ in request["inputs"]]

        class OneAndHalf(V2ModelServer):
            def load(self):
                pass

            def predict(self, request):
                return [1.5 for _ in request["inputs"]]

        fn = _serving_fn()
        fn.set_topology("router", class_name="VotingEnsemble",
                        vote_type="regression")
        fn.add_model("a", class_name=Half)
        fn.add_model("b", class_name=OneAndHalf)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/infer", body={"inputs": [0]})
        assert resp["outputs"] == [1.0]

    def test_parallel_run_merge(self):
        fn = _serving_fn()
        fn.set_topology("router", class_name="ParallelRun",
                        executor_type="thread")
        fn.add_model("a", class_name=ConstModel, value=1)
        fn.add_model("b", class_name=ConstModel, value=2)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/infer", body={"inputs": [9]})
        assert resp["results"]["a"]["outputs"] == [1]
        assert resp["results"]["b"]["outputs"] == [2]


class TestGraphServerDict:
    def test_roundtrip(self):
        fn = _serving_fn()
        fn.add_model("m1", class_name="EchoModel", model_path="")
        struct = fn.to_dict()
        fn2 = mlrun_amd.new_function(runtime=struct)
        assert fn2.kind == "serving"
        server = fn2.to_mock_server(namespace={"EchoModel": EchoModel})
        resp = server.test("/v2/models/m1/infer", body={"inputs": [1]})
        assert resp["outputs"] == [2]


class TestHTTPDeploy:
    def test_deploy_and_invoke(self):
        fn = _serving_fn()
        fn.add_model("m1", class_name=EchoModel)
        address = fn.deploy()
        assert address.startswith("http://")
        resp = fn.invoke("/v2/models/m1/infer", body={"inputs": [10]})
        assert resp["outputs"] == [20]
        import requests

        health = requests.get(address + "/healthz", timeout=5).json()
        assert health["status"] == "ok"
        fn.stop()


class TestMonitoringIntegration:
    def test_tracked_model_pushes_events(self):
        from mlrun_amd.model_monitoring import get_stream_processor

        fn = _serving_fn()
        fn.set_tracking()
        fn.add_model("tracked", class_name=EchoModel)
        server = fn.to_mock_server(track_models=True)
        for _ in range(5):
            server.test("/v2/models/tracked/infer", body={"inputs": [1]})
        processor = get_stream_processor("default")
        stats = processor.endpoint_stats("tracked")
        assert stats["300"]["count"] == 5
        assert stats["300"]["avg_latency_ms"] >= 0


class TestAPIGateway:
    def test_canary_and_auth(self):
        from mlrun_amd.runtimes.api_gateway import APIGateway

        fn_a = _serving_fn()
        fn_a.add_model("m", class_name=ConstModel, value=1)
        addr_a = fn_a.deploy()
        fn_b = _serving_fn()
        fn_b.add_model("m", class_name=ConstModel, value=2)
        addr_b = fn_b.deploy()
        gateway = APIGateway(name="gw").with_canary(
            [addr_a, addr_b], [50, 50]).with_basic_auth("user", "pw")
        gateway.deploy()
        try:
            # unauthorized without credentials
            status, _ = gateway.invoke("/v2/models/m/infer",
                                       body={"inputs": [0]})
            assert status == 401
            seen = set()
            for _ in range(20):
                status, out = gateway.invoke(
                    "/v2/models/m/infer", body={"inputs": [0]},
                    credentials=("user", "pw"))
                assert status == 200
                seen.add(out["outputs"][0])
            assert seen == {1, 2}  # both canary legs took traffic
        finally:
            gateway.stop()
            fn_a.stop()
            fn_b.stop()


class TestV1Serving:
    def test_v1_protocol(self):
        from mlrun_amd.serving.v1_serving import MLModelServer

        class V1Echo(MLModelServer):
            def load(self):
                self.model = True

            def predict(self, request):
                return [x + 1 for x in request["instances"]]

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(V1Echo(name="v1m"), name="model")
        server = fn.to_mock_server()
        resp = server.test("/predict", body={"instances": [1, 2]})
        assert resp["predictions"] == [2, 3]
        assert resp["model_name"] == "v1m"


class TestTracing:
    def test_spans_collected(self):
        from mlrun_amd.utils import tracing

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=double_handler, name="d1") \
            .to(handler=add_one, name="a1")
        server = fn.to_mock_server()
        tracing.enable(True)
        try:
            with tracing.collect() as spans:
                server.test("/", body={"x": 1})
            names = [s["name"] for s in spans]
            assert "step:d1" in names and "step:a1" in names
            assert all(s["ms"] >= 0 for s in spans)
        finally:
            tracing.enable(False)

    def test_disabled_no_overhead_path(self):
        from mlrun_amd.utils import tracing

        assert not tracing.is_enabled()
        with tracing.collect() as spans:
            with tracing.span("x"):
                pass
        assert spans == []


class TestEventPaths:
    def test_input_and_result_path(self):
        """input_path feeds the handler a body subfield; result_path
        writes its output to another subfield (reference TaskStep
        event-path semantics)."""
        def double(value):
            return value * 2

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=double, name="d",
                 input_path="request.x", result_path="response.y")
        server = fn.to_mock_server()
        resp = server.test("/", body={"request": {"x": 21}})
        assert resp["response"]["y"] == 42
        assert resp["request"]["x"] == 21

    def test_input_path_only_writes_back(self):
        def incr(value):
            return value + 1

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=incr, name="i", input_path="a.b")
        server = fn.to_mock_server()
        resp = server.test("/", body={"a": {"b": 1}, "other": True})
        assert resp["a"]["b"] == 2 and resp["other"] is True


class TestRAG:
    def test_vector_index_topk_matches_reference(self):
        import torch

        from mlrun_amd.serving import VectorIndex

        index = VectorIndex(dim=64)
        gen = torch.Generator().manual_seed(0)
        emb = torch.randn(32, 64, generator=gen)
        index.add(emb, [{"id": i} for i in range(32)])
        q = torch.randn(64, generator=gen)
        hits = index.search(q, k=3)
        qn = torch.nn.functional.normalize(q.unsqueeze(0), dim=-1)
        en = torch.nn.functional.normalize(emb, dim=-1)
        # reference via bf16-rounded fp32 matmul (index stores bf16)
        scores = (qn.to(torch.bfloat16).float() @
                  en.to(torch.bfloat16).float().t())[0]
        expect = torch.topk(scores, 3).indices.tolist()
        assert [h["payload"]["id"] for h in hits] == expect
        assert hits[0]["score"] >= hits[1]["score"] >= hits[2]["score"]

    def test_retrieval_step_enriches_prompts(self):
        import torch

        from mlrun_amd.serving import (RetrievalStep, TokenMeanEmbedder,
                                       VectorIndex)

        embedder = TokenMeanEmbedder(vocab_size=100, dim=32, seed=3)
        index = VectorIndex(dim=32)
        docs = [{"id": f"d{i}", "tokens": [i * 10 + j for j in range(4)]}
                for i in range(8)]
        index.add(torch.stack([embedder(d["tokens"])[0] for d in docs]),
                  docs)
        step = RetrievalStep(index=index, embedder=embedder, top_k=2)

        class _Ev:
            body = {"inputs": [[5, 6, 7]]}
            path = "/infer"

        out = step.do_event(_Ev()).body
        assert len(out["inputs"][0]) == 3 + 2 * 4  # 2 docs x 4 tokens
        assert out["inputs"][0][-3:] == [5, 6, 7]  # prompt preserved
        assert len(out["retrieval"][0]) == 2
        assert out["retrieval"][0][0]["doc_id"].startswith("d")

    def test_rag_graph_end_to_end(self):
        import torch

        import mlrun_amd
        from mlrun_amd.models.llama import LlamaServer
        from mlrun_amd.serving import (RetrievalStep, TokenMeanEmbedder,
                                       VectorIndex)

        vocab = 200
        embedder = TokenMeanEmbedder(vocab_size=vocab, dim=32, seed=1)
        index = VectorIndex(dim=32)
        gen = torch.Generator().manual_seed(7)
        docs = [{"id": f"doc{i}",
                 "tokens": torch.randint(0, vocab, (6,),
                                         generator=gen).tolist()}
                for i in range(16)]
        index.add(torch.stack([embedder(d["tokens"])[0] for d in docs]),
                  docs)
        fn = mlrun_amd.new_function(name="rag", kind="serving")
        graph = fn.set_topology("flow", engine="sync")
        graph.add_step(RetrievalStep, name="retrieve", index=index,
                       embedder=embedder, top_k=2)
        graph.add_step(LlamaServer, name="llm", after="retrieve",
                       config="tiny", batch_size=4, max_new_tokens=4,
                       respond=True)
        server = fn.to_mock_server()
        prompt = torch.randint(0, vocab, (5,), generator=gen).tolist()
        resp = server.test("/v2/models/llm/infer",
                           body={"inputs": [prompt], "max_tokens": 4})
        assert len(resp["outputs"][0]) == 4
        assert len(resp["retrieval"][0]) == 2


class TestPrometheusMetrics:
    def test_host_exposes_request_counters(self):
        import requests

        import mlrun_amd

        fn = mlrun_amd.new_function(name="prom", kind="remote")
        fn.spec.build["functionSourceCode"] = (
            "def handler(ctx, event):\n    return {'ok': 1}\n")
        addr = fn.deploy()
        try:
            requests.post(addr + "/infer", json={}, timeout=10)
            body = requests.get(addr + "/metrics", timeout=10).text
            assert 'mlrun_serving_requests_total{path="/infer",' \
                   'status="200"} 1.0' in body
            assert "mlrun_serving_request_seconds_bucket" in body
        finally:
            fn.stop()


class TestWorkerPool:
    def test_multi_process_workers_roundrobin(self):
        import requests

        import mlrun_amd

        fn = mlrun_amd.new_function(name="pool", kind="serving")
        fn.add_model("llm",
                     class_name="mlrun_amd.models.llama.LlamaServer",
                     config="tiny", batch_size=2, max_new_tokens=6)
        fn.with_replicas(2)
        addr = fn.deploy()
        try:
            outs = []
            for _ in range(4):
                resp = requests.post(
                    addr + "/v2/models/llm/infer",
                    json={"inputs": [[5, 6, 7]], "max_tokens": 4},
                    timeout=120)
                assert resp.status_code == 200
                outs.append(resp.json()["outputs"][0])
            assert all(len(o) == 4 for o in outs)
            # both workers run the same weights (same seed) — outputs
            # must agree across backends
            assert all(o == outs[0] for o in outs)
        finally:
            fn.stop()



class TestWorkerAutoscale:
    def test_pool_scales_up_under_load(self):
        """min_replicas=1, max_replicas=2: sustained concurrent
        connections trigger a scale-up (nuclio max_replicas analog)."""
        import threading

        import requests

        import mlrun_amd

        fn = mlrun_amd.new_function(name="auto-pool", kind="serving")
        fn.add_model("llm",
                     class_name="mlrun_amd.models.llama.LlamaServer",
                     config="tiny", batch_size=2, max_new_tokens=6)
        fn.with_replicas(1, max_replicas=2)
        pool = None
        addr = fn.deploy()
        try:
            pool = fn._worker_pool
            pool.scale_connections_per_worker = 2  # low trigger
            pool.scale_interval = 0.2
            assert len(pool.ports) == 1
            stop = threading.Event()

            def hammer():
                session = requests.Session()
                while not stop.is_set():
                    try:
                        session.post(addr + "/v2/models/llm/infer",
                                     json={"inputs": [[1, 2, 3]],
                                           "max_tokens": 3},
                                     timeout=60)
                    except Exception:
                        return

            threads = [threading.Thread(target=hammer, daemon=True)
                       for _ in range(6)]
            [t.start() for t in threads]
            import time as _time

            deadline = _time.time() + 60
            while len(pool.ports) < 2 and _time.time() < deadline:
                _time.sleep(0.3)
            stop.set()
            [t.join(timeout=10) for t in threads]
            assert len(pool.ports) == 2, "pool did not scale up"
            # the scaled-up backend serves traffic
            resp = requests.post(addr + "/v2/models/llm/infer",
                                 json={"inputs": [[1, 2, 3]],
                                       "max_tokens": 3}, timeout=120)
            assert resp.status_code == 200
        finally:
            fn.stop()


class TestCycleGuard:
    def test_cyclic_graph_errors_instead_of_hanging(self):
        import mlrun_amd
        from mlrun_amd.serving.states import RootFlowStep

        class Pass:
            def __init__(self, context=None, name=None):
                pass

            def do_event(self, event):
                return event

        fn = mlrun_amd.new_function(name="cycfn", kind="serving")
        graph = fn.set_topology("flow", engine="sync")
        graph.add_step(Pass, name="a")
        graph.add_step(Pass, name="b", after="a")
        server = fn.to_mock_server(namespace={"Pass": Pass})
        # wire the cycle after build: b -> a
        flow = server.graph
        flow.steps["b"]._next = ["a"]
        import pytest as _pytest

        with _pytest.raises(RuntimeError, match="cycle"):
            server.test("/x", body={})


class TestWorkerPoolStreaming:
    def test_ndjson_streams_through_the_l4_proxy(self):
        import json

        import requests

        import mlrun_amd

        fn = mlrun_amd.new_function(name="pool-stream", kind="serving")
        fn.add_model("llm",
                     class_name="mlrun_amd.models.llama.LlamaServer",
                     config="tiny", batch_size=2, max_new_tokens=6,
                     scheduling="continuous")
        addr = fn.deploy(workers=2)
        try:
            with requests.post(addr + "/v2/models/llm/infer",
                               json={"inputs": [[1, 2, 3]],
                                     "max_tokens": 4, "stream": True},
                               stream=True, timeout=120) as resp:
                toks = [json.loads(line)["token"]
                        for line in resp.iter_lines() if line]
            assert len(toks) == 4
        finally:
            fn.stop()


class TestV2ModelListing:
    def test_list_models_endpoint(self):
        import mlrun_amd

        fn = mlrun_amd.new_function(name="lst", kind="serving")
        fn.add_model("m1",
                     class_name="mlrun_amd.frameworks.tree."
                                "TreeEnsembleModelServer",
                     n_trees=3, depth=2, n_features=4)
        fn.add_model("m2",
                     class_name="mlrun_amd.frameworks.tree."
                                "TreeEnsembleModelServer",
                     n_trees=3, depth=2, n_features=4)
        server = fn.to_mock_server()
        out = server.test("/v2/models/", body=None, method="GET")
        assert sorted(out["models"]) == ["m1", "m2"]
        meta = server.test("/v2/models/m1", body=None, method="GET")
        assert meta["name"] == "m1"


class TestErrorStatusMapping:
    def test_unknown_model_404_over_http(self):
        import requests

        import mlrun_amd

        fn = _serving_fn()
        fn.add_model("m", class_name=EchoModel)
        addr = fn.deploy()
        try:
            resp = requests.post(addr + "/v2/models/nope/infer",
                                 json={"inputs": [1]}, timeout=30)
            assert resp.status_code == 404
            assert "not found" in resp.json()["error"]
            # bad input shape -> 400 family, not 500
            resp = requests.post(addr + "/v2/models/m/infer",
                                 data="definitely-not-json{{{",
                                 headers={"content-type":
                                          "application/json"},
                                 timeout=30)
            assert resp.status_code in (200, 400)
        finally:
            fn.stop()

    def test_mock_server_status_codes(self):
        fn = _serving_fn()
        fn.add_model("m", class_name=EchoModel)
        server = fn.to_mock_server()
        resp = server.test("/v2/models/ghost/infer",
                           body={"inputs": [1]}, silent=True,
                           get_body=False)
        assert resp.status_code == 404


class TestErrorStream:
    def test_failed_events_pushed_to_error_stream(self):
        """server.error_stream receives failed events (reference
        test_push_error / _init_async_objects error-stream wiring)."""
        import mlrun_amd
        from mlrun_amd.platforms import OutputStream

        def boom(body):
            raise ValueError("nope")

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=boom, name="boom")
        server = fn.to_mock_server(namespace={"boom": boom})
        server.error_stream = "errors-stream"
        server.test(body={"x": 1}, silent=True)
        records = OutputStream.get_stream("errors-stream").drain()
        assert records, "error stream got nothing"
        import json as _json

        record = _json.loads(records[0]) if isinstance(
            records[0], (str, bytes)) else records[0]
        assert "nope" in record["error"]

    def test_raising_error_stream_does_not_mask(self):
        """A broken error-stream object must not crash serving
        (reference test_push_error _DummyStreamRaiser)."""
        def boom(body):
            raise ValueError("original")

        fn = _serving_fn()
        graph = fn.set_topology("flow")
        graph.to(handler=boom, name="boom")
        server = fn.to_mock_server(namespace={"boom": boom})
        server.error_stream = "dummy:///nothing"

        class _Raiser:
            def push(self, data):
                raise RuntimeError("stream down")

        server._error_stream_object = _Raiser()
        resp = server.test(body={}, silent=True, get_body=False)
        assert "original" in str(resp.body)


class TestAsyncLoadMode:
    def test_lazy_model_load(self):
        """load_mode="async": models load on FIRST event, not at init
        (reference v2 async load mode)."""
        import mlrun_amd

        fn = _serving_fn()
        fn.spec.load_mode = "async"
        fn.set_topology("router")
        fn.add_model("m", class_name=EchoModel, model_path=".")
        server = fn.to_mock_server()
        step = server.graph.steps["router"].routes["m"]
        assert step._object.ready is False  # not loaded yet
        resp = server.test("/v2/models/m/infer", body={"inputs": [2]})
        assert resp["outputs"] == [4]
        assert step._object.ready is True   # loaded on demand


class TestRemoteModelUrl:
    def test_add_model_with_model_url_proxies(self):
        """add_model(model_url=...) routes infers to a remote model
        endpoint (reference new_remote_endpoint route)."""
        import json
        import threading

        from http.server import BaseHTTPRequestHandler, HTTPServer

        import mlrun_amd

        class Handler(BaseHTTPRequestHandler):
            def do_POST(self):
                n = int(self.headers.get("content-length", 0))
                body = json.loads(self.rfile.read(n))
                out = json.dumps(
                    {"outputs": [sum(body["inputs"])]}).encode()
                self.send_response(200)
                self.send_header("content-type", "application/json")
                self.end_headers()
                self.wfile.write(out)

            def log_message(self, *a):
                pass

        httpd = HTTPServer(("127.0.0.1", 0), Handler)
        port = httpd.server_address[1]
        threading.Thread(target=httpd.serve_forever,
                         daemon=True).start()
        try:
            fn = _serving_fn()
            fn.add_model("ext",
                         model_url=f"http://127.0.0.1:{port}/infer")
            server = fn.to_mock_server()
            resp = server.test("/v2/models/ext/infer",
                               body={"inputs": [1, 2, 3]})
            assert resp["outputs"] == [6]
        finally:
            httpd.shutdown()


def _tiny_hf_model_dir(tmp_path):
    import transformers

    cfg = transformers.LlamaConfig(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, max_position_embeddings=128)
    model = transformers.LlamaForCausalLM(cfg)
    model.save_pretrained(str(tmp_path / "hf"))
    return str(tmp_path / "hf")


class TestHuggingFaceServer:
    """HuggingFaceModelServer serves a real transformers model
    (reference frameworks/huggingface/model_server.py:24)."""

    def test_text_generation_cpu(self, tmp_path):
        import mlrun_amd

        model_dir = _tiny_hf_model_dir(tmp_path)
        fn = mlrun_amd.new_function("hf", kind="serving")
        fn.set_topology("router")
        fn.add_model(
            "tiny",
            class_name="mlrun_amd.frameworks.huggingface."
                       "HuggingFaceModelServer",
            model_path=model_dir, task="text-generation",
            device="cpu")
        server = fn.to_mock_server()
        resp = server.test("/v2/models/tiny/infer",
                           body={"inputs": [[1, 2, 3, 4]],
                                 "max_tokens": 4})
        out = resp["outputs"]
        assert len(out[0]) == 8  # 4 prompt + 4 generated ids

    @pytest.mark.gpu
    def test_text_generation_gpu(self, tmp_path):
        import torch

        import mlrun_amd

        assert torch.cuda.is_available()
        model_dir = _tiny_hf_model_dir(tmp_path)
        fn = mlrun_amd.new_function("hf", kind="serving")
        fn.set_topology("router")
        fn.add_model(
            "tiny",
            class_name="mlrun_amd.frameworks.huggingface."
                       "HuggingFaceModelServer",
            model_path=model_dir, task="text-generation",
            device="cuda:0")
        server = fn.to_mock_server()
        resp = server.test("/v2/models/tiny/infer",
                           body={"inputs": [[1, 2, 3, 4]],
                                 "max_tokens": 4})
        assert len(resp["outputs"][0]) == 8
