#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Serving-graph engine overhead: the reference benchmark harness shape
(hack/benchmarks/model_serving_benchmark_local.py — router ->
V2ModelServer -> postprocess driven through the mock server) with the
model stubbed, so the number is the pure step-engine cost per event."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import mlrun_amd  # noqa: E402
from mlrun_amd.serving import V2ModelServer  # noqa: E402


class StubModel(V2ModelServer):
    def load(self):
        self.model = True

    def predict(self, request):
        return request["inputs"]


def main():
    fn = mlrun_amd.new_function(name="overhead", kind="serving")
    fn.add_model("m1", class_name=StubModel)
    fn.add_model("m2", class_name=StubModel)
    server = fn.to_mock_server()
    body = {"inputs": [1.0, 2.0, 3.0]}
    path = "/v2/models/m1/infer"
    for _ in range(1000):
        server.test(path, body=dict(body))
    n = 100_000
    t0 = time.perf_counter()
    for _ in range(n):
        server.test(path, body=dict(body))
    dt = time.perf_counter() - t0
    print(f"graph engine: {n / dt:,.0f} events/s "
          f"({dt / n * 1e6:.1f} us/event) through router->model->"
          f"response")
    # deeper flow: 4 chained steps
    fn2 = mlrun_amd.new_function(name="flow4", kind="serving")
    graph = fn2.set_topology("flow")
    step = graph.to(handler=lambda b: b, name="s0")
    for i in range(1, 4):
        step = step.to(handler=lambda b: b, name=f"s{i}")
    server2 = fn2.to_mock_server()
    for _ in range(1000):
        server2.test("/", body=dict(body))
    t0 = time.perf_counter()
    for _ in range(n):
        server2.test("/", body=dict(body))
    dt = time.perf_counter() - t0
    print(f"4-step flow:  {n / dt:,.0f} events/s "
          f"({dt / n * 1e6:.1f} us/event)")


if __name__ == "__main__":
    main()
