#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Async serving-graph example: responder mid-flow + background
branches + bounded-queue hand-off (engine="async").

    python examples/async_graph.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import mlrun_amd


class Enrich:
    def __init__(self, context=None, name=None, **kw):
        self.name = name

    def do(self, event):
        event.body = {**event.body, "enriched_by": self.name}
        return event


class Audit:
    """Background branch: runs AFTER the response went out."""

    log = []

    def __init__(self, context=None, name=None, **kw):
        pass

    def do(self, event):
        time.sleep(0.05)  # pretend to write an audit record
        Audit.log.append(event.body)
        return event


def main():
    fn = mlrun_amd.new_function("async-demo", kind="serving")
    graph = fn.set_topology("flow", engine="async")
    responder = graph.to(Enrich, name="enrich")
    responder.respond()                      # response leaves HERE
    responder.to(Audit, name="audit")        # ... audit continues async

    server = fn.to_mock_server()
    t0 = time.perf_counter()
    resp = server.test("/score", body={"user": "u1"})
    latency_ms = (time.perf_counter() - t0) * 1000
    print(f"response (after {latency_ms:.1f} ms):", resp)
    assert not Audit.log, "audit should still be running"
    server.wait_for_completion()
    print("audit log after wait_for_completion:", Audit.log)
    assert Audit.log
    server.graph.shutdown()
    print("OK — responder returned before the audit branch finished")


if __name__ == "__main__":
    main()
