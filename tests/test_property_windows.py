# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Property-based checks (hypothesis) of the window-aggregation ring
against a brute-force bucket model, and of the drift metrics'
mathematical properties."""

import math

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from mlrun_amd.feature_store.online import WindowRing

EVENT = st.tuples(
    st.integers(min_value=0, max_value=7),          # key
    st.floats(min_value=-100, max_value=100,
              allow_nan=False, allow_infinity=False),  # value
    st.integers(min_value=0, max_value=3599),       # ts offset seconds
)


def brute_force(events, period, n_periods, window_seconds, now_ts):
    """Bucket-quantized sliding window over raw events (the semantics
    window_values documents): include every bucket intersecting
    [now - window, now], drop events older than the ring span."""
    current = now_ts // period
    window_periods = max(window_seconds // period, 1) + 1
    sums, counts = {}, {}
    for key, value, ts in events:
        bucket = ts // period
        if bucket <= current - n_periods:
            continue  # outside the ring span
        if bucket > current or bucket <= current - window_periods:
            continue
        sums[key] = sums.get(key, 0.0) + value
        counts[key] = counts.get(key, 0) + 1
    return sums, counts


class TestWindowRingProperties:
    @settings(max_examples=60, deadline=None)
    @given(st.lists(EVENT, min_size=1, max_size=80),
           st.sampled_from([60, 300]),
           st.sampled_from([300, 900, 3600]))
    def test_matches_brute_force_buckets(self, events, period,
                                         window_seconds):
        n_periods = max(window_seconds // period, 1) + 2
        ring = WindowRing(period, n_periods, capacity=16)
        # ingest in timestamp order (the online path sees ordered
        # batches; stale-drop handles mild disorder separately)
        events = sorted(events, key=lambda e: e[2])
        keys = torch.tensor([e[0] for e in events])
        vals = torch.tensor([e[1] for e in events])
        ts = torch.tensor([float(e[2]) for e in events])
        ring.ingest(keys, vals, ts)
        now_ts = float(max(e[2] for e in events))
        got = ring.window_values(window_seconds, now_ts)
        sums, counts = brute_force(events, period, n_periods,
                                   window_seconds, int(now_ts))
        for key in range(8):
            expect_sum = sums.get(key, 0.0)
            expect_count = counts.get(key, 0)
            assert got["count"][key].item() == expect_count, (
                key, events)
            assert math.isclose(got["sum"][key].item(), expect_sum,
                                rel_tol=1e-4, abs_tol=1e-3), (key, events)

    @settings(max_examples=30, deadline=None)
    @given(st.lists(st.floats(min_value=-50, max_value=50,
                              allow_nan=False), min_size=2,
                    max_size=40))
    def test_stddev_matches_torch(self, values):
        ring = WindowRing(60, 4, capacity=4)
        n = len(values)
        ring.ingest(torch.zeros(n, dtype=torch.long),
                    torch.tensor(values), torch.full((n,), 30.0))
        got = ring.window_values(60, 30.0)
        expect = torch.tensor(values).var(unbiased=True).item()
        assert math.isclose(got["stdvar"][0].item(), expect,
                            rel_tol=1e-3, abs_tol=1e-3)


class TestDriftMetricProperties:
    @settings(max_examples=50, deadline=None)
    @given(st.lists(st.floats(min_value=0.01, max_value=10,
                              allow_nan=False), min_size=2, max_size=12),
           st.lists(st.floats(min_value=0.01, max_value=10,
                              allow_nan=False), min_size=2, max_size=12))
    def test_metric_bounds_and_symmetry(self, p, q):
        from mlrun_amd.model_monitoring.drift import (
            hellinger_distance,
            total_variation_distance,
        )

        size = min(len(p), len(q))
        p, q = p[:size], q[:size]
        tvd = total_variation_distance(p, q)
        hell = hellinger_distance(p, q)
        assert 0.0 <= tvd <= 1.0 + 1e-9
        assert 0.0 <= hell <= 1.0 + 1e-9
        # symmetry
        assert math.isclose(tvd, total_variation_distance(q, p),
                            abs_tol=1e-9)
        assert math.isclose(hell, hellinger_distance(q, p),
                            abs_tol=1e-7)


class TestContinuousSchedulerProperty:
    @settings(max_examples=10, deadline=None)
    @given(st.lists(
        st.tuples(st.lists(st.integers(min_value=1, max_value=900),
                           min_size=1, max_size=12),
                  st.integers(min_value=1, max_value=7)),
        min_size=1, max_size=9))
    def test_continuous_matches_batch_for_any_workload(self, requests):
        """For ANY set of (prompt, max_new) requests, token-level
        continuous scheduling produces exactly the greedy outputs of
        isolated batch generation."""
        from mlrun_amd.models.llama import LlamaConfig, LlamaServer

        cfg = LlamaConfig.tiny()
        if not hasattr(self, "_servers"):
            cont = LlamaServer(name="pc", config=cfg, batch_size=3,
                               max_new_tokens=8,
                               scheduling="continuous",
                               use_graph=False)
            cont.load()
            batch = LlamaServer(name="pb", config=cfg, batch_size=3,
                                max_new_tokens=8, use_graph=False)
            batch.load()
            batch.engines[0].weights.load_state_dict(
                cont.engines[0].weights.state_dict())
            self.__class__._servers = (cont, batch)
        cont, batch = self._servers

        class _Ev:
            path = "/infer"
            id = "t"

        import threading

        results = {}

        def one(i, prompt, max_new):
            ev = _Ev()
            ev.body = {"inputs": [prompt], "max_tokens": max_new}
            results[i] = cont.do_event(ev).body["outputs"][0]

        threads = [threading.Thread(target=one, args=(i, p, m))
                   for i, (p, m) in enumerate(requests)]
        [t.start() for t in threads]
        [t.join(timeout=60) for t in threads]
        for i, (prompt, max_new) in enumerate(requests):
            ev = _Ev()
            ev.body = {"inputs": [prompt], "max_tokens": max_new}
            expect = batch.do_event(ev).body["outputs"][0]
            assert results[i] == expect, (i, prompt, max_new)
