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
    [now - window, now], drop events older than the ring span.
    Returns per-key lists of (ts, value) inside the window."""
    current = now_ts // period
    window_periods = max(window_seconds // period, 1) + 1
    covered = {}
    for key, value, ts in events:
        bucket = ts // period
        if bucket <= current - n_periods:
            continue  # outside the ring span
        if bucket > current or bucket <= current - window_periods:
            continue
        covered.setdefault(key, []).append((ts, value))
    return covered


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
        covered = brute_force(events, period, n_periods,
                              window_seconds, int(now_ts))
        for key in range(8):
            inside = covered.get(key, [])
            values = [v for _, v in inside]
            expect_count = len(values)
            assert got["count"][key].item() == expect_count, (
                key, events)
            assert math.isclose(got["sum"][key].item(), sum(values),
                                rel_tol=1e-4, abs_tol=1e-3), (key, events)
            if not values:
                for op in ("min", "max", "first", "last"):
                    assert math.isnan(got[op][key].item()), (op, key)
                continue
            # ALL TEN reference ops against the brute-force oracle
            assert math.isclose(got["min"][key].item(), min(values),
                                rel_tol=1e-6, abs_tol=1e-6)
            assert math.isclose(got["max"][key].item(), max(values),
                                rel_tol=1e-6, abs_tol=1e-6)
            assert math.isclose(
                got["sqr"][key].item(), sum(v * v for v in values),
                rel_tol=1e-4, abs_tol=1e-2)
            assert math.isclose(
                got["avg"][key].item(), sum(values) / len(values),
                rel_tol=1e-4, abs_tol=1e-3)
            # first/last by event time (stable ties resolved by the
            # packed value order — pick matching oracle values)
            first_ts = min(t for t, _ in inside)
            last_ts = max(t for t, _ in inside)
            first_candidates = [v for t, v in inside if t == first_ts]
            last_candidates = [v for t, v in inside if t == last_ts]
            assert any(math.isclose(got["first"][key].item(), v,
                                    rel_tol=1e-6, abs_tol=1e-6)
                       for v in first_candidates), (key, events)
            assert any(math.isclose(got["last"][key].item(), v,
                                    rel_tol=1e-6, abs_tol=1e-6)
                       for v in last_candidates), (key, events)
            if len(values) > 1:
                mean = sum(values) / len(values)
                var = sum((v - mean) ** 2 for v in values) / \
                    (len(values) - 1)
                assert math.isclose(got["stdvar"][key].item(), var,
                                    rel_tol=1e-3, abs_tol=1e-2)
                assert math.isclose(got["stddev"][key].item(),
                                    math.sqrt(var),
                                    rel_tol=1e-3, abs_tol=1e-2)

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
