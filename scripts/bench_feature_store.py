#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature-store benchmark (baseline config 3): ingest events/sec and
online feature-service lookups/sec, CPU ring vs GPU ring."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import numpy as np  # noqa: E402
import pandas as pd  # noqa: E402
import torch  # noqa: E402


def run(device_label: str, device, n_keys=100_000, n_events=1_000_000,
        lookup_batch=4096, lookup_iters=20):
    from mlrun_amd import feature_store as fstore
    from mlrun_amd.feature_store.online import OnlineTable

    fstore.reset_online_tables()
    fset = fstore.FeatureSet(f"bench-{device_label}", entities=["key"],
                             timestamp_key="ts")
    fset.add_aggregation("amount", ["sum", "count", "avg"], ["1h"], "5m")
    table = OnlineTable(fset, device=device)

    rng = np.random.default_rng(0)
    now = time.time()
    df = pd.DataFrame({
        "key": rng.integers(0, n_keys, n_events).astype(str),
        "amount": rng.normal(10, 3, n_events).astype("float32"),
        "ts": pd.to_datetime(now - rng.uniform(0, 3000, n_events),
                             unit="s"),
    })

    t0 = time.perf_counter()
    table.ingest_batch(df)
    if device != "cpu":
        torch.cuda.synchronize()
    ingest_dt = time.perf_counter() - t0

    keys = [{"key": str(k)} for k in rng.integers(0, n_keys, lookup_batch)]
    table.get(keys[:8])  # warmup
    t0 = time.perf_counter()
    for _ in range(lookup_iters):
        records = table.get(keys)
    if device != "cpu":
        torch.cuda.synchronize()
    lookup_dt = (time.perf_counter() - t0) / lookup_iters

    # columnar fast path (OnlineVectorService as_list over aggregates)
    names = ["amount_sum_1h", "amount_avg_1h", "amount_count_1h"]
    table.get_agg_matrix(keys[:8], names)
    t0 = time.perf_counter()
    for _ in range(lookup_iters):
        table.get_agg_matrix(keys, names)
    if device != "cpu":
        torch.cuda.synchronize()
    col_dt = (time.perf_counter() - t0) / lookup_iters
    print(f"[{device_label}] columnar get: "
          f"{lookup_batch / col_dt:,.0f} lookups/s "
          f"({col_dt * 1000:.2f} ms per batch)", flush=True)

    print(f"[{device_label}] ingest: {n_events / ingest_dt:,.0f} events/s "
          f"({ingest_dt * 1000:.0f} ms for {n_events:,}); "
          f"online get: {lookup_batch / lookup_dt:,.0f} lookups/s "
          f"({lookup_dt * 1000:.2f} ms per {lookup_batch}-row batch)",
          flush=True)
    assert records[0]["amount_count_1h"] is None or \
        records[0]["amount_count_1h"] >= 0
    return n_events / ingest_dt, lookup_batch / lookup_dt


if __name__ == "__main__":
    run("cpu", "cpu", n_events=200_000)
    if torch.cuda.is_available():
        run("gpu", "cuda:0")
