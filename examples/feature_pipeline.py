#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature-store pipeline: transform graph -> windowed aggregations ->
online service (baseline config 3 shape)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import numpy as np  # noqa: E402
import pandas as pd  # noqa: E402

import mlrun_amd.feature_store as fstore  # noqa: E402


def main():
    rng = np.random.default_rng(0)
    n = 10000
    now = time.time()
    events = pd.DataFrame({
        "customer": rng.choice([f"c{i}" for i in range(50)], n),
        "amount": rng.uniform(1, 500, n).round(2),
        "city": rng.choice(["NY", "SF", "TLV"], n),
        "ts": pd.to_datetime(now - rng.uniform(0, 3600, n), unit="s"),
    })

    fset = fstore.FeatureSet("transactions", entities=["customer"],
                             timestamp_key="ts")
    fset.graph.to(fstore.OneHotEncoder(
        mapping={"city": ["NY", "SF", "TLV"]}), name="onehot")
    fset.add_aggregation("amount", ["sum", "avg", "count", "max"],
                         windows=["10m", "1h"], period="5m")
    fstore.ingest(fset, events)
    print("ingested", n, "events; targets:",
          [t["kind"] for t in fset.status.targets])

    vector = fstore.FeatureVector(
        "fraud-features",
        features=["transactions.amount_sum_1h",
                  "transactions.amount_avg_10m",
                  "transactions.amount_max_1h"])
    vector.metadata.project = "default"
    svc = fstore.get_online_feature_service(
        vector, impute_policy={"amount_sum_1h": 0.0})
    rows = svc.get([{"customer": "c1"}, {"customer": "c2"}])
    for row in rows:
        print(row)


if __name__ == "__main__":
    main()
