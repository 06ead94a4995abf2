# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Online feature store: HBM-resident window aggregations + latest-value
KV table.

The reference's online path is a storey Table over V3IO/Redis with
per-key running aggregates (datastore/targets.py:1415).  MI355X-native
design: per-(featureset, column) ring buffers of per-period partial
aggregates live as torch tensors — on the GPU when available (288 GB
HBM easily holds millions of keys), CPU tensors otherwise — ingested
and reduced by the HIP window kernels (ops.window_ingest/window_reduce)
in one batched launch per column, not per-event asyncio emits.
"""

import threading
import time
import typing

import numpy as np
import torch

from .feature_set import FeatureSet, parse_span


class WindowRing:
    """Ring of per-period partials for one aggregated column.

    ring[key, period, 0] = sum, [1] = count; min/max/first/last are
    tracked in auxiliary per-key tensors (they don't decompose over
    ring periods for sliding windows; kept as running values like the
    reference's running aggregates)."""

    def __init__(self, period_seconds: int, n_periods: int, device="cpu",
                 capacity: int = 1024):
        self.period_seconds = period_seconds
        self.n_periods = n_periods
        self.device = device
        self.capacity = capacity
        self.ring = torch.zeros(capacity, n_periods, 4, dtype=torch.float32,
                                device=device)
        # running (non-windowed) aggregates per key:
        # [min, max, first, last, count, sum, sumsq]
        self.running = torch.zeros(capacity, 7, dtype=torch.float32)
        self.running[:, 0] = float("inf")
        self.running[:, 1] = float("-inf")
        # per-(key, period) sum of squares for stdvar/stddev windows
        self.ring_sq = torch.zeros(capacity, n_periods, 4,
                                   dtype=torch.float32, device=device)
        self.last_period = -1

    def grow(self, capacity: int):
        if capacity <= self.capacity:
            return
        new_ring = torch.zeros(capacity, self.n_periods, 4,
                               dtype=torch.float32, device=self.device)
        new_ring[:self.capacity] = self.ring
        self.ring = new_ring
        new_sq = torch.zeros(capacity, self.n_periods, 4,
                             dtype=torch.float32, device=self.device)
        new_sq[:self.capacity] = self.ring_sq
        self.ring_sq = new_sq
        new_running = torch.zeros(capacity, 7, dtype=torch.float32)
        new_running[:, 0] = float("inf")
        new_running[:, 1] = float("-inf")
        new_running[:self.capacity] = self.running
        self.running = new_running
        self.capacity = capacity

    def _expire_old_periods(self, current_period: int):
        """Zero ring cells for periods that wrapped since last ingest."""
        if self.last_period < 0:
            self.last_period = current_period
            return
        gap = current_period - self.last_period
        if gap <= 0:
            return
        if gap >= self.n_periods:
            self.ring.zero_()
            self.ring_sq.zero_()
        else:
            for p in range(self.last_period + 1, current_period + 1):
                idx = p % self.n_periods
                self.ring[:, idx].zero_()
                self.ring_sq[:, idx].zero_()
        self.last_period = current_period

    def ingest(self, key_ids: torch.Tensor, values: torch.Tensor,
               timestamps: torch.Tensor):
        """Batched fold of events into the ring (HIP kernel on GPU)."""
        from .. import ops

        period_idx_abs = (timestamps.long() //
                          self.period_seconds)
        current = int(period_idx_abs.max())
        self._expire_old_periods(current)
        # drop late events older than the ring span (their slot would
        # alias a live period)
        fresh = period_idx_abs > (current - self.n_periods)
        keys32 = key_ids[fresh].to(torch.int32)
        pidx = (period_idx_abs[fresh] % self.n_periods).to(torch.int32)
        vals = values[fresh].to(torch.float32)
        if self.ring.is_cuda:
            keys32 = keys32.cuda(self.ring.device)
            pidx = pidx.to(self.ring.device)
            vals_dev = vals.to(self.ring.device)
        else:
            vals_dev = vals
        ops.window_ingest(self.ring, keys32, vals_dev, pidx)
        ops.window_ingest(self.ring_sq, keys32, vals_dev * vals_dev, pidx)
        # running aggregates over the FULL batch (storey keeps running
        # first/last/min/max even for late events) — fully vectorized
        # via sorted segments + reduceat (no per-key python loop)
        k = key_ids.numpy()
        v = values.to(torch.float32).numpy()
        order = np.argsort(k, kind="stable")
        k_sorted, v_sorted = k[order], v[order]
        uniq, starts = np.unique(k_sorted, return_index=True)
        ends = np.append(starts[1:], len(k_sorted))
        seg_min = np.minimum.reduceat(v_sorted, starts)
        seg_max = np.maximum.reduceat(v_sorted, starts)
        seg_sum = np.add.reduceat(v_sorted, starts)
        seg_sumsq = np.add.reduceat(v_sorted * v_sorted, starts)
        seg_count = (ends - starts).astype(np.float32)
        run = self.running.numpy()
        fresh_keys = run[uniq, 4] == 0
        run[uniq, 0] = np.minimum(run[uniq, 0], seg_min)
        run[uniq, 1] = np.maximum(run[uniq, 1], seg_max)
        run[uniq[fresh_keys], 2] = v_sorted[starts[fresh_keys]]  # first
        run[uniq, 3] = v_sorted[ends - 1]                        # last
        run[uniq, 4] += seg_count
        run[uniq, 5] += seg_sum
        run[uniq, 6] += seg_sumsq

    def window_values(self, window_seconds: int, now_ts: float) -> dict:
        """Reduce the ring for one window length -> tensors keyed by op
        (sum/count/avg/stdvar/stddev), each [capacity]."""
        from .. import ops

        # cover every bucket intersecting [now - window, now]: the
        # current partial bucket + window/period full buckets (events in
        # the oldest partial bucket are included — bucket-quantized
        # sliding window, storey-compatible)
        window_periods = max(window_seconds // self.period_seconds, 1) + 1
        # a window longer than the ring span would re-count wrapped
        # buckets — clamp to one pass over the ring
        window_periods = min(window_periods, self.n_periods)
        current_period = int(now_ts // self.period_seconds)
        self._expire_old_periods(current_period)
        out = ops.window_reduce(self.ring, window_periods,
                                current_period % self.n_periods)
        out_sq = ops.window_reduce(self.ring_sq, window_periods,
                                   current_period % self.n_periods)
        if out.is_cuda:
            out = out.cpu()
            out_sq = out_sq.cpu()
        total, count, avg = out[:, 0], out[:, 1], out[:, 2]
        sumsq = out_sq[:, 0]
        n = count.clamp(min=1.0)
        var = (sumsq / n - (total / n) ** 2).clamp(min=0.0)
        unbiased = torch.where(count > 1, var * count / (count - 1).clamp(
            min=1.0), torch.zeros_like(var))
        return {
            "sum": total,
            "count": count,
            "avg": avg,
            "sqr": sumsq,
            "stdvar": unbiased,
            "stddev": unbiased.sqrt(),
        }


class OnlineTable:
    """Per-featureset online state: latest row per key + window rings."""

    def __init__(self, feature_set: FeatureSet, device=None):
        self.feature_set = feature_set
        self.device = device or ("cuda:0" if torch.cuda.is_available()
                                 else "cpu")
        self.key_index: typing.Dict[typing.Any, int] = {}
        self.latest: typing.Dict[typing.Any, dict] = {}
        self.rings: typing.Dict[str, WindowRing] = {}
        self._lock = threading.Lock()
        for agg in feature_set.spec.aggregations:
            period = agg.period or agg.windows[0]
            period_s = parse_span(period)
            max_window = max(parse_span(w) for w in agg.windows)
            n_periods = max(max_window // period_s, 1) + 1
            self.rings[agg.name] = WindowRing(period_s, n_periods,
                                              device=self.device)

    def _key_of(self, row: dict):
        entities = self.feature_set.entity_names()
        if len(entities) == 1:
            return row[entities[0]]
        return tuple(row[e] for e in entities)

    def _key_ids(self, keys: list) -> torch.Tensor:
        import pandas as pd

        codes, uniques = pd.factorize(pd.Index(keys), sort=False)
        # map each unique key through (and extend) the dictionary once
        lut = np.empty(len(uniques), dtype=np.int64)
        for i, key in enumerate(uniques):
            idx = self.key_index.get(key)
            if idx is None:
                idx = len(self.key_index)
                self.key_index[key] = idx
            lut[i] = idx
        needed = len(self.key_index)
        for ring in self.rings.values():
            if needed > ring.capacity:
                ring.grow(max(needed, ring.capacity * 2))
        return torch.from_numpy(lut[codes])

    def ingest_batch(self, df):
        """Fold a dataframe batch: update latest rows + window rings."""
        import pandas as pd

        fset = self.feature_set
        ts_key = fset.spec.timestamp_key
        entities = fset.entity_names()
        with self._lock:
            if len(entities) == 1:
                keys = df[entities[0]].tolist()
            else:
                keys = list(zip(*(df[e] for e in entities)))
            key_ids = self._key_ids(keys)
            if ts_key and ts_key in df.columns:
                ts = pd.to_datetime(df[ts_key]).astype("int64") // 10 ** 9
                timestamps = torch.tensor(ts.values, dtype=torch.int64)
            else:
                timestamps = torch.full((len(df),), int(time.time()),
                                        dtype=torch.int64)
            for agg in fset.spec.aggregations:
                if agg.column not in df.columns:
                    continue
                values = torch.tensor(
                    df[agg.column].astype("float32").values)
                self.rings[agg.name].ingest(key_ids, values, timestamps)
            # latest row per key: one pandas pass, not a python loop
            if ts_key and ts_key in df.columns:
                ordered = df.sort_values(ts_key, kind="stable")
            else:
                ordered = df
            last_rows = ordered.drop_duplicates(subset=entities,
                                                keep="last")
            last_keys = (last_rows[entities[0]].tolist()
                         if len(entities) == 1 else
                         list(zip(*(last_rows[e] for e in entities))))
            for key, row in zip(last_keys,
                                last_rows.to_dict(orient="records")):
                self.latest[key] = row
        return len(df)

    def get_agg_matrix(self, entity_rows: typing.List[dict],
                       feature_names: typing.List[str],
                       now_ts: float = None):
        """Columnar fast path: aggregate features only -> float32
        matrix [rows, features] (NaN for unknown keys).  One ring
        reduce per (agg, window) serves the whole batch; rows resolve
        by fancy indexing — no per-row dicts."""
        import numpy as np

        now_ts = now_ts or time.time()
        fset = self.feature_set
        with self._lock:
            ids = np.array([self.key_index.get(self._key_of(row), -1)
                            for row in entity_rows], dtype=np.int64)
            known = ids >= 0
            safe = np.where(known, ids, 0)
            out = np.full((len(entity_rows), len(feature_names)), np.nan,
                          dtype=np.float32)
            reduced_cache = {}
            for j, name in enumerate(feature_names):
                resolved = None
                for agg in fset.spec.aggregations:
                    for window in agg.windows:
                        for op in agg.operations:
                            if f"{agg.name}_{op}_{window}" == name:
                                resolved = (agg, window, op)
                if resolved is None:
                    continue
                agg, window, op = resolved
                ring = self.rings[agg.name]
                ckey = (agg.name, window)
                if ckey not in reduced_cache:
                    reduced_cache[ckey] = ring.window_values(
                        parse_span(window), now_ts)
                vals = reduced_cache[ckey]
                if op in vals:
                    col = vals[op].numpy()[safe]
                else:
                    mapping = {"min": 0, "max": 1, "first": 2, "last": 3}
                    col = ring.running.numpy()[safe, mapping[op]]
                out[:, j] = np.where(known, col, np.nan)
            return out

    def get(self, entity_rows: typing.List[dict], now_ts: float = None
            ) -> typing.List[dict]:
        """Batched online lookup: latest values + window aggregates."""
        now_ts = now_ts or time.time()
        fset = self.feature_set
        with self._lock:
            # one ring reduce per (agg, window) serves the whole batch
            reduced = {}
            for agg in fset.spec.aggregations:
                ring = self.rings[agg.name]
                for window in agg.windows:
                    reduced[(agg.name, window)] = ring.window_values(
                        parse_span(window), now_ts)
            out = []
            for row in entity_rows:
                key = self._key_of(row)
                idx = self.key_index.get(key)
                record: dict = {}
                latest = self.latest.get(key)
                if latest:
                    record.update(latest)
                for agg in fset.spec.aggregations:
                    ring = self.rings[agg.name]
                    for window in agg.windows:
                        vals = reduced[(agg.name, window)]
                        for op in agg.operations:
                            name = f"{agg.name}_{op}_{window}"
                            if idx is None:
                                record[name] = None
                            elif op in vals:
                                record[name] = float(vals[op][idx])
                            else:  # min/max/first/last: running values
                                run = ring.running[idx]
                                mapping = {"min": 0, "max": 1, "first": 2,
                                           "last": 3}
                                value = float(run[mapping[op]])
                                if op in ("min", "max") and \
                                        float(run[4]) == 0:
                                    value = None
                                record[name] = value
                out.append(record)
            return out


_tables: dict = {}
_tables_lock = threading.Lock()


def get_online_table(feature_set: FeatureSet, device=None) -> OnlineTable:
    key = feature_set.fullname
    with _tables_lock:
        if key not in _tables:
            _tables[key] = OnlineTable(feature_set, device=device)
        return _tables[key]


def reset_online_tables():
    with _tables_lock:
        _tables.clear()
