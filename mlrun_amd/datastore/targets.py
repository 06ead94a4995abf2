# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature-store / ingestion targets.

Parity target: reference mlrun/datastore/targets.py (BaseStoreTarget
:382, ParquetTarget :800, CSVTarget, NoSqlTarget :1409).  The NoSql
(online KV) target maps to the GPU-resident OnlineTable engine.
"""

import os

from ..config import config
from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj
from ..utils import now_iso


class BaseStoreTarget(ModelObj):
    kind = "target"
    is_online = False
    is_offline = False

    def __init__(self, name: str = "", path: str = None, attributes=None,
                 partitioned: bool = False, key_bucketing_number=None,
                 partition_cols=None, time_partitioning_granularity=None):
        self.name = name or self.kind
        self.path = path
        self.attributes = attributes or {}
        self.partitioned = partitioned
        self.partition_cols = partition_cols
        self.time_partitioning_granularity = time_partitioning_granularity

    def default_path(self, feature_set) -> str:
        base = str(config.feature_store.data_prefix or "") or os.path.join(
            config.base_dir, "feature-store")
        return os.path.join(base, feature_set.metadata.project or "default",
                            f"{feature_set.metadata.name}.{self.kind}")

    def write_dataframe(self, df, feature_set) -> str:
        raise NotImplementedError

    def as_df(self, columns=None):
        raise NotImplementedError

    def status_entry(self, path: str) -> dict:
        return {"name": self.name, "kind": self.kind, "path": path,
                "updated": now_iso()}


class ParquetTarget(BaseStoreTarget):
    kind = "parquet"
    is_offline = True

    def write_dataframe(self, df, feature_set) -> str:
        import pandas as pd

        path = self.path or self.default_path(feature_set)
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        if self.partitioned and self.partition_cols:
            df.to_parquet(path, partition_cols=self.partition_cols)
        elif os.path.isfile(path) and self.attributes.get("append"):
            existing = pd.read_parquet(path)
            pd.concat([existing, df], ignore_index=True).to_parquet(path)
        else:
            df.to_parquet(path)
        self.path = path
        return path

    def as_df(self, columns=None):
        import pandas as pd

        return pd.read_parquet(self.path, columns=columns)


class CSVTarget(BaseStoreTarget):
    kind = "csv"
    is_offline = True

    def write_dataframe(self, df, feature_set) -> str:
        path = self.path or self.default_path(feature_set)
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        df.to_csv(path, index=False)
        self.path = path
        return path

    def as_df(self, columns=None):
        import pandas as pd

        df = pd.read_csv(self.path)
        return df[columns] if columns else df


class NoSqlTarget(BaseStoreTarget):
    """Online KV + window-aggregation target = the GPU OnlineTable
    (reference NoSqlTarget/RedisNoSqlTarget over storey Table)."""

    kind = "nosql"
    is_online = True

    def write_dataframe(self, df, feature_set) -> str:
        from ..feature_store.online import get_online_table

        table = get_online_table(feature_set)
        table.ingest_batch(df)
        self.path = f"online://{feature_set.fullname}"
        return self.path


class StreamTarget(BaseStoreTarget):
    """Push rows into an in-process StreamSource (queue-step analog of
    the reference's KafkaTarget)."""

    kind = "stream"
    is_online = True

    def __init__(self, stream=None, **kwargs):
        super().__init__(**kwargs)
        self._stream = stream

    def write_dataframe(self, df, feature_set) -> str:
        if self._stream is not None:
            self._stream.push(df.to_dict(orient="records"))
        self.path = f"stream://{self.name}"
        return self.path


class SQLTarget(BaseStoreTarget):
    kind = "sql"
    is_offline = True

    def write_dataframe(self, df, feature_set) -> str:
        import sqlite3

        path = self.path or self.default_path(feature_set) + ".db"
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        conn = sqlite3.connect(path)
        df.to_sql(self.attributes.get("table",
                                      feature_set.metadata.name),
                  conn, if_exists="replace", index=False)
        conn.close()
        self.path = path
        return path


class KafkaTarget(BaseStoreTarget):
    """Stream-fan-out target (reference KafkaTarget,
    targets.py:1634).  Node-locally a kafka topic is an OutputStream
    (platforms.py) keyed by the topic path — consumers attach via
    StreamSource/get_stream; when the ``kafka`` client IS importable
    and brokers are configured, rows publish to real Kafka instead."""

    kind = "kafka"
    is_online = True

    def __init__(self, path=None, brokers=None, **kwargs):
        super().__init__(path=path, **kwargs)
        self.brokers = brokers

    def write_dataframe(self, df, feature_set) -> str:
        topic = (self.path or feature_set.metadata.name).replace(
            "kafka://", "")
        records = df.to_dict(orient="records")
        if self.brokers:
            try:
                from kafka import KafkaProducer  # optional client
            except ImportError as exc:
                raise ImportError(
                    "kafka-python is not installed; omit `brokers` to "
                    "use the node-local stream analog") from exc
            import json as _json

            producer = KafkaProducer(bootstrap_servers=self.brokers)
            for record in records:
                producer.send(topic, _json.dumps(
                    record, default=str).encode())
            producer.flush()
        else:
            from ..platforms import OutputStream

            OutputStream(f"kafka://{topic}").push(records)
        self.path = f"kafka://{topic}"
        return self.path


class RedisNoSqlTarget(BaseStoreTarget):
    """Online KV rows in Redis (reference RedisNoSqlTarget,
    targets.py:1482).  Requires the ``redis`` client (not shipped in
    this image): fails loudly with the node-local alternative
    (NoSqlTarget = GPU OnlineTable) rather than degrading."""

    kind = "redisnosql"
    is_online = True

    def write_dataframe(self, df, feature_set) -> str:
        try:
            import redis
        except ImportError as exc:
            raise ImportError(
                "redis client not installed — use NoSqlTarget (the "
                "HBM-resident online table) for node-local online "
                "storage") from exc
        url = (self.path or "redis://localhost:6379").replace(
            "rediss://", "redis://")
        client = redis.Redis.from_url(url)
        entities = feature_set.entity_names()
        prefix = f"{feature_set.fullname}:"
        import json as _json

        for record in df.to_dict(orient="records"):
            key = prefix + "|".join(str(record.get(e)) for e in entities)
            client.hset(key, mapping={
                k: _json.dumps(v, default=str)
                for k, v in record.items()})
        self.path = url
        return url


class TSDBTarget(BaseStoreTarget):
    """Append-only time-series parquet log (the node-local TSDB the
    reference writes through v3io-frames/TDEngine — model monitoring's
    results log uses the same layout)."""

    kind = "tsdb"
    is_offline = True

    def write_dataframe(self, df, feature_set) -> str:
        import time as _time

        base = self.path or self.default_path(feature_set) + "_tsdb"
        os.makedirs(base, exist_ok=True)
        path = os.path.join(base, f"ts-{int(_time.time() * 1000)}"
                            f".parquet")
        df.to_parquet(path)
        self.path = base
        return base


_target_kinds = {cls.kind: cls for cls in
                 [ParquetTarget, CSVTarget, NoSqlTarget, StreamTarget,
                  SQLTarget, KafkaTarget, RedisNoSqlTarget, TSDBTarget]}


def get_target_from_spec(spec) -> BaseStoreTarget:
    if isinstance(spec, BaseStoreTarget):
        return spec
    if isinstance(spec, str):
        cls = _target_kinds.get(spec)
        if cls is None:
            raise MLRunInvalidArgumentError(f"unknown target kind {spec}")
        return cls()
    if isinstance(spec, dict):
        cls = _target_kinds.get(spec.get("kind", "parquet"))
        return cls.from_dict(spec)
    raise MLRunInvalidArgumentError(f"cannot build target from {spec!r}")
