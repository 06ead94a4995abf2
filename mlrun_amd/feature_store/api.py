# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature-store API: ingest / get_offline_features /
get_online_feature_service.

Parity target: reference mlrun/feature_store/api.py (ingest :450,
get_offline_features :99, get_online_feature_service :296) +
retrieval/local_merger.py (pandas offline merge).
"""

import os
import typing

from ..config import config
from ..errors import MLRunInvalidArgumentError, MLRunNotFoundError
from ..utils import logger, now_iso
from .feature_set import FeatureSet
from .online import get_online_table
from .vector import FeatureVector, OnlineVectorService


def _resolve_feature_set(ref) -> FeatureSet:
    if isinstance(ref, FeatureSet):
        return ref
    if isinstance(ref, str):
        from ..db import get_run_db

        name = ref
        project = "default"
        if ref.startswith("store://feature-sets/"):
            body = ref[len("store://feature-sets/"):]
            project, _, name = body.partition("/")
        elif "/" in ref:
            project, _, name = ref.partition("/")
        name = name.split(":")[0]
        struct = get_run_db().get_feature_set(name, project)
        return FeatureSet.from_dict(struct)
    raise MLRunInvalidArgumentError("cannot resolve feature set")


def ingest(featureset: typing.Union[FeatureSet, str] = None, source=None,
           targets: list = None, namespace=None, return_df: bool = True,
           infer_options=None, overwrite=None, mlrun_context=None,
           run_config=None, spark_context=None):
    """Batch-ingest a source (DataFrame / csv / parquet path) through
    the featureset's transform graph into its targets (parquet offline
    + the online window/KV table)."""
    import pandas as pd

    from ..datastore.sources import BaseSource

    if spark_context is not None:
        raise MLRunInvalidArgumentError(
            "spark ingestion is replaced by the node-local batch/GPU "
            "engine (SURVEY §7); drop spark_context")
    if run_config is not None:
        # run-as-job (reference RunConfig): route through the job
        # runtime wrapper
        return run_ingestion_job(featureset,
                                 source if isinstance(source, str)
                                 else "")
    fset = _resolve_feature_set(featureset)
    if source is None:
        source = fset.spec.source
    if isinstance(source, BaseSource):
        df = source.to_dataframe()
    elif isinstance(source, str):
        if source.endswith(".csv"):
            df = pd.read_csv(source)
        elif source.endswith((".parquet", ".pq")):
            df = pd.read_parquet(source)
        else:
            raise MLRunInvalidArgumentError(f"unsupported source {source}")
    elif isinstance(source, pd.DataFrame):
        df = source.copy()
    else:
        raise MLRunInvalidArgumentError(
            "source must be a DataFrame, path, or Source object")

    # run the transform graph (batch steps)
    df = _run_graph(fset, df)

    # infer features from columns not yet declared
    known = set(fset.feature_names()) | set(fset.entity_names())
    from .feature_set import Feature

    for col in df.columns:
        if col not in known and col != fset.spec.timestamp_key:
            kind = "float" if df[col].dtype.kind in "if" else "str"
            fset.spec.features.append(Feature(name=col, value_type=kind))

    targets = targets or fset.spec.targets or ["parquet", "nosql"]
    # aggregations always fold into the online window table; the offline
    # (parquet) target is then enriched with the as-of-ingest aggregate
    # columns so vectors can reference them offline too
    if fset.spec.aggregations:
        table = get_online_table(fset)
        table.ingest_batch(df)
        entities = fset.entity_names()
        unique_keys = df[entities].drop_duplicates().to_dict(
            orient="records")
        agg_records = table.get(unique_keys)
        agg_cols = [f.name for f in fset.spec.features if f.aggregate]
        agg_df = pd.DataFrame([
            {**key, **{c: rec.get(c) for c in agg_cols}}
            for key, rec in zip(unique_keys, agg_records)])
        df = df.merge(agg_df, on=entities, how="left")
    from ..datastore.targets import BaseStoreTarget, get_target_from_spec

    fset.status.targets = []
    for target in targets:
        if isinstance(target, (BaseStoreTarget, dict)) or \
                (isinstance(target, str) and target not in
                 ("parquet", "offline", "nosql", "online")):
            tgt = get_target_from_spec(target)
            if tgt.kind == "nosql" and fset.spec.aggregations:
                path = f"online://{fset.fullname}"  # folded above
            else:
                path = tgt.write_dataframe(df, fset)
            fset.status.targets.append(tgt.status_entry(path))
            continue
        kind = target if isinstance(target, str) else target.get("kind")
        if kind in ("parquet", "offline"):
            path = _parquet_target_path(fset)
            os.makedirs(os.path.dirname(path), exist_ok=True)
            if os.path.isfile(path) and not overwrite:
                existing = pd.read_parquet(path)
                pd.concat([existing, df], ignore_index=True).to_parquet(path)
            else:
                df.to_parquet(path)
            fset.status.targets.append(
                {"name": "parquet", "kind": "parquet", "path": path,
                 "updated": now_iso()})
        elif kind in ("nosql", "online"):
            table = get_online_table(fset)
            if not fset.spec.aggregations:
                table.ingest_batch(df)  # aggregated sets folded above
            fset.status.targets.append(
                {"name": "nosql", "kind": "nosql",
                 "path": f"online://{fset.fullname}", "updated": now_iso()})
    # store feature statistics (used by $mean/$max impute policies and
    # model-monitoring reference data — reference infer_options=Stats)
    try:
        from ..data_types import InferOptions, get_df_stats

        fset.status.stats = get_df_stats(
            df.select_dtypes("number"), InferOptions.Stats)
    except Exception:
        pass
    fset.status.state = "ready"
    try:
        fset.save()
    except Exception as exc:
        logger.warning("feature set save failed", error=str(exc))
    return df if return_df else None


def _run_graph(fset: FeatureSet, df):
    graph = fset.graph
    if not graph.steps:
        return df
    from ..serving.server import Event, GraphContext

    graph.init_object(GraphContext(), {})
    event = Event(body=df)
    result = graph.run(event)
    return result.body if result is not None else df


def _parquet_target_path(fset: FeatureSet) -> str:
    base = str(config.feature_store.data_prefix or "") or os.path.join(
        config.base_dir, "feature-store")
    return os.path.join(base, fset.metadata.project or "default",
                        f"{fset.metadata.name}.parquet")


class IngestionService:
    """Continuous ingestion: drains a StreamSource into the feature
    set's targets on an interval (the reference's deploy_ingestion_
    service / run_ingestion_job analog, node-local)."""

    def __init__(self, feature_set: FeatureSet, stream,
                 interval_seconds: float = 1.0, targets=None,
                 max_batch: int = 4096):
        import threading

        self.feature_set = feature_set
        self.stream = stream
        self.interval = interval_seconds
        self.targets = targets or ["nosql"]
        self.max_batch = max_batch
        self.events_ingested = 0
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name=f"ingest-{feature_set.name}")

    def start(self):
        self._thread.start()
        return self

    def _loop(self):
        import pandas as pd

        while not self._stop.wait(self.interval):
            self.drain_once()

    def drain_once(self) -> int:
        import pandas as pd

        events = self.stream.drain(self.max_batch)
        if not events:
            return 0
        df = pd.DataFrame(events)
        ingest(self.feature_set, df, targets=self.targets,
               return_df=False)
        self.events_ingested += len(events)
        return len(events)

    def stop(self):
        self._stop.set()
        if self._thread.is_alive():
            self._thread.join(timeout=5)
        self.drain_once()  # final drain


def run_ingestion_job(featureset, source_path: str, name: str = "",
                      schedule: str = None, run_config=None):
    """Batch-ingest as a (schedulable) JOB run (reference api.py
    run_ingestion_job): materializes a wrapper function that calls
    ``ingest`` on the stored feature set, then runs it now or stores
    a cron schedule for the service scheduler."""
    import mlrun_amd

    fset = featureset if isinstance(featureset, FeatureSet) else \
        _resolve_feature_set(featureset)
    fset.save()
    project = fset.metadata.project or "default"
    name = name or f"{fset.metadata.name}-ingest"
    code = (
        "from mlrun_amd import feature_store as fstore\n\n\n"
        "def handler(context, featureset_uri: str = "
        f"{project + '/' + fset.metadata.name!r}, "
        f"source_path: str = {source_path!r}):\n"
        "    import pandas as pd\n\n"
        "    read = pd.read_parquet if source_path.endswith("
        "'.parquet') else pd.read_csv\n"
        "    df = fstore.ingest(featureset_uri, read(source_path))\n"
        "    context.log_result('rows', int(len(df)))\n")
    fn = mlrun_amd.new_function(name=name, kind="job",
                                project=project)
    fn.with_code(body=code)
    return fn.run(handler="handler", name=name, schedule=schedule,
                  local=schedule is None, watch=schedule is None)


def deploy_ingestion_service(featureset, source=None,
                             interval_seconds: float = 1.0, targets=None,
                             start: bool = True) -> IngestionService:
    """Continuously ingest a StreamSource into the feature set
    (reference api.deploy_ingestion_service)."""
    from ..datastore.sources import StreamSource

    fset = _resolve_feature_set(featureset)
    if source is None:
        source = StreamSource(name=f"{fset.name}-stream")
    service = IngestionService(fset, source,
                               interval_seconds=interval_seconds,
                               targets=targets)
    if start:
        service.start()
    return service


def preview(featureset, source, limit: int = 20):
    """Dry-run the transform graph on a sample (reference: preview)."""
    import pandas as pd

    fset = _resolve_feature_set(featureset)
    if isinstance(source, pd.DataFrame):
        df = source.head(limit)
    else:
        df = pd.read_csv(source, nrows=limit) if str(source).endswith(
            ".csv") else pd.read_parquet(source).head(limit)
    out = _run_graph(fset, df)
    fset.status.preview = out.head(limit).values.tolist()
    return out


class OfflineVectorResponse:
    def __init__(self, df, vector=None):
        self._df = df
        self.vector = vector
        self.status = "completed"

    def to_dataframe(self):
        return self._df

    def to_parquet(self, path, **kwargs):
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        self._df.to_parquet(path, **kwargs)
        return path

    def to_csv(self, path, **kwargs):
        self._df.to_csv(path, index=False, **kwargs)
        return path


def get_offline_features(feature_vector, entity_rows=None,
                         entity_timestamp_column=None, target=None,
                         drop_columns=None, with_indexes=False,
                         update_stats=False, join_graph=None,
                         start_time=None, end_time=None,
                         timestamp_for_filtering: str = None,
                         query: str = None, order_by=None,
                         additional_filters: list = None,
                         engine: str = None, engine_args: dict = None,
                         run_config=None,
                         spark_service=None) -> OfflineVectorResponse:
    """Join features from the parquet targets of the referenced sets
    (pandas merger — reference retrieval/local_merger.py).

    Reference query surface: start_time/end_time window on
    ``timestamp_for_filtering`` (or each set's timestamp key), a
    pandas ``query`` expression, ``order_by`` columns, and
    ``additional_filters`` as (column, op, value) tuples with op in
    =/==/!=/in/not in/>/>=/</<=.  dask/spark engines are replaced by
    the node-local pandas merger: any other ``engine``
    raises; ``run_config`` (run-as-job) is not needed node-locally."""
    import pandas as pd

    if engine not in (None, "", "local", "pandas"):
        raise MLRunInvalidArgumentError(
            f"engine {engine!r} is not available in the node-local "
            f"build — the pandas merger runs in-process (dask/spark "
            f"engines are replaced; SURVEY §7 design)")
    if spark_service:
        raise MLRunInvalidArgumentError(
            "spark_service is not supported (spark engines are "
            "replaced by the node-local design)")
    if run_config is not None:
        raise MLRunInvalidArgumentError(
            "run_config (run-as-job) is unnecessary node-locally — "
            "call get_offline_features in-process, or wrap it in a "
            "job runtime yourself")

    vector = FeatureVector.resolve(feature_vector)
    if join_graph is None:
        join_graph = getattr(vector.spec, "join_graph", None)
    asof = entity_timestamp_column is not None and entity_rows is not None
    merged = None
    entity_cols: list = []
    grouped = vector.grouped_features()
    if join_graph is not None:
        order = join_graph.order()
        grouped = sorted(
            grouped, key=lambda g: order.index(g[0])
            if g[0] in order else len(order))
    for fs_name, columns, aliases in grouped:
        fset = _resolve_feature_set(
            f"{vector.metadata.project or 'default'}/{fs_name}")
        path = _parquet_target_path(fset)
        if not os.path.isfile(path):
            raise MLRunNotFoundError(
                f"feature set {fs_name} has no parquet target (ingest it "
                f"first)")
        df = pd.read_parquet(path)
        entities = fset.entity_names()
        entity_cols = entities
        ts_key = fset.spec.timestamp_key
        if columns != ["*"]:
            missing = [c for c in columns if c not in df.columns]
            if missing:
                raise MLRunInvalidArgumentError(
                    f"features {missing} not found in {fs_name}")
            keep = entities + columns
            if ts_key and ts_key in df.columns and ts_key not in keep:
                keep = keep + [ts_key]
            df = df[keep]
        if aliases:
            df = df.rename(columns=aliases)
        ts = fset.spec.timestamp_key
        if asof and ts and ts in df.columns:
            # point-in-time (as-of) join against the entity frame's
            # timestamp column (reference entity_timestamp_column)
            left = entity_rows.sort_values(entity_timestamp_column)
            right = df.sort_values(ts)
            joined = pd.merge_asof(
                left, right, left_on=entity_timestamp_column,
                right_on=ts, by=entities,
                direction="backward")
            if ts != entity_timestamp_column:
                joined = joined.drop(columns=[ts], errors="ignore")
            merged = joined if merged is None else merged.merge(
                joined, on=list(entity_rows.columns), how="inner")
            continue
        # latest row per entity for the join (offline snapshot)
        if ts and ts in df.columns:
            df = df.sort_values(ts).groupby(entities, as_index=False).last()
            if ts not in (columns if columns != ["*"] else df.columns):
                df = df.drop(columns=[ts], errors="ignore")
        else:
            df = df.groupby(entities, as_index=False).last()
        how = join_graph.how_for(fs_name) if join_graph is not None \
            else "inner"
        merged = df if merged is None else merged.merge(
            df, on=entities, how=how)
    if merged is None:
        raise MLRunInvalidArgumentError("vector references no features")
    if entity_rows is not None and not asof:
        merged = entity_rows.merge(merged, on=entity_cols, how="left")
    if vector.spec.label_feature:
        pass
    # reference query surface: time window, row filters, query, order
    ts_col = timestamp_for_filtering or entity_timestamp_column
    if (start_time is not None or end_time is not None) and ts_col and \
            ts_col in merged.columns:
        ts_series = pd.to_datetime(merged[ts_col])
        if start_time is not None:
            merged = merged[ts_series >= pd.Timestamp(start_time)]
            ts_series = ts_series[ts_series >= pd.Timestamp(start_time)]
        if end_time is not None:
            merged = merged[ts_series <= pd.Timestamp(end_time)]
    for filt in additional_filters or []:
        col, op, value = filt
        if col not in merged.columns:
            continue
        if op in ("=", "=="):
            merged = merged[merged[col] == value]
        elif op == "!=":
            merged = merged[merged[col] != value]
        elif op == "in":
            merged = merged[merged[col].isin(value)]
        elif op == "not in":
            merged = merged[~merged[col].isin(value)]
        elif op == ">":
            merged = merged[merged[col] > value]
        elif op == ">=":
            merged = merged[merged[col] >= value]
        elif op == "<":
            merged = merged[merged[col] < value]
        elif op == "<=":
            merged = merged[merged[col] <= value]
        else:
            raise MLRunInvalidArgumentError(
                f"unsupported filter op {op!r}")
    if query:
        merged = merged.query(query)
    if order_by:
        cols = [order_by] if isinstance(order_by, str) else list(order_by)
        merged = merged.sort_values([c for c in cols
                                     if c in merged.columns])
    if drop_columns:
        merged = merged.drop(columns=[c for c in drop_columns
                                      if c in merged.columns])
    if not with_indexes:
        merged = merged.reset_index(drop=True)
    if target:
        merged.to_parquet(target)
    return OfflineVectorResponse(merged, vector)


def get_online_feature_service(feature_vector, impute_policy: dict = None,
                               fixed_window_type=None,
                               entity_keys=None, update_stats=False,
                               run_config=None) -> OnlineVectorService:
    """Start an online lookup service over the vector's feature sets
    (reference api.py:296).  update_stats refreshes the vector's
    feature-stats from the referenced sets; run_config (deploy-as-
    nuclio) is unnecessary node-locally — the service runs
    in-process."""
    if run_config is not None:
        raise MLRunInvalidArgumentError(
            "run_config is not supported: the online service is "
            "in-process (node-local design); use fn.deploy() for an "
            "HTTP-fronted service")
    vector = FeatureVector.resolve(feature_vector)
    if update_stats:
        stats = {}
        for fs_name, _, _ in vector.grouped_features():
            fset = _resolve_feature_set(
                f"{vector.metadata.project or 'default'}/{fs_name}")
            fset_stats = (fset.status.to_dict() or {}).get("stats") \
                if hasattr(fset.status, "to_dict") else None
            if fset_stats:
                stats.update(fset_stats)
        vector.status.stats = stats
        try:
            vector.save()
        except Exception:
            pass  # vector may be ad-hoc (not stored)
    tables = {}
    for fs_name, _, _ in vector.grouped_features():
        fset = _resolve_feature_set(
            f"{vector.metadata.project or 'default'}/{fs_name}")
        tables[fs_name] = get_online_table(fset)
    return OnlineVectorService(vector, tables,
                               impute_policy=impute_policy or {})


def get_feature_set(uri, project: str = None) -> FeatureSet:
    """Get a feature set object from the db by uri
    ({project}/{name}[:tag]) — reference api.py:1223."""
    if project and "/" not in str(uri) and \
            not str(uri).startswith("store://"):
        uri = f"{project}/{uri}"
    return _resolve_feature_set(uri)


def get_feature_vector(uri, project: str = None):
    """Get a feature vector object from the db by uri
    ({project}/{name}[:tag]) — reference api.py:1232."""
    from ..db import get_run_db
    from .vector import FeatureVector

    name = str(uri)
    proj = project or "default"
    if name.startswith("store://feature-vectors/"):
        body = name[len("store://feature-vectors/"):]
        proj, _, name = body.partition("/")
    elif "/" in name:
        proj, _, name = name.partition("/")
    name, _, tag = name.partition(":")
    struct = get_run_db().get_feature_vector(name, proj, tag=tag or None)
    return FeatureVector.from_dict(struct)


def delete_feature_set(name, project: str = "", tag: str = None,
                       uid: str = None, force: bool = False):
    """Delete a FeatureSet from the DB (reference api.py:1241); with
    force=False refuses when materialized targets still exist."""
    from ..db import get_run_db
    from ..errors import MLRunInvalidArgumentError

    if tag and uid:
        raise MLRunInvalidArgumentError(
            "both tag and uid must not be specified")
    db = get_run_db()
    if not force:
        try:
            struct = db.get_feature_set(name, project, tag=tag)
        except Exception:
            struct = None
        targets = ((struct or {}).get("status") or {}).get("targets")
        if targets:
            raise MLRunInvalidArgumentError(
                f"feature set {name} has materialized targets; delete "
                "them first or pass force=True")
    db.delete_feature_set(name, project, tag=tag)


def delete_feature_vector(name, project: str = "", tag: str = None,
                          uid: str = None):
    """Delete a FeatureVector from the DB (reference api.py:1264)."""
    from ..db import get_run_db
    from ..errors import MLRunInvalidArgumentError

    if tag and uid:
        raise MLRunInvalidArgumentError(
            "both tag and uid must not be specified")
    get_run_db().delete_feature_vector(name, project, tag=tag)


def deploy_ingestion_service_v2(featureset, source=None, targets=None,
                                name: str = None, run_config=None,
                                verbose=False):
    """V2 alias of deploy_ingestion_service (reference api.py) —
    returns (endpoint/url, function object)."""
    service = deploy_ingestion_service(featureset, source=source,
                                       targets=targets)
    return getattr(service, "endpoint", ""), service
