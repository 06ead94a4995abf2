# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model-monitoring writer: persist application results.

Parity target: reference mlrun/model_monitoring/writer.py (the nuclio
writer function consuming the app-results stream and persisting to KV +
TSDB).  Node-local: results land in
- the model-endpoint record in the run DB (current state per app), and
- an append-only parquet results log under the monitoring dir (the
  TSDB analog — time-stamped rows queryable with pandas).
"""

import os
import time
import typing

from ..config import config
from ..utils import logger, now_iso


class ModelMonitoringWriter:
    def __init__(self, project: str = "default", db=None,
                 results_dir: str = ""):
        self.project = project
        self._db = db
        self.results_dir = results_dir or os.path.join(
            config.base_dir, "monitoring", project, "app_results")
        self._pending: typing.List[dict] = []

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def write(self, endpoint_id: str, application_name: str,
              results: list, window_stats: dict = None):
        """Persist one application run's results for one endpoint."""
        now = now_iso()
        rows = []
        app_state: dict = {}
        for result in results:
            row = result.to_dict()
            row.update({"endpoint_id": endpoint_id,
                        "application_name": application_name,
                        "time": now})
            rows.append(row)
            key = row.get("result_name") or row.get("metric_name")
            app_state[key] = row.get("result_value",
                                     row.get("metric_value"))
            if "result_status" in row:
                app_state[f"{key}_status"] = row["result_status"]
        self._pending.extend(rows)
        # current-state store: fold into the endpoint record
        try:
            db = self._get_db()
            record = {}
            try:
                record = db.get_model_endpoint(self.project,
                                               endpoint_id) or {}
            except Exception:
                pass
            status = record.setdefault("status", {})
            app_results = status.setdefault("app_results", {})
            app_results[application_name] = {"updated": now, **app_state}
            if window_stats:
                status["stats"] = window_stats
            record.setdefault("metadata", {}).update(
                {"project": self.project, "uid": endpoint_id})
            record.setdefault("kind", "model-endpoint")
            db.store_model_endpoint(self.project, endpoint_id, record)
        except Exception as exc:
            logger.warning("writer endpoint update failed",
                           error=str(exc))
        if len(self._pending) >= 64:
            self.flush()
        return rows

    def flush(self):
        """Append pending result rows to the parquet results log."""
        batch, self._pending = self._pending, []
        if not batch:
            return None
        try:
            import json

            import pandas as pd

            os.makedirs(self.results_dir, exist_ok=True)
            df = pd.DataFrame(batch)
            if "result_extra_data" in df:
                df["result_extra_data"] = df["result_extra_data"].map(
                    lambda v: json.dumps(v, default=str))
            path = os.path.join(self.results_dir,
                                f"results-{int(time.time() * 1000)}"
                                f".parquet")
            df.to_parquet(path)
            return path
        except Exception as exc:
            logger.warning("writer parquet flush failed", error=str(exc))
            return None

    def read_results(self, endpoint_id: str = None):
        """Load the results log (TSDB-analog query path)."""
        import pandas as pd

        self.flush()
        if not os.path.isdir(self.results_dir):
            return pd.DataFrame()
        frames = [pd.read_parquet(os.path.join(self.results_dir, f))
                  for f in sorted(os.listdir(self.results_dir))
                  if f.endswith(".parquet")]
        if not frames:
            return pd.DataFrame()
        df = pd.concat(frames, ignore_index=True)
        if endpoint_id is not None:
            df = df[df["endpoint_id"] == endpoint_id]
        return df
