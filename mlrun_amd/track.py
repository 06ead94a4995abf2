# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Third-party experiment trackers auto-logged into runs.

Parity target: reference mlrun/track (tracker_manager.py:34,
trackers/mlflow_tracker.py:35 — pre_run/post_run hooks that import an
external tracker's artifacts/params/metrics into the run).  mlflow is
not installed in the MI355X image, so the MLFlowTracker activates only
when importable; the TrackerManager + Tracker protocol are the
extension points.
"""

import typing

from .utils import logger


class Tracker:
    """Protocol: enrich the run before/after the user handler runs."""

    enabled = False

    @classmethod
    def is_enabled(cls) -> bool:
        return cls.enabled

    def pre_run(self, context):
        pass

    def post_run(self, context):
        pass


class MLFlowTracker(Tracker):
    """Imports mlflow's active-run params/metrics/artifacts into the
    MLRun context (enabled only when mlflow is importable)."""

    def __init__(self):
        try:
            import mlflow  # noqa: F401

            self.enabled = True
        except ImportError:
            self.enabled = False

    def is_enabled(self) -> bool:
        return self.enabled

    def pre_run(self, context):
        if not self.enabled:
            return
        import mlflow

        mlflow.start_run(run_name=context.name)

    def post_run(self, context):
        if not self.enabled:
            return
        import mlflow

        run = mlflow.active_run()
        if run is None:
            return
        client = mlflow.tracking.MlflowClient()
        data = client.get_run(run.info.run_id).data
        for key, value in (data.params or {}).items():
            context.set_label(f"mlflow-{key}", value)
        for key, value in (data.metrics or {}).items():
            context.log_result(key, value)
        mlflow.end_run()


class TrackerManager:
    def __init__(self):
        self._trackers: typing.List[Tracker] = []

    def add_tracker(self, tracker: Tracker):
        if tracker.is_enabled():
            self._trackers.append(tracker)
        return self

    def pre_run(self, context):
        for tracker in self._trackers:
            try:
                tracker.pre_run(context)
            except Exception as exc:
                logger.warning("tracker pre_run failed", error=str(exc))

    def post_run(self, context):
        for tracker in self._trackers:
            try:
                tracker.post_run(context)
            except Exception as exc:
                logger.warning("tracker post_run failed", error=str(exc))


_manager: typing.Optional[TrackerManager] = None


def get_trackers_manager() -> TrackerManager:
    global _manager

    if _manager is None:
        _manager = TrackerManager()
        _manager.add_tracker(MLFlowTracker())
    return _manager
