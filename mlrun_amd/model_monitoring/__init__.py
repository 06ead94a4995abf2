# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model monitoring: endpoint records, event stream stats, drift apps.

Parity target: reference mlrun/model_monitoring (stream_processing.py
EventStreamProcessor, controller, histogram_data_drift).  Implemented
in mlrun_amd as: per-endpoint event stream -> window stats (count/avg/
latency percentiles) -> parquet + in-memory TSDB; drift application
computing TVD/Hellinger/KL vs. the reference histogram.
"""

from .stream import (  # noqa: F401
    EventStreamProcessor,
    ModelMonitoringEvent,
    get_stream_processor,
)
from .drift import histogram_drift_metrics  # noqa: F401
from .controller import enable_model_monitoring, MonitoringController  # noqa: F401
from .apps import (  # noqa: F401
    HistogramDataDriftApplication,
    LatencyPerformanceApplication,
    ModelMonitoringApplicationBase,
    ModelMonitoringApplicationMetric,
    ModelMonitoringApplicationResult,
    MonitoringApplicationContext,
    ResultKindApp,
    ResultStatusApp,
)
from .writer import ModelMonitoringWriter  # noqa: F401
from .model_endpoint import (  # noqa: F401
    ModelEndpoint,
    ModelEndpointMetadata,
    ModelEndpointSpec,
    ModelEndpointStatus,
    TrackingPolicy,
    get_stream_path,
)


def get_store_object(project: str = "", **kwargs):
    """Endpoint-record store for a project (reference
    model_monitoring/db get_store_object): the node-local build keeps
    endpoint records in the run DB."""
    from ..db import get_run_db

    return get_run_db()


def get_tsdb_connector(project: str = "", **kwargs):
    """Time-series store connector (reference get_tsdb_connector):
    the node-local build records window stats in the stream
    processor's in-memory TSDB."""
    return get_stream_processor(project=project)
