# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Serving-graph engine: step DAG + graph server + model servers."""

from .states import (  # noqa: F401
    BaseStep,
    ErrorStep,
    FlowStep,
    GraphError,
    MonitoringApplicationStep,
    QueueStep,
    RootFlowStep,
    RouterStep,
    TaskStep,
)
from .server import (  # noqa: F401
    Event,
    GraphContext,
    GraphServer,
    GraphServerHost,
    MockEvent,
    create_graph_server,
)
from .v1_serving import MLModelServer, new_v1_model_server  # noqa: F401
from .v2_serving import V2ModelServer  # noqa: F401
from .routers import (  # noqa: F401
    BaseModelRouter,
    EnrichmentModelRouter,
    ModelRouter,
    ParallelRun,
    VotingEnsemble,
)
from .remote import BatchHttpRequests, RemoteStep  # noqa: F401
from .rag import RetrievalStep, TokenMeanEmbedder, VectorIndex  # noqa: F401
