# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature store: feature sets with GPU-resident window aggregations,
feature vectors, offline merge + online service."""

from .feature_set import (  # noqa: F401
    AGGREGATION_OPS,
    Entity,
    Feature,
    FeatureAggregation,
    FeatureSet,
    parse_span,
)
from .vector import FeatureVector, JoinGraph, OnlineVectorService  # noqa: F401
from .online import OnlineTable, get_online_table, reset_online_tables  # noqa: F401
from .api import (  # noqa: F401
    IngestionService,
    deploy_ingestion_service,
    get_offline_features,
    get_online_feature_service,
    ingest,
    preview,
    run_ingestion_job,
)
from .steps import (  # noqa: F401
    DateExtractor,
    DropFeatures,
    FeaturesetValidator,
    Imputer,
    MapValues,
    MLRunStep,
    OneHotEncoder,
    SetEventMetadata,
)
