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
from .vector import (  # noqa: F401
    FeatureVector,
    FixedWindowType,
    JoinGraph,
    OnlineVectorService,
)
from .online import OnlineTable, get_online_table, reset_online_tables  # noqa: F401
from .api import (  # noqa: F401
    IngestionService,
    OfflineVectorResponse,
    delete_feature_set,
    delete_feature_vector,
    deploy_ingestion_service,
    deploy_ingestion_service_v2,
    get_feature_set,
    get_feature_vector,
    get_offline_features,
    get_online_feature_service,
    ingest,
    preview,
    run_ingestion_job,
)
from .common import RunConfig  # noqa: F401
from ..data_types import InferOptions, ValueType  # noqa: F401
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
