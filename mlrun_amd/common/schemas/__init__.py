# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Pydantic wire schemas shared by client and service.

Parity target: reference mlrun/common/schemas (32 modules of pydantic
models) — condensed to the entities this framework's API carries.
Validation is available to API users; internally the service stores
the validated dicts.
"""

import typing
from enum import Enum

from pydantic import BaseModel, Field


class RunState(str, Enum):
    created = "created"
    pending = "pending"
    running = "running"
    completed = "completed"
    error = "error"
    aborted = "aborted"
    aborting = "aborting"
    skipped = "skipped"
    unknown = "unknown"


class ObjectMetadata(BaseModel):
    name: typing.Optional[str] = None
    uid: typing.Optional[str] = None
    project: typing.Optional[str] = "default"
    tag: typing.Optional[str] = None
    labels: typing.Dict[str, str] = Field(default_factory=dict)
    annotations: typing.Dict[str, str] = Field(default_factory=dict)
    updated: typing.Optional[str] = None


class RunSpecSchema(BaseModel):
    function: typing.Optional[str] = None
    handler: typing.Optional[str] = None
    parameters: dict = Field(default_factory=dict)
    inputs: typing.Dict[str, str] = Field(default_factory=dict)
    outputs: typing.List[str] = Field(default_factory=list)
    output_path: typing.Optional[str] = None
    hyperparams: dict = Field(default_factory=dict)
    notifications: typing.List[dict] = Field(default_factory=list)
    state_thresholds: dict = Field(default_factory=dict)


class RunStatusSchema(BaseModel):
    state: RunState = RunState.created
    error: typing.Optional[str] = None
    results: typing.Optional[dict] = None
    artifacts: typing.Optional[typing.List[dict]] = None
    start_time: typing.Optional[str] = None
    last_update: typing.Optional[str] = None


class RunSchema(BaseModel):
    kind: str = "run"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: RunSpecSchema = Field(default_factory=RunSpecSchema)
    status: RunStatusSchema = Field(default_factory=RunStatusSchema)


class ArtifactSchema(BaseModel):
    kind: str = "artifact"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)


class FunctionSchema(BaseModel):
    kind: str = "job"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)


class ProjectSchema(BaseModel):
    kind: str = "project"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)


class ScheduleSchema(BaseModel):
    name: str
    kind: str = "job"
    cron_trigger: str
    task: dict = Field(default_factory=dict)
    next_run_time: typing.Optional[str] = None
    last_run_uri: typing.Optional[str] = None


class FeatureSetSchema(BaseModel):
    kind: str = "FeatureSet"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)


class FeatureVectorSchema(BaseModel):
    kind: str = "FeatureVector"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)


class ModelEndpointSchema(BaseModel):
    kind: str = "model-endpoint"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)


class AlertSeverity(str, Enum):
    low = "low"
    medium = "medium"
    high = "high"
    critical = "critical"


class AlertConfigSchema(BaseModel):
    project: typing.Optional[str] = "default"
    name: str
    summary: str = ""
    severity: AlertSeverity = AlertSeverity.medium
    trigger: dict = Field(default_factory=dict)
    criteria: dict = Field(default_factory=dict)
    notifications: typing.List[dict] = Field(default_factory=list)
    reset_policy: str = "auto"


class BackgroundTaskSchema(BaseModel):
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    status: dict = Field(default_factory=dict)


class SubmitJobBody(BaseModel):
    task: RunSchema
    schedule: typing.Optional[str] = None


def validate_run(struct: dict) -> dict:
    return RunSchema(**struct).model_dump(exclude_none=True)


def validate_artifact(struct: dict) -> dict:
    return ArtifactSchema(**struct).model_dump(exclude_none=True)


# ---------------------------------------------------------------------
# Round-2 expansion: typed sub-schemas + constants for the wire
# entities the reference models across its 32 schema modules
# (mlrun/common/schemas).  Sub-models use extra="allow" so documents
# with forward-compatible fields still validate; malformed SHAPES
# (wrong types, missing required names) 422.
# ---------------------------------------------------------------------
from pydantic import ConfigDict


class _Open(BaseModel):
    model_config = ConfigDict(extra="allow")


class NotificationKind(str, Enum):
    console = "console"
    ipython = "ipython"
    slack = "slack"
    git = "git"
    webhook = "webhook"
    mail = "mail"


class NotificationSchema(_Open):
    kind: NotificationKind = NotificationKind.console
    name: str = ""
    message: str = ""
    severity: str = "info"
    when: typing.List[str] = Field(default_factory=list)
    condition: str = ""
    params: dict = Field(default_factory=dict)
    secret_params: dict = Field(default_factory=dict)


class EntitySchema(_Open):
    name: str
    value_type: str = "str"
    labels: typing.Dict[str, str] = Field(default_factory=dict)


class FeatureSchema(_Open):
    name: str
    value_type: str = "float"
    labels: typing.Dict[str, str] = Field(default_factory=dict)


class AggregationSchema(_Open):
    name: str
    column: str
    operations: typing.List[str]
    windows: typing.List[str]
    period: typing.Optional[str] = None


class FeatureSetSpecSchema(_Open):
    entities: typing.List[EntitySchema] = Field(default_factory=list)
    features: typing.List[FeatureSchema] = Field(default_factory=list)
    aggregations: typing.List[AggregationSchema] = \
        Field(default_factory=list)
    timestamp_key: typing.Optional[str] = None
    targets: typing.List[dict] = Field(default_factory=list)
    description: typing.Optional[str] = None


class FeatureVectorSpecSchema(_Open):
    features: typing.List[str] = Field(default_factory=list)
    label_feature: typing.Optional[str] = None
    description: typing.Optional[str] = None


class FunctionSpecSchema(_Open):
    command: str = ""
    args: typing.List[str] = Field(default_factory=list)
    image: str = ""
    handler: typing.Optional[str] = None
    default_handler: typing.Optional[str] = None
    description: str = ""
    replicas: typing.Optional[int] = None
    resources: dict = Field(default_factory=dict)


class ArtifactSpecSchema(_Open):
    target_path: typing.Optional[str] = None
    size: typing.Optional[int] = None
    db_key: typing.Optional[str] = None
    producer: typing.Optional[dict] = None
    format: typing.Optional[str] = None


class ModelEndpointSpecSchema(_Open):
    model: typing.Optional[str] = None
    model_class: typing.Optional[str] = None
    function_uri: typing.Optional[str] = None
    monitoring_mode: str = "enabled"


class ModelEndpointStatusSchema(_Open):
    state: str = "ready"
    drift_status: typing.Optional[str] = None
    drift_metrics: typing.Optional[dict] = None
    stats: typing.Optional[dict] = None
    app_results: typing.Optional[dict] = None
    last_request: typing.Optional[str] = None


class ProjectSpecSchema(_Open):
    description: typing.Optional[str] = None
    goals: typing.Optional[str] = None
    source: typing.Optional[str] = None
    owner: typing.Optional[str] = None
    functions: typing.List[dict] = Field(default_factory=list)
    workflows: typing.List[dict] = Field(default_factory=list)
    artifacts: typing.List[dict] = Field(default_factory=list)
    params: dict = Field(default_factory=dict)


class BackgroundTaskState(str, Enum):
    created = "created"
    running = "running"
    succeeded = "succeeded"
    failed = "failed"


class HubSourceSchema(_Open):
    name: typing.Optional[str] = None
    spec: dict = Field(default_factory=dict)


class APIGatewaySchema(_Open):
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: dict = Field(default_factory=dict)


class DatastoreProfileSchema(_Open):
    name: str
    type: str = "generic"
    public: dict = Field(default_factory=dict)
    private: dict = Field(default_factory=dict)


class EventSchema(_Open):
    kind: typing.Optional[str] = None
    entity: dict = Field(default_factory=dict)
    value_dict: dict = Field(default_factory=dict)


class AlertTemplateSchema(_Open):
    template_name: typing.Optional[str] = None
    template_description: str = ""
    summary: str = ""
    severity: str = "medium"
    trigger: dict = Field(default_factory=dict)
    criteria: dict = Field(default_factory=dict)
    reset_policy: str = "auto"


class PaginationInfo(_Open):
    page: typing.Optional[int] = None
    page_size: typing.Optional[int] = None
    page_token: typing.Optional[str] = None
    total: typing.Optional[int] = None


class WorkflowSpecSchema(_Open):
    name: str = ""
    path: typing.Optional[str] = None
    code: typing.Optional[str] = None
    handler: typing.Optional[str] = None
    args: dict = Field(default_factory=dict)
    schedule: typing.Optional[str] = None
    engine: str = "local"


class ArtifactCategory(str, Enum):
    model = "model"
    dataset = "dataset"
    document = "document"
    other = "other"


class ScheduleKinds(str, Enum):
    job = "job"
    pipeline = "pipeline"


# typed wrappers over the earlier open dict-shaped models
class FeatureSetSchemaV2(BaseModel):
    kind: str = "FeatureSet"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: FeatureSetSpecSchema = Field(
        default_factory=FeatureSetSpecSchema)
    status: dict = Field(default_factory=dict)


class ModelEndpointSchemaV2(BaseModel):
    kind: str = "model-endpoint"
    metadata: ObjectMetadata = Field(default_factory=ObjectMetadata)
    spec: ModelEndpointSpecSchema = Field(
        default_factory=ModelEndpointSpecSchema)
    status: ModelEndpointStatusSchema = Field(
        default_factory=ModelEndpointStatusSchema)


def validate_feature_set(struct: dict) -> dict:
    return FeatureSetSchemaV2(**struct).model_dump(exclude_none=True)


def validate_model_endpoint(struct: dict) -> dict:
    return ModelEndpointSchemaV2(**struct).model_dump(exclude_none=True)


def validate_notification(struct: dict) -> dict:
    return NotificationSchema(**struct).model_dump(exclude_none=True)


def validate_alert_config(struct: dict) -> dict:
    validated = AlertConfigSchema(**{k: v for k, v in struct.items()
                                     if k != "entities"})
    out = validated.model_dump(exclude_none=True)
    # client AlertConfig carries extra fields (entities, description,
    # state, id, count, created) — preserve them verbatim
    for key, value in struct.items():
        if key not in out and value is not None:
            out[key] = value
    for spec in out.get("notifications", []):
        inner = spec.get("notification", spec) if isinstance(
            spec, dict) else spec
        validate_notification(inner)
    return out


def validate_datastore_profile(struct: dict) -> dict:
    return DatastoreProfileSchema(**struct).model_dump(
        exclude_none=True)
