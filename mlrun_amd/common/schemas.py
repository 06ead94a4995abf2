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
