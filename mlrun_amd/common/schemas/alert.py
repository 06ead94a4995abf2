# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Alert wire objects (reference mlrun/common/schemas/alert.py):
event kinds, entities, triggers, criteria and notifications used by
client-side ``mlrun.alerts.AlertConfig`` and the alerts API."""

import typing
from datetime import datetime
from enum import Enum

import pydantic


class EventEntityKind(str, Enum):
    MODEL_ENDPOINT_RESULT = "model-endpoint-result"
    MODEL_MONITORING_APPLICATION = "model-monitoring-application"
    JOB = "job"


class EventEntities(pydantic.BaseModel):
    kind: EventEntityKind
    project: str
    ids: typing.List[str]

    @pydantic.field_validator("ids")
    @classmethod
    def _one_id(cls, v):
        if len(v) != 1:
            raise ValueError("entity ids must contain exactly one id")
        return v


class EventKind(str, Enum):
    DATA_DRIFT_DETECTED = "data-drift-detected"
    DATA_DRIFT_SUSPECTED = "data-drift-suspected"
    CONCEPT_DRIFT_DETECTED = "concept-drift-detected"
    CONCEPT_DRIFT_SUSPECTED = "concept-drift-suspected"
    MODEL_PERFORMANCE_DETECTED = "model-performance-detected"
    MODEL_PERFORMANCE_SUSPECTED = "model-performance-suspected"
    SYSTEM_PERFORMANCE_DETECTED = "system-performance-detected"
    SYSTEM_PERFORMANCE_SUSPECTED = "system-performance-suspected"
    MM_APP_ANOMALY_DETECTED = "mm-app-anomaly-detected"
    MM_APP_ANOMALY_SUSPECTED = "mm-app-anomaly-suspected"
    MM_APP_FAILED = "mm-app-failed"
    FAILED = "failed"


# which entity kinds each event kind may attach to
_event_kind_entity_map = {
    kind: [EventEntityKind.MODEL_ENDPOINT_RESULT] for kind in EventKind}
_event_kind_entity_map[EventKind.MM_APP_FAILED] = [
    EventEntityKind.MODEL_MONITORING_APPLICATION]
_event_kind_entity_map[EventKind.FAILED] = [EventEntityKind.JOB]


class Event(pydantic.BaseModel):
    kind: EventKind
    timestamp: typing.Optional[typing.Union[str, datetime]] = None
    entity: EventEntities
    value_dict: typing.Optional[dict] = pydantic.Field(
        default_factory=dict)

    def is_valid(self) -> bool:
        return self.entity.kind in _event_kind_entity_map[self.kind]


class AlertActiveState(str, Enum):
    ACTIVE = "active"
    INACTIVE = "inactive"


class AlertSeverity(str, Enum):
    LOW = "low"
    MEDIUM = "medium"
    HIGH = "high"


class AlertTrigger(pydantic.BaseModel):
    events: typing.List[EventKind] = []
    prometheus_alert: typing.Optional[str] = None

    def __eq__(self, other):
        return (self.prometheus_alert == other.prometheus_alert
                and self.events == other.events)


class AlertCriteria(pydantic.BaseModel):
    count: int = 1
    period: typing.Optional[str] = None

    def __eq__(self, other):
        return self.count == other.count and self.period == other.period


class ResetPolicy(str, Enum):
    MANUAL = "manual"
    AUTO = "auto"


class AlertNotification(pydantic.BaseModel):
    notification: dict
    cooldown_period: typing.Optional[str] = None


class AlertTemplate(pydantic.BaseModel):
    template_id: typing.Optional[int] = None
    template_name: typing.Optional[str] = None
    template_description: typing.Optional[str] = ""
    summary: typing.Optional[str] = ""
    severity: typing.Optional[AlertSeverity] = None
    trigger: typing.Optional[AlertTrigger] = None
    criteria: typing.Optional[AlertCriteria] = None
    reset_policy: typing.Optional[ResetPolicy] = None
