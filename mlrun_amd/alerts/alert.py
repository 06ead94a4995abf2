# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Client-side alert config (reference mlrun/alerts/alert.py:22).

``AlertConfig`` is the user-facing object passed to
``project.store_alert_config`` / ``db.store_alert_config``: it carries
the trigger (event kinds), firing criteria (count within period),
entities, notifications and reset policy, optionally seeded from a
server-side alert template."""

import typing

from ..common.schemas import alert as alert_objects
from ..errors import MLRunBadRequestError, MLRunInvalidArgumentError
from ..model import ModelObj


def _to_dict(obj):
    if obj is None or isinstance(obj, (dict, list, str, int, float)):
        return obj
    if hasattr(obj, "model_dump"):
        return obj.model_dump(mode="json", exclude_none=True)
    if hasattr(obj, "dict"):
        return obj.dict()
    if hasattr(obj, "to_dict"):
        return obj.to_dict()
    return obj


class AlertConfig(ModelObj):
    """Alert configuration: fire notifications when matching events
    accumulate (reference alerts/alert.py:22).

    Example::

        import mlrun.common.schemas.alert as alert_objects
        from mlrun.alerts import AlertConfig

        alert = AlertConfig(
            project="my-project",
            name="drift-alert",
            summary="a drift was detected",
            severity=alert_objects.AlertSeverity.LOW,
            entities=alert_objects.EventEntities(
                kind=alert_objects.EventEntityKind.MODEL_ENDPOINT_RESULT,
                project="my-project", ids=[endpoint_id]),
            trigger=alert_objects.AlertTrigger(
                events=[alert_objects.EventKind.DATA_DRIFT_DETECTED]),
            criteria=alert_objects.AlertCriteria(count=3, period="1h"),
            notifications=[alert_objects.AlertNotification(
                notification=notification_dict)],
        )
        project.store_alert_config(alert)
    """

    _dict_fields = ["project", "name", "description", "summary",
                    "severity", "reset_policy", "state", "count",
                    "created"]

    def __init__(self, project: str = None, name: str = None,
                 template=None, description: str = None,
                 summary: str = None, severity=None, trigger=None,
                 criteria=None, reset_policy=None,
                 notifications: list = None, entities=None,
                 id: int = None, state=None, created: str = None,
                 count: int = None):
        self.project = project
        self.name = name
        self.description = description
        self.summary = summary
        self.severity = severity
        self.trigger = trigger
        self.criteria = criteria
        self.reset_policy = reset_policy
        self.notifications = notifications or []
        self.entities = entities
        self.id = id
        self.state = state
        self.created = created
        self.count = count
        if template:
            self._apply_template(template)

    def validate_required_fields(self):
        if not self.name:
            raise MLRunInvalidArgumentError("Alert name must be provided")

    def to_dict(self, fields: list = None, exclude: list = None,
                strip: bool = False) -> dict:
        if self.entities is None:
            raise MLRunBadRequestError("Alert entity field is missing")
        if not self.notifications:
            raise MLRunBadRequestError(
                "Alert must have at least one notification")
        struct = super().to_dict(self._dict_fields)
        struct["entities"] = _to_dict(self.entities)
        struct["notifications"] = [_to_dict(n)
                                   for n in self.notifications]
        if self.trigger is not None:
            struct["trigger"] = _to_dict(self.trigger)
        if self.criteria is not None:
            struct["criteria"] = _to_dict(self.criteria)
        if isinstance(struct.get("severity"), alert_objects.AlertSeverity):
            struct["severity"] = struct["severity"].value
        if isinstance(struct.get("reset_policy"),
                      alert_objects.ResetPolicy):
            struct["reset_policy"] = struct["reset_policy"].value
        return struct

    @classmethod
    def from_dict(cls, struct: dict = None, fields: list = None,
                  deprecated_fields: dict = None):
        struct = struct or {}
        new_obj = super().from_dict(struct, fields=fields)
        if struct.get("entities"):
            new_obj.entities = alert_objects.EventEntities.model_validate(
                struct["entities"])
        if struct.get("notifications"):
            new_obj.notifications = [
                alert_objects.AlertNotification.model_validate(n)
                for n in struct["notifications"]]
        if struct.get("trigger"):
            new_obj.trigger = alert_objects.AlertTrigger.model_validate(
                struct["trigger"])
        if struct.get("criteria"):
            new_obj.criteria = alert_objects.AlertCriteria.model_validate(
                struct["criteria"])
        return new_obj

    def with_notifications(self, notifications: typing.List[
            "alert_objects.AlertNotification"]):
        if not isinstance(notifications, list) or not all(
                isinstance(n, alert_objects.AlertNotification)
                for n in notifications):
            raise ValueError(
                "Notifications parameter must be a list of "
                "AlertNotification")
        self.notifications.extend(notifications)
        return self

    def with_entities(self, entities: "alert_objects.EventEntities"):
        if not isinstance(entities, alert_objects.EventEntities):
            raise ValueError(
                "Entities parameter must be of type: EventEntities")
        self.entities = entities
        return self

    def _apply_template(self, template):
        """Seed unset fields from a template (by name via the run db,
        or an AlertTemplate/dict); user fields win."""
        if isinstance(template, str):
            from ..db import get_run_db

            template = get_run_db().get_alert_template(template)
        if isinstance(template, dict):
            template = alert_objects.AlertTemplate.model_validate(
                {k: v for k, v in template.items()
                 if k in alert_objects.AlertTemplate.model_fields})
        self.summary = self.summary or template.summary
        self.severity = self.severity or template.severity
        self.criteria = self.criteria or template.criteria
        self.trigger = self.trigger or template.trigger
        self.reset_policy = self.reset_policy or template.reset_policy
