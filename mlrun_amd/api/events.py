# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Alerts & events: match incoming events against stored alert
configs, bump state, fire notifications.

Parity target: reference server/api/crud/alerts.py:32 Alerts
(process_event :150) + mlrun/alerts/alert.py AlertConfig.
"""

import typing

from ..model import ModelObj
from ..utils import logger, now_iso


class AlertConfig(ModelObj):
    """Client-side alert config object (reference alerts/alert.py:22)."""

    def __init__(self, project=None, name=None, summary=None, severity=None,
                 trigger=None, criteria=None, notifications=None,
                 reset_policy=None, entities=None):
        self.project = project
        self.name = name
        self.summary = summary or ""
        self.severity = severity or "medium"
        # trigger: {"events": ["model-drift", ...]}
        self.trigger = trigger or {}
        # criteria: {"count": N, "period": "10m"} - fire after N events
        self.criteria = criteria or {}
        self.notifications = notifications or []
        self.reset_policy = reset_policy or "auto"
        self.entities = entities or {}

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {k: getattr(self, k) for k in
                ("project", "name", "summary", "severity", "trigger",
                 "criteria", "notifications", "reset_policy", "entities")}


def process_event(project: str, event_kind: str, event: dict,
                  db=None) -> typing.List[str]:
    """Match an event against alert configs; returns fired alert names
    (reference crud/alerts.py:150)."""
    if db is None:
        from ..db import get_run_db

        db = get_run_db()
    fired = []
    for alert in db.list_alert_configs(project):
        trigger = alert.get("trigger") or {}
        events = trigger.get("events") or []
        if events and event_kind not in events:
            continue
        name = alert.get("name")
        count = db.bump_alert_state(project, name) if hasattr(
            db, "bump_alert_state") else 1
        required = int((alert.get("criteria") or {}).get("count", 1) or 1)
        if count < required:
            continue
        fired.append(name)
        _push_alert_notifications(alert, event_kind, event)
        if (alert.get("reset_policy") or "auto") == "auto" and \
                hasattr(db, "reset_alert_state"):
            # auto: firing rearms the alert (another `count` events
            # must arrive before it fires again)
            db.reset_alert_state(project, name)
    return fired


def _push_alert_notifications(alert: dict, event_kind: str, event: dict):
    from ..utils.notifications import get_notification_class

    for spec in alert.get("notifications") or []:
        if isinstance(spec, dict) and "notification" in spec:
            spec = spec["notification"]
        kind = spec.get("kind", "console")
        try:
            notification = get_notification_class(kind)(
                spec.get("name", ""), spec.get("params") or {})
            notification.push(
                f"alert {alert.get('name')}: {alert.get('summary') or event_kind}",
                alert.get("severity", "medium"),
                [{"event": event_kind, "body": event, "time": now_iso()}])
        except Exception as exc:
            logger.warning("alert notification failed", error=str(exc))
