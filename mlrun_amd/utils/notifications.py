# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Notification push: console / ipython / webhook / slack / mail kinds.

Parity target: reference mlrun/utils/notifications/notification_pusher.py:96
NotificationPusher + notification/ kinds.  Network-less kinds (console)
always work; webhook/slack post JSON over HTTP when a URL is reachable.
"""

import json
import typing

from ..model import RunObject, RunStates
from . import logger, now_iso


class NotificationBase:
    kind = "base"

    def __init__(self, name="", params=None):
        self.name = name
        self.params = params or {}

    def push(self, message: str, severity: str, runs: list):
        raise NotImplementedError


class ConsoleNotification(NotificationBase):
    kind = "console"

    def push(self, message, severity, runs):
        print(f"[{severity}] {message}")
        for run in runs:
            meta = run.get("metadata", {})
            status = run.get("status", {})
            print(f"  run {meta.get('name')} ({meta.get('uid')}): "
                  f"state={status.get('state')} "
                  f"results={status.get('results')}")


class IPythonNotification(ConsoleNotification):
    kind = "ipython"


class WebhookNotification(NotificationBase):
    kind = "webhook"

    def push(self, message, severity, runs):
        import requests

        url = self.params.get("url")
        if not url:
            logger.warning("webhook notification has no url")
            return
        body = {"message": message, "severity": severity, "runs": runs,
                "time": now_iso()}
        body.update(self.params.get("override_body") or {})
        requests.post(url, json=body,
                      headers=self.params.get("headers") or {},
                      timeout=10, verify=self.params.get("verify_ssl", True))


class SlackNotification(NotificationBase):
    kind = "slack"

    def push(self, message, severity, runs):
        import requests

        url = self.params.get("webhook")
        if not url:
            logger.warning("slack notification has no webhook")
            return
        lines = [f"*{severity}* {message}"]
        for run in runs:
            meta = run.get("metadata", {})
            status = run.get("status", {})
            lines.append(f"• {meta.get('name')}: {status.get('state')}")
        requests.post(url, json={"text": "\n".join(lines)}, timeout=10)


class MailNotification(NotificationBase):
    kind = "mail"

    def push(self, message, severity, runs):
        import smtplib
        from email.message import EmailMessage

        from ..config import config

        server = self.params.get("server") or config.notifications.smtp.server
        if not server:
            logger.warning("mail notification has no smtp server")
            return
        msg = EmailMessage()
        msg["Subject"] = f"[{severity}] {message}"
        msg["From"] = self.params.get("sender") or \
            config.notifications.smtp.sender
        msg["To"] = self.params.get("to", "")
        msg.set_content(json.dumps(runs, indent=2, default=str))
        with smtplib.SMTP(server) as smtp:
            smtp.send_message(msg)


class GitNotification(NotificationBase):
    kind = "git"

    def push(self, message, severity, runs):
        logger.info("git notification", message=message, runs=len(runs))


_kinds = {
    cls.kind: cls
    for cls in [ConsoleNotification, IPythonNotification,
                WebhookNotification, SlackNotification, MailNotification,
                GitNotification]
}


def get_notification_class(kind: str):
    from ..errors import MLRunInvalidArgumentError

    if kind not in _kinds:
        raise MLRunInvalidArgumentError(f"unsupported notification kind {kind}")
    return _kinds[kind]


class NotificationPusher:
    """Push a run's configured notifications for its terminal state."""

    def __init__(self, runs: typing.List[RunObject]):
        self._runs = runs

    def push(self, db=None):
        for run in self._runs:
            state = run.status.state
            for spec in run.spec.notifications or []:
                if hasattr(spec, "to_dict"):
                    spec = spec.to_dict()
                elif not isinstance(spec, dict):
                    spec = {"kind": str(spec)}
                when = spec.get("when") or [RunStates.completed,
                                            RunStates.error]
                if state not in when:
                    continue
                condition = spec.get("condition")
                if condition:
                    from .safe_eval import safe_eval

                    try:
                        ok = bool(safe_eval(condition,
                                            {"run": run.to_dict()}))
                    except Exception:
                        ok = True
                    if not ok:
                        continue
                kind = spec.get("kind", "console")
                try:
                    notification = get_notification_class(kind)(
                        spec.get("name", ""), spec.get("params") or
                        spec.get("secret_params") or {})
                    notification.push(
                        spec.get("message") or
                        f"run {run.metadata.name} {state}",
                        spec.get("severity", "info"), [run.to_dict()])
                    status = "sent"
                except Exception as exc:
                    logger.warning("notification failed", kind=kind,
                                   error=str(exc))
                    status = "error"
                run.status.notifications[spec.get("name") or kind] = {
                    "status": status, "sent_time": now_iso()}
