# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Cron scheduler for stored run schedules.

Parity target: reference server/api/utils/scheduler.py:48 (APScheduler-
backed, DB-stored schedules, min-interval guard :634, reload-on-start
:767, misfire catch-up).  Self-contained cron parser + ticking thread —
no external scheduler dependency.
"""

import datetime
import threading
import typing

from ..config import config
from ..errors import MLRunInvalidArgumentError
from ..utils import logger


def _parse_field(field: str, low: int, high: int) -> typing.Set[int]:
    values: typing.Set[int] = set()
    for part in field.split(","):
        part = part.strip()
        step = 1
        if "/" in part:
            part, step_s = part.split("/", 1)
            step = int(step_s)
        if part in ("*", ""):
            start, end = low, high
        elif "-" in part:
            start_s, end_s = part.split("-", 1)
            start, end = int(start_s), int(end_s)
        else:
            start = end = int(part)
        if start < low or end > high or start > end:
            raise MLRunInvalidArgumentError(
                f"cron field value {part!r} out of range [{low},{high}]")
        values.update(range(start, end + 1, step))
    return values


def _parse_weekday_field(field: str) -> typing.Set[int]:
    """Standard-cron day-of-week (Sun=0 or 7, Mon=1 .. Sat=6) mapped to
    Python ``dt.weekday()`` numbering (Mon=0 .. Sun=6)."""
    cron_values = _parse_field(field, 0, 7)
    return {(v - 1) % 7 for v in cron_values}


class CronTrigger:
    """5-field cron: minute hour day-of-month month day-of-week."""

    def __init__(self, expression: str):
        fields = expression.split()
        if len(fields) != 5:
            raise MLRunInvalidArgumentError(
                f"invalid cron expression {expression!r} (need 5 fields)")
        self.expression = expression
        self.minutes = _parse_field(fields[0], 0, 59)
        self.hours = _parse_field(fields[1], 0, 23)
        self.days = _parse_field(fields[2], 1, 31)
        self.months = _parse_field(fields[3], 1, 12)
        self.weekdays = _parse_weekday_field(fields[4])

    def matches(self, dt: datetime.datetime) -> bool:
        return (dt.minute in self.minutes and dt.hour in self.hours and
                dt.day in self.days and dt.month in self.months and
                dt.weekday() in self.weekdays)

    def next_fire_time(self, after: datetime.datetime) -> datetime.datetime:
        """Next matching minute within 366 days."""
        candidate = after.replace(second=0, microsecond=0) + \
            datetime.timedelta(minutes=1)
        for _ in range(366 * 24 * 60):
            if self.matches(candidate):
                return candidate
            candidate += datetime.timedelta(minutes=1)
        raise MLRunInvalidArgumentError(
            f"cron {self.expression!r} never fires")

    def min_interval_seconds(self) -> float:
        """Approximate smallest gap between fires (for the guard)."""
        if len(self.minutes) > 1:
            sorted_m = sorted(self.minutes)
            gaps = [b - a for a, b in zip(sorted_m, sorted_m[1:])]
            return min(gaps) * 60
        return 3600.0


class Scheduler:
    """Ticking scheduler over DB-stored schedules (leader-only in the
    reference; node-local here — one service instance per node)."""

    def __init__(self, db, tick_seconds: float = None):
        self._db = db
        self._tick = tick_seconds or float(config.scheduler.tick_seconds)
        self._thread: typing.Optional[threading.Thread] = None
        self._stop = threading.Event()
        self._triggers: typing.Dict[tuple, CronTrigger] = {}
        self._last_fired: typing.Dict[tuple, datetime.datetime] = {}
        self.reload()

    def reload(self):
        """Re-read schedules from the DB (reference: reload on start)."""
        triggers = {}
        try:
            for project in [p.get("metadata", {}).get("name", "default")
                            for p in self._db.list_projects()] + ["default"]:
                for sched in self._db.list_schedules(project):
                    cron = sched.get("cron_trigger")
                    name = sched.get("name")
                    if not cron or not name:
                        continue
                    try:
                        trigger = CronTrigger(cron)
                    except MLRunInvalidArgumentError:
                        continue
                    min_interval = float(
                        config.scheduler.min_allowed_interval_seconds)
                    if trigger.min_interval_seconds() < min_interval:
                        logger.warning("schedule below min interval, "
                                       "skipping", name=name)
                        continue
                    triggers[(project, name)] = trigger
        except Exception as exc:
            logger.warning("scheduler reload failed", error=str(exc))
        self._triggers = triggers

    def start(self):
        if self._thread is not None:
            return
        self._catch_up_misfires()
        self._stop.clear()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="scheduler")
        self._thread.start()

    def _catch_up_misfires(self, grace_seconds: int = 3600):
        """Invoke schedules whose stored next_run_time passed while
        the service was down (APScheduler misfire analog)."""
        now = datetime.datetime.now()
        for (project, name) in list(self._triggers.keys()):
            try:
                sched = self._db.get_schedule(project, name)
            except Exception:
                continue
            next_run = sched.get("next_run_time")
            if not next_run:
                continue
            try:
                next_dt = datetime.datetime.fromisoformat(next_run)
            except ValueError:
                continue
            if next_dt < now and \
                    (now - next_dt).total_seconds() <= grace_seconds:
                logger.info("catching up missed schedule", schedule=name)
                try:
                    self.invoke(project, name)
                except Exception as exc:
                    logger.warning("misfire catch-up failed",
                                   schedule=name, error=str(exc))

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    def _loop(self):
        while not self._stop.wait(self._tick):
            now = datetime.datetime.now()
            minute = now.replace(second=0, microsecond=0)
            for key, trigger in list(self._triggers.items()):
                if not trigger.matches(minute):
                    continue
                if self._last_fired.get(key) == minute:
                    continue
                self._last_fired[key] = minute
                project, name = key
                try:
                    self.invoke(project, name)
                except Exception as exc:
                    logger.error("scheduled invocation failed",
                                 schedule=name, error=str(exc))

    def invoke(self, project: str, name: str):
        """Run a schedule's task now (reference invoke_schedule :428)."""
        sched = self._db.get_schedule(project, name)
        task = sched.get("task")
        if not task:
            raise MLRunInvalidArgumentError(
                f"schedule {name} has no task body")
        from ..model import RunObject, generate_uid

        run = RunObject.from_dict(task)
        run.metadata.uid = generate_uid()
        result = self._db.submit_job(run)
        uid = (result.get("data", {}).get("metadata", {}) or {}).get("uid",
                                                                     "")
        self._db.update_schedule(project, name, {
            "last_run_uri": f"{project}/{uid}",
            "next_run_time": None,
        })
        try:
            trigger = self._triggers.get((project, name))
            if trigger:
                nxt = trigger.next_fire_time(datetime.datetime.now())
                self._db.update_schedule(project, name,
                                         {"next_run_time": nxt.isoformat()})
        except Exception as exc:
            logger.debug("next_run_time update failed", error=str(exc))
        logger.info("schedule invoked", schedule=name, run=uid)
        return result
