# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Utilities: structured logger, time/uid helpers, dict helpers.

Parity target: reference mlrun/utils/logger.py:157 (Logger with
human/JSON formatters) and mlrun/utils/helpers.py, re-written fresh.
"""

import json
import logging
import re
import sys
import typing
from datetime import datetime, timezone

from ..config import config


class _HumanFormatter(logging.Formatter):
    def format(self, record):
        more = getattr(record, "with_fields", None)
        more_str = f" {more}" if more else ""
        now = datetime.fromtimestamp(record.created, tz=timezone.utc)
        return (f"> {now.isoformat(timespec='milliseconds')} "
                f"[{record.levelname.lower()}] {record.getMessage()}{more_str}")


class _JSONFormatter(logging.Formatter):
    def format(self, record):
        rec = {
            "datetime": datetime.fromtimestamp(
                record.created, tz=timezone.utc).isoformat(),
            "level": record.levelname.lower(),
            "message": record.getMessage(),
            "with": getattr(record, "with_fields", {}) or {},
        }
        return json.dumps(rec, default=str)


class Logger:
    """Structured logger: logger.info("msg", key=value, ...)."""

    def __init__(self, level=None, name="mlrun_amd", stream=None):
        self._logger = logging.getLogger(name)
        self._logger.propagate = False
        self._handler = logging.StreamHandler(stream or sys.stdout)
        fmt = (_JSONFormatter() if str(config.log_format).lower() == "json"
               else _HumanFormatter())
        self._handler.setFormatter(fmt)
        if not self._logger.handlers:
            self._logger.addHandler(self._handler)
        self.set_level(level or config.log_level or "INFO")

    def set_level(self, level):
        if isinstance(level, str):
            level = getattr(logging, level.upper(), logging.INFO)
        self._logger.setLevel(level)

    @property
    def level(self):
        return self._logger.level

    def _log(self, level, message, **kwargs):
        self._logger.log(level, message, extra={"with_fields": kwargs})

    def debug(self, message, **kwargs):
        self._log(logging.DEBUG, message, **kwargs)

    def info(self, message, **kwargs):
        self._log(logging.INFO, message, **kwargs)

    def warning(self, message, **kwargs):
        self._log(logging.WARNING, message, **kwargs)

    warn = warning

    def error(self, message, **kwargs):
        self._log(logging.ERROR, message, **kwargs)

    def exception(self, message, **kwargs):
        self._logger.exception(message, extra={"with_fields": kwargs})


logger = Logger()


def now_date() -> datetime:
    return datetime.now(timezone.utc)


def now_iso() -> str:
    return now_date().isoformat()


def to_date_str(dt) -> typing.Optional[str]:
    if dt is None:
        return None
    if isinstance(dt, str):
        return dt
    return dt.isoformat()


def parse_time(value) -> typing.Optional[datetime]:
    if value is None or isinstance(value, datetime):
        return value
    try:
        return datetime.fromisoformat(value)
    except (ValueError, TypeError):
        return None


_name_re = re.compile(r"^[a-zA-Z0-9][a-zA-Z0-9\-_.]*$")


def verify_field_regex(name: str, value: str, allow_empty=False):
    from ..errors import MLRunInvalidArgumentError

    if not value:
        if allow_empty:
            return
        raise MLRunInvalidArgumentError(f"{name} must not be empty")
    if not _name_re.match(value):
        raise MLRunInvalidArgumentError(
            f"{name}={value!r} is invalid (must match {_name_re.pattern})")


def normalize_name(name: str) -> str:
    """Normalize a function/run name: lowercase, underscores -> dashes."""
    return re.sub(r"[^a-z0-9\-.]", "-", str(name).lower().replace("_", "-"))


def dict_to_list(struct: dict) -> list:
    return [f"{k}={v}" for k, v in (struct or {}).items()]


def list_to_dict(lines: typing.Iterable) -> dict:
    out = {}
    for line in lines or []:
        if "=" in line:
            key, value = line.split("=", 1)
            out[key.strip()] = value.strip()
    return out


def update_in(obj: dict, key: typing.Union[str, list], value):
    """Set a nested key ("a.b.c" or list path) in a dict tree."""
    parts = key.split(".") if isinstance(key, str) else key
    for part in parts[:-1]:
        obj = obj.setdefault(part, {})
    obj[parts[-1]] = value


def get_in(obj: dict, key: typing.Union[str, list], default=None):
    parts = key.split(".") if isinstance(key, str) else key
    for part in parts:
        if not isinstance(obj, dict) or part not in obj:
            return default
        obj = obj[part]
    return obj


def fill_artifact_path_template(template: str, project: str = None) -> str:
    if not template:
        return template
    return template.replace("{{project}}", project or config.default_project)


def is_relative_path(path: str) -> bool:
    if not path:
        return False
    return not ("://" in path or path.startswith("/"))
