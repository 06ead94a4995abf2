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
import os
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


# --------------------------------------------------------------------
# reference utils/helpers.py surface: the commonly-used public helpers
# (serialization, dynamic loading, uri building, retry)


def dict_to_yaml(struct: dict) -> str:
    import yaml as _yaml

    return _yaml.safe_dump(struct, default_flow_style=False,
                           sort_keys=False)


def dict_to_json(struct: dict) -> str:
    import json as _json

    return _json.dumps(struct, default=str)


def dict_to_str(struct: dict, separator: str = ",") -> str:
    if not struct:
        return ""
    return separator.join(f"{k}={v}" for k, v in struct.items())


def as_list(element) -> list:
    return element if isinstance(element, list) else [element]


def as_number(field_name, field_value):
    if isinstance(field_value, str) and not field_value.isnumeric():
        raise ValueError(f"{field_name} must be numeric (str/int types)")
    return int(field_value)


def flatten(df, col: str, prefix: str = ""):
    """Expand a dict column of a DataFrame into prefixed columns
    (reference helpers.py flatten)."""
    import pandas as pd

    params = []
    for row in df[col]:
        if row:
            for key in row.keys():
                if key not in params:
                    params.append(key)
    for param in params:
        df[prefix + param] = df[col].apply(
            lambda x: x.get(param, "") if x else "")
    return df.drop(columns=[col])


def list2dict(lines: list) -> dict:
    """['k=v', ...] -> {k: v} (reference helpers.py list2dict)."""
    out = {}
    for line in lines:
        i = line.find("=")
        if i == -1:
            continue
        key, value = line[:i].strip(), line[i + 1:].strip()
        if key is None:
            raise ValueError("cannot find key in line (key=value)")
        value = os.path.expandvars(value)
        out[key] = value
    return out


def datetime_to_iso(time_obj) -> typing.Optional[str]:
    return time_obj.isoformat() if time_obj else None


def datetime_from_iso(time_str: str):
    if not time_str:
        return None
    from datetime import datetime

    return datetime.fromisoformat(str(time_str).replace("Z", "+00:00"))


def str_to_timestamp(time_str, now_time=None):
    """Parse 'now', 'now + 2h', iso strings or pandas-parsable dates to
    a Timestamp (reference helpers.py str_to_timestamp)."""
    import pandas as pd

    if not isinstance(time_str, str):
        return time_str
    trimmed = time_str.strip().lower()
    if trimmed.startswith("now"):
        now_time = now_time or pd.Timestamp.now()
        rest = trimmed[len("now"):].strip()
        if not rest:
            return now_time
        sign = 1
        if rest[0] == "+":
            rest = rest[1:]
        elif rest[0] == "-":
            sign = -1
            rest = rest[1:]
        return now_time + sign * pd.Timedelta(rest.strip())
    return pd.Timestamp(time_str)


def get_class(class_name, namespace=None):
    """Resolve a class by name from a namespace or dotted path
    (reference helpers.py get_class)."""
    if isinstance(class_name, type):
        return class_name
    if namespace and class_name in namespace:
        return namespace[class_name]
    if "." in str(class_name):
        import importlib

        module_name, _, cls = str(class_name).rpartition(".")
        module = importlib.import_module(module_name)
        return getattr(module, cls)
    raise ValueError(f"class {class_name} not found")


create_class = get_class


def get_function(function, namespace=None):
    """Resolve a function by name (reference helpers.py
    get_function)."""
    if callable(function):
        return function
    if namespace and function in namespace:
        return namespace[function]
    if "." in str(function):
        return get_class(function)
    raise ValueError(f"function {function} not found in namespace")


create_function = get_function


def fill_object_hash(object_dict: dict, uid_property_name: str = "hash",
                     tag: str = "") -> str:
    """Deterministic content hash of an object dict, ignoring volatile
    fields (reference helpers.py fill_object_hash)."""
    import hashlib
    import json as _json

    import copy as _copy

    obj = _copy.deepcopy(object_dict)
    metadata = obj.setdefault("metadata", {})
    tag = tag or metadata.get("tag")
    status = obj.pop("status", None)  # noqa: F841  volatile
    metadata.pop("updated", None)
    metadata.pop("tag", None)
    metadata.pop(uid_property_name, None)
    data = _json.dumps(obj, sort_keys=True, default=str).encode()
    digest = hashlib.sha1(data).hexdigest()
    metadata[uid_property_name] = digest
    if tag:
        metadata["tag"] = tag
    if status is not None:
        object_dict["status"] = status
    object_dict["metadata"] = metadata
    return digest


def fill_function_hash(function_dict: dict, tag: str = "") -> str:
    return fill_object_hash(function_dict, "hash", tag)


def generate_object_uri(project: str, name: str, tag: str = None,
                        hash_key: str = None) -> str:
    uri = f"{project}/{name}"
    if tag:
        uri += f":{tag}"
    elif hash_key:
        uri += f"@{hash_key}"
    return uri


def generate_artifact_uri(project: str, key: str, tag: str = None,
                          iter: int = None, tree: str = None) -> str:
    uri = f"{project}/{key}"
    if iter is not None:
        uri = f"{uri}#{iter}"
    if tag is not None:
        uri = f"{uri}:{tag}"
    if tree is not None:
        uri = f"{uri}@{tree}"
    return uri


def parse_artifact_uri(uri: str, default_project: str = ""):
    """'<project>/<key>[#iter][:tag][@tree]' -> (project, key, iter,
    tag, tree) — reference helpers.py parse_artifact_uri."""
    import re as _re

    pattern = (r"^((?P<project>.*)/)?(?P<key>.*?)"
               r"(\#(?P<iteration>.*?))?(:(?P<tag>.*?))?"
               r"(@(?P<tree>.*))?$")
    match = _re.match(pattern, uri)
    if not match:
        raise ValueError(f"cannot parse artifact uri {uri}")
    group_dict = match.groupdict()
    iteration = group_dict["iteration"]
    if iteration is not None:
        try:
            iteration = int(iteration)
        except ValueError:
            raise ValueError(
                f"illegal store path {uri}, iteration must be integer")
    return (group_dict["project"] or default_project,
            group_dict["key"], iteration or 0, group_dict["tag"],
            group_dict["tree"])


class StorePrefix:
    """store:// uri kind prefixes (reference helpers.py
    StorePrefix)."""

    Artifact = "artifacts"
    Model = "models"
    Dataset = "datasets"
    FeatureSet = "feature-sets"
    FeatureVector = "feature-vectors"

    @staticmethod
    def is_artifact(prefix) -> bool:
        return prefix in [StorePrefix.Artifact, StorePrefix.Model,
                          StorePrefix.Dataset]

    @staticmethod
    def kind_to_prefix(kind: str) -> str:
        kind_map = {"model": StorePrefix.Model,
                    "dataset": StorePrefix.Dataset}
        return kind_map.get(kind, StorePrefix.Artifact)


def is_yaml_path(url: str) -> bool:
    return str(url).endswith(".yaml") or str(url).endswith(".yml")


def is_safe_path(base: str, filepath: str,
                 is_symlink: bool = False) -> bool:
    """True when filepath resolves inside base (reference helpers.py
    is_safe_path — path-traversal guard)."""
    resolve = os.path.realpath if is_symlink else os.path.abspath
    return os.path.commonpath(
        [os.path.abspath(base)]) == os.path.commonpath(
        [os.path.abspath(base), resolve(filepath)])


def template_artifact_path(artifact_path: str, project: str,
                           run_uid: str = "") -> str:
    """Expand {{project}} / {{run.uid}} templates in an artifact path
    (reference helpers.py template_artifact_path)."""
    if not artifact_path:
        return artifact_path
    return artifact_path.replace("{{project}}", project or "default") \
        .replace("{{run.project}}", project or "default") \
        .replace("{{run.uid}}", run_uid or "")


def retry_until_successful(backoff, timeout: float, logger_, verbose,
                           function, *args, **kwargs):
    """Call function until it succeeds or timeout passes (reference
    helpers.py retry_until_successful); backoff is seconds between
    tries (or an iterable of delays)."""
    import time as _time

    start = _time.monotonic()
    last_exception = None
    delays = backoff if hasattr(backoff, "__iter__") else None
    while _time.monotonic() - start < timeout:
        try:
            return function(*args, **kwargs)
        except Exception as exc:  # noqa: BLE001
            last_exception = exc
            if verbose and logger_:
                logger_.debug(f"retrying {getattr(function, '__name__', '?')}",
                              error=str(exc))
            delay = next(iter(delays), 1) if delays else backoff
            _time.sleep(delay)
    raise Exception(
        f"failed to execute command by the given deadline: "
        f"last_exception: {last_exception}")


def iterate_list_by_chunks(iterable, chunk_size: int):
    """Yield successive chunk_size-lists from iterable (reference
    helpers.py iterate_list_by_chunks)."""
    if chunk_size <= 0 or not iterable:
        yield list(iterable or [])
        return
    chunk = []
    for item in iterable:
        chunk.append(item)
        if len(chunk) == chunk_size:
            yield chunk
            chunk = []
    if chunk:
        yield chunk


def to_non_empty_values_dict(struct: dict) -> dict:
    return {k: v for k, v in (struct or {}).items() if v}


def merge_dicts_with_precedence(*dicts) -> dict:
    """Merge dicts; later dicts win (reference helpers.py
    merge_dicts_with_precedence)."""
    out = {}
    for struct in dicts:
        if struct:
            out.update(struct)
    return out
