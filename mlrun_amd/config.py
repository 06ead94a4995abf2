# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Configuration system.

A nested default-config dict exposed as an attribute-access ``Config``
object, with environment-variable overrides under the ``MLRUN_`` prefix
(``MLRUN_A__B=x`` sets ``config.a.b = "x"``), mirroring the behavior of
the reference config system (reference: mlrun/config.py:52 default dict,
:763 Config, :1379 read_env) re-implemented for the node-local MI355X
deployment model (no k8s / nuclio sections; GPU-engine sections instead).
"""

import base64
import binascii
import copy
import json
import os
import threading
import typing

ENV_PREFIX = "MLRUN_"

default_config = {
    "version": "0.1.0",
    # where the run-DB lives: "" -> local SQLite db under base_dir,
    # "http://host:port" -> remote API service, "nop" -> no-op DB
    "dbpath": "",
    "base_dir": os.path.expanduser("~/.mlrun_amd"),
    "default_project": "default",
    # url prefix for get_sample_path (reference config.py:181)
    "default_samples_path": "https://s3.wasabisys.com/iguazio/",
    "artifact_path": "",  # default artifact path template
    "log_level": "INFO",
    "log_format": "human",  # human | json
    "namespace": "",
    "ui_url": "",
    "igz_version": "",
    "kfp_url": "",
    "httpdb": {
        "port": 8080,
        "dirpath": "",  # sqlite + logs location for the API service
        "dsn": "",  # sqlalchemy DSN override
        "logs_path": "",
        "max_log_size_bytes": 1024 * 1024 * 8,
        "retry_api_call_on_exception": "enabled",
        "http_connection_timeout": 20,
        # comma-separated path/scheme prefixes the /files and /filestat
        # endpoints may serve; empty -> derived from base_dir +
        # artifact_path (reference routes files.py through per-path
        # authorization — this is the node-local analog)
        "files_allowed_paths": "",
    },
    "runs": {
        "monitoring_interval": 30,  # seconds between run-monitor sweeps
        "state_thresholds": {
            # seconds a run may stay in a state before being aborted
            "pending": 3600,
            "running": 24 * 3600,
        },
    },
    "scheduler": {
        "min_allowed_interval_seconds": 10,
        "tick_seconds": 1.0,
    },
    "function_defaults": {
        "image": "",
        "kind": "job",
    },
    "gpu": {
        # MI355X node model: number of local GPUs to schedule over
        "devices_per_node": 8,
        "arch": "gfx950",
        "hbm_gb": 288,
        "require_native_ops": "auto",  # auto|true|false: fail if HIP ext missing
    },
    "distributed": {
        # RCCL-over-xGMI defaults (analog of the reference's NCCL env
        # defaults at mpijob/abstract.py:88-95, retuned for xGMI rings)
        "backend": "nccl",  # "nccl" is RCCL on ROCm
        "master_addr": "127.0.0.1",
        "master_port": 29400,
        "rccl_env": {
            # 7 p2p xGMI links per GPU -> many concurrent channels
            "NCCL_MIN_NCHANNELS": "28",
            "NCCL_PROTO": "Simple",
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        },
        "bucket_cap_mb": 64,  # gradient bucket size for per-link-bound rings
    },
    "serving": {
        "default_batch_size": 16,
        "max_queue": 1024,
        "graph_capture": "auto",  # hipGraph capture of step chains: auto|off
        "stream_args": {},
    },
    "feature_store": {
        "default_targets": "parquet,nosql",
        "data_prefix": "",  # defaults to {base_dir}/feature-store
        "flush_interval_secs": 300,
    },
    "model_endpoint_monitoring": {
        "sample_percent": 100,
        "parquet_batching_max_events": 1024,
        # window-ring placement: cpu | auto (GPU when available) |
        # explicit device — the rings reuse the feature-store HIP
        # window kernels
        "device": "cpu",
    },
    "model_monitoring": {
        # drift classification thresholds (reference
        # histogram_data_drift defaults)
        "drift_thresholds": {"detected": 0.7, "possible": 0.5},
    },
    "notifications": {
        "smtp": {"server": "", "sender": ""},
        # default pushers applied when a run declares none
        "default_kinds": "console",
    },
    "secret_stores": {
        "env_file": "",
    },
    "packagers": {"enabled": True, "pack_results": True},
    "background_tasks": {
        "default_timeout": 600,
        # terminal tasks older than this are swept (reference
        # background-task cleanup loop)
        "ttl_seconds": 6 * 3600,
    },
    "artifacts": {
        # mirror of the reference artifacts block (config.py
        # "artifacts"): hashing + target-path generation behavior
        "calculate_hash": True,
        "generate_target_path_from_artifact_hash": False,
        "artifact_max_size_mb": 1024,
        # artifact tags accepted by the tag endpoints
        "allowed_tag_chars": "a-zA-Z0-9-_.",
    },
    "pagination": {
        "default_page_size": 20,
        # pagination_cache rows idle longer than this are cleaned
        "cache_ttl_seconds": 3600,
    },
    "alerts": {
        "max_per_project": 1000,
        # reset criteria count after this many seconds without events
        "event_window_seconds": 600,
    },
    "projects": {
        # summary recompute cadence for the /project-summaries loop
        "summaries_interval": 30,
        "default_owner": "",
    },
    "workflows": {
        "default_timeout": 3600,
        "engine": "local",  # the node-local runner (KFP out of scope)
    },
    "logs": {
        "decode": {"errors": "replace"},
        "pull_state_interval": 3,   # run-log poll cadence (watch_log)
        "pipelines_redirect": True,
    },
    "function": {
        # per-run spec guards (reference function block: limits the
        # server enforces at submit time)
        "spec": {
            "max_parameters": 1024,
            "max_notifications": 16,
        },
    },
    "runtimes": {
        # node-local runtime behavior knobs
        "job": {"venv_cache": True},
        "mpijob": {
            "gang_abort_grace_seconds": 5,
            "heartbeat_interval": 5,
        },
        "serving": {
            "worker_ready_timeout": 180,
            "autoscale_connections_per_worker": 16,
        },
    },
    "hub": {
        "default_source": "default",
    },
    "tracing": {
        # rocTX range emission for serving steps (utils/tracing.py)
        "enabled": "auto",
    },
}


class Config:
    """Attribute-access wrapper over a nested config dict."""

    _initialized = False

    def __init__(self, cfg: typing.Optional[dict] = None):
        self._cfg = cfg if cfg is not None else {}

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        try:
            val = self._cfg[name]
        except KeyError:
            raise AttributeError(f"config has no attribute {name!r}") from None
        if isinstance(val, dict):
            return Config(val)
        return val

    def __setattr__(self, name, value):
        if name.startswith("_"):
            object.__setattr__(self, name, value)
        else:
            self._cfg[name] = value

    def __getitem__(self, name):
        return self.__getattr__(name)

    def get(self, name, default=None):
        val = self._cfg.get(name, default)
        if isinstance(val, dict):
            return Config(val)
        return val

    def to_dict(self) -> dict:
        return copy.deepcopy(self._cfg)

    def update(self, overrides: dict):
        _merge(self._cfg, overrides)

    def dump_yaml(self) -> str:
        import yaml

        return yaml.safe_dump(self._cfg, default_flow_style=False)

    def reload(self):
        """Re-read defaults + environment into this config object."""
        new = copy.deepcopy(default_config)
        _merge(new, read_env(os.environ))
        self._cfg.clear()
        self._cfg.update(new)


def _merge(base: dict, overrides: dict):
    for key, value in overrides.items():
        if isinstance(value, dict) and isinstance(base.get(key), dict):
            _merge(base[key], value)
        else:
            base[key] = value


def _convert(value: str):
    """Interpret env-var strings: json literals, base64 ("b64:"), plain."""
    if value.startswith("b64:"):
        try:
            return base64.b64decode(value[4:]).decode()
        except (binascii.Error, UnicodeDecodeError):
            return value
    try:
        return json.loads(value)
    except (ValueError, TypeError):
        return value


def read_env(env: typing.Mapping, prefix: str = ENV_PREFIX) -> dict:
    """Collect MLRUN_* env vars into a nested override dict.

    ``MLRUN_HTTPDB__PORT=9090`` -> ``{"httpdb": {"port": 9090}}``.
    """
    out: dict = {}
    for key, value in env.items():
        if not key.startswith(prefix):
            continue
        path = key[len(prefix):].lower().split("__")
        node = out
        for part in path[:-1]:
            node = node.setdefault(part, {})
        node[path[-1]] = _convert(value)
    # well-known aliases
    if "MLRUN_DBPATH" in env:
        out["dbpath"] = env["MLRUN_DBPATH"]
    return out


_load_lock = threading.Lock()
config = Config(copy.deepcopy(default_config))


def _populate():
    with _load_lock:
        config.update(read_env(os.environ))
        config._initialized = True


_populate()
