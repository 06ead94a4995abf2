# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Run-DB factory.  ``get_run_db()`` dispatches on config.dbpath:

- ``""`` / ``"local"`` / filesystem path -> node-local SQLite DB
- ``http(s)://...``                      -> HTTP client to the API service
- ``"nop"``                              -> NopDB (offline)

Parity: reference mlrun/db/__init__.py get_run_db + run-db singleton.
"""

import threading

from ..config import config
from .base import RunDBInterface
from .httpdb import HTTPRunDB
from .nopdb import NopDB
from .sqldb import SQLRunDB

_lock = threading.Lock()
_run_db = None
_run_db_url = None
_run_db_pinned = False


def create_run_db(url: str = "", secrets=None) -> RunDBInterface:
    url = url if url is not None else ""
    if url == "nop":
        return NopDB()
    if url.startswith("http://") or url.startswith("https://"):
        return HTTPRunDB(url).connect(secrets)
    return SQLRunDB(url if url not in ("", "local") else "")


def get_run_db(url: str = None, secrets=None, force_reconnect=False) -> RunDBInterface:
    """Return the process-wide run DB (created on first use)."""
    global _run_db, _run_db_url

    global _run_db_pinned
    explicit_url = url is not None
    url = url if url is not None else str(config.dbpath or "")
    with _lock:
        if _run_db_pinned and not explicit_url and not force_reconnect:
            return _run_db
        if _run_db is None or force_reconnect or url != _run_db_url:
            _run_db = create_run_db(url, secrets)
            _run_db_url = url
            _run_db_pinned = False
        return _run_db


def set_run_db(db: RunDBInterface):
    """Install (pin) a DB instance; get_run_db() returns it until a
    reconnect or an explicit url is requested (test/mocking hook)."""
    global _run_db, _run_db_url, _run_db_pinned

    with _lock:
        _run_db = db
        _run_db_url = getattr(db, "dsn", getattr(db, "base_url", "injected"))
        _run_db_pinned = True


class RunDBError(Exception):
    """Generic run-DB failure (reference db/base.py:29)."""


def get_or_set_dburl(default: str = "") -> str:
    """Return config.dbpath, setting it (and MLRUN_DBPATH) to default
    first when unset (reference db/__init__.py:20)."""
    import os

    if not config.dbpath and default:
        config.dbpath = default
        os.environ["MLRUN_DBPATH"] = default
    return config.dbpath
