# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Shared test fixtures.

Mirrors the reference's test seams (tests/common_fixtures.py):
- config_test_base resets config + installs a fresh per-test SQLite DB
  under tmp_path (the analog of RunDBMock — here the real SQLRunDB on a
  throwaway file, which exercises the same code path the service uses)
- ``gpu`` marker for tests needing a real MI355X
"""

import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run on gpurun box)")


@pytest.fixture(autouse=True)
def config_test_base(tmp_path, monkeypatch):
    """Fresh config + DB + artifact path per test."""
    import mlrun_amd
    from mlrun_amd.config import config
    import mlrun_amd.db as dbmod

    monkeypatch.setenv("MLRUN_BASE_DIR", str(tmp_path))
    config.reload()
    config.base_dir = str(tmp_path)
    config.artifact_path = str(tmp_path / "artifacts")
    config.dbpath = ""
    # reset the db singleton
    dbmod._run_db = None
    dbmod._run_db_url = None
    # reset store manager cache
    from mlrun_amd.datastore import store_manager

    store_manager._stores.clear()
    store_manager._db = None
    # fresh monitoring stream processors (the "default" processor is a
    # process-global: ring stats would leak between tests otherwise)
    from mlrun_amd.model_monitoring import stream as _stream_mod

    _stream_mod._processors.clear()
    yield config
    dbmod._run_db = None
    dbmod._run_db_url = None


@pytest.fixture
def rundb(config_test_base):
    from mlrun_amd.db import get_run_db

    return get_run_db()
