# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""SQLite-backed run DB.

Design: unlike the reference (which splits a 6k-LoC SQLAlchemy DAL in
the server from an HTTP client in the SDK — server/api/db/sqldb/db.py),
the MI355X-native framework is node-local, so the same SQLite DB class
serves BOTH as the in-process "local" DB for client work AND as the
storage layer of the FastAPI service (mlrun_amd/api).  Tables mirror
the reference's entity set (runs, artifacts, functions, projects, logs,
schedules, feature sets/vectors, model endpoints, background tasks,
alerts — reference server/api/db/sqldb/models.py:195-760) with JSON
document columns instead of wide ORM models.

Thread-safe: one connection per thread (WAL mode), plus a process-level
write lock for multi-thread writers.
"""

import json
import os
import sqlite3
import threading

from ..errors import (MLRunConflictError, MLRunInvalidArgumentError,
                      MLRunNotFoundError)
from ..model import RunStates
from ..utils import now_iso
from .base import RunDBInterface
from .extras import RunDBExtras

_SCHEMA = """
CREATE TABLE IF NOT EXISTS runs (
    project TEXT NOT NULL, uid TEXT NOT NULL, iteration INTEGER NOT NULL,
    name TEXT, state TEXT, start_time TEXT, updated TEXT, body TEXT,
    PRIMARY KEY (project, uid, iteration)
);
CREATE TABLE IF NOT EXISTS artifacts (
    project TEXT NOT NULL, key TEXT NOT NULL, tree TEXT NOT NULL DEFAULT '',
    iteration INTEGER NOT NULL DEFAULT 0, tag TEXT DEFAULT '',
    kind TEXT, uid TEXT, updated TEXT, body TEXT,
    PRIMARY KEY (project, key, tree, iteration)
);
CREATE TABLE IF NOT EXISTS artifact_tags (
    project TEXT NOT NULL, key TEXT NOT NULL, tag TEXT NOT NULL,
    tree TEXT, iteration INTEGER DEFAULT 0,
    PRIMARY KEY (project, key, tag)
);
CREATE TABLE IF NOT EXISTS functions (
    project TEXT NOT NULL, name TEXT NOT NULL, tag TEXT NOT NULL DEFAULT 'latest',
    hash_key TEXT, updated TEXT, body TEXT,
    PRIMARY KEY (project, name, tag)
);
CREATE TABLE IF NOT EXISTS function_versions (
    project TEXT NOT NULL, name TEXT NOT NULL, hash_key TEXT NOT NULL,
    updated TEXT, body TEXT,
    PRIMARY KEY (project, name, hash_key)
);
CREATE TABLE IF NOT EXISTS projects (
    name TEXT PRIMARY KEY, state TEXT, created TEXT, body TEXT
);
CREATE TABLE IF NOT EXISTS logs (
    project TEXT NOT NULL, uid TEXT NOT NULL, body BLOB,
    PRIMARY KEY (project, uid)
);
CREATE TABLE IF NOT EXISTS schedules (
    project TEXT NOT NULL, name TEXT NOT NULL, kind TEXT, cron TEXT,
    creation_time TEXT, next_run_time TEXT, last_run_uri TEXT, body TEXT,
    PRIMARY KEY (project, name)
);
CREATE TABLE IF NOT EXISTS feature_sets (
    project TEXT NOT NULL, name TEXT NOT NULL, tag TEXT NOT NULL DEFAULT 'latest',
    updated TEXT, body TEXT,
    PRIMARY KEY (project, name, tag)
);
CREATE TABLE IF NOT EXISTS feature_vectors (
    project TEXT NOT NULL, name TEXT NOT NULL, tag TEXT NOT NULL DEFAULT 'latest',
    updated TEXT, body TEXT,
    PRIMARY KEY (project, name, tag)
);
CREATE TABLE IF NOT EXISTS model_endpoints (
    project TEXT NOT NULL, endpoint_id TEXT NOT NULL, model TEXT,
    function TEXT, updated TEXT, body TEXT,
    PRIMARY KEY (project, endpoint_id)
);
CREATE TABLE IF NOT EXISTS background_tasks (
    project TEXT NOT NULL, name TEXT NOT NULL, state TEXT, updated TEXT,
    body TEXT,
    PRIMARY KEY (project, name)
);
CREATE TABLE IF NOT EXISTS alert_configs (
    project TEXT NOT NULL, name TEXT NOT NULL, updated TEXT, body TEXT,
    PRIMARY KEY (project, name)
);
CREATE TABLE IF NOT EXISTS alert_states (
    project TEXT NOT NULL, name TEXT NOT NULL, count INTEGER DEFAULT 0,
    last_updated TEXT, active INTEGER DEFAULT 0,
    PRIMARY KEY (project, name)
);
CREATE TABLE IF NOT EXISTS datastore_profiles (
    project TEXT NOT NULL, name TEXT NOT NULL, type TEXT, updated TEXT,
    body TEXT,
    PRIMARY KEY (project, name)
);
CREATE TABLE IF NOT EXISTS alert_templates (
    name TEXT PRIMARY KEY, updated TEXT, body TEXT
);
CREATE TABLE IF NOT EXISTS project_secrets (
    project TEXT NOT NULL, key TEXT NOT NULL, value TEXT,
    PRIMARY KEY (project, key)
);
CREATE TABLE IF NOT EXISTS hub_sources (
    name TEXT PRIMARY KEY, idx INTEGER, updated TEXT, body TEXT
);
CREATE INDEX IF NOT EXISTS idx_runs_name ON runs(project, name);
CREATE INDEX IF NOT EXISTS idx_runs_state ON runs(project, state);
CREATE INDEX IF NOT EXISTS idx_artifacts_key ON artifacts(project, key);
"""

# ---------------------------------------------------------------------
# Versioned schema migrations (reference: server/api/db/sqldb alembic
# tree, 2,532 LoC).  The version lives in SQLite's PRAGMA user_version;
# a round-1 DB file reports 0 and every migration is applied in order,
# so old on-disk DBs open cleanly after an upgrade.  Each migration
# must be idempotent (guard ALTERs by inspecting the live schema) —
# _SCHEMA above only creates missing objects and never alters.
# ---------------------------------------------------------------------
SCHEMA_VERSION = 4


def _has_column(conn, table: str, column: str) -> bool:
    return any(r[1] == column for r in
               conn.execute(f"PRAGMA table_info({table})"))


def _migration_1(conn):
    """Run bookkeeping columns the round-1 schema lacked:
    requested_logs (log-collector pull tracking, reference models.py
    Run.requested_logs) + updated-time indexes for list pagination."""
    if not _has_column(conn, "runs", "requested_logs"):
        conn.execute("ALTER TABLE runs ADD COLUMN "
                     "requested_logs INTEGER DEFAULT 0")
    conn.execute("CREATE INDEX IF NOT EXISTS idx_runs_updated "
                 "ON runs(project, updated)")
    conn.execute("CREATE INDEX IF NOT EXISTS idx_artifacts_updated "
                 "ON artifacts(project, updated)")


def _migration_2(conn):
    """Pagination cache (reference models.py PaginationCache): stores
    paginated-query cursors so clients can page with a token."""
    conn.execute("""
        CREATE TABLE IF NOT EXISTS pagination_cache (
            key TEXT PRIMARY KEY, method TEXT, current_page INTEGER,
            page_size INTEGER, kwargs TEXT, last_accessed TEXT)""")


def _migration_3(conn):
    """Time-window trackers (reference models.py TimeWindowTracker):
    high-water marks for periodic sweeps (runs monitor, summaries)."""
    conn.execute("""
        CREATE TABLE IF NOT EXISTS time_window_trackers (
            key TEXT PRIMARY KEY, timestamp TEXT, max_allowed_period
            INTEGER)""")


def _migration_4(conn):
    """API-gateway configs (reference models.py has no table — the
    reference stores gateways in nuclio; node-locally they are DB
    documents served by /projects/{p}/api-gateways)."""
    conn.execute("""
        CREATE TABLE IF NOT EXISTS api_gateways (
            project TEXT NOT NULL, name TEXT NOT NULL, updated TEXT,
            body TEXT, PRIMARY KEY (project, name))""")


_MIGRATIONS = {1: _migration_1, 2: _migration_2, 3: _migration_3,
               4: _migration_4}


def _match_labels(body: dict, labels) -> bool:
    if not labels:
        return True
    obj_labels = (body.get("metadata", {}) or {}).get("labels", {}) or {}
    if isinstance(labels, dict):
        items = labels.items()
    else:
        items = []
        for lbl in labels:
            if "=" in lbl:
                items.append(tuple(lbl.split("=", 1)))
            else:
                items.append((lbl, None))
    for key, val in items:
        if key not in obj_labels:
            return False
        if val is not None and str(obj_labels[key]) != str(val):
            return False
    return True


class SQLRunDB(RunDBExtras, RunDBInterface):
    """Node-local SQLite run DB (kind="local")."""

    kind = "local"

    def __init__(self, dsn: str = ""):
        if not dsn or dsn == "local":
            from ..config import config

            base = config.httpdb.dirpath or os.path.join(config.base_dir, "db")
            os.makedirs(base, exist_ok=True)
            dsn = os.path.join(base, "mlrun.db")
        self.dsn = dsn
        self._local = threading.local()
        self._write_lock = threading.Lock()
        self._init_schema()

    # -- connection management --
    def _conn(self) -> sqlite3.Connection:
        conn = getattr(self._local, "conn", None)
        if conn is None:
            if self.dsn != ":memory:":
                os.makedirs(os.path.dirname(os.path.abspath(self.dsn)),
                            exist_ok=True)
            conn = sqlite3.connect(self.dsn, timeout=30)
            conn.row_factory = sqlite3.Row
            conn.execute("PRAGMA journal_mode=WAL")
            conn.execute("PRAGMA synchronous=NORMAL")
            self._local.conn = conn
        return conn

    def _init_schema(self):
        with self._write_lock:
            conn = self._conn()
            conn.executescript(_SCHEMA)
            conn.commit()
            self._migrate_locked(conn)

    def _migrate_locked(self, conn) -> list:
        """Apply pending schema migrations; returns applied versions."""
        current = conn.execute("PRAGMA user_version").fetchone()[0]
        applied = []
        for version in range(current + 1, SCHEMA_VERSION + 1):
            _MIGRATIONS[version](conn)
            conn.execute(f"PRAGMA user_version = {version}")
            conn.commit()
            applied.append(version)
        if applied:
            from ..utils import logger

            logger.info("schema migrated", dsn=self.dsn,
                        from_version=current, to_version=SCHEMA_VERSION)
        return applied

    def trigger_migrations(self) -> dict:
        """Explicit migration entry (reference httpdb
        trigger_migrations -> operations/migrations endpoint)."""
        with self._write_lock:
            conn = self._conn()
            applied = self._migrate_locked(conn)
            current = conn.execute("PRAGMA user_version").fetchone()[0]
        return {"schema_version": current, "applied": applied}

    def _execute(self, sql, params=()):
        with self._write_lock:
            cur = self._conn().execute(sql, params)
            self._conn().commit()
            return cur

    def _query(self, sql, params=()):
        return self._conn().execute(sql, params).fetchall()

    def connect(self, secrets=None):
        return self

    # ------------------------------------------------------------- runs
    def store_run(self, struct, uid, project="", iter=0):
        project = project or "default"
        if hasattr(struct, "to_dict"):
            struct = struct.to_dict()
        state = struct.get("status", {}).get("state", RunStates.created)
        name = struct.get("metadata", {}).get("name", "")
        start = struct.get("status", {}).get("start_time") or now_iso()
        self._execute(
            "INSERT OR REPLACE INTO runs "
            "(project, uid, iteration, name, state, start_time, updated, body) "
            "VALUES (?,?,?,?,?,?,?,?)",
            (project, uid, iter, name, state, start, now_iso(),
             json.dumps(struct, default=str)))
        return struct

    def update_run(self, updates: dict, uid, project="", iter=0):
        project = project or "default"
        run = self.read_run(uid, project, iter)
        from ..utils import update_in

        for key, val in (updates or {}).items():
            update_in(run, key, val)
        self.store_run(run, uid, project, iter)
        return run

    def read_run(self, uid, project="", iter=0):
        project = project or "default"
        rows = self._query(
            "SELECT body FROM runs WHERE project=? AND uid=? AND iteration=?",
            (project, uid, iter or 0))
        if not rows:
            raise MLRunNotFoundError(f"run {project}/{uid} not found")
        return json.loads(rows[0]["body"])

    def list_runs(self, name="", uid=None, project="", labels=None, state=None,
                  sort=True, last=0, iter=False, start_time_from=None,
                  start_time_to=None, partition_by=None,
                  rows_per_partition: int = 1, partition_sort_by="updated",
                  partition_order="desc", max_partitions: int = 0,
                  offset: int = 0, limit: int = 0):
        """List runs with the reference's richer query surface
        (httpdb.py list_runs): label selectors, iteration expansion,
        time windows, OFFSET/LIMIT pagination and partition-by
        ("name"/"project"): per-partition top-N by the sort field —
        the reference's newest-run-per-name query shape."""
        project = project or "default"
        sql = "SELECT body FROM runs WHERE project=?"
        params: list = [project]
        if name:
            sql += " AND name LIKE ?"
            params.append(f"%{name.replace('~', '')}%" if name.startswith("~")
                          else name)
        if uid:
            if isinstance(uid, (list, tuple)):
                sql += f" AND uid IN ({','.join('?' * len(uid))})"
                params.extend(uid)
            else:
                sql += " AND uid=?"
                params.append(uid)
        if state:
            sql += " AND state=?"
            params.append(state)
        if not iter:
            sql += " AND iteration=0"
        if start_time_from:
            sql += " AND start_time>=?"
            params.append(str(start_time_from))
        if start_time_to:
            sql += " AND start_time<=?"
            params.append(str(start_time_to))
        if partition_by:
            # window-function partition query (SQLite >= 3.25): top
            # rows_per_partition per name/project by partition_sort_by
            field = {"updated": "updated",
                     "created": "start_time",
                     "start_time": "start_time"}.get(
                partition_sort_by or "updated", "updated")
            order = "DESC" if (partition_order or "desc").lower() == \
                "desc" else "ASC"
            part_col = {"name": "name", "project": "project"}.get(
                partition_by)
            if part_col is None:
                raise MLRunInvalidArgumentError(
                    f"unsupported partition_by {partition_by!r}")
            inner = sql.replace(
                "SELECT body FROM runs",
                f"SELECT body, ROW_NUMBER() OVER (PARTITION BY "
                f"{part_col} ORDER BY {field} {order}) AS rn FROM runs",
                1)
            sql = (f"SELECT body FROM ({inner}) "
                   f"WHERE rn <= {int(rows_per_partition)}")
        elif sort:
            sql += " ORDER BY start_time DESC"
        if last:
            sql += f" LIMIT {int(last)}"
        elif limit:
            sql += f" LIMIT {int(limit)} OFFSET {int(offset)}"
        elif offset:
            sql += f" LIMIT -1 OFFSET {int(offset)}"
        out = []
        for row in self._query(sql, params):
            body = json.loads(row["body"])
            if _match_labels(body, labels):
                out.append(body)
        if partition_by and max_partitions:
            seen: dict = {}
            capped = []
            for body in out:
                part = (body.get("metadata", {}) or {}).get(
                    "name" if partition_by == "name" else "project", "")
                bucket = seen.setdefault(part, [])
                if len(seen) > int(max_partitions) and not bucket:
                    continue
                bucket.append(body)
                capped.append(body)
            out = capped
        from ..lists import RunList

        return RunList(out)

    def del_run(self, uid, project="", iter=0):
        project = project or "default"
        self._execute("DELETE FROM runs WHERE project=? AND uid=? AND iteration=?",
                      (project, uid, iter or 0))

    def del_runs(self, name="", project="", labels=None, state=None, days_ago=0):
        project = project or "default"
        for run in self.list_runs(name=name, project=project, labels=labels,
                                  state=state, iter=True):
            meta = run.get("metadata", {})
            self.del_run(meta.get("uid"), project, meta.get("iteration") or 0)

    def abort_run(self, uid, project="", iter=0, status_text=""):
        current = self.read_run(uid, project, iter)
        if RunStates.is_terminal(current.get("status", {}).get("state", "")):
            return  # terminal runs are not abortable (reference crud)
        self.update_run(
            {"status.state": RunStates.aborted,
             "status.status_text": status_text or "aborted by request",
             "status.last_update": now_iso()},
            uid, project, iter)

    # ------------------------------------------------------------- logs
    def store_log(self, uid, project="", body=None, append=False):
        project = project or "default"
        if body is None:
            return
        if isinstance(body, str):
            body = body.encode()
        if append:
            rows = self._query("SELECT body FROM logs WHERE project=? AND uid=?",
                               (project, uid))
            if rows:
                body = bytes(rows[0]["body"] or b"") + body
        self._execute("INSERT OR REPLACE INTO logs (project, uid, body) "
                      "VALUES (?,?,?)", (project, uid, body))

    def get_log(self, uid, project="", offset=0, size=0):
        project = project or "default"
        rows = self._query("SELECT body FROM logs WHERE project=? AND uid=?",
                           (project, uid))
        state = ""
        try:
            run = self.read_run(uid, project)
            state = run.get("status", {}).get("state", "")
        except MLRunNotFoundError:
            pass
        if not rows:
            return state, b""
        body = bytes(rows[0]["body"] or b"")
        if offset:
            body = body[offset:]
        if size:
            body = body[:size]
        return state, body

    # -------------------------------------------------------- artifacts
    def store_artifact(self, key, artifact, uid=None, iter=None, tag="",
                       project="", tree=None):
        project = project or "default"
        if hasattr(artifact, "to_dict"):
            artifact = artifact.to_dict()
        tree = tree or artifact.get("metadata", {}).get("tree") or ""
        iter = iter or 0
        kind = artifact.get("kind", "artifact")
        artifact.setdefault("metadata", {})["key"] = key
        artifact["metadata"]["project"] = project
        artifact["metadata"]["tree"] = tree
        if iter:
            artifact["metadata"]["iter"] = iter
        self._execute(
            "INSERT OR REPLACE INTO artifacts "
            "(project, key, tree, iteration, tag, kind, uid, updated, body) "
            "VALUES (?,?,?,?,?,?,?,?,?)",
            (project, key, tree, iter, tag or "", kind, uid, now_iso(),
             json.dumps(artifact, default=str)))
        for a_tag in {tag or "latest", "latest"}:
            self._execute(
                "INSERT OR REPLACE INTO artifact_tags "
                "(project, key, tag, tree, iteration) VALUES (?,?,?,?,?)",
                (project, key, a_tag, tree, iter))
        return artifact

    def store_datastore_profile(self, project, profile: dict):
        name = profile.get("name")
        self._execute(
            "INSERT OR REPLACE INTO datastore_profiles "
            "(project, name, type, updated, body) VALUES (?,?,?,?,?)",
            (project or "default", name, profile.get("type", ""),
             now_iso(), json.dumps(profile, default=str)))

    def get_datastore_profile(self, project, name):
        rows = self._query(
            "SELECT body FROM datastore_profiles WHERE project=? AND "
            "name=?", (project or "default", name))
        if not rows:
            raise MLRunNotFoundError(
                f"datastore profile {name} not found")
        return json.loads(rows[0]["body"])

    def list_datastore_profiles(self, project):
        return [json.loads(r["body"]) for r in self._query(
            "SELECT body FROM datastore_profiles WHERE project=? "
            "ORDER BY name", (project or "default",))]

    def delete_datastore_profile(self, project, name):
        self._execute(
            "DELETE FROM datastore_profiles WHERE project=? AND name=?",
            (project or "default", name))

    def store_alert_template(self, name, template: dict):
        self._execute(
            "INSERT OR REPLACE INTO alert_templates "
            "(name, updated, body) VALUES (?,?,?)",
            (name, now_iso(), json.dumps(template, default=str)))

    def get_alert_template(self, name):
        rows = self._query(
            "SELECT body FROM alert_templates WHERE name=?", (name,))
        if not rows:
            raise MLRunNotFoundError(f"alert template {name} not found")
        return json.loads(rows[0]["body"])

    def list_alert_templates(self):
        return [json.loads(r["body"]) for r in self._query(
            "SELECT body FROM alert_templates ORDER BY name")]

    def delete_alert_template(self, name):
        self._execute("DELETE FROM alert_templates WHERE name=?",
                      (name,))

    def tag_artifact(self, project, key, tree, tag, iteration=0):
        """Attach a tag to a stored artifact version (reference tags
        endpoint)."""
        self._execute(
            "INSERT OR REPLACE INTO artifact_tags "
            "(project, key, tag, tree, iteration) VALUES (?,?,?,?,?)",
            (project or "default", key, tag, tree or "", iteration))

    def delete_artifact_tag(self, project, key, tag):
        self._execute(
            "DELETE FROM artifact_tags WHERE project=? AND key=? AND "
            "tag=?", (project or "default", key, tag))

    def list_artifact_tags(self, project, key=""):
        if key:
            rows = self._query(
                "SELECT DISTINCT tag FROM artifact_tags WHERE "
                "project=? AND key=? ORDER BY tag",
                (project or "default", key))
        else:
            rows = self._query(
                "SELECT DISTINCT tag FROM artifact_tags WHERE "
                "project=? ORDER BY tag", (project or "default",))
        return [r["tag"] for r in rows]

    def read_artifact(self, key, tag="", iter=None, project="", tree=None,
                      uid=None):
        project = project or "default"
        tree = tree or uid
        if tree:
            rows = self._query(
                "SELECT body FROM artifacts WHERE project=? AND key=? AND "
                "tree=? ORDER BY iteration LIMIT 1", (project, key, tree))
        else:
            tag = tag or "latest"
            tags = self._query(
                "SELECT tree, iteration FROM artifact_tags WHERE project=? "
                "AND key=? AND tag=?", (project, key, tag))
            if not tags:
                raise MLRunNotFoundError(
                    f"artifact {project}/{key}:{tag} not found")
            rows = self._query(
                "SELECT body FROM artifacts WHERE project=? AND key=? AND "
                "tree=? AND iteration=?",
                (project, key, tags[0]["tree"] or "", tags[0]["iteration"]))
        if not rows:
            raise MLRunNotFoundError(f"artifact {project}/{key} not found")
        return json.loads(rows[0]["body"])

    _ARTIFACT_CATEGORIES = {
        # reference common/schemas ArtifactCategories kind groupings
        "model": ("model",),
        "dataset": ("dataset",),
        "document": ("document",),
        "other": ("", "artifact", "plot", "chart", "table", "link"),
    }

    def list_artifacts(self, name="", project="", tag="", labels=None,
                       since=None, until=None, kind=None, category=None,
                       iter=None, tree=None, limit: int = 0,
                       offset: int = 0):
        """Artifact listing with the v2 query surface (reference
        endpoints/artifacts_v2.py): kind/category filters, time
        window, tree (producer id), OFFSET/LIMIT pagination."""
        project = project or "default"
        sql = "SELECT body FROM artifacts WHERE project=?"
        params: list = [project]
        if name:
            sql += " AND key LIKE ?"
            params.append(f"%{name.strip('~')}%" if name.startswith("~") else name)
        if kind:
            sql += " AND kind=?"
            params.append(kind)
        if category:
            kinds = self._ARTIFACT_CATEGORIES.get(category)
            if kinds is None:
                raise MLRunInvalidArgumentError(
                    f"unknown artifact category {category!r}")
            sql += f" AND kind IN ({','.join('?' * len(kinds))})"
            params.extend(kinds)
        if tree:
            sql += " AND tree=?"
            params.append(tree)
        if iter is not None:
            sql += " AND iteration=?"
            params.append(int(iter))
        if since:
            sql += " AND updated>=?"
            params.append(str(since))
        if until:
            sql += " AND updated<=?"
            params.append(str(until))
        if tag and tag not in ("*", "latest"):
            sql += (" AND key IN (SELECT key FROM artifact_tags "
                    "WHERE project=? AND tag=?)")
            params.extend([project, tag])
        sql += " ORDER BY updated DESC"
        if limit:
            sql += f" LIMIT {int(limit)} OFFSET {int(offset)}"
        elif offset:
            sql += f" LIMIT -1 OFFSET {int(offset)}"
        out = []
        for row in self._query(sql, params):
            body = json.loads(row["body"])
            if _match_labels(body, labels):
                out.append(body)
        from ..lists import ArtifactList

        result = ArtifactList(out)
        result.tag = tag
        return result

    def del_artifact(self, key, tag="", project="", uid=None, tree=None):
        project = project or "default"
        self._execute("DELETE FROM artifacts WHERE project=? AND key=?",
                      (project, key))
        self._execute("DELETE FROM artifact_tags WHERE project=? AND key=?",
                      (project, key))

    def del_artifacts(self, name="", project="", tag="", labels=None):
        for artifact in self.list_artifacts(name=name, project=project,
                                            tag=tag, labels=labels):
            key = artifact.get("metadata", {}).get("key")
            if key:
                self.del_artifact(key, project=project)

    # -------------------------------------------------------- functions
    def store_function(self, function, name, project="", tag="",
                       versioned=False):
        project = project or "default"
        tag = tag or "latest"
        if hasattr(function, "to_dict"):
            function = function.to_dict()
        import hashlib

        body = json.dumps(function, default=str, sort_keys=True)
        hash_key = hashlib.sha1(body.encode()).hexdigest()
        self._execute(
            "INSERT OR REPLACE INTO functions "
            "(project, name, tag, hash_key, updated, body) VALUES (?,?,?,?,?,?)",
            (project, name, tag, hash_key, now_iso(), body))
        if versioned:
            self._execute(
                "INSERT OR REPLACE INTO function_versions "
                "(project, name, hash_key, updated, body) VALUES (?,?,?,?,?)",
                (project, name, hash_key, now_iso(), body))
        return hash_key

    def get_function(self, name, project="", tag="", hash_key=""):
        project = project or "default"
        if hash_key:
            rows = self._query(
                "SELECT body FROM function_versions WHERE project=? AND "
                "name=? AND hash_key=?", (project, name, hash_key))
        else:
            rows = self._query(
                "SELECT body FROM functions WHERE project=? AND name=? AND tag=?",
                (project, name, tag or "latest"))
        if not rows:
            raise MLRunNotFoundError(f"function {project}/{name} not found")
        return json.loads(rows[0]["body"])

    def list_functions(self, name=None, project="", tag="", labels=None):
        project = project or "default"
        sql = "SELECT body FROM functions WHERE project=?"
        params: list = [project]
        if name:
            sql += " AND name=?"
            params.append(name)
        if tag:
            sql += " AND tag=?"
            params.append(tag)
        out = []
        for row in self._query(sql, params):
            body = json.loads(row["body"])
            if _match_labels(body, labels):
                out.append(body)
        return out

    def delete_function(self, name, project=""):
        project = project or "default"
        self._execute("DELETE FROM functions WHERE project=? AND name=?",
                      (project, name))
        self._execute("DELETE FROM function_versions WHERE project=? AND name=?",
                      (project, name))

    # --------------------------------------------------------- projects
    def create_project(self, project):
        if hasattr(project, "to_dict"):
            project = project.to_dict()
        if isinstance(project, str):
            project = {"metadata": {"name": project}}
        name = project.get("metadata", {}).get("name")
        if not name:
            raise ValueError("project has no metadata.name")
        existing = self._query("SELECT name FROM projects WHERE name=?", (name,))
        if existing:
            raise MLRunConflictError(f"project {name} already exists")
        self._execute(
            "INSERT INTO projects (name, state, created, body) VALUES (?,?,?,?)",
            (name, "online", now_iso(), json.dumps(project, default=str)))
        return project

    def get_project(self, name):
        rows = self._query("SELECT body FROM projects WHERE name=?", (name,))
        if not rows:
            raise MLRunNotFoundError(f"project {name} not found")
        return json.loads(rows[0]["body"])

    # ------------------------------------------------------- secrets
    def store_project_secrets(self, project: str, secrets: dict,
                              provider: str = "kubernetes"):
        """Store project secrets (values retrievable server-side only
        via get_project_secret; the list endpoint exposes KEYS, like
        the reference k8s-secret flow)."""
        for key, value in (secrets or {}).items():
            self._execute(
                "INSERT OR REPLACE INTO project_secrets "
                "(project, key, value) VALUES (?, ?, ?)",
                (project, key, str(value)))

    def list_project_secret_keys(self, project: str,
                                 provider: str = "kubernetes") -> list:
        return [row["key"] for row in self._query(
            "SELECT key FROM project_secrets WHERE project = ? "
            "ORDER BY key", (project,))]

    def get_project_secret(self, project: str, key: str):
        rows = self._query(
            "SELECT value FROM project_secrets WHERE project = ? "
            "AND key = ?", (project, key))
        return rows[0]["value"] if rows else None

    def delete_project_secrets(self, project: str, keys: list = None,
                               provider: str = "kubernetes"):
        if keys:
            for key in keys:
                self._execute(
                    "DELETE FROM project_secrets WHERE project = ? "
                    "AND key = ?", (project, key))
        else:
            self._execute(
                "DELETE FROM project_secrets WHERE project = ?",
                (project,))

    def list_projects(self, owner=None, format_=None, labels=None, state=None):
        out = []
        for row in self._query("SELECT body FROM projects"):
            body = json.loads(row["body"])
            if _match_labels(body, labels):
                out.append(body)
        return out

    def store_project(self, name, project):
        if hasattr(project, "to_dict"):
            project = project.to_dict()
        self._execute(
            "INSERT OR REPLACE INTO projects (name, state, created, body) "
            "VALUES (?,?,?,?)",
            (name, "online", now_iso(), json.dumps(project, default=str)))
        return project

    def delete_project(self, name, deletion_strategy=None):
        if deletion_strategy in (None, "cascade", "cascading"):
            for table in ["runs", "artifacts", "artifact_tags", "functions",
                          "function_versions", "logs", "schedules",
                          "feature_sets", "feature_vectors", "model_endpoints",
                          "background_tasks", "alert_configs", "alert_states"]:
                self._execute(f"DELETE FROM {table} WHERE project=?", (name,))
        self._execute("DELETE FROM projects WHERE name=?", (name,))

    # -------------------------------------------------------- schedules
    def create_schedule(self, project, schedule: dict):
        project = project or "default"
        name = schedule.get("name")
        existing = self._query(
            "SELECT name FROM schedules WHERE project=? AND name=?",
            (project, name))
        if existing:
            raise MLRunConflictError(f"schedule {project}/{name} exists")
        self._execute(
            "INSERT INTO schedules (project, name, kind, cron, creation_time, "
            "next_run_time, last_run_uri, body) VALUES (?,?,?,?,?,?,?,?)",
            (project, name, schedule.get("kind", "job"),
             schedule.get("cron_trigger", ""), now_iso(),
             schedule.get("next_run_time"), None,
             json.dumps(schedule, default=str)))

    def update_schedule(self, project, name, schedule: dict):
        project = project or "default"
        current = self.get_schedule(project, name)
        current.update(schedule)
        self._execute(
            "UPDATE schedules SET kind=?, cron=?, next_run_time=?, "
            "last_run_uri=?, body=? WHERE project=? AND name=?",
            (current.get("kind", "job"), current.get("cron_trigger", ""),
             current.get("next_run_time"), current.get("last_run_uri"),
             json.dumps(current, default=str), project, name))

    def get_schedule(self, project, name):
        rows = self._query(
            "SELECT body, next_run_time, last_run_uri FROM schedules "
            "WHERE project=? AND name=?", (project or "default", name))
        if not rows:
            raise MLRunNotFoundError(f"schedule {project}/{name} not found")
        body = json.loads(rows[0]["body"])
        body["next_run_time"] = rows[0]["next_run_time"]
        body["last_run_uri"] = rows[0]["last_run_uri"]
        return body

    def list_schedules(self, project, name=""):
        sql = "SELECT body, next_run_time, last_run_uri FROM schedules WHERE project=?"
        params = [project or "default"]
        if name:
            sql += " AND name LIKE ?"
            params.append(f"%{name}%")
        out = []
        for row in self._query(sql, params):
            body = json.loads(row["body"])
            body["next_run_time"] = row["next_run_time"]
            body["last_run_uri"] = row["last_run_uri"]
            out.append(body)
        return out

    def delete_schedule(self, project, name):
        self._execute("DELETE FROM schedules WHERE project=? AND name=?",
                      (project or "default", name))

    # ----------------------------------------------------- feature store
    def _store_tagged(self, table, body, name, project, tag):
        project = project or "default"
        if hasattr(body, "to_dict"):
            body = body.to_dict()
        name = name or body.get("metadata", {}).get("name")
        tag = tag or body.get("metadata", {}).get("tag") or "latest"
        self._execute(
            f"INSERT OR REPLACE INTO {table} (project, name, tag, updated, body) "
            "VALUES (?,?,?,?,?)",
            (project, name, tag, now_iso(), json.dumps(body, default=str)))
        if tag != "latest":
            self._execute(
                f"INSERT OR REPLACE INTO {table} (project, name, tag, updated, "
                "body) VALUES (?,?,?,?,?)",
                (project, name, "latest", now_iso(),
                 json.dumps(body, default=str)))
        return body

    def _get_tagged(self, table, name, project, tag):
        rows = self._query(
            f"SELECT body FROM {table} WHERE project=? AND name=? AND tag=?",
            (project or "default", name, tag or "latest"))
        if not rows:
            raise MLRunNotFoundError(f"{table[:-1]} {project}/{name} not found")
        return json.loads(rows[0]["body"])

    def _list_tagged(self, table, project, name, labels):
        sql = f"SELECT body FROM {table} WHERE project=? AND tag='latest'"
        params = [project or "default"]
        if name:
            sql += " AND name LIKE ?"
            params.append(f"%{name}%")
        out = []
        for row in self._query(sql, params):
            body = json.loads(row["body"])
            if _match_labels(body, labels):
                out.append(body)
        return out

    def store_feature_set(self, feature_set, name=None, project="", tag=None,
                          versioned=False):
        return self._store_tagged("feature_sets", feature_set, name, project, tag)

    def get_feature_set(self, name, project="", tag=None):
        return self._get_tagged("feature_sets", name, project, tag)

    def list_feature_sets(self, project="", name=None, tag=None, labels=None):
        return self._list_tagged("feature_sets", project, name, labels)

    def delete_feature_set(self, name, project="", tag=None):
        self._execute("DELETE FROM feature_sets WHERE project=? AND name=?",
                      (project or "default", name))

    def store_feature_vector(self, feature_vector, name=None, project="",
                             tag=None, versioned=False):
        return self._store_tagged("feature_vectors", feature_vector, name,
                                  project, tag)

    def get_feature_vector(self, name, project="", tag=None):
        return self._get_tagged("feature_vectors", name, project, tag)

    def list_feature_vectors(self, project="", name=None, tag=None, labels=None):
        return self._list_tagged("feature_vectors", project, name, labels)

    def delete_feature_vector(self, name, project="", tag=None):
        self._execute("DELETE FROM feature_vectors WHERE project=? AND name=?",
                      (project or "default", name))

    # --------------------------------------------------- model endpoints
    def store_model_endpoint(self, project, endpoint_id, endpoint: dict):
        project = project or "default"
        if hasattr(endpoint, "to_dict"):
            endpoint = endpoint.to_dict()
        self._execute(
            "INSERT OR REPLACE INTO model_endpoints "
            "(project, endpoint_id, model, function, updated, body) "
            "VALUES (?,?,?,?,?,?)",
            (project, endpoint_id,
             endpoint.get("spec", {}).get("model", ""),
             endpoint.get("spec", {}).get("function_uri", ""), now_iso(),
             json.dumps(endpoint, default=str)))

    def get_model_endpoint(self, project, endpoint_id):
        rows = self._query(
            "SELECT body FROM model_endpoints WHERE project=? AND endpoint_id=?",
            (project or "default", endpoint_id))
        if not rows:
            raise MLRunNotFoundError(
                f"model endpoint {project}/{endpoint_id} not found")
        return json.loads(rows[0]["body"])

    def list_model_endpoints(self, project, model=None, function=None,
                             labels=None):
        sql = "SELECT body FROM model_endpoints WHERE project=?"
        params = [project or "default"]
        if model:
            sql += " AND model=?"
            params.append(model)
        if function:
            sql += " AND function=?"
            params.append(function)
        out = []
        for row in self._query(sql, params):
            body = json.loads(row["body"])
            if _match_labels(body, labels):
                out.append(body)
        return out

    def delete_model_endpoint(self, project, endpoint_id):
        self._execute(
            "DELETE FROM model_endpoints WHERE project=? AND endpoint_id=?",
            (project or "default", endpoint_id))

    # -------------------------------------------------- background tasks
    def store_background_task(self, project, task: dict):
        project = project or "default"
        name = task.get("metadata", {}).get("name") or task.get("name")
        state = task.get("status", {}).get("state", "running")
        self._execute(
            "INSERT OR REPLACE INTO background_tasks "
            "(project, name, state, updated, body) VALUES (?,?,?,?,?)",
            (project, name, state, now_iso(), json.dumps(task, default=str)))

    def get_background_task(self, project, name):
        rows = self._query(
            "SELECT body FROM background_tasks WHERE project=? AND name=?",
            (project or "default", name))
        if not rows:
            raise MLRunNotFoundError(
                f"background task {project}/{name} not found")
        return json.loads(rows[0]["body"])

    def list_background_tasks(self, project):
        return [json.loads(row["body"]) for row in self._query(
            "SELECT body FROM background_tasks WHERE project=?",
            (project or "default",))]

    # ------------------------------------------------------------ alerts
    def store_alert_config(self, project, name, alert: dict):
        project = project or "default"
        if hasattr(alert, "to_dict"):
            alert = alert.to_dict()
        self._execute(
            "INSERT OR REPLACE INTO alert_configs (project, name, updated, body) "
            "VALUES (?,?,?,?)",
            (project, name, now_iso(), json.dumps(alert, default=str)))
        self._execute(
            "INSERT OR IGNORE INTO alert_states "
            "(project, name, count, last_updated, active) VALUES (?,?,0,?,0)",
            (project, name, now_iso()))

    def get_alert_config(self, project, name):
        rows = self._query(
            "SELECT body FROM alert_configs WHERE project=? AND name=?",
            (project or "default", name))
        if not rows:
            raise MLRunNotFoundError(f"alert {project}/{name} not found")
        return json.loads(rows[0]["body"])

    def list_alert_configs(self, project):
        return [json.loads(row["body"]) for row in self._query(
            "SELECT body FROM alert_configs WHERE project=?",
            (project or "default",))]

    def delete_alert_config(self, project, name):
        project = project or "default"
        self._execute("DELETE FROM alert_configs WHERE project=? AND name=?",
                      (project, name))
        self._execute("DELETE FROM alert_states WHERE project=? AND name=?",
                      (project, name))

    def bump_alert_state(self, project, name):
        project = project or "default"
        self._execute(
            "UPDATE alert_states SET count=count+1, last_updated=?, active=1 "
            "WHERE project=? AND name=?", (now_iso(), project, name))
        rows = self._query(
            "SELECT count FROM alert_states WHERE project=? AND name=?",
            (project, name))
        return rows[0]["count"] if rows else 0

    def reset_alert_state(self, project, name):
        """Reset an alert's event counter (manual reset API, and the
        auto reset applied after an alert fires)."""
        self._execute(
            "UPDATE alert_states SET count=0, active=0, last_updated=? "
            "WHERE project=? AND name=?",
            (now_iso(), project or "default", name))

    def get_alert_state(self, project, name):
        rows = self._query(
            "SELECT count, last_updated, active FROM alert_states "
            "WHERE project=? AND name=?", (project or "default", name))
        if not rows:
            return {"count": 0, "active": False}
        return {"count": rows[0]["count"],
                "last_updated": rows[0]["last_updated"],
                "active": bool(rows[0]["active"])}

    # ----------------------------------------------------------- submit
    def submit_job(self, runspec, schedule=None):
        """Execute the run in-process via the server-side launcher."""
        from ..launcher import ServerSideLauncher

        if schedule:
            sched = {"name": runspec.metadata.name,
                     "kind": "job",
                     "cron_trigger": schedule,
                     "task": runspec.to_dict()}
            self.create_schedule(runspec.metadata.project, sched)
            return {"schedule": sched}
        launcher = ServerSideLauncher(db=self)
        return launcher.launch_task(runspec)

    # ------------------------------------------- pagination / trackers
    def paginated_list(self, method: str, page_token: str = None,
                       page: int = 1, page_size: int = 20, **kwargs):
        """Token-based pagination over any list_* method (reference
        utils/pagination.py + PaginationCache model).  First call:
        pass method/page_size/filters -> returns (items, token).
        Subsequent calls: pass page_token alone to get the next page;
        token is None when exhausted."""
        import hashlib

        if page_token:
            rows = self._query(
                "SELECT method, current_page, page_size, kwargs "
                "FROM pagination_cache WHERE key=?", (page_token,))
            if not rows:
                raise MLRunNotFoundError(
                    f"pagination token {page_token!r} expired")
            method = rows[0]["method"]
            page = rows[0]["current_page"] + 1
            page_size = rows[0]["page_size"]
            kwargs = json.loads(rows[0]["kwargs"])
        else:
            token_src = json.dumps([method, page_size, kwargs],
                                   sort_keys=True, default=str)
            page_token = hashlib.sha1(token_src.encode()).hexdigest()[:20]
        lister = getattr(self, method)
        items = lister(limit=page_size + 1,
                       offset=(page - 1) * page_size, **kwargs)
        has_more = len(items) > page_size
        items = items[:page_size]
        if has_more:
            self._execute(
                "INSERT OR REPLACE INTO pagination_cache "
                "(key, method, current_page, page_size, kwargs, "
                "last_accessed) VALUES (?,?,?,?,?,?)",
                (page_token, method, page, page_size,
                 json.dumps(kwargs, default=str), now_iso()))
            return items, page_token
        self._execute("DELETE FROM pagination_cache WHERE key=?",
                      (page_token,))
        return items, None

    def clean_pagination_cache(self, older_than_iso: str = None):
        if older_than_iso:
            self._execute("DELETE FROM pagination_cache WHERE "
                          "last_accessed < ?", (older_than_iso,))
        else:
            self._execute("DELETE FROM pagination_cache")

    def get_time_window_tracker(self, key: str):
        rows = self._query(
            "SELECT timestamp FROM time_window_trackers WHERE key=?",
            (key,))
        return rows[0]["timestamp"] if rows else None

    def store_time_window_tracker(self, key: str, timestamp: str = None):
        """High-water mark for periodic sweeps (reference
        TimeWindowTracker model)."""
        self._execute(
            "INSERT OR REPLACE INTO time_window_trackers "
            "(key, timestamp) VALUES (?,?)", (key, timestamp or now_iso()))

    # -------------------------------------------------- feature search
    def list_features(self, project, name=None, tag=None, entities=None,
                      labels=None):
        """Search features ACROSS feature sets (reference
        endpoints/feature_store.py list_features): returns
        feature + its owning feature-set digest."""
        out = []
        for fset in self.list_feature_sets(project):
            meta = fset.get("metadata", {})
            spec = fset.get("spec", {})
            if not _match_labels(fset, labels):
                continue
            set_entities = [e.get("name") for e in
                            spec.get("entities", [])]
            if entities and not set(entities) & set(set_entities):
                continue
            for feature in spec.get("features", []):
                fname = feature.get("name", "")
                if name and name.strip("~") not in fname:
                    continue
                out.append({
                    "feature": feature,
                    "feature_set_digest": {
                        "metadata": {"name": meta.get("name"),
                                     "tag": meta.get("tag", "latest")},
                        "spec": {"entities": spec.get("entities", [])},
                    }})
        return out

    def list_entities(self, project, name=None, tag=None, labels=None):
        """Search entities across feature sets (reference
        list_entities)."""
        out = []
        for fset in self.list_feature_sets(project):
            meta = fset.get("metadata", {})
            if not _match_labels(fset, labels):
                continue
            for entity in fset.get("spec", {}).get("entities", []):
                ename = entity.get("name", "")
                if name and name.strip("~") not in ename:
                    continue
                out.append({
                    "entity": entity,
                    "feature_set_digest": {
                        "metadata": {"name": meta.get("name"),
                                     "tag": meta.get("tag", "latest")},
                    }})
        return out

    # ----------------------------------------------------- api gateways
    def store_api_gateway(self, project, name, body: dict):
        self._execute(
            "INSERT OR REPLACE INTO api_gateways "
            "(project, name, updated, body) VALUES (?,?,?,?)",
            (project or "default", name, now_iso(),
             json.dumps(body, default=str)))
        return body

    def get_api_gateway(self, project, name):
        rows = self._query(
            "SELECT body FROM api_gateways WHERE project=? AND name=?",
            (project or "default", name))
        if not rows:
            raise MLRunNotFoundError(f"api gateway {name} not found")
        return json.loads(rows[0]["body"])

    def list_api_gateways(self, project):
        rows = self._query(
            "SELECT body FROM api_gateways WHERE project=?",
            (project or "default",))
        return [json.loads(r["body"]) for r in rows]

    def delete_api_gateway(self, project, name):
        self._execute(
            "DELETE FROM api_gateways WHERE project=? AND name=?",
            (project or "default", name))

    # ------------------------------------------------ project summaries
    def compute_project_summary(self, project: str) -> dict:
        """Per-project counts (reference ProjectSummary model +
        the periodic summary-calculation loop, server main.py:630)."""
        def _count(sql, params):
            return self._query(sql, params)[0][0]

        counts = {
            "runs_completed_recent_count": _count(
                "SELECT COUNT(*) FROM runs WHERE project=? AND "
                "state='completed'", (project,)),
            "runs_failed_recent_count": _count(
                "SELECT COUNT(*) FROM runs WHERE project=? AND "
                "state IN ('error','failed','aborted')", (project,)),
            "runs_running_count": _count(
                "SELECT COUNT(*) FROM runs WHERE project=? AND "
                "state='running'", (project,)),
            "files_count": _count(
                "SELECT COUNT(*) FROM artifacts WHERE project=? AND "
                "(kind NOT IN ('model','dataset') OR kind IS NULL)",
                (project,)),
            "models_count": _count(
                "SELECT COUNT(*) FROM artifacts WHERE project=? AND "
                "kind='model'", (project,)),
            "datasets_count": _count(
                "SELECT COUNT(*) FROM artifacts WHERE project=? AND "
                "kind='dataset'", (project,)),
            "feature_sets_count": _count(
                "SELECT COUNT(*) FROM feature_sets WHERE project=?",
                (project,)),
            "schedules_count": _count(
                "SELECT COUNT(*) FROM schedules WHERE project=?",
                (project,)),
            "model_endpoints_count": _count(
                "SELECT COUNT(*) FROM model_endpoints WHERE project=?",
                (project,)),
        }
        return {"name": project, "updated": now_iso(), **counts}

    def list_project_summaries(self) -> list:
        names = [p.get("metadata", {}).get("name", p.get("name", ""))
                 for p in self.list_projects()]
        return [self.compute_project_summary(n) for n in names if n]
