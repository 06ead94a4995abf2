# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""The mlrun_amd API service: FastAPI app over the node-local SQLite
run DB.

Parity target: reference server/api/main.py:93 (app assembly) + the
endpoint modules under server/api/api/endpoints/ (runs, artifacts,
functions, projects, logs, schedules, feature-store, model-endpoints,
background tasks, submit).  Periodic loops (runs monitoring :608,
scheduler ticks) run as in-process asyncio tasks instead of k8s
informers.
"""

import asyncio
import json
import os
import typing

from fastapi import (FastAPI, HTTPException, Query, Request,
                     Response)
from fastapi.responses import JSONResponse

from ..config import config
from ..db.sqldb import SQLRunDB
from ..errors import MLRunBaseError, err_to_status
from ..utils import logger, now_iso
from .scheduler import Scheduler


def _validate_or_422(validator, body):
    """Run a pydantic wire-schema validator; malformed shapes 422
    (reference: every endpoint validates its pydantic body)."""
    try:
        validator(body)
    except Exception as exc:
        raise HTTPException(status_code=422, detail=str(exc))


def _deep_update(target: dict, patch: dict):
    """Recursive dict merge (PATCH semantics — reference strategies)."""
    for key, value in patch.items():
        if isinstance(value, dict) and isinstance(target.get(key), dict):
            _deep_update(target[key], value)
        else:
            target[key] = value
    return target


def create_app(db: SQLRunDB = None, with_scheduler: bool = True) -> FastAPI:
    db = db or SQLRunDB(config.httpdb.dsn or "")
    import contextlib

    @contextlib.asynccontextmanager
    async def lifespan(app_):
        if app_.state.scheduler:
            app_.state.scheduler.start()
        app_.state.monitor_task = asyncio.create_task(_runs_monitor(db))
        try:
            yield
        finally:
            if app_.state.scheduler:
                app_.state.scheduler.stop()
            task = getattr(app_.state, "monitor_task", None)
            if task:
                task.cancel()

    app = FastAPI(title="mlrun-amd-api", version="v1",
                  lifespan=lifespan)
    app.state.db = db
    app.state.scheduler = Scheduler(db) if with_scheduler else None

    @app.exception_handler(MLRunBaseError)
    async def mlrun_error_handler(request: Request, exc: MLRunBaseError):
        return JSONResponse(status_code=err_to_status(exc),
                            content={"detail": str(exc)})

    # ------------------------------------------------------------ misc
    @app.get("/api/v1/healthz")
    async def healthz():
        from .. import __version__

        return {"status": "ok", "version": __version__}

    @app.get("/api/v1/client-spec")
    async def client_spec():
        return {"version": config.version,
                "default_project": config.default_project}

    # ------------------------------------------------------------ runs
    @app.post("/api/v1/run/{project}/{uid}")
    async def store_run(project: str, uid: str, request: Request,
                        iter: int = 0):
        body = await request.json()
        db.store_run(body, uid, project, iter=iter)
        return {}

    @app.patch("/api/v1/run/{project}/{uid}")
    async def update_run(project: str, uid: str, request: Request,
                         iter: int = 0):
        body = await request.json()
        return db.update_run(body, uid, project, iter=iter)

    @app.get("/api/v1/run/{project}/{uid}")
    async def read_run(project: str, uid: str, iter: int = 0):
        return {"data": db.read_run(uid, project, iter=iter)}

    @app.delete("/api/v1/run/{project}/{uid}")
    async def del_run(project: str, uid: str, iter: int = 0):
        db.del_run(uid, project, iter=iter)
        return {}

    @app.post("/api/v1/run/{project}/{uid}/abort")
    async def abort_run(project: str, uid: str, request: Request):
        try:
            body = await request.json()
        except Exception:
            body = {}
        db.abort_run(uid, project,
                     status_text=body.get("status_text", ""))
        return {}

    @app.get("/api/v1/runs")
    async def list_runs(project: str = "default", name: str = "",
                        state: str = "", sort: int = 1, last: int = 0,
                        iter: int = 0, page: int = 0, page_size: int = 0,
                        page_token: str = "",
                        partition_by: str = "",
                        rows_per_partition: int = 1,
                        partition_sort_by: str = "updated",
                        partition_order: str = "desc",
                        max_partitions: int = 0,
                        uid: typing.List[str] = Query(None),
                        label: typing.List[str] = Query(None)):
        if page_token or (page_size and not page):
            items, token = db.paginated_list(
                "list_runs", page_token=page_token or None,
                page_size=page_size or 20,
                **({} if page_token else dict(
                    name=name, project=project, labels=label,
                    state=state or None, iter=bool(iter),
                    partition_by=partition_by or None,
                    rows_per_partition=rows_per_partition,
                    partition_sort_by=partition_sort_by,
                    partition_order=partition_order)))
            return {"runs": items,
                    "pagination": {"page_token": token,
                                   "page_size": page_size}}
        runs = db.list_runs(name=name, uid=uid, project=project,
                            labels=label, state=state or None,
                            sort=bool(sort), last=last, iter=bool(iter),
                            partition_by=partition_by or None,
                            rows_per_partition=rows_per_partition,
                            partition_sort_by=partition_sort_by,
                            partition_order=partition_order,
                            max_partitions=max_partitions)
        total = len(runs)
        if page_size:
            start = max(page - 1, 0) * page_size
            runs = runs[start:start + page_size]
            return {"runs": runs,
                    "pagination": {"page": max(page, 1),
                                   "page_size": page_size,
                                   "total": total}}
        return {"runs": runs}

    # ------------------------------------------------------------ logs
    @app.post("/api/v1/log/{project}/{uid}")
    async def store_log(project: str, uid: str, request: Request,
                        append: int = 0):
        body = await request.body()
        db.store_log(uid, project, body, append=bool(append))
        return {}

    @app.get("/api/v1/log/{project}/{uid}")
    async def get_log(project: str, uid: str, offset: int = 0,
                      size: int = 0):
        state, data = db.get_log(uid, project, offset=offset, size=size)
        return Response(content=data,
                        headers={"x-mlrun-run-state": state or ""},
                        media_type="application/octet-stream")

    # -------------------------------------------------------- artifacts
    @app.post("/api/v1/artifact/{project}/{key:path}")
    async def store_artifact(project: str, key: str, request: Request,
                             tag: str = "", iter: int = 0, tree: str = "",
                             uid: str = ""):
        body = await request.json()
        return db.store_artifact(key, body, uid=uid or None, iter=iter,
                                 tag=tag, project=project,
                                 tree=tree or None)

    @app.get("/api/v1/artifact/{project}/{key:path}")
    async def read_artifact(project: str, key: str, tag: str = "",
                            iter: int = 0, tree: str = "", uid: str = ""):
        return {"data": db.read_artifact(key, tag=tag, iter=iter,
                                         project=project,
                                         tree=tree or None,
                                         uid=uid or None)}

    @app.delete("/api/v1/artifact/{project}/{key:path}")
    async def del_artifact(project: str, key: str, tag: str = ""):
        db.del_artifact(key, tag=tag, project=project)
        return {}

    @app.get("/api/v1/artifacts")
    async def list_artifacts(project: str = "default", name: str = "",
                             tag: str = "", kind: str = "", tree: str = "",
                             category: str = "", since: str = "",
                             until: str = "", limit: int = 0,
                             offset: int = 0, page_token: str = "",
                             page_size: int = 0,
                             label: typing.List[str] = Query(None)):
        """Artifacts listing with the v2 query surface (reference
        endpoints/artifacts_v2.py): category/time filters + token
        pagination."""
        if page_token or page_size:
            items, token = db.paginated_list(
                "list_artifacts", page_token=page_token or None,
                page_size=page_size or 20,
                **({} if page_token else dict(
                    name=name, project=project, tag=tag, labels=label,
                    kind=kind or None, category=category or None,
                    tree=tree or None, since=since or None,
                    until=until or None)))
            return {"artifacts": items,
                    "pagination": {"page_token": token,
                                   "page_size": page_size}}
        artifacts = db.list_artifacts(name=name, project=project, tag=tag,
                                      labels=label, kind=kind or None,
                                      category=category or None,
                                      tree=tree or None,
                                      since=since or None,
                                      until=until or None,
                                      limit=limit, offset=offset)
        return {"artifacts": artifacts}

    # -------------------------------------------------------- functions
    @app.post("/api/v1/func/{project}/{name}")
    async def store_function(project: str, name: str, request: Request,
                             tag: str = "", versioned: int = 0):
        body = await request.json()
        hash_key = db.store_function(body, name, project, tag=tag,
                                     versioned=bool(versioned))
        return {"hash_key": hash_key}

    @app.get("/api/v1/func/{project}/{name}")
    async def get_function(project: str, name: str, tag: str = "",
                           hash_key: str = ""):
        return {"func": db.get_function(name, project, tag=tag,
                                        hash_key=hash_key)}

    @app.delete("/api/v1/func/{project}/{name}")
    async def delete_function(project: str, name: str):
        db.delete_function(name, project)
        return {}

    @app.get("/api/v1/funcs")
    async def list_functions(project: str = "default", name: str = "",
                             tag: str = "",
                             label: typing.List[str] = Query(None)):
        return {"funcs": db.list_functions(name=name or None,
                                           project=project, tag=tag,
                                           labels=label)}

    # --------------------------------------------------------- projects
    @app.post("/api/v1/projects")
    async def create_project(request: Request):
        body = await request.json()
        return db.create_project(body)

    @app.get("/api/v1/projects/{name}")
    async def get_project(name: str):
        return db.get_project(name)

    @app.put("/api/v1/projects/{name}")
    async def store_project(name: str, request: Request):
        body = await request.json()
        return db.store_project(name, body)

    @app.get("/api/v1/projects")
    async def list_projects():
        return {"projects": db.list_projects()}

    @app.delete("/api/v1/projects/{name}")
    async def delete_project(name: str, deletion_strategy: str = ""):
        db.delete_project(name, deletion_strategy or None)
        return {}

    # -------------------------------------------------------- schedules
    @app.post("/api/v1/projects/{project}/schedules")
    async def create_schedule(project: str, request: Request):
        import pydantic

        from ..common.schemas import ScheduleSchema

        body = await request.json()
        try:  # wire-schema validation (pydantic; 422 on bad payloads)
            schema = ScheduleSchema(**body)
            body = schema.model_dump() \
                if hasattr(schema, "model_dump") else schema.dict()
        except pydantic.ValidationError as exc:
            raise HTTPException(status_code=422, detail=str(exc))
        db.create_schedule(project, body)
        if app.state.scheduler:
            app.state.scheduler.reload()
        return {}

    @app.put("/api/v1/projects/{project}/schedules/{name}")
    async def update_schedule(project: str, name: str, request: Request):
        body = await request.json()
        db.update_schedule(project, name, body)
        if app.state.scheduler:
            app.state.scheduler.reload()
        return {}

    @app.get("/api/v1/projects/{project}/schedules/{name}")
    async def get_schedule(project: str, name: str):
        return db.get_schedule(project, name)

    @app.get("/api/v1/projects/{project}/schedules")
    async def list_schedules(project: str, name: str = ""):
        return {"schedules": db.list_schedules(project, name=name)}

    @app.delete("/api/v1/projects/{project}/schedules/{name}")
    async def delete_schedule(project: str, name: str):
        db.delete_schedule(project, name)
        if app.state.scheduler:
            app.state.scheduler.reload()
        return {}

    @app.post("/api/v1/projects/{project}/schedules/{name}/invoke")
    async def invoke_schedule(project: str, name: str):
        scheduler = app.state.scheduler or Scheduler(db)
        await asyncio.to_thread(scheduler.invoke, project, name)
        return {}

    # ----------------------------------------------------- feature store
    @app.put("/api/v1/projects/{project}/feature-sets/{name}")
    async def store_feature_set(project: str, name: str, request: Request,
                                tag: str = ""):
        body = await request.json()
        from ..common.schemas import validate_feature_set

        _validate_or_422(validate_feature_set, body)
        return db.store_feature_set(body, name=name, project=project,
                                    tag=tag or None)

    @app.get("/api/v1/projects/{project}/feature-sets/{name}")
    async def get_feature_set(project: str, name: str, tag: str = ""):
        return db.get_feature_set(name, project, tag=tag or None)

    @app.get("/api/v1/projects/{project}/feature-sets")
    async def list_feature_sets(project: str, name: str = ""):
        return {"feature_sets": db.list_feature_sets(project,
                                                     name=name or None)}

    @app.delete("/api/v1/projects/{project}/feature-sets/{name}")
    async def delete_feature_set(project: str, name: str):
        db.delete_feature_set(name, project)
        return {}

    @app.put("/api/v1/projects/{project}/feature-vectors/{name}")
    async def store_feature_vector(project: str, name: str, request: Request,
                                   tag: str = ""):
        body = await request.json()
        return db.store_feature_vector(body, name=name, project=project,
                                       tag=tag or None)

    @app.get("/api/v1/projects/{project}/feature-vectors/{name}")
    async def get_feature_vector(project: str, name: str, tag: str = ""):
        return db.get_feature_vector(name, project, tag=tag or None)

    @app.get("/api/v1/projects/{project}/feature-vectors")
    async def list_feature_vectors(project: str, name: str = ""):
        return {"feature_vectors": db.list_feature_vectors(
            project, name=name or None)}

    @app.delete("/api/v1/projects/{project}/feature-vectors/{name}")
    async def delete_feature_vector(project: str, name: str):
        db.delete_feature_vector(name, project)
        return {}

    # --------------------------------------------------- model endpoints
    @app.put("/api/v1/projects/{project}/model-endpoints/{endpoint_id}")
    async def store_model_endpoint(project: str, endpoint_id: str,
                                   request: Request):
        body = await request.json()
        from ..common.schemas import validate_model_endpoint

        _validate_or_422(validate_model_endpoint, body)
        db.store_model_endpoint(project, endpoint_id, body)
        return {}

    @app.get("/api/v1/projects/{project}/model-endpoints/{endpoint_id}")
    async def get_model_endpoint(project: str, endpoint_id: str):
        return db.get_model_endpoint(project, endpoint_id)

    @app.get("/api/v1/projects/{project}/model-endpoints")
    async def list_model_endpoints(project: str, model: str = "",
                                   function: str = ""):
        return {"endpoints": db.list_model_endpoints(
            project, model=model or None, function=function or None)}

    @app.delete("/api/v1/projects/{project}/model-endpoints/{endpoint_id}")
    async def delete_model_endpoint(project: str, endpoint_id: str):
        db.delete_model_endpoint(project, endpoint_id)
        return {}

    @app.get("/api/v1/projects/{project}/model-endpoints/{endpoint_id}"
             "/metrics")
    async def model_endpoint_metrics(project: str, endpoint_id: str):
        """Time series of window stats from the monitoring TSDB
        (reference: grafana-proxy / TSDB connectors)."""
        from ..model_monitoring import get_stream_processor

        processor = get_stream_processor(project)
        series = processor.tsdb_series(endpoint_id)
        return {"endpoint_id": endpoint_id,
                "series": [{"time": t, "stats": stats}
                           for t, stats in series],
                "current": processor.endpoint_stats(endpoint_id)}

    @app.post("/api/v1/projects/{project}/alerts/{name}/reset")
    async def reset_alert(project: str, name: str):
        db.reset_alert_state(project, name)
        return {}

    # ------------------------------------------------------------ hub
    @app.get("/api/v1/hub/sources")
    async def hub_sources():
        from ..hub import list_hub_sources

        return {"sources": list_hub_sources()}

    @app.get("/api/v1/hub/sources/{source}/items")
    async def hub_catalog(source: str):
        from ..hub import get_hub_catalog

        return {"catalog": get_hub_catalog(source)}

    @app.get("/api/v1/hub/sources/{source}/items/{name}")
    async def hub_item(source: str, name: str):
        import yaml as _yaml

        from ..hub import get_hub_catalog

        for item in get_hub_catalog(source):
            if item["name"] == name:
                with open(item["path"]) as stream:
                    return {"item": item,
                            "spec": _yaml.safe_load(stream)}
        raise HTTPException(status_code=404,
                            detail=f"hub item {name} not found")

    # -------------------------------------------------------- secrets
    @app.post("/api/v1/projects/{project}/secrets")
    async def store_project_secrets(project: str, body: dict):
        db.store_project_secrets(project, body.get("secrets") or {})
        return {}

    @app.get("/api/v1/projects/{project}/secret-keys")
    async def list_secret_keys(project: str):
        return {"secret_keys": db.list_project_secret_keys(project)}

    @app.delete("/api/v1/projects/{project}/secrets")
    async def delete_project_secrets(project: str,
                                     secrets: str = ""):
        keys = [k for k in secrets.split(",") if k] or None
        db.delete_project_secrets(project, keys)
        return {}

    # ---------------------------------------------- datastore profiles
    @app.put("/api/v1/projects/{project}/datastore-profiles")
    async def store_datastore_profile(project: str, body: dict):
        """Named datastore configs (reference datastore_profile.py —
        connection parameters for s3/redis/... targets, minus the
        secrets which live in the project secret store)."""
        from ..common.schemas import validate_datastore_profile

        _validate_or_422(validate_datastore_profile, body)
        db.store_datastore_profile(project, body)
        return body

    @app.get("/api/v1/projects/{project}/datastore-profiles")
    async def list_datastore_profiles(project: str):
        return {"profiles": db.list_datastore_profiles(project)}

    @app.get("/api/v1/projects/{project}/datastore-profiles/{name}")
    async def get_datastore_profile(project: str, name: str):
        return db.get_datastore_profile(project, name)

    @app.delete("/api/v1/projects/{project}/datastore-profiles/{name}")
    async def delete_datastore_profile(project: str, name: str):
        db.delete_datastore_profile(project, name)
        return {}

    # ------------------------------------------------ alert templates
    @app.put("/api/v1/alert-templates/{name}")
    async def store_alert_template(name: str, body: dict):
        """System-wide reusable alert templates (reference
        alert_template.py); instantiate by merging into an
        AlertConfig."""
        body["template_name"] = name
        db.store_alert_template(name, body)
        return body

    @app.get("/api/v1/alert-templates")
    async def list_alert_templates():
        return {"templates": db.list_alert_templates()}

    @app.get("/api/v1/alert-templates/{name}")
    async def get_alert_template(name: str):
        return db.get_alert_template(name)

    @app.delete("/api/v1/alert-templates/{name}")
    async def delete_alert_template(name: str):
        db.delete_alert_template(name)
        return {}

    # ----------------------------------------------------------- tags
    @app.get("/api/v1/projects/{project}/tags")
    async def list_tags(project: str, key: str = ""):
        return {"tags": db.list_artifact_tags(project, key)}

    @app.put("/api/v1/projects/{project}/tags/{tag}")
    async def tag_objects(project: str, tag: str, body: dict):
        """Attach a tag to artifact versions (reference tags.py):
        body = {"identifiers": [{"key", "tree"|"uid", "iter"?}]}."""
        for ident in body.get("identifiers") or []:
            db.tag_artifact(project, ident.get("key"),
                            ident.get("tree") or ident.get("uid"),
                            tag, ident.get("iter", 0))
        return {}

    @app.delete("/api/v1/projects/{project}/tags/{tag}")
    async def delete_tag(project: str, tag: str, key: str):
        db.delete_artifact_tag(project, key, tag)
        return {}

    # ---------------------------------------------------------- files
    def _authorize_file_path(path: str):
        """Restrict /files and /filestat to configured data prefixes
        (reference server routes files.py through per-path
        authorization; here the allowlist is base_dir + artifact_path +
        the service's own dirpath + any ``httpdb.files_allowed_paths``
        entries)."""
        from ..config import config as _cfg

        allowed = [p for p in (
            _cfg.httpdb.files_allowed_paths or "").split(",") if p]
        for candidate in (_cfg.base_dir, _cfg.artifact_path,
                          _cfg.httpdb.dirpath, _cfg.httpdb.logs_path):
            if candidate:
                allowed.append(candidate)
        # non-file schemes (memory://, store://) resolve through their
        # own stores and never touch the host filesystem directly
        if "://" in path and not path.startswith("file://"):
            return
        local = path[len("file://"):] if path.startswith("file://") else path
        real = os.path.realpath(local)
        for prefix in allowed:
            if "://" in prefix:
                if path.startswith(prefix):
                    return
                continue
            rp = os.path.realpath(os.path.expanduser(prefix))
            if real == rp or real.startswith(rp.rstrip("/") + "/"):
                return
        raise HTTPException(
            status_code=403,
            detail=f"path {path!r} is outside the allowed data prefixes "
                   f"(configure httpdb.files_allowed_paths)")

    @app.get("/api/v1/files")
    async def get_file(path: str, size: int = 0, offset: int = 0):
        """Serve object bytes through the datastore layer (reference
        files.py GET /files?path=...)."""
        from ..datastore import store_manager

        _authorize_file_path(path)
        try:
            data = store_manager.object(path).get(size=size or None,
                                                  offset=offset)
        except FileNotFoundError:
            raise HTTPException(status_code=404,
                                detail=f"{path} not found")
        if isinstance(data, str):
            data = data.encode()
        return Response(content=data,
                        media_type="application/octet-stream")

    @app.get("/api/v1/filestat")
    async def file_stat(path: str):
        from ..datastore import store_manager

        _authorize_file_path(path)
        try:
            return store_manager.object(path).stat()
        except FileNotFoundError:
            raise HTTPException(status_code=404,
                                detail=f"{path} not found")

    # ------------------------------------------------------ workflows
    @app.get("/api/v1/projects/{project}/workflows")
    async def list_workflows(project: str):
        body = db.get_project(project)
        spec = (body or {}).get("spec", {})
        return {"workflows": spec.get("workflows", [])}

    @app.post("/api/v1/projects/{project}/workflows/{name}/submit")
    async def submit_workflow(project: str, name: str,
                              request: Request):
        """Server-side workflow execution (reference _RemoteRunner /
        workflows endpoint): loads the stored project and runs the
        named workflow in a background task."""
        import threading

        try:
            payload = await request.json()
        except Exception:
            payload = {}
        body = db.get_project(project)
        if not body:
            raise HTTPException(status_code=404,
                                detail=f"project {project} not found")
        from ..model import generate_uid
        from ..projects import MlrunProject

        proj = MlrunProject.from_dict(body)
        task_name = f"workflow-{name}-{generate_uid()[:8]}"
        db.store_background_task(project, {
            "name": task_name, "status": {"state": "running"}})

        def run_it():
            try:
                status = proj.run(
                    name=name,
                    arguments=payload.get("arguments") or {})
                db.store_background_task(project, {
                    "name": task_name,
                    "status": {"state": "succeeded"
                               if status.state == "completed"
                               else "failed"}})
            except Exception as exc:
                logger.error("workflow run failed", error=str(exc))
                db.store_background_task(project, {
                    "name": task_name,
                    "status": {"state": "failed"}})

        threading.Thread(target=run_it, daemon=True,
                         name=task_name).start()
        return {"name": name, "project": project,
                "background_task": task_name}

    # ------------------------------------------------------ pipelines
    @app.get("/api/v1/projects/{project}/pipelines")
    async def list_pipelines(project: str):
        from ..projects.pipelines import list_pipeline_runs

        return {"runs": [r.to_dict()
                         for r in list_pipeline_runs(project)]}

    @app.get("/api/v1/projects/{project}/pipelines/{run_id}")
    async def get_pipeline_run(project: str, run_id: str):
        from ..projects.pipelines import get_pipeline

        try:
            return get_pipeline(run_id).to_dict()
        except Exception:
            raise HTTPException(status_code=404,
                                detail=f"pipeline {run_id} not found")

    @app.get("/metrics")
    async def prometheus_metrics():
        """Prometheus scrape endpoint for the API service."""
        try:
            from prometheus_client import (CONTENT_TYPE_LATEST,
                                           generate_latest)
            from fastapi import Response as _Resp

            return _Resp(generate_latest(),
                         media_type=CONTENT_TYPE_LATEST)
        except ImportError:
            return {"error": "prometheus_client not installed"}

    @app.get("/api/v1/monitoring/memory")
    async def memory_report():
        """Process + GPU memory report (reference:
        server/api/utils/memory_reports.py)."""
        import resource

        import psutil

        proc = psutil.Process()
        report = {
            "rss_bytes": proc.memory_info().rss,
            "vms_bytes": proc.memory_info().vms,
            "peak_rss_bytes": resource.getrusage(
                resource.RUSAGE_SELF).ru_maxrss * 1024,
            "open_files": len(proc.open_files()),
            "threads": proc.num_threads(),
            "gpus": [],
        }
        try:
            import torch

            if torch.cuda.is_available():
                for i in range(torch.cuda.device_count()):
                    free, total = torch.cuda.mem_get_info(i)
                    report["gpus"].append({
                        "device": i, "free_bytes": free,
                        "total_bytes": total,
                        "allocated_bytes":
                            torch.cuda.memory_allocated(i),
                        "reserved_bytes":
                            torch.cuda.memory_reserved(i),
                    })
        except Exception:
            pass
        return report

    # ------------------------------------------------------------ alerts
    @app.put("/api/v1/projects/{project}/alerts/{name}")
    async def store_alert(project: str, name: str, request: Request):
        body = await request.json()
        from ..common.schemas import validate_alert_config

        _validate_or_422(validate_alert_config, body)
        db.store_alert_config(project, name, body)
        return {}

    @app.get("/api/v1/projects/{project}/alerts/{name}")
    async def get_alert(project: str, name: str):
        return db.get_alert_config(project, name)

    @app.get("/api/v1/projects/{project}/alerts")
    async def list_alerts(project: str):
        return {"alerts": db.list_alert_configs(project)}

    @app.delete("/api/v1/projects/{project}/alerts/{name}")
    async def delete_alert(project: str, name: str):
        db.delete_alert_config(project, name)
        return {}

    @app.post("/api/v1/projects/{project}/events/{name}")
    async def post_event(project: str, name: str, request: Request):
        body = await request.json()
        from .events import process_event

        results = await asyncio.to_thread(process_event, project, name,
                                          body, db)
        return {"alerts_fired": results}

    # -------------------------------------------------- background tasks
    @app.get("/api/v1/projects/{project}/background-tasks/{name}")
    async def get_background_task(project: str, name: str):
        return db.get_background_task(project, name)

    @app.get("/api/v1/projects/{project}/background-tasks")
    async def list_background_tasks(project: str):
        return {"background_tasks": db.list_background_tasks(project)}

    # -------------------------------------------------- runtime resources
    @app.get("/api/v1/projects/{project}/runtime-resources")
    async def runtime_resources(project: str):
        """Node-local analog of the reference's runtime-resources
        listing: live GPU leases + running runs."""
        from ..parallel.scheduler import get_gpu_allocator

        allocator = get_gpu_allocator()
        running = db.list_runs(project=project, state="running")
        return {
            "gpu": {"total": allocator.total,
                    "in_use": {str(k): v for k, v in
                               allocator.usage().items()},
                    "available": allocator.available()},
            "runs": [{"uid": r.get("metadata", {}).get("uid"),
                      "name": r.get("metadata", {}).get("name")}
                     for r in running],
        }

    # ------------------------------------------------------------ submit
    # ------------------------------------------------ function deploy
    def _load_function(project: str, name: str, tag: str = ""):
        from ..run import new_function

        body = db.get_function(name, project, tag=tag)
        return new_function(runtime=body)

    @app.post("/api/v1/build/function")
    async def build_function(request: Request):
        """Server-side function build (reference build/function +
        kaniko): node-locally a venv/wheel build via utils/builder,
        tracked as a background task."""
        payload = await request.json()
        func = payload.get("function") or {}
        project = func.get("metadata", {}).get("project", "default")
        name = func.get("metadata", {}).get("name", "")
        with_mlrun = payload.get("with_mlrun", False)
        task_name = f"build-{name}"
        db.store_background_task(project, {
            "metadata": {"name": task_name, "project": project},
            "status": {"state": "running"}})

        def _build():
            try:
                from ..run import new_function
                from ..utils.builder import build_runtime

                fn = new_function(runtime=func)
                build_runtime(fn, with_mlrun=with_mlrun, install=False)
                db.store_function(fn.to_dict(), name, project)
                db.store_background_task(project, {
                    "metadata": {"name": task_name, "project": project},
                    "status": {"state": "succeeded"}})
            except Exception as exc:
                db.store_background_task(project, {
                    "metadata": {"name": task_name, "project": project},
                    "status": {"state": "failed", "error": str(exc)}})

        import threading as _threading

        _threading.Thread(target=_build, daemon=True).start()
        return {"data": func, "ready": False,
                "background_task": task_name}

    @app.get("/api/v1/build/status")
    async def build_status(project: str = "default", name: str = "",
                           tag: str = ""):
        task = db.get_background_task(project, f"build-{name}")
        state = (task.get("status") or {}).get("state", "unknown")
        return {"ready": state == "succeeded", "state": state,
                "error": (task.get("status") or {}).get("error", "")}

    @app.post("/api/v1/start/function")
    async def start_function(request: Request):
        """Deploy a STORED function as a live local host (the node-
        local analog of the reference's nuclio deploy): serving/remote
        kinds get a FastAPI host; the address lands in the function
        status."""
        payload = await request.json()
        func = payload.get("function") or {}
        project = func.get("metadata", {}).get("project", "default")
        name = func.get("metadata", {}).get("name", "")

        def _deploy():
            from ..run import new_function

            fn = new_function(runtime=func)
            address = fn.deploy()
            body = fn.to_dict()
            body.setdefault("status", {})["address"] = address
            body["status"]["state"] = "ready"
            db.store_function(body, name, project)
            return address

        address = await asyncio.to_thread(_deploy)
        return {"data": {"address": address, "name": name,
                         "state": "ready"}}

    @app.get("/api/v1/projects/{project}/functions/{name}/status")
    async def function_status(project: str, name: str, tag: str = ""):
        body = db.get_function(name, project, tag=tag)
        return {"status": body.get("status", {})}

    # ------------------------------------------------- grafana proxy
    @app.get("/api/v1/grafana-proxy/model-endpoints")
    async def grafana_proxy_health():
        return {"status": "ok"}  # datasource test endpoint

    @app.post("/api/v1/grafana-proxy/model-endpoints/search")
    async def grafana_search(body: dict = None):
        """Metric-name discovery (reference grafana_proxy.py /search):
        returns the monitoring metric names recorded per endpoint."""
        from ..model_monitoring import get_stream_processor

        names = set()
        for project in [p.get("metadata", {}).get("name", "default")
                        for p in db.list_projects()] + ["default"]:
            processor = get_stream_processor(project)
            for endpoint_id in processor._endpoint_ids:
                names.add(endpoint_id)
        return sorted(names)

    @app.post("/api/v1/grafana-proxy/model-endpoints/query")
    async def grafana_query(body: dict):
        """Grafana simple-json timeseries query (reference
        grafana_proxy.py): targets name endpoint ids; datapoints come
        from the in-memory monitoring TSDB snapshots."""
        from ..model_monitoring import get_stream_processor

        out = []
        project = body.get("project", "default")
        processor = get_stream_processor(project)
        for target in body.get("targets") or []:
            endpoint_id = target.get("target") if isinstance(
                target, dict) else str(target)
            metric = (target.get("metric", "count")
                      if isinstance(target, dict) else "count")
            window = str(target.get("window", "300")) if isinstance(
                target, dict) else "300"
            series = []
            for ts, snapshot in processor.tsdb_series(endpoint_id):
                value = (snapshot.get(window) or {}).get(metric, 0)
                series.append([value, int(ts * 1000)])
            out.append({"target": f"{endpoint_id}.{metric}.{window}",
                        "datapoints": series})
        return out

    # ---------------------------------------------- project summaries
    @app.get("/api/v1/project-summaries")
    async def list_project_summaries():
        """Per-project entity counts (reference projects summaries
        endpoint + the periodic summary loop, server main.py:630 —
        node-locally computed on demand; counts are indexed SQL)."""
        return {"project_summaries": db.list_project_summaries()}

    @app.get("/api/v1/project-summaries/{project}")
    async def get_project_summary(project: str):
        db.get_project(project)  # 404 on unknown project
        return db.compute_project_summary(project)

    # ------------------------------------------ migrations/operations
    @app.post("/api/v1/operations/migrations")
    async def trigger_migrations():
        """Apply pending schema migrations (reference
        operations.py trigger_migrations)."""
        return db.trigger_migrations()

    # ------------------------------------------------ features search
    @app.get("/api/v1/projects/{project}/features")
    async def list_features(project: str, name: str = "",
                            entity: typing.List[str] = Query(None),
                            label: typing.List[str] = Query(None)):
        return {"features": db.list_features(
            project, name=name or None, entities=entity, labels=label)}

    @app.get("/api/v1/projects/{project}/entities")
    async def list_entities(project: str, name: str = "",
                            label: typing.List[str] = Query(None)):
        return {"entities": db.list_entities(project, name=name or None,
                                             labels=label)}

    # ------------------------------------------------- patch variants
    @app.patch("/api/v1/projects/{name}")
    async def patch_project(name: str, body: dict):
        current = db.get_project(name)
        from ..utils import update_in

        for key, value in (body or {}).items():
            if isinstance(value, dict) and isinstance(
                    current.get(key), dict):
                current[key].update(value)
            else:
                update_in(current, key, value)
        db.store_project(name, current)
        return current

    @app.patch("/api/v1/projects/{project}/feature-sets/{name}")
    async def patch_feature_set(project: str, name: str, body: dict,
                                tag: str = "latest"):
        current = db.get_feature_set(name, project, tag)
        _deep_update(current, body or {})
        db.store_feature_set(current, name, project, tag=tag)
        return current

    @app.patch("/api/v1/projects/{project}/feature-vectors/{name}")
    async def patch_feature_vector(project: str, name: str, body: dict,
                                   tag: str = "latest"):
        current = db.get_feature_vector(name, project, tag)
        _deep_update(current, body or {})
        db.store_feature_vector(current, name, project, tag=tag)
        return current

    @app.patch("/api/v1/projects/{project}/model-endpoints/{endpoint_id}")
    async def patch_model_endpoint(project: str, endpoint_id: str,
                                   body: dict):
        current = db.get_model_endpoint(project, endpoint_id)
        _deep_update(current, body or {})
        db.store_model_endpoint(project, endpoint_id, current)
        return current

    # ------------------------------------------------------ bulk dels
    @app.delete("/api/v1/runs")
    async def del_runs(project: str = "", name: str = "",
                       state: str = ""):
        db.del_runs(name=name, project=project, state=state or None)
        return {}

    @app.delete("/api/v1/artifacts")
    async def del_artifacts(project: str = "", name: str = "",
                            tag: str = ""):
        db.del_artifacts(name=name, project=project, tag=tag)
        return {}

    # --------------------------------------------------- hub source CRUD
    @app.put("/api/v1/hub/sources/{name}")
    async def store_hub_source(name: str, body: dict):
        """Register a hub source (reference hub.py store_source): body
        {"path": ..., "order": N}; registered in the hub registry AND
        persisted."""
        from .. import hub

        path = (body.get("spec", body) or {}).get("path", "")
        order = int((body.get("spec", body) or {}).get("order", -1))
        hub.add_hub_source(name, path, order)
        db._execute(
            "INSERT OR REPLACE INTO hub_sources "
            "(name, idx, updated, body) VALUES (?,?,?,?)",
            (name, order, now_iso(),
             json.dumps(body, default=str)))
        return body

    @app.get("/api/v1/hub/sources/{name}")
    async def get_hub_source(name: str):
        rows = db._query("SELECT body FROM hub_sources WHERE name=?",
                         (name,))
        if rows:
            return json.loads(rows[0]["body"])
        from ..hub import list_hub_sources

        for source in list_hub_sources():
            if source["name"] == name:
                return source
        raise HTTPException(status_code=404,
                            detail=f"hub source {name} not found")

    @app.delete("/api/v1/hub/sources/{name}")
    async def delete_hub_source(name: str):
        from .. import hub

        hub._sources.pop(name, None)
        db._execute("DELETE FROM hub_sources WHERE name=?", (name,))
        return {}

    # --------------------------------------------------- api gateways
    @app.put("/api/v1/projects/{project}/api-gateways/{name}")
    async def store_api_gateway(project: str, name: str, body: dict):
        return db.store_api_gateway(project, name, body)

    @app.get("/api/v1/projects/{project}/api-gateways/{name}")
    async def get_api_gateway(project: str, name: str):
        return db.get_api_gateway(project, name)

    @app.get("/api/v1/projects/{project}/api-gateways")
    async def list_api_gateways(project: str):
        return {"api_gateways": db.list_api_gateways(project)}

    @app.delete("/api/v1/projects/{project}/api-gateways/{name}")
    async def delete_api_gateway(project: str, name: str):
        db.delete_api_gateway(project, name)
        return {}

    # -------------------------------------------------- notifications
    @app.put("/api/v1/projects/{project}/runs/{uid}/notifications")
    async def set_run_notifications(project: str, uid: str, body: dict):
        """Replace a run's notification list (reference
        set_run_notifications)."""
        run = db.read_run(uid, project)
        run.setdefault("spec", {})["notifications"] = \
            body.get("notifications", [])
        db.store_run(run, uid, project)
        return {}

    @app.put("/api/v1/projects/{project}/schedules/{name}/notifications")
    async def set_schedule_notifications(project: str, name: str,
                                         body: dict):
        sched = db.get_schedule(project, name)
        task = sched.get("task") or {}
        task.setdefault("spec", {})["notifications"] = \
            body.get("notifications", [])
        db.update_schedule(project, name, {"task": task})
        return {}

    @app.post("/api/v1/submit_job")
    async def submit_job(request: Request):
        body = await request.json()
        task = body.get("task", body)
        schedule = body.get("schedule")
        from ..model import RunObject

        run = RunObject.from_dict(task)
        result = await asyncio.to_thread(db.submit_job, run, schedule)
        return result if "data" in result else {"data": result}

    return app


def check_stuck_runs(db: SQLRunDB, now=None) -> list:
    """One monitoring sweep: abort runs stuck past the configured
    state thresholds (parity: reference runtime_handlers/base.py:1387
    state-threshold logic).  Returns the aborted run uids."""
    import datetime

    from ..utils import parse_time

    thresholds = config.runs.state_thresholds.to_dict()
    now = now or datetime.datetime.now(datetime.timezone.utc)
    aborted = []
    projects = [p.get("metadata", {}).get("name", "default")
                for p in db.list_projects()] or ["default"]
    for project in projects:
        for state in ("running", "pending"):
            limit = thresholds.get(state)
            if not limit:
                continue
            for run in db.list_runs(project=project, state=state):
                started = run.get("status", {}).get("start_time")
                if not started:
                    continue
                start_dt = parse_time(started)
                if start_dt is None:
                    continue
                if start_dt.tzinfo is None:
                    start_dt = start_dt.replace(
                        tzinfo=datetime.timezone.utc)
                age = (now - start_dt).total_seconds()
                if age > float(limit):
                    uid = run.get("metadata", {}).get("uid")
                    logger.warning("aborting run past state threshold",
                                   uid=uid, state=state, age=age)
                    db.abort_run(
                        uid, project,
                        status_text=f"aborted by monitor: {state} "
                                    f"for {int(age)}s > {limit}s")
                    aborted.append(uid)
    return aborted


def sweep_service_caches(db: SQLRunDB, now=None) -> dict:
    """Periodic hygiene (reference cleanup loops, server main.py:599):
    expire idle pagination-cache rows and terminal background tasks
    past their TTL.  Returns what was swept (tested directly)."""
    import datetime

    now = now or datetime.datetime.now()
    ttl = int(config.pagination.cache_ttl_seconds)
    cutoff = (now - datetime.timedelta(seconds=ttl)).isoformat()
    db.clean_pagination_cache(older_than_iso=cutoff)
    task_ttl = int(config.background_tasks.ttl_seconds)
    task_cutoff = (now - datetime.timedelta(
        seconds=task_ttl)).isoformat()
    swept = []
    projects = [r[0] for r in db._query(
        "SELECT DISTINCT project FROM background_tasks")]
    for project in projects:
        for task in db.list_background_tasks(project):
            state = (task.get("status") or {}).get("state", "")
            updated = (task.get("metadata") or {}).get("updated") or \
                (task.get("status") or {}).get("updated", "")
            if state in ("succeeded", "failed") and updated and \
                    updated < task_cutoff:
                name = (task.get("metadata") or {}).get("name", "")
                db._execute(
                    "DELETE FROM background_tasks WHERE project=? AND "
                    "name=?", (project, name))
                swept.append(f"{project}/{name}")
    return {"background_tasks": swept, "pagination_cutoff": cutoff}


async def _runs_monitor(db: SQLRunDB):
    """Periodic run monitoring loop (reference main.py:608) + service
    cache hygiene sweeps."""
    interval = int(config.runs.monitoring_interval)
    ticks = 0
    while True:
        try:
            await asyncio.sleep(interval)
            check_stuck_runs(db)
            ticks += 1
            if ticks % 10 == 0:  # hygiene at 1/10th cadence
                sweep_service_caches(db)
        except asyncio.CancelledError:
            return
        except Exception as exc:
            logger.warning("runs monitor iteration failed", error=str(exc))


def serve(port: int = None, db=None):
    """Run the API service (blocking)."""
    import uvicorn

    app = create_app(db)
    uvicorn.run(app, host="0.0.0.0",
                port=port or int(config.httpdb.port), log_level="warning")
