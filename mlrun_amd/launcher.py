# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Launchers: the strategy objects that take (function, run template)
and execute them — locally in-process, or submitted to the API service.

Parity target: reference mlrun/launcher (BaseLauncher base.py:35,
ClientLocalLauncher local.py:29, ClientRemoteLauncher remote.py:33) and
server/api/launcher.py:40 ServerSideLauncher.  Hyperparameter fan-out
(_run_many, reference runtimes/base.py:508) lives here, with a
multi-worker parallel mode replacing the reference's Dask-based one.
"""

import concurrent.futures
import os
import typing

from .config import config
from .execution import MLClientCtx
from .model import RunObject, RunStates, generate_uid
from .runtimes.generators import get_generator, selector
from .utils import logger


class BaseLauncher:
    def launch(self, runtime, run: RunObject, schedule=None, watch=True,
               **kwargs) -> RunObject:
        raise NotImplementedError

    @staticmethod
    def _enrich_run(runtime, run: RunObject):
        if not run.metadata.uid:
            run.metadata.uid = generate_uid()
        if not run.spec.output_path:
            run.spec.output_path = config.artifact_path or os.path.join(
                config.base_dir, "artifacts")
        run.spec.output_path = run.spec.output_path.replace(
            "{{project}}", run.metadata.project or "default")
        run.spec.output_path = run.spec.output_path.replace(
            "{{run.uid}}", run.metadata.uid)

    @staticmethod
    def _push_notifications(run: RunObject):
        from .utils.notifications import NotificationPusher

        if run.spec.notifications:
            try:
                NotificationPusher([run]).push()
            except Exception as exc:
                logger.warning("failed pushing notifications", error=str(exc))


class ClientLocalLauncher(BaseLauncher):
    """Execute the run in this process (or a subprocess for command
    mode), tracking state in the run DB."""

    def __init__(self, db=None):
        self._db = db

    def _get_db(self):
        if self._db is None:
            from .db import get_run_db

            self._db = get_run_db()
        return self._db

    def launch(self, runtime, run: RunObject, schedule=None, watch=True,
               **kwargs) -> RunObject:
        if schedule:
            db = self._get_db()
            runtime._store_function(run, db)
            return self._schedule(db, runtime, run, schedule)
        self._enrich_run(runtime, run)
        db = self._get_db()
        runtime._store_function(run, db)
        result = execute_run(runtime, run, db)
        # MLClientCtx doesn't round-trip the notification spec — restore it
        result.spec.notifications = run.spec.notifications
        self._push_notifications(result)
        if result.spec.notifications:
            db.store_run(result.to_dict(), result.metadata.uid,
                         result.metadata.project,
                         iter=result.metadata.iteration or 0)
        return result

    def _schedule(self, db, runtime, run: RunObject, schedule) -> RunObject:
        sched = {
            "name": run.metadata.name,
            "kind": "job",
            "cron_trigger": schedule,
            "task": run.to_dict(),
        }
        db.create_schedule(run.metadata.project, sched)
        run.status.state = RunStates.created
        run.status.status_text = f"scheduled ({schedule})"
        return run


class ClientRemoteLauncher(BaseLauncher):
    """Submit the run to the API service over HTTP (POST /submit_job)."""

    def launch(self, runtime, run: RunObject, schedule=None, watch=True,
               **kwargs) -> RunObject:
        from .db import get_run_db

        self._enrich_run(runtime, run)
        db = get_run_db()
        runtime._store_function(run, db)
        run.spec.function = runtime.uri
        resp = db.submit_job(run, schedule=schedule)
        if isinstance(resp, dict) and "data" in resp:
            resp = resp["data"]
        if isinstance(resp, dict) and resp.get("metadata"):
            result = RunObject.from_dict(resp)
        else:
            result = run
        if watch and not schedule:
            result.wait_for_completion(raise_on_failure=False)
        return result


class ServerSideLauncher(BaseLauncher):
    """Launcher used inside the API service (parity: reference
    server/api/launcher.py:40)."""

    def __init__(self, db=None):
        self._db = db

    def launch_task(self, run: RunObject) -> dict:
        """Rebuild the function object from the DB and execute."""
        from .run import new_function

        db = self._db
        function_uri = run.spec.function
        runtime = None
        if function_uri:
            project, rest = (function_uri.split("/", 1) + [""])[:2] \
                if "/" in function_uri else ("default", function_uri)
            name, _, ref = rest.partition("@")
            if not ref:
                name, _, tag = rest.partition(":")
            else:
                tag = ""
            try:
                struct = db.get_function(name, project, tag=tag or "latest",
                                         hash_key=ref)
                runtime = new_function(runtime=struct)
            except Exception:
                runtime = None
        if runtime is None:
            runtime = new_function(name=run.metadata.name, kind="local")
        self._enrich_run(runtime, run)
        result = execute_run(runtime, run, db)
        return {"data": result.to_dict()}

    def launch(self, runtime, run: RunObject, schedule=None, watch=True,
               **kwargs) -> RunObject:
        self._enrich_run(runtime, run)
        return execute_run(runtime, run, self._db)


class LauncherFactory:
    @staticmethod
    def create_launcher(is_remote=False, local=None, db=None) -> BaseLauncher:
        if local is False and is_remote:
            return ClientRemoteLauncher()
        if local or not is_remote:
            return ClientLocalLauncher(db=db)
        return ClientRemoteLauncher()


# ------------------------------------------------------------- executor


def execute_run(runtime, run: RunObject, db) -> RunObject:
    """Run a task (or a hyperparameter sweep of tasks) via the runtime's
    _run, committing state to the DB."""
    generator = get_generator(run.spec)
    execution = MLClientCtx.from_dict(run.to_dict(), rundb=db,
                                      autocommit=False, host=None)
    if generator is None:
        result = _execute_single(runtime, run, execution)
        return _to_run_object(result)

    # hyperparameter sweep: child runs per iteration
    results: typing.List[dict] = []
    errors = 0
    tasks = list(generator.generate(run))
    parallel = generator.options.parallel_runs or 0

    def _run_child(task: RunObject) -> dict:
        child_exec = MLClientCtx.from_dict(task.to_dict(), rundb=db,
                                           autocommit=False)
        return _execute_single(runtime, task, child_exec)

    if parallel > 1:
        with concurrent.futures.ThreadPoolExecutor(max_workers=parallel) as ex:
            futures = {ex.submit(_run_child, t): t for t in tasks}
            for future in concurrent.futures.as_completed(futures):
                result = future.result()
                results.append(result)
                if result.get("status", {}).get("state") == RunStates.error:
                    errors += 1
    else:
        for task in tasks:
            result = _run_child(task)
            results.append(result)
            state = result.get("status", {}).get("state")
            if state == RunStates.error:
                errors += 1
                if generator.max_errors and errors >= generator.max_errors:
                    logger.warning("max errors reached, stopping sweep")
                    break
            stop = generator.eval_stop_condition(
                result.get("status", {}).get("results", {}) or {})
            if stop:
                logger.info("stop condition met, stopping sweep")
                break

    best_id, best_task = selector(
        results, (generator.options.selector or "") if generator.options
        else "")
    summary = [
        {"iter": r.get("metadata", {}).get("iteration"),
         "state": r.get("status", {}).get("state"),
         "results": r.get("status", {}).get("results", {})}
        for r in results]
    execution.log_iteration_results(best_id, summary, best_task or {})
    state = RunStates.completed if errors == 0 else RunStates.error
    if errors:
        execution.set_state(
            error=f"{errors} of {len(results)} iterations failed")
    else:
        execution.set_state(state)
    return _to_run_object(execution.to_dict())


def _execute_single(runtime, run: RunObject, execution: MLClientCtx) -> dict:
    try:
        return runtime._run(run, execution)
    except Exception as exc:
        logger.error("run raised", error=str(exc))
        execution.set_state(error=str(exc))
        return execution.to_dict()


def _to_run_object(result) -> RunObject:
    if isinstance(result, RunObject):
        return result
    return RunObject.from_dict(result)
