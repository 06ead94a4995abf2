# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""CLI: python -m mlrun_amd <command>

Parity target: reference mlrun/__main__.py (run :191, build :485,
deploy :634, get, logs, project :964, clean :1117, config, version) —
re-built on argparse (no click dependency) for the node-local service.
"""

import argparse
import json
import sys

from . import __version__
from .config import config


def cmd_version(args):
    print(f"mlrun_amd version {__version__}")


def cmd_config(args):
    print(config.dump_yaml())


def _parse_hyperparams(spec: str):
    """Accept JSON ({"p1": [1,2]}) or key=json-list (p1=[1,2];p2=[..])
    like the reference's -x arguments."""
    if not spec:
        return None
    spec = spec.strip()
    if spec.startswith("{"):
        return json.loads(spec)
    params = {}
    for part in spec.split(";"):
        key, _, value = part.partition("=")
        params[key.strip()] = json.loads(value)
    return params


def cmd_run(args):
    import mlrun_amd

    params = {}
    for param in args.param or []:
        key, _, value = param.partition("=")
        try:
            params[key] = json.loads(value)
        except ValueError:
            params[key] = value
    inputs = {}
    for inp in args.inputs or []:
        key, _, value = inp.partition("=")
        inputs[key] = value
    kind = args.kind or ("local" if args.local else "job")
    fn = mlrun_amd.new_function(name=args.name or "", kind=kind,
                                command=args.command or "")
    if args.replicas and hasattr(fn, "with_replicas"):
        fn.with_replicas(args.replicas)
    if args.gpus:
        fn.with_limits(gpus=args.gpus)
    run = fn.run(name=args.name or "", handler=args.handler,
                 params=params, inputs=inputs, project=args.project or "",
                 artifact_path=args.artifact_path or "",
                 schedule=args.schedule, watch=not args.no_wait,
                 hyperparams=_parse_hyperparams(args.hyperparam))
    state = run.status.state
    print(f"run {run.metadata.uid} finished: {state}")
    if run.status.results:
        for key, value in run.status.results.items():
            print(f"  {key}: {value}")
    if state == "error":
        print(f"  error: {run.status.error}")
        sys.exit(1)


def cmd_get(args):
    from .db import get_run_db

    db = get_run_db()
    kind = args.kind
    project = args.project or "default"
    if kind in ("run", "runs"):
        items = db.list_runs(project=project, name=args.name or "",
                             last=args.last or 0)
        for run in items:
            meta, status = run.get("metadata", {}), run.get("status", {})
            print(f"{meta.get('uid', '')[:8]}  {meta.get('name', ''):24s}"
                  f"  {status.get('state', ''):10s}"
                  f"  {status.get('start_time', '')}")
    elif kind in ("artifact", "artifacts"):
        for artifact in db.list_artifacts(project=project,
                                          name=args.name or ""):
            meta = artifact.get("metadata", {})
            print(f"{meta.get('key', ''):24s} {artifact.get('kind', ''):8s}"
                  f" {artifact.get('spec', {}).get('target_path', '')}")
    elif kind in ("function", "func", "functions"):
        for func in db.list_functions(project=project):
            meta = func.get("metadata", {})
            print(f"{meta.get('name', ''):24s} {func.get('kind', '')}")
    elif kind in ("project", "projects"):
        for proj in db.list_projects():
            print(proj.get("metadata", {}).get("name"))
    elif kind in ("schedule", "schedules"):
        for sched in db.list_schedules(project):
            print(f"{sched.get('name', ''):24s} "
                  f"{sched.get('cron_trigger', '')}")
    else:
        print(f"unsupported kind {kind}")
        sys.exit(1)


def cmd_logs(args):
    from .db import get_run_db

    state, text = get_run_db().get_log(args.uid, args.project or "default")
    if isinstance(text, bytes):
        text = text.decode(errors="replace")
    print(text)
    if state:
        print(f"final state: {state}")


def cmd_project(args):
    import mlrun_amd

    project = mlrun_amd.get_or_create_project(args.name,
                                              context=args.context or "./")
    if args.run_workflow:
        status = project.run(name=args.run_workflow,
                             arguments=json.loads(args.arguments)
                             if args.arguments else {})
        print(f"workflow {args.run_workflow}: {status.state} "
              f"({len(status.runs)} runs)")
    else:
        print(f"project {project.name} loaded "
              f"({len(project.spec.functions)} functions)")


def cmd_clean(args):
    from .db import get_run_db

    db = get_run_db()
    project = args.project or "default"
    db.del_runs(project=project, state=args.state or None)
    print(f"cleaned runs in {project}")


def cmd_build(args):
    """Build a function "image" dir (reference `mlrun build` :485;
    kaniko replaced by the node-local directory builder)."""
    import mlrun_amd
    from .utils.builder import build_runtime

    if args.spec:
        fn = mlrun_amd.import_function(args.spec)
    else:
        fn = mlrun_amd.new_function(name=args.name or "fn", kind="job",
                                    command=args.command or "")
    if args.requirements:
        fn.spec.build.setdefault("requirements", []).extend(
            args.requirements)
    if args.source:
        fn.spec.build["source"] = args.source
    build_runtime(fn, with_mlrun=args.with_mlrun,
                  install=args.install)
    print(f"image built at {fn.spec.image}")


def cmd_deploy(args):
    import mlrun_amd

    fn = mlrun_amd.import_function(args.spec) if args.spec else \
        mlrun_amd.new_function(name=args.name or "srv", kind="serving",
                               command=args.command or "")
    address = fn.deploy()
    print(f"deployed at {address}")
    if args.wait:
        try:
            import time

            while True:
                time.sleep(1)
        except KeyboardInterrupt:
            fn.stop()


def cmd_db(args):
    from .api.main import serve

    print(f"starting mlrun-amd API service on port "
          f"{args.port or config.httpdb.port}")
    serve(port=args.port)


def cmd_watch_stream(args):
    from .model_monitoring import get_stream_processor

    processor = get_stream_processor(args.project or "default")
    stats = processor.endpoint_stats(args.endpoint)
    print(json.dumps(stats, indent=2, default=str))


def cmd_migrate(args):
    """Apply pending run-DB schema migrations (reference
    `mlrun migrate` / trigger_migrations)."""
    from .db import get_run_db

    db = get_run_db(args.dbpath or None)
    result = db.trigger_migrations()
    print(json.dumps(result, indent=2))


def cmd_summary(args):
    from .db import get_run_db

    db = get_run_db()
    if args.project:
        print(json.dumps(db.compute_project_summary(args.project),
                         indent=2, default=str))
    else:
        print(json.dumps(db.list_project_summaries(), indent=2,
                         default=str))


def main(argv=None):
    parser = argparse.ArgumentParser(
        prog="mlrun_amd",
        description="MI355X-native MLOps framework CLI")
    sub = parser.add_subparsers(dest="cmd")

    p = sub.add_parser("run", help="execute a task")
    p.add_argument("command", nargs="?", help="python file to run")
    p.add_argument("--name", default="")
    p.add_argument("--handler", default=None)
    p.add_argument("--kind", default="")
    p.add_argument("--project", default="")
    p.add_argument("--param", "-p", action="append")
    p.add_argument("--inputs", "-i", action="append")
    p.add_argument("--hyperparam", default="")
    p.add_argument("--artifact-path", default="")
    p.add_argument("--schedule", default=None)
    p.add_argument("--local", action="store_true")
    p.add_argument("--replicas", type=int, default=0)
    p.add_argument("--gpus", type=int, default=0)
    p.add_argument("--no-wait", action="store_true")
    p.set_defaults(func=cmd_run)

    p = sub.add_parser("get", help="list objects")
    p.add_argument("kind")
    p.add_argument("name", nargs="?", default="")
    p.add_argument("--project", default="")
    p.add_argument("--last", type=int, default=0)
    p.set_defaults(func=cmd_get)

    p = sub.add_parser("logs", help="show run logs")
    p.add_argument("uid")
    p.add_argument("--project", default="")
    p.set_defaults(func=cmd_logs)

    p = sub.add_parser("project", help="load/run a project")
    p.add_argument("name")
    p.add_argument("--context", default="./")
    p.add_argument("--run-workflow", default="")
    p.add_argument("--arguments", default="")
    p.set_defaults(func=cmd_project)

    p = sub.add_parser("clean", help="delete runs")
    p.add_argument("--project", default="")
    p.add_argument("--state", default="")
    p.set_defaults(func=cmd_clean)

    p = sub.add_parser("build", help="build a function image dir")
    p.add_argument("--spec", default="")
    p.add_argument("--name", default="")
    p.add_argument("--command", default="")
    p.add_argument("--source", default="")
    p.add_argument("--requirements", "-r", action="append")
    p.add_argument("--with-mlrun", action="store_true")
    p.add_argument("--install", action="store_true",
                   help="resolve requirements from the offline wheelhouse")
    p.set_defaults(func=cmd_build)

    p = sub.add_parser("deploy", help="deploy a serving function")
    p.add_argument("--spec", default="")
    p.add_argument("--name", default="")
    p.add_argument("--command", default="")
    p.add_argument("--wait", action="store_true")
    p.set_defaults(func=cmd_deploy)

    p = sub.add_parser("db", help="run the API service")
    p.add_argument("--port", type=int, default=0)
    p.set_defaults(func=cmd_db)

    p = sub.add_parser("watch-stream", help="show monitoring stream stats")
    p.add_argument("endpoint")
    p.add_argument("--project", default="")
    p.set_defaults(func=cmd_watch_stream)

    p = sub.add_parser("migrate",
                       help="apply pending run-DB schema migrations")
    p.add_argument("--dbpath", default="")
    p.set_defaults(func=cmd_migrate)

    p = sub.add_parser("summary", help="per-project entity counts")
    p.add_argument("project", nargs="?", default="")
    p.set_defaults(func=cmd_summary)

    sub.add_parser("version").set_defaults(func=cmd_version)
    sub.add_parser("config").set_defaults(func=cmd_config)

    args = parser.parse_args(argv)
    if not getattr(args, "func", None):
        parser.print_help()
        return 0
    return args.func(args)


if __name__ == "__main__":
    sys.exit(main() or 0)
