# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Batch "job" runtime — node-local replacement for the reference's
KubejobRuntime (reference mlrun/runtimes/kubejob.py).

Instead of building a pod, a job executes as a managed local process
(1 per run) with the requested GPU devices exposed via
HIP_VISIBLE_DEVICES, scheduled by the node-local GPU allocator
(mlrun_amd/parallel/scheduler.py).
"""

import os

from ..model import RunObject
from .base import BaseRuntime
from .local import LocalRuntime


class KubejobRuntime(BaseRuntime):
    """kind="job": batch run, optionally on reserved GPUs."""

    kind = "job"
    _is_remote = True  # submits to the service when dbpath is remote

    def with_code(self, from_file="", body=None):
        if from_file:
            with open(from_file) as fp:
                body = fp.read()
        self.spec.build["functionSourceCode"] = body
        return self

    def deploy(self, watch=True, with_mlrun=None, skip_deployed=False,
               is_kfp=False, mlrun_version_specifier=None, builder_env=None,
               show_on_failure=False):
        """No container builds in the node-local model — the builder
        materializes an image DIRECTORY (source + requirements +
        run.sh; utils/builder.py, reference build_runtime :644)."""
        if skip_deployed and self.is_deployed():
            return True
        build = self.spec.build or {}
        if build.get("functionSourceCode") or build.get("source") or \
                build.get("requirements"):
            from ..utils.builder import build_runtime

            return build_runtime(self, with_mlrun=with_mlrun)
        self.status.state = "ready"
        return True

    def is_deployed(self):
        return bool(self.spec.command or
                    self.spec.build.get("functionSourceCode"))

    def _run(self, run: RunObject, execution) -> dict:
        """Execute: reserve GPUs (if requested), run in-process (handler)
        or as subprocess (command mode)."""
        gpus = int(self.spec.resources.gpus or 0)
        env_patch = {}
        lease = None
        if gpus:
            from ..parallel.scheduler import get_gpu_allocator

            allocator = get_gpu_allocator()
            lease = allocator.acquire(gpus, owner=run.metadata.uid)
            env_patch["HIP_VISIBLE_DEVICES"] = ",".join(
                str(d) for d in lease.devices)
            env_patch["CUDA_VISIBLE_DEVICES"] = env_patch["HIP_VISIBLE_DEVICES"]
        old_env = {}
        try:
            for key, value in {**self.spec.build.get("env", {}),
                               **env_patch}.items():
                old_env[key] = os.environ.get(key)
                os.environ[key] = str(value)
            self.deploy(watch=False)
            local = LocalRuntime.from_dict(self.to_dict())
            local.spec.command = self.spec.command
            return local._run(run, execution)
        finally:
            for key, value in old_env.items():
                if value is None:
                    os.environ.pop(key, None)
                else:
                    os.environ[key] = value
            if lease is not None:
                lease.release()

    def build_config(self, image="", base_image="", commands=None,
                     requirements=None, secret=None, source=None,
                     extra=None, load_source_on_run=None):
        self.spec.build.update({
            k: v for k, v in {
                "image": image, "base_image": base_image,
                "commands": commands, "requirements": requirements,
                "source": source,
            }.items() if v})
        return self
