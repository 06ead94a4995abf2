# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""No-op run DB (offline mode).  Parity: reference mlrun/db/nopdb.py:31."""

from ..errors import MLRunNotFoundError
from .base import RunDBInterface


class NopDB(RunDBInterface):
    kind = "nop"

    def __init__(self, url=None):
        self.url = url

    def connect(self, secrets=None):
        return self

    def store_run(self, struct, uid, project="", iter=0):
        return struct

    def update_run(self, updates, uid, project="", iter=0):
        return {}

    def read_run(self, uid, project="", iter=0):
        raise MLRunNotFoundError("nop db has no runs")

    def list_runs(self, *args, **kwargs):
        return []

    def del_run(self, uid, project="", iter=0):
        pass

    def store_artifact(self, key, artifact, uid=None, iter=None, tag="",
                       project="", tree=None):
        return artifact

    def read_artifact(self, key, tag="", iter=None, project="", tree=None,
                      uid=None):
        raise MLRunNotFoundError("nop db has no artifacts")

    def list_artifacts(self, *args, **kwargs):
        return []

    def del_artifact(self, key, tag="", project="", uid=None, tree=None):
        pass

    def store_function(self, function, name, project="", tag="",
                       versioned=False):
        return ""

    def get_function(self, name, project="", tag="", hash_key=""):
        raise MLRunNotFoundError("nop db has no functions")

    def list_functions(self, name=None, project="", tag="", labels=None):
        return []
