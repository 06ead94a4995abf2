# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature-store run configuration (reference
feature_store/common.py:175 RunConfig): holds the function + job spec
used when ingestion/merge tasks run as jobs instead of inline."""

import copy as _copy

from ..errors import MLRunInvalidArgumentError


class RunConfig:
    """Function + run spec for feature-store jobs (ingest as job,
    offline merge as job, online service deployment).

    example::

        # config for local run emulation
        config = RunConfig(local=True)

        # config using a .py file with an image
        config = RunConfig("mycode.py", image="mlrun/mlrun",
                           kind="job")

        # config using a function object
        function = mlrun.new_function("ingest", kind="job")
        config = RunConfig(function)
    """

    def __init__(self, function=None, local: bool = None,
                 image: str = None, kind: str = None, handler: str = None,
                 parameters: dict = None, watch: bool = None, owner=None,
                 credentials=None, code: str = None, requirements=None,
                 extra_spec: dict = None, auth_info=None):
        self._function = None
        self._modifiers = []
        self.secret_sources = []
        self.function = function
        self.local = local
        self.image = image
        self.kind = kind
        self.handler = handler
        self.parameters = parameters or {}
        self.watch = True if watch is None else watch
        self.owner = owner
        self.credentials = credentials
        self.code = code or ""
        self.requirements = requirements
        self.extra_spec = extra_spec
        self.auth_info = auth_info

    @property
    def function(self):
        return self._function

    @function.setter
    def function(self, function):
        if function and not (isinstance(function, str)
                             or hasattr(function, "apply")):
            raise MLRunInvalidArgumentError(
                "function must be a uri/path (string) or mlrun "
                "function object")
        self._function = function

    def apply(self, modifier):
        """Register a function modifier (e.g. auto_mount())."""
        self._modifiers.append(modifier)
        return self

    def with_secret(self, kind, source):
        """Register a secrets source (file/env/inline dict)."""
        self.secret_sources.append({"kind": kind, "source": source})
        return self

    def to_function(self, default_kind: str = None,
                    default_image: str = None):
        """Materialise the runtime function object for this config."""
        from ..run import code_to_function, import_function, new_function

        function = self.function
        if hasattr(function, "apply"):
            function = _copy.copy(function)
        elif isinstance(function, str) and (
                function.endswith(".py") or function.endswith(".ipynb")):
            function = code_to_function(
                "", filename=function,
                kind=self.kind or default_kind or "job",
                image=self.image or default_image,
                handler=self.handler)
        elif isinstance(function, str) and function:
            function = import_function(function)
        else:
            function = new_function(
                kind=self.kind or default_kind or "job",
                image=self.image or default_image)
        if self.image and hasattr(function, "spec"):
            function.spec.image = function.spec.image or self.image
        for modifier in self._modifiers:
            function.apply(modifier)
        for source in self.secret_sources:
            if hasattr(function, "with_secrets"):
                function.with_secrets(source["kind"], source["source"])
        return function

    def copy(self):
        return _copy.deepcopy(self)
