# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Layered secret sources (env / inline / file).

Parity target: reference mlrun/secrets.py (SecretsStore :22,
get_secret_or_env :149) — vault/azure/k8s providers of the reference
have no node-local analog and raise if requested.
"""

import os
import typing

from .errors import MLRunInvalidArgumentError


class SecretsStore:
    def __init__(self):
        self._secrets: dict = {}
        self.env_prefix = "MLRUN_SECRET_"

    def add_source(self, kind: str, source=None, prefix: str = ""):
        if kind == "inline":
            if not isinstance(source, dict):
                raise MLRunInvalidArgumentError(
                    "inline source must be a dict")
            for key, value in source.items():
                self._secrets[prefix + key] = str(value)
        elif kind == "env":
            names = source.split(",") if isinstance(source, str) else \
                (source or [])
            for name in names:
                name = name.strip()
                if name in os.environ:
                    self._secrets[prefix + name] = os.environ[name]
        elif kind == "file":
            from .utils import list_to_dict

            with open(source) as fp:
                for key, value in list_to_dict(fp.readlines()).items():
                    if key and not key.startswith("#"):
                        self._secrets[prefix + key] = value
        elif kind in ("vault", "azure_vault", "kubernetes"):
            raise MLRunInvalidArgumentError(
                f"secret provider {kind!r} is not available in the "
                f"node-local deployment (use env/file/inline)")
        else:
            raise MLRunInvalidArgumentError(f"unknown secret kind {kind}")
        return self

    def get(self, key: str, default=None):
        if key in self._secrets:
            return self._secrets[key]
        if self.env_prefix + key in os.environ:
            return os.environ[self.env_prefix + key]
        return os.environ.get(key, default)

    def to_serial(self) -> list:
        return [{"kind": "inline", "source": dict(self._secrets)}]

    def to_dict(self, struct: dict = None) -> dict:
        """Serialize into a run-spec dict (reference secrets.py
        to_dict — populates spec.secret_sources)."""
        struct = struct if struct is not None else {}
        struct["secret_sources"] = self.to_serial()
        return struct

    def items(self):
        return dict(self._secrets)

    @classmethod
    def from_list(cls, sources: typing.List[dict]) -> "SecretsStore":
        store = cls()
        for source in sources or []:
            store.add_source(source.get("kind", "inline"),
                             source.get("source"))
        return store


def get_secret_or_env(key: str, secret_provider=None, default=None,
                      prefix: str = None):
    """Resolve a secret: provider -> MLRUN_SECRET_ env -> plain env
    (parity: reference secrets.py:149)."""
    if prefix:
        key = f"{prefix}_{key}"
    if secret_provider is not None:
        value = None
        if isinstance(secret_provider, dict):
            value = secret_provider.get(key)
        elif isinstance(secret_provider, SecretsStore):
            value = secret_provider.get(key)
        elif callable(secret_provider):
            value = secret_provider(key)
        if value:
            return value
    return os.environ.get(f"MLRUN_SECRET_{key}",
                          os.environ.get(key, default))
