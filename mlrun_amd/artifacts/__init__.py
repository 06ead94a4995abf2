# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Artifacts: object model + manager + model/dataset kinds.

Parity targets (reference): mlrun/artifacts/base.py:179 Artifact,
manager.py:117 ArtifactManager, model.py:124 ModelArtifact (on-disk
format: model file(s) + model_spec.yaml + extra_data — kept compatible),
dataset.py:144 DatasetArtifact.
"""

import os
import pathlib
import typing

import yaml

from ..config import config
from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj
from ..utils import now_iso

MODEL_SPEC_FILENAME = "model_spec.yaml"


class ArtifactMetadata(ModelObj):
    def __init__(self, key=None, project=None, iter=None, tree=None, tag=None,
                 hash=None, labels=None, updated=None, description=None):
        self.key = key
        self.project = project
        self.iter = iter
        self.tree = tree
        self.tag = tag
        self.hash = hash
        self.labels = labels or {}
        self.updated = updated
        self.description = description


class ArtifactSpec(ModelObj):
    def __init__(self, src_path=None, target_path=None, viewer=None,
                 format=None, size=None, db_key=None, extra_data=None,
                 unpackaging_instructions=None, producer=None, sources=None,
                 license=None, encoding=None):
        self.src_path = src_path
        self.target_path = target_path
        self.viewer = viewer
        self.format = format
        self.size = size
        self.db_key = db_key
        self.extra_data = extra_data or {}
        self.unpackaging_instructions = unpackaging_instructions
        self.producer = producer
        self.sources = sources or []
        self.license = license
        self.encoding = encoding


class ArtifactStatus(ModelObj):
    def __init__(self, state=None, stats=None, preview=None, header=None):
        self.state = state or "created"
        self.stats = stats
        self.preview = preview
        self.header = header


class Artifact(ModelObj):
    kind = "artifact"
    _store_prefix = "artifacts"

    def __init__(self, key=None, body=None, viewer=None, is_dir=False,
                 format=None, size=None, target_path=None, project=None,
                 src_path=None, metadata: ArtifactMetadata = None,
                 spec: ArtifactSpec = None):
        self._metadata = None
        self.metadata = metadata or ArtifactMetadata(key=key, project=project)
        self._spec = None
        self.spec = spec or ArtifactSpec(src_path=src_path,
                                         target_path=target_path,
                                         viewer=viewer, format=format,
                                         size=size)
        self.status = ArtifactStatus()
        self._body = body
        self.is_dir = is_dir

    @property
    def metadata(self) -> ArtifactMetadata:
        return self._metadata

    @metadata.setter
    def metadata(self, value):
        self._metadata = self._verify_dict(value, "metadata", ArtifactMetadata)

    @property
    def spec(self) -> ArtifactSpec:
        return self._spec

    @spec.setter
    def spec(self, value):
        self._spec = self._verify_dict(value, "spec", ArtifactSpec)

    # convenience passthroughs (reference exposes these flat)
    @property
    def key(self):
        return self.metadata.key

    @key.setter
    def key(self, value):
        self.metadata.key = value

    @property
    def target_path(self):
        return self.spec.target_path

    @target_path.setter
    def target_path(self, value):
        self.spec.target_path = value

    @property
    def extra_data(self):
        return self.spec.extra_data

    def get_body(self):
        return self._body

    def before_log(self):
        pass

    def get_store_url(self, with_tag=True) -> str:
        uri = f"store://{self._store_prefix}/{self.metadata.project}/{self.metadata.key}"
        if with_tag and self.metadata.tree:
            uri += f"@{self.metadata.tree}"
        return uri

    @property
    def uri(self) -> str:
        return self.get_store_url()

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "kind": self.kind,
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
            "status": self.status.to_dict(),
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        kind = struct.get("kind", "artifact")
        target_cls = artifact_types.get(kind, cls)
        obj = target_cls()
        obj.metadata = struct.get("metadata")
        obj.spec = struct.get("spec")
        status = struct.get("status")
        if status:
            obj.status = ArtifactStatus.from_dict(status)
        return obj


class DatasetArtifact(Artifact):
    kind = "dataset"
    _store_prefix = "datasets"

    def __init__(self, key=None, df=None, preview=None, format="parquet",
                 stats=None, target_path=None, project=None, **kwargs):
        super().__init__(key=key, target_path=target_path, project=project,
                         format=format, **kwargs)
        self._df = df
        self.preview_rows = preview or 20
        self.compute_stats = stats

    def get_body(self):
        return self._df

    def before_log(self):
        if self._df is None:
            return
        df = self._df
        header = list(df.columns) if hasattr(df, "columns") else []
        self.status.header = header
        try:
            self.status.preview = df.head(self.preview_rows).values.tolist()
        except Exception:
            pass
        if self.compute_stats:
            try:
                self.status.stats = {
                    col: {k: _py(v) for k, v in stats.items()}
                    for col, stats in
                    df.describe(include="all").to_dict().items()}
            except Exception:
                pass

    def write_body(self, target_path: str):
        fmt = self.spec.format or "parquet"
        os.makedirs(os.path.dirname(os.path.abspath(target_path)) or ".",
                    exist_ok=True)
        if fmt in ("parquet", "pq"):
            self._df.to_parquet(target_path)
        elif fmt == "csv":
            self._df.to_csv(target_path, index=False)
        else:
            raise MLRunInvalidArgumentError(f"unsupported dataset format {fmt}")


def _py(value):
    try:
        import numpy as np

        if isinstance(value, np.generic):
            return value.item()
    except ImportError:
        pass
    return value


class ModelArtifact(Artifact):
    """Model artifact: a directory of model file(s) + model_spec.yaml +
    extra_data items.  On-disk layout kept compatible with the reference
    (artifacts/model.py:32) so V2ModelServer.get_model can resolve it."""

    kind = "model"
    _store_prefix = "models"

    def __init__(self, key=None, body=None, model_file=None, model_dir=None,
                 metrics=None, parameters=None, inputs=None, outputs=None,
                 framework=None, algorithm=None, feature_vector=None,
                 feature_weights=None, target_path=None, project=None,
                 **kwargs):
        super().__init__(key=key, body=body, target_path=target_path,
                         project=project, **kwargs)
        self.model_file = model_file
        self.model_dir = model_dir
        self.metrics = metrics or {}
        self.parameters = parameters or {}
        self.inputs = inputs or []
        self.outputs = outputs or []
        self.framework = framework
        self.algorithm = algorithm
        self.feature_vector = feature_vector
        self.feature_weights = feature_weights
        self.feature_stats = None

    def to_dict(self, fields=None, exclude=None, strip=False):
        struct = super().to_dict()
        struct["spec"].update({
            k: v for k, v in {
                "model_file": self.model_file,
                "metrics": self.metrics,
                "parameters": self.parameters,
                "inputs": self.inputs,
                "outputs": self.outputs,
                "framework": self.framework,
                "algorithm": self.algorithm,
                "feature_vector": self.feature_vector,
                "feature_weights": self.feature_weights,
            }.items() if v
        })
        return struct

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        obj = super().from_dict(struct)
        spec = (struct or {}).get("spec", {})
        for field in ["model_file", "metrics", "parameters", "inputs",
                      "outputs", "framework", "algorithm", "feature_vector",
                      "feature_weights"]:
            if field in spec:
                setattr(obj, field, spec[field])
        return obj

    def spec_yaml(self) -> str:
        return yaml.safe_dump(self.to_dict(), default_flow_style=False)

    def export_spec(self, target_dir: str):
        os.makedirs(target_dir, exist_ok=True)
        with open(os.path.join(target_dir, MODEL_SPEC_FILENAME), "w") as fp:
            fp.write(self.spec_yaml())


class PlotArtifact(Artifact):
    kind = "plot"

    def __init__(self, key=None, body=None, **kwargs):
        super().__init__(key=key, body=body, **kwargs)
        self.spec.viewer = "chart"


class DirArtifact(Artifact):
    """Directory artifact (reference artifacts/base.py DirArtifact):
    target_path is a directory tree; upload copies it recursively."""

    kind = "dir"

    def write_body(self, target_path: str):
        import shutil

        src = self.spec.src_path
        if not src or not os.path.isdir(src):
            raise MLRunInvalidArgumentError(
                f"dir artifact needs a src_path directory (got {src!r})")
        if os.path.abspath(src) != os.path.abspath(target_path):
            shutil.copytree(src, target_path, dirs_exist_ok=True)


class TableArtifact(Artifact):
    """Tabular artifact from a DataFrame or csv body
    (reference artifacts/plots.py TableArtifact)."""

    kind = "table"

    def __init__(self, key=None, body=None, df=None, viewer=None,
                 visible=False, format=None, header=None, **kwargs):
        if df is not None:
            format = format or "csv"
            body = df
        super().__init__(key=key, body=body, format=format, **kwargs)
        self._df = df
        self.header = header
        self.spec.viewer = viewer or "table"
        self.visible = visible

    def get_body(self):
        if self._df is not None:
            return self._df.to_csv(index=False)
        body = self.spec.get_body() if hasattr(self.spec, "get_body") \
            else self._body
        return body


artifact_types = {
    "artifact": Artifact,
    "": Artifact,
    "dataset": DatasetArtifact,
    "model": ModelArtifact,
    "plot": PlotArtifact,
    "dir": DirArtifact,
    "table": TableArtifact,
}


def dict_to_artifact(struct: dict) -> Artifact:
    """Rebuild a typed artifact object from its dict form (reference
    artifacts/__init__.py dict_to_artifact)."""
    return Artifact.from_dict(struct)


def get_artifact_meta(artifact):
    """Resolve an artifact path / DataItem to (artifact object,
    extra_data dict) — reference artifacts/base.py:801."""
    import yaml as _yaml

    from ..datastore import is_store_uri, store_manager

    if hasattr(artifact, "artifact_url"):
        artifact = artifact.artifact_url or artifact.url
    if isinstance(artifact, Artifact):
        spec = artifact
    elif is_store_uri(artifact):
        struct, _ = store_manager.get_store_artifact(artifact)
        spec = dict_to_artifact(struct)
    elif str(artifact).lower().endswith((".yaml", ".yml")):
        data = store_manager.object(url=artifact).get()
        spec = dict_to_artifact(_yaml.safe_load(data))
    else:
        raise MLRunInvalidArgumentError(
            f"cant resolve artifact file for {artifact}")
    extra_dataitems = {}
    from ..run import get_dataitem

    for key, item in (getattr(spec.spec, "extra_data", None) or {}).items():
        extra_dataitems[key] = get_dataitem(item) if isinstance(
            item, str) else item
    return spec, extra_dataitems


def _resolve_stored_artifact(artifact, expected_kind: str):
    from ..datastore import is_store_uri, store_manager

    if hasattr(artifact, "artifact_url"):
        artifact = artifact.artifact_url or artifact.url
    if isinstance(artifact, Artifact):
        spec = artifact
    elif isinstance(artifact, str) and is_store_uri(artifact):
        struct, _ = store_manager.get_store_artifact(artifact)
        spec = dict_to_artifact(struct)
    else:
        raise MLRunInvalidArgumentError(
            f"{expected_kind} path must be a {expected_kind} store "
            "object/URL/DataItem")
    if spec.kind != expected_kind:
        raise MLRunInvalidArgumentError(
            f"store artifact ({artifact}) is not {expected_kind} kind")
    return spec


def _store_artifact_object(spec):
    from ..db import get_run_db

    meta = spec.metadata
    get_run_db().store_artifact(
        meta.key, spec.to_dict(), tag=meta.tag or "latest",
        project=meta.project, tree=meta.tree)


def update_dataset_meta(artifact, from_df=None, schema: dict = None,
                        header: list = None, preview: list = None,
                        stats: dict = None, extra_data: dict = None,
                        column_metadata: dict = None, labels: dict = None,
                        ignore_preview_limits: bool = False):
    """Edit/add metadata on a stored dataset artifact (reference
    artifacts/dataset.py:396)."""
    spec = _resolve_stored_artifact(artifact, "dataset")
    if from_df is not None:
        spec._df = from_df
        spec.before_log()
    if header:
        spec.status.header = header
    if preview:
        spec.status.preview = preview
    if stats:
        spec.status.stats = stats
    if schema:
        spec.spec.schema = schema
    if column_metadata:
        spec.spec.column_metadata = column_metadata
    if extra_data:
        existing = dict(getattr(spec.spec, "extra_data", None) or {})
        existing.update(extra_data)
        spec.spec.extra_data = existing
    if labels:
        spec.metadata.labels.update(labels)
    _store_artifact_object(spec)
    return spec


def update_model(model_artifact, parameters: dict = None,
                 metrics: dict = None, extra_data: dict = None,
                 inputs: list = None, outputs: list = None,
                 feature_vector: str = None, feature_weights: list = None,
                 key_prefix: str = "", labels: dict = None,
                 write_spec_copy: bool = True, store_object: bool = True):
    """Edit/add attributes on a stored model artifact (reference
    artifacts/model.py:515)."""
    spec = _resolve_stored_artifact(model_artifact, "model")
    for key, val in (parameters or {}).items():
        spec.parameters[key] = val
    for key, val in (metrics or {}).items():
        spec.metrics[key_prefix + key] = val
    for key, val in (labels or {}).items():
        spec.metadata.labels[key] = val
    if inputs:
        spec.inputs = inputs
    if outputs:
        spec.outputs = outputs
    if feature_vector:
        spec.feature_vector = feature_vector
    if feature_weights:
        spec.feature_weights = feature_weights
    if extra_data:
        existing = dict(getattr(spec.spec, "extra_data", None) or {})
        for key, item in extra_data.items():
            if hasattr(item, "target_path"):
                item = item.target_path
            existing[key_prefix + key] = item
        spec.spec.extra_data = existing
    target_dir = os.path.dirname(spec.spec.target_path or "")
    if write_spec_copy and target_dir and os.path.isdir(target_dir):
        import yaml as _yaml

        with open(os.path.join(target_dir, "model_spec.yaml"), "w") as f:
            _yaml.safe_dump(spec.to_dict(), f, default_flow_style=False)
    if store_object:
        _store_artifact_object(spec)
    return spec


class ArtifactProducer:
    def __init__(self, kind, project, name, tag=None, owner=None, uid=None):
        self.kind = kind
        self.project = project
        self.name = name
        self.tag = tag
        self.owner = owner
        self.uid = uid or ""
        self.iteration = 0
        self.inputs = {}

    def get_meta(self):
        return {"kind": self.kind, "name": self.name, "tag": self.tag,
                "uri": f"{self.project}/{self.uid}",
                "owner": self.owner}


class ArtifactManager:
    """Materialize + register artifacts (parity: reference
    artifacts/manager.py:117)."""

    def __init__(self, db=None):
        self._db = db
        self.artifacts: dict = {}

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def artifact_list(self, full=False) -> list:
        return [artifact.to_dict() for artifact in self.artifacts.values()]

    def log_artifact(self, producer: ArtifactProducer, item, body=None,
                     target_path="", tag="", viewer="", local_path="",
                     artifact_path=None, format=None, upload=None,
                     labels=None, db_key=None, **kwargs) -> Artifact:
        if isinstance(item, str):
            key = item
            if body is not None and not isinstance(body, (str, bytes)) and \
                    hasattr(body, "to_parquet"):
                item = DatasetArtifact(key, df=body)
            else:
                item = Artifact(key, body)
        else:
            key = item.key
        item.metadata.project = producer.project
        item.metadata.tree = producer.uid or item.metadata.tree
        item.metadata.iter = producer.iteration or None
        item.metadata.updated = now_iso()
        if tag:
            item.metadata.tag = tag
        if labels:
            item.metadata.labels.update(labels)
        if viewer:
            item.spec.viewer = viewer
        if format:
            item.spec.format = format
        if local_path:
            item.spec.src_path = local_path
        item.spec.producer = producer.get_meta()
        item.spec.db_key = db_key or key

        # resolve target path
        if target_path:
            item.spec.target_path = target_path
        elif not item.spec.target_path:
            base = artifact_path or config.artifact_path or os.path.join(
                config.base_dir, "artifacts")
            base = base.replace("{{project}}", producer.project or "default")
            suffix = ""
            src = item.spec.src_path or ""
            if item.spec.format:
                suffix = f".{item.spec.format}"
            elif src and not os.path.isdir(src):
                suffix = pathlib.Path(src).suffix
            iter_part = f"{producer.iteration}/" if producer.iteration else ""
            item.spec.target_path = os.path.join(
                base, producer.project or "default",
                f"{iter_part}{key}{suffix}")

        item.before_log()

        # materialize the body / copy the source
        upload = upload if upload is not None else True
        if upload:
            self._materialize(item, body)

        self.artifacts[key] = item
        self._get_db().store_artifact(
            key, item.to_dict(), iter=producer.iteration,
            tag=item.metadata.tag or "latest", project=producer.project,
            tree=item.metadata.tree)
        return item

    def _materialize(self, item: Artifact, body):
        from ..datastore import store_manager

        target = item.spec.target_path
        if not target:
            return
        if isinstance(item, DatasetArtifact) and item.get_body() is not None:
            item.write_body(target)
            try:
                item.spec.size = os.path.getsize(target)
            except OSError:
                pass
            return
        body = body if body is not None else item.get_body()
        if body is not None:
            if not isinstance(body, (str, bytes)):
                body = str(body)
            store_manager.object(target).put(body)
            item.spec.size = len(body)
        elif item.spec.src_path:
            src = item.spec.src_path
            if os.path.isdir(src):
                os.makedirs(target, exist_ok=True)
                import shutil

                shutil.copytree(src, target, dirs_exist_ok=True)
                item.is_dir = True
            elif os.path.isfile(src):
                store_manager.object(target).upload(src)
                item.spec.size = os.path.getsize(src)

    def log_model(self, producer: ArtifactProducer, key, body=None,
                  model_file=None, model_dir=None, artifact_path=None,
                  framework=None, algorithm=None, metrics=None,
                  parameters=None, inputs=None, outputs=None, tag="",
                  extra_data=None, labels=None, upload=None,
                  feature_vector=None, feature_weights=None,
                  **kwargs) -> ModelArtifact:
        model = ModelArtifact(
            key, body=body, model_file=model_file, model_dir=model_dir,
            framework=framework, algorithm=algorithm, metrics=metrics,
            parameters=parameters, inputs=inputs, outputs=outputs,
            feature_vector=feature_vector,
            feature_weights=feature_weights,
            project=producer.project)
        if model_dir and not model_file:
            # pick first file in dir as the model file
            files = [f for f in os.listdir(model_dir)
                     if os.path.isfile(os.path.join(model_dir, f))
                     and f != MODEL_SPEC_FILENAME]
            if files:
                model.model_file = files[0]
        if model_dir:
            model.spec.src_path = model_dir
        elif model_file and os.path.sep in str(model_file):
            model.spec.src_path = os.path.dirname(model_file)
            model.model_file = os.path.basename(model_file)

        # model target is a DIRECTORY
        base = artifact_path or config.artifact_path or os.path.join(
            config.base_dir, "artifacts")
        base = base.replace("{{project}}", producer.project or "default")
        model.spec.target_path = os.path.join(
            base, producer.project or "default", key) + "/"

        if body is not None and not model.model_file:
            model.model_file = f"{key}.bin"

        upload = upload if upload is not None else True
        if upload:
            target_dir = model.spec.target_path.rstrip("/")
            os.makedirs(target_dir, exist_ok=True)
            if body is not None:
                if isinstance(body, str):
                    body = body.encode()
                with open(os.path.join(target_dir, model.model_file), "wb") as fp:
                    fp.write(body)
            elif model_dir and os.path.isdir(model_dir):
                import shutil

                src_abs = os.path.abspath(model_dir)
                dst_abs = os.path.abspath(target_dir)
                if dst_abs.startswith(src_abs + os.sep):
                    raise MLRunInvalidArgumentError(
                        f"model target dir {target_dir} is inside the "
                        f"source dir {model_dir}; use a different "
                        "artifact_path")
                shutil.copytree(model_dir, target_dir,
                                dirs_exist_ok=True)
            elif model_file and os.path.isfile(model_file):
                # only the named model file travels (its directory may
                # contain unrelated files — or the target itself)
                import shutil

                dst = os.path.join(target_dir, model.model_file)
                if os.path.abspath(model_file) != os.path.abspath(dst):
                    shutil.copyfile(model_file, dst)
            for name, value in (extra_data or {}).items():
                if isinstance(value, (str, bytes)):
                    data = value.encode() if isinstance(value, str) else value
                    extra_path = os.path.join(target_dir, name)
                    with open(extra_path, "wb") as fp:
                        fp.write(data)
                    model.spec.extra_data[name] = extra_path
                else:
                    model.spec.extra_data[name] = str(value)
            model.export_spec(target_dir)

        return self.log_artifact(producer, model, tag=tag, labels=labels,
                                 upload=False,
                                 artifact_path=artifact_path)


def get_model(model_dir_or_uri: str, suffix: str = "") -> typing.Tuple[
        str, ModelArtifact, dict]:
    """Resolve a model from a store:// uri or directory.

    Returns (model_file_path, ModelArtifact, extra_data_items) — parity
    with reference artifacts/model.py:412 get_model.
    """
    from ..datastore import store_manager

    extra_dataitems: dict = {}
    if model_dir_or_uri.startswith("store://"):
        item = store_manager.object(model_dir_or_uri)
        artifact = ModelArtifact.from_dict(item.meta)
        target = artifact.spec.target_path or ""
        model_file = os.path.join(target, artifact.model_file or "")
        for name, path in (artifact.spec.extra_data or {}).items():
            extra_dataitems[name] = store_manager.object(path, key=name)
        return model_file, artifact, extra_dataitems

    model_dir = model_dir_or_uri
    spec_path = os.path.join(model_dir, MODEL_SPEC_FILENAME)
    if os.path.isfile(spec_path):
        with open(spec_path) as fp:
            artifact = ModelArtifact.from_dict(yaml.safe_load(fp))
        model_file = os.path.join(model_dir, artifact.model_file or "")
        for name, path in (artifact.spec.extra_data or {}).items():
            extra_dataitems[name] = store_manager.object(path, key=name)
        return model_file, artifact, extra_dataitems
    if os.path.isfile(model_dir_or_uri):
        return model_dir_or_uri, ModelArtifact(
            key=os.path.basename(model_dir_or_uri),
            model_file=os.path.basename(model_dir_or_uri)), {}
    # bare dir: find a model file
    if os.path.isdir(model_dir):
        files = [f for f in os.listdir(model_dir)
                 if suffix and f.endswith(suffix)] or [
            f for f in os.listdir(model_dir)
            if os.path.isfile(os.path.join(model_dir, f))]
        if files:
            return os.path.join(model_dir, files[0]), ModelArtifact(
                key=files[0], model_file=files[0]), {}
    raise MLRunInvalidArgumentError(
        f"cannot resolve model from {model_dir_or_uri}")


class PlotlyArtifact(Artifact):
    """Plotly-figure artifact stored as html (reference
    artifacts/plots.py:77).  Requires plotly at construction time;
    the stored form is plain html so readers don't need plotly."""

    kind = "plotly"

    def __init__(self, figure=None, key: str = None,
                 target_path: str = None, **kwargs):
        super().__init__(key=key, target_path=target_path,
                         format="html", **kwargs)
        self.spec.viewer = "plotly"
        self._figure = figure
        if figure is not None:
            module = type(figure).__module__
            if not module.startswith("plotly"):
                raise MLRunInvalidArgumentError(
                    "PlotlyArtifact requires a plotly Figure object "
                    f"(got {module}.{type(figure).__name__})")

    def get_body(self):
        if self._figure is not None:
            return self._figure.to_html()
        return super().get_body()


artifact_types["plotly"] = PlotlyArtifact
