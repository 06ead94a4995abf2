# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Datastore layer: URI-scheme object stores + the DataItem handle.

Parity target: reference mlrun/datastore (schema_to_store dispatch
datastore/datastore.py:56, DataItem base.py:424, store:// resolution
store_resources.py:144).  The MI355X deployment is node-local, so the
built-in stores are file/memory; the scheme registry is open for
extension (register_store).
"""

import io
import os
import shutil
import tempfile
import threading
import typing
from urllib.parse import urlparse

from ..errors import MLRunInvalidArgumentError, MLRunNotFoundError

# ---------------------------------------------------------------- stores


class DataStore:
    """Base store: get/put/stat/listdir over a scheme."""

    kind = ""

    def __init__(self, parent, name, scheme, endpoint=""):
        self.parent = parent
        self.name = name
        self.kind = scheme
        self.endpoint = endpoint

    def _join(self, key: str) -> str:
        return key

    def get(self, key: str, size=None, offset=0) -> bytes:
        raise NotImplementedError

    def put(self, key: str, data, append=False):
        raise NotImplementedError

    def stat(self, key: str):
        raise NotImplementedError

    def listdir(self, key: str) -> list:
        raise NotImplementedError

    def download(self, remote_path: str, local_path: str):
        data = self.get(remote_path)
        os.makedirs(os.path.dirname(os.path.abspath(local_path)), exist_ok=True)
        with open(local_path, "wb") as fp:
            fp.write(data)

    def upload(self, key: str, src_path: str):
        with open(src_path, "rb") as fp:
            self.put(key, fp.read())

    def as_df(self, key: str, columns=None, df_module=None, format="",
              **kwargs):
        import pandas as pd

        fmt = format or os.path.splitext(key)[1].lstrip(".")
        data = io.BytesIO(self.get(key))
        if fmt in ("csv", ""):
            df = pd.read_csv(data, **kwargs)
        elif fmt in ("parquet", "pq"):
            df = pd.read_parquet(data, **kwargs)
        elif fmt == "json":
            df = pd.read_json(data, **kwargs)
        else:
            raise MLRunInvalidArgumentError(f"unsupported dataframe format {fmt}")
        if columns:
            df = df[columns]
        return df

    def rm(self, key: str, recursive=False):
        raise NotImplementedError


class FileStore(DataStore):
    kind = "file"

    def _join(self, key: str) -> str:
        if self.endpoint:
            return os.path.join(self.endpoint, key.lstrip("/"))
        return key

    def get(self, key, size=None, offset=0):
        path = self._join(key)
        if not os.path.isfile(path):
            raise MLRunNotFoundError(f"file {path} not found")
        with open(path, "rb") as fp:
            if offset:
                fp.seek(offset)
            return fp.read(size) if size else fp.read()

    def put(self, key, data, append=False):
        path = self._join(key)
        os.makedirs(os.path.dirname(os.path.abspath(path)) or ".", exist_ok=True)
        mode = "ab" if append else "wb"
        if isinstance(data, str):
            data = data.encode()
        with open(path, mode) as fp:
            fp.write(data)

    def stat(self, key):
        path = self._join(key)
        if not os.path.exists(path):
            raise MLRunNotFoundError(f"file {path} not found")
        st = os.stat(path)
        return {"size": st.st_size, "modified": st.st_mtime}

    def listdir(self, key):
        path = self._join(key)
        if not os.path.isdir(path):
            return []
        out = []
        for root, _, files in os.walk(path):
            for fname in files:
                out.append(os.path.relpath(os.path.join(root, fname), path))
        return out

    def download(self, remote_path, local_path):
        src = self._join(remote_path)
        os.makedirs(os.path.dirname(os.path.abspath(local_path)) or ".",
                    exist_ok=True)
        if os.path.abspath(src) != os.path.abspath(local_path):
            shutil.copyfile(src, local_path)

    def rm(self, key, recursive=False):
        path = self._join(key)
        if os.path.isdir(path) and recursive:
            shutil.rmtree(path, ignore_errors=True)
        elif os.path.isfile(path):
            os.remove(path)


class MemoryStore(DataStore):
    """In-process memory store (memory://) — used by tests & queues."""

    kind = "memory"
    _items: dict = {}
    _lock = threading.Lock()

    def get(self, key, size=None, offset=0):
        with MemoryStore._lock:
            if key not in MemoryStore._items:
                raise MLRunNotFoundError(f"memory://{key} not found")
            data = MemoryStore._items[key]
        if isinstance(data, str):
            data = data.encode()
        if not isinstance(data, bytes):
            return data
        if offset:
            data = data[offset:]
        return data[:size] if size else data

    def put(self, key, data, append=False):
        with MemoryStore._lock:
            if append and key in MemoryStore._items:
                prev = MemoryStore._items[key]
                if isinstance(prev, str):
                    prev = prev.encode()
                if isinstance(data, str):
                    data = data.encode()
                data = prev + data
            MemoryStore._items[key] = data

    def put_object(self, key, obj):
        with MemoryStore._lock:
            MemoryStore._items[key] = obj

    def stat(self, key):
        data = self.get(key)
        return {"size": len(data) if isinstance(data, (bytes, str)) else 0,
                "modified": 0}

    def listdir(self, key):
        with MemoryStore._lock:
            return [k for k in MemoryStore._items if k.startswith(key)]

    def rm(self, key, recursive=False):
        with MemoryStore._lock:
            if recursive:
                for k in list(MemoryStore._items):
                    if k.startswith(key):
                        del MemoryStore._items[k]
            else:
                MemoryStore._items.pop(key, None)

    def as_df(self, key, columns=None, df_module=None, format="", **kwargs):
        with MemoryStore._lock:
            obj = MemoryStore._items.get(key)
        import pandas as pd

        if isinstance(obj, pd.DataFrame):
            return obj[columns] if columns else obj
        return super().as_df(key, columns, df_module, format, **kwargs)


class FsspecStore(DataStore):
    """Object-store adapter over fsspec (reference: the s3/az/gcs/
    dbfs/hdfs/http store classes of mlrun/datastore — one adapter
    here; each scheme works when its fsspec protocol package is
    installed, with a clear error otherwise)."""

    def __init__(self, parent, name, scheme, endpoint=""):
        super().__init__(parent, name, scheme, endpoint)
        import fsspec

        try:
            self._fs = fsspec.filesystem(scheme)
        except (ImportError, ValueError) as exc:
            from ..errors import MLRunMissingDependencyError

            raise MLRunMissingDependencyError(
                f"fsspec protocol {scheme!r} is unavailable in this "
                f"image: {exc}")

    def _full(self, key):
        base = f"{self.endpoint}/" if self.endpoint else ""
        return f"{base}{key.lstrip('/')}" if base else key

    def get(self, key, size=None, offset=0):
        with self._fs.open(self._full(key), "rb") as stream:
            if offset:
                stream.seek(offset)
            return stream.read(size) if size else stream.read()

    def put(self, key, data, append=False):
        if isinstance(data, str):
            data = data.encode()
        mode = "ab" if append else "wb"
        with self._fs.open(self._full(key), mode) as stream:
            stream.write(data)

    def stat(self, key):
        info = self._fs.info(self._full(key))
        return {"size": info.get("size", 0),
                "modified": info.get("mtime", 0)}

    def listdir(self, key):
        return [str(p) for p in self._fs.ls(self._full(key))]

    def rm(self, key, recursive=False):
        self._fs.rm(self._full(key), recursive=recursive)


_schemes: dict = {}


def register_store(scheme: str, cls):
    _schemes[scheme] = cls


register_store("file", FileStore)
register_store("", FileStore)
register_store("memory", MemoryStore)
# object-store schemes ride the fsspec adapter (protocol packages
# optional; errors are explicit when missing)
for _scheme in ("s3", "gs", "gcs", "az", "abfs", "http", "https",
                "hdfs", "webhdfs", "dbfs", "oss", "ftp", "sftp",
                "redis"):
    register_store(_scheme, FsspecStore)


def schema_to_store(scheme: str):
    if scheme not in _schemes:
        raise MLRunInvalidArgumentError(
            f"unsupported data store scheme {scheme!r} "
            f"(registered: {sorted(_schemes)})")
    return _schemes[scheme]


def parse_url(url: str):
    parsed = urlparse(url)
    scheme = parsed.scheme.lower()
    endpoint = parsed.netloc
    path = parsed.path
    if scheme == "file" and endpoint:
        path = endpoint + path
        endpoint = ""
    if not scheme:
        path = url
    return scheme, endpoint, path


# ------------------------------------------------------------- DataItem


class DataItem:
    """Handle over a data URI: lazy get/put/local/as_df/show."""

    def __init__(self, key: str, store: DataStore, subpath: str, url: str = "",
                 meta=None, artifact_url=None):
        self._store = store
        self._key = key
        self._url = url
        self._path = subpath
        self._meta = meta
        self._artifact_url = artifact_url
        self._local_path = ""

    @property
    def key(self):
        return self._key

    @property
    def kind(self):
        return self._store.kind

    @property
    def meta(self):
        return self._meta

    @property
    def artifact_url(self):
        return self._artifact_url or self._url

    @property
    def url(self):
        return self._url

    @property
    def suffix(self):
        return os.path.splitext(self._path)[1]

    @property
    def store(self):
        return self._store

    def get(self, size=None, offset=0, encoding=""):
        body = self._store.get(self._path, size=size, offset=offset)
        if encoding and isinstance(body, bytes):
            return body.decode(encoding)
        return body

    def download(self, target_path: str):
        self._store.download(self._path, target_path)

    def put(self, data, append=False):
        self._store.put(self._path, data, append=append)

    def delete(self):
        self._store.rm(self._path)

    def upload(self, src_path: str):
        self._store.upload(self._path, src_path)

    def stat(self):
        return self._store.stat(self._path)

    def listdir(self):
        return self._store.listdir(self._path)

    def ls(self):
        """Alias of listdir (reference DataItem.ls)."""
        return self.listdir()

    def open(self, mode: str = "rb"):
        """Open the (localized) item as a file object (reference
        DataItem.open)."""
        return open(self.local(), mode)

    def get_artifact_type(self):
        """Kind of the backing artifact, if this item resolves one."""
        if self._meta and isinstance(self._meta, dict):
            return self._meta.get("kind")
        return None

    def local(self) -> str:
        """Download to a local temp file (or return the path if local)."""
        if self.kind == "file":
            return self._store._join(self._path)
        if self._local_path:
            return self._local_path
        suffix = self.suffix or ".tmp"
        temp = tempfile.NamedTemporaryFile(suffix=suffix, delete=False)
        self._local_path = temp.name
        temp.close()
        self.download(self._local_path)
        return self._local_path

    def remove_local(self):
        if self.kind == "file":
            return
        if self._local_path:
            os.remove(self._local_path)
            self._local_path = ""

    def as_df(self, columns=None, df_module=None, format="", **kwargs):
        return self._store.as_df(self._path, columns=columns,
                                 df_module=df_module, format=format, **kwargs)

    def show(self, format=None):
        print(self.get(encoding="utf-8"))

    def __str__(self):
        return self.url or self._path

    def __repr__(self):
        return f"DataItem({self.url or self._path!r})"


# --------------------------------------------------------- store manager


class StoreManager:
    def __init__(self, secrets=None, db=None):
        self._stores: dict = {}
        self._db = db
        self._secrets = secrets or {}

    def set(self, secrets=None, db=None):
        if db:
            self._db = db
        if secrets:
            self._secrets.update(secrets)
        return self

    def _get_db(self):
        if not self._db:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def get_or_create_store(self, url: str) -> typing.Tuple[DataStore, str]:
        scheme, endpoint, path = parse_url(url)
        store_key = f"{scheme}://{endpoint}"
        if store_key not in self._stores:
            cls = schema_to_store(scheme)
            self._stores[store_key] = cls(self, scheme or "file",
                                          scheme or "file", endpoint)
        return self._stores[store_key], path

    def object(self, url: str, key: str = "", project: str = "") -> DataItem:
        meta = artifact_url = None
        if url.startswith("store://"):
            artifact_url = url
            meta, url = self.get_store_artifact(url, project)
        store, subpath = self.get_or_create_store(url)
        return DataItem(key, store, subpath, url, meta=meta,
                        artifact_url=artifact_url)

    def get_store_artifact(self, url: str, project: str = ""):
        """Resolve store://artifacts/<project>/<key>[:<tag>|@<tree>] ->
        (artifact dict, target url)."""
        kind, project_, key, tag, tree, iteration = parse_store_uri(url, project)
        db = self._get_db()
        if kind in ("artifacts", "artifact", "models", "model", "datasets",
                    "dataset"):
            artifact = db.read_artifact(key, tag=tag, project=project_,
                                        tree=tree, iter=iteration)
            spec = artifact.get("spec", artifact)
            target = spec.get("target_path", "")
            return artifact, target
        raise MLRunInvalidArgumentError(f"unsupported store uri kind {kind}")


def parse_store_uri(url: str, default_project: str = ""):
    """Parse store://<kind>/<project>/<key>[#iter][:<tag>][@<tree>]."""
    if not url.startswith("store://"):
        raise MLRunInvalidArgumentError(f"not a store uri: {url}")
    body = url[len("store://"):]
    parts = body.split("/", 2)
    if len(parts) == 3:
        kind, project, key = parts
    elif len(parts) == 2:
        kind, key = parts
        project = default_project or "default"
    else:
        kind, project, key = "artifacts", default_project or "default", parts[0]
    tag = tree = None
    iteration = 0
    if "@" in key:
        key, tree = key.rsplit("@", 1)
    if ":" in key:
        key, tag = key.rsplit(":", 1)
    if "#" in key:
        key, it = key.rsplit("#", 1)
        iteration = int(it)
    return kind, project, key, tag, tree, iteration


store_manager = StoreManager()


def get_store_resource_uri_item(url: str, project: str = "") -> DataItem:
    return store_manager.object(url, project=project)


def get_object(url: str, secrets=None, size=None, offset=0) -> bytes:
    return store_manager.object(url).get(size=size, offset=offset)


def get_dataitem(url: str, secrets=None) -> DataItem:
    return store_manager.object(url)


def is_store_uri(url) -> bool:
    """True when the url is a store:// artifact reference
    (reference store_resources.py:28)."""
    return isinstance(url, str) and url.startswith("store://")


def get_store_uri(kind: str, uri: str) -> str:
    """Build a store://<kind>/<uri> reference
    (reference store_resources.py:48)."""
    return f"store://{kind}/{uri}"


def parse_kafka_url(url: str, brokers=None):
    """Split kafka://[broker[,broker]]/topic into (topic, brokers)
    (reference datastore/utils.py:28)."""
    from urllib.parse import urlparse

    brokers = brokers or []
    if isinstance(brokers, str):
        brokers = brokers.split(",")
    parsed = urlparse(url)
    topic = parsed.path.strip("/")
    if parsed.netloc:
        brokers = brokers or parsed.netloc.split(",")
    return topic, brokers


def get_stream_pusher(stream_path: str, **kwargs):
    """Stream pusher from a path/url (reference datastore
    __init__.py:98): kafka:// -> KafkaOutputStream, http(s):// ->
    HTTPOutputStream, dummy:// -> in-memory, else file-backed
    OutputStream."""
    from ..platforms import (HTTPOutputStream, KafkaOutputStream,
                             OutputStream)

    kafka_brokers = kwargs.get("kafka_brokers")
    if stream_path.startswith("kafka://") or kafka_brokers:
        topic, brokers = parse_kafka_url(stream_path, kafka_brokers)
        return KafkaOutputStream(
            topic, brokers, kwargs.get("kafka_producer_options"))
    if stream_path.startswith(("http://", "https://")):
        return HTTPOutputStream(stream_path)
    if stream_path.startswith("dummy://"):
        return OutputStream(stream_path)
    return OutputStream(stream_path, **{
        k: v for k, v in kwargs.items() if k in ("shards", "create")})


from .sources import (  # noqa: F401,E402
    BaseSource,
    CSVSource,
    DataFrameSource,
    KafkaSource,
    ParquetSource,
    SQLSource,
    StreamSource,
)
from .targets import (  # noqa: F401,E402
    BaseStoreTarget,
    CSVTarget,
    KafkaTarget,
    NoSqlTarget,
    ParquetTarget,
    RedisNoSqlTarget,
    SQLTarget,
    StreamTarget,
    TSDBTarget,
)


def parse_path(url: str):
    """Split a scheme url into (endpoint, path) (reference
    datastore/__init__.py parse_path)."""
    parsed = urlparse(url)
    endpoint = parsed.netloc
    return endpoint, parsed.path.lstrip("/")


def get_kafka_brokers_from_dict(options: dict, pop: bool = False):
    """Extract kafka brokers from an options dict (reference
    datastore/utils.py)."""
    if not isinstance(options, dict):
        return None
    key = "kafka_brokers"
    value = options.pop(key, None) if pop else options.get(key)
    return value


def uri_to_ipython(link: str) -> str:
    """Render a data uri as a notebook link target (reference
    datastore/__init__.py uri_to_ipython)."""
    if is_store_uri(link):
        return ""
    return link


def get_store_resource(uri: str, db=None, secrets=None, project: str = "",
                       data_store_secrets=None):
    """Resolve a store:// uri to its object (artifact / feature set /
    feature vector) — reference store_resources.py:144."""
    kind, project_, name, tag, tree, iteration = parse_store_uri(
        uri, project)
    if db is None:
        from ..db import get_run_db

        db = get_run_db(secrets=secrets) if secrets else get_run_db()
    if kind in ("feature-sets", "feature_sets"):
        from ..feature_store.feature_set import FeatureSet

        return FeatureSet.from_dict(
            db.get_feature_set(name, project_, tag=tag))
    if kind in ("feature-vectors", "feature_vectors"):
        from ..feature_store.vector import FeatureVector

        return FeatureVector.from_dict(
            db.get_feature_vector(name, project_, tag=tag))
    if kind in ("artifacts", "models", "datasets", "artifact", "model",
                "dataset"):
        from ..artifacts import dict_to_artifact

        return dict_to_artifact(
            db.read_artifact(name, tag=tag, project=project_, tree=tree,
                             iter=iteration))
    raise MLRunInvalidArgumentError(
        f"unsupported store uri kind {kind!r} in {uri}")


class HttpSource:
    """Read a remote http(s) object as a source (reference
    sources.py HttpSource) — node-local build fetches via requests."""

    kind = "http"

    def __init__(self, name: str = "", path: str = None, **kwargs):
        self.name = name
        self.path = path

    def to_dataframe(self, **kwargs):
        import io as _io

        import pandas as pd
        import requests

        resp = requests.get(self.path, timeout=30)
        resp.raise_for_status()
        if self.path.endswith(".parquet") or self.path.endswith(".pq"):
            return pd.read_parquet(_io.BytesIO(resp.content))
        return pd.read_csv(_io.BytesIO(resp.content))


in_memory_store = MemoryStore(None, "memory", "memory")


def set_in_memory_item(key: str, value):
    """Store an object under memory://<key> (reference
    datastore/__init__.py set_in_memory_item)."""
    in_memory_store.put_object(key.lstrip("/"), value)
    return DataItem(key, in_memory_store, key.lstrip("/"),
                    f"memory://{key.lstrip('/')}")


def get_in_memory_items() -> dict:
    return dict(MemoryStore._items)


from ..platforms import (  # noqa: F401,E402
    HTTPOutputStream,
    KafkaOutputStream,
    OutputStream,
)
