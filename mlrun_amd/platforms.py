# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Platform glue: output streams + mount helpers.

Parity target: reference mlrun/platforms/iguazio.py (OutputStream :80,
KafkaOutputStream :177, V3ioStreamClient :232) and the mount_v3io /
auto_mount volume helpers.  Node-locally, streams are in-process
bounded queues (shared with serving QueueSteps) persisted optionally
to parquet, and mounts are no-ops (the node's filesystem is the
volume).
"""

import json
import queue
import threading



class OutputStream:
    """Push records to a named node-local stream (consumers attach via
    get_stream / StreamSource)."""

    _streams: dict = {}
    _lock = threading.Lock()

    def __init__(self, stream_path: str, shards: int = 1,
                 create: bool = True, max_events: int = 65536):
        self.path = stream_path
        with OutputStream._lock:
            if stream_path not in OutputStream._streams:
                OutputStream._streams[stream_path] = queue.Queue(
                    maxsize=max_events)
            self._queue = OutputStream._streams[stream_path]

    def push(self, data):
        records = data if isinstance(data, list) else [data]
        for record in records:
            if isinstance(record, (dict, list)):
                record = json.dumps(record, default=str)
            try:
                self._queue.put_nowait(record)
            except queue.Full:
                self._queue.get_nowait()  # drop-oldest backpressure
                self._queue.put_nowait(record)

    def drain(self, max_batch: int = 4096) -> list:
        out = []
        while len(out) < max_batch:
            try:
                out.append(self._queue.get_nowait())
            except queue.Empty:
                break
        return out

    @classmethod
    def get_stream(cls, stream_path: str) -> "OutputStream":
        return cls(stream_path)


class KafkaOutputStream(OutputStream):
    """Alias kind — the node-local build has no Kafka; kept for spec
    compatibility (push/drain semantics are identical).  Accepts the
    reference (topic, brokers, producer_options) signature."""

    def __init__(self, topic: str, brokers=None,
                 producer_options: dict = None, **kwargs):
        self.brokers = brokers if isinstance(brokers, list) else (
            (brokers or "").split(",") if brokers else [])
        self.producer_options = producer_options or {}
        super().__init__(topic, **kwargs)


class HTTPOutputStream(OutputStream):
    """Push records by POSTing them to an HTTP endpoint (reference
    platforms/iguazio.py HTTPOutputStream)."""

    def __init__(self, stream_path: str, **kwargs):
        self.path = stream_path
        self._queue = None

    def push(self, data):
        import requests

        records = data if isinstance(data, list) else [data]
        for record in records:
            if isinstance(record, (dict, list)):
                resp = requests.post(self.path, json=record, timeout=10)
            else:
                resp = requests.post(self.path, data=record, timeout=10)
            resp.raise_for_status()

    def drain(self, max_batch: int = 4096) -> list:
        raise NotImplementedError(
            "HTTPOutputStream is write-only (consume at the endpoint)")


def mount_v3io(*args, **kwargs):
    """No-op modifier (the node filesystem is already mounted)."""

    def apply(runtime):
        return runtime

    return apply


def auto_mount(*args, **kwargs):
    return mount_v3io()


def v3io_cred(*args, **kwargs):
    return mount_v3io()


class VolumeMount:
    """Spec-compat volume mount record (node-local paths)."""

    def __init__(self, path: str, sub_path: str = ""):
        self.path = path
        self.sub_path = sub_path
