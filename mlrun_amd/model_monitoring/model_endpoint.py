# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Client-side model-endpoint record object (reference
model_monitoring/model_endpoint.py:69 ModelEndpoint) and the
TrackingPolicy monitoring configuration (reference
model_monitoring/tracking_policy.py:22)."""

import hashlib

from ..model import ModelObj


class ModelEndpointMetadata(ModelObj):
    def __init__(self, project: str = None, labels: dict = None,
                 uid: str = None):
        self.project = project
        self.labels = labels or {}
        self.uid = uid


class ModelEndpointSpec(ModelObj):
    def __init__(self, function_uri: str = "", model: str = "",
                 model_class: str = "", model_uri: str = "",
                 feature_names: list = None, label_names: list = None,
                 stream_path: str = "", algorithm: str = "",
                 monitor_configuration: dict = None, active: bool = True,
                 monitoring_mode: str = "disabled"):
        self.function_uri = function_uri
        self.model = model
        self.model_class = model_class
        self.model_uri = model_uri
        self.feature_names = feature_names or []
        self.label_names = label_names or []
        self.stream_path = stream_path
        self.algorithm = algorithm
        self.monitor_configuration = monitor_configuration or {}
        self.active = active
        self.monitoring_mode = monitoring_mode


class ModelEndpointStatus(ModelObj):
    def __init__(self, feature_stats: dict = None,
                 current_stats: dict = None, first_request: str = "",
                 last_request: str = "", error_count: int = 0,
                 drift_status: str = "", drift_measures: dict = None,
                 metrics: dict = None, features: list = None,
                 children: list = None, children_uids: list = None,
                 endpoint_type: str = "node-ep",
                 monitoring_feature_set_uri: str = "", state: str = ""):
        self.feature_stats = feature_stats or {}
        self.current_stats = current_stats or {}
        self.first_request = first_request
        self.last_request = last_request
        self.error_count = error_count
        self.drift_status = drift_status
        self.drift_measures = drift_measures or {}
        self.metrics = metrics or {}
        self.features = features or []
        self.children = children or []
        self.children_uids = children_uids or []
        self.endpoint_type = endpoint_type
        self.monitoring_feature_set_uri = monitoring_feature_set_uri
        self.state = state


class ModelEndpoint(ModelObj):
    """Model-endpoint record: identity (function+model), monitored
    feature schema and live status (reference
    model_endpoint.py:69)."""

    kind = "model-endpoint"
    _dict_fields = ["kind", "metadata", "spec", "status"]

    def __init__(self):
        self.metadata = ModelEndpointMetadata()
        self.spec = ModelEndpointSpec()
        self.status = ModelEndpointStatus()

    def create_endpoint_id(self) -> str:
        """Deterministic endpoint uid from function_uri + model
        (reference: hashed endpoint id)."""
        if not self.spec.function_uri or not self.spec.model:
            from ..errors import MLRunInvalidArgumentError

            raise MLRunInvalidArgumentError(
                "function_uri and model must be set to generate an "
                "endpoint id")
        digest = hashlib.sha1(
            f"{self.spec.function_uri}_{self.spec.model}".encode()
        ).hexdigest()
        self.metadata.uid = digest
        return digest

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        obj = cls()
        obj.metadata = ModelEndpointMetadata.from_dict(
            struct.get("metadata") or {})
        obj.spec = ModelEndpointSpec.from_dict(struct.get("spec") or {})
        obj.status = ModelEndpointStatus.from_dict(
            struct.get("status") or {})
        return obj


class TrackingPolicy(ModelObj):
    """Model-monitoring scheduling/config policy (reference
    tracking_policy.py:22): batch-job cadence + stream base period."""

    _dict_fields = ["default_batch_intervals", "default_batch_image",
                    "stream_image", "base_period",
                    "default_controller_image"]

    def __init__(self, default_batch_intervals: str = "0 */1 * * *",
                 default_batch_image: str = "mlrun/mlrun",
                 stream_image: str = "mlrun/mlrun",
                 base_period: int = 10,
                 default_controller_image: str = "mlrun/mlrun"):
        self.default_batch_intervals = default_batch_intervals
        self.default_batch_image = default_batch_image
        self.stream_image = stream_image
        self.base_period = base_period
        self.default_controller_image = default_controller_image


def get_stream_path(project: str, function_name: str = "stream",
                    stream_uri: str = None) -> str:
    """Monitoring stream path for a project/application (reference
    helpers.py:45): env/secret override, else the node-local stream
    name used by OutputStream/get_stream_pusher."""
    from ..secrets import get_secret_or_env

    stream_uri = stream_uri or get_secret_or_env(
        "STREAM_PATH") or f"monitoring/{project}/{function_name}"
    return stream_uri
