# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Data-drift metrics: TVD / Hellinger / KL over feature histograms.

Parity target: reference
mlrun/model_monitoring/applications/histogram_data_drift.py.
"""

import typing

import numpy as np


def _normalize(hist: np.ndarray) -> np.ndarray:
    hist = np.asarray(hist, dtype=np.float64)
    total = hist.sum()
    if total <= 0:
        return np.full_like(hist, 1.0 / max(len(hist), 1))
    return hist / total


def total_variation_distance(p, q) -> float:
    p, q = _normalize(p), _normalize(q)
    return float(0.5 * np.abs(p - q).sum())


def hellinger_distance(p, q) -> float:
    p, q = _normalize(p), _normalize(q)
    return float(np.sqrt(max(0.0, 1.0 - np.sum(np.sqrt(p * q)))))


def kl_divergence(p, q, eps: float = 1e-10) -> float:
    p, q = _normalize(p), _normalize(q)
    p = np.clip(p, eps, None)
    q = np.clip(q, eps, None)
    return float(np.sum(p * np.log(p / q)))


def histogram_drift_metrics(reference_hist, current_hist) -> dict:
    """Compute the drift metric triple + the combined drift score
    (mean of TVD and Hellinger, as in the reference app)."""
    tvd = total_variation_distance(reference_hist, current_hist)
    hellinger = hellinger_distance(reference_hist, current_hist)
    kld = kl_divergence(reference_hist, current_hist)
    return {
        "tvd": tvd,
        "hellinger": hellinger,
        "kld": kld,
        "drift_score": (tvd + hellinger) / 2.0,
    }


def compute_feature_drift(reference_samples: typing.Sequence,
                          current_samples: typing.Sequence,
                          bins: int = 20) -> dict:
    """Histogram two sample sets over a shared range and compute drift."""
    ref = np.asarray(reference_samples, dtype=np.float64).ravel()
    cur = np.asarray(current_samples, dtype=np.float64).ravel()
    if ref.size == 0 or cur.size == 0:
        return {"tvd": 0.0, "hellinger": 0.0, "kld": 0.0, "drift_score": 0.0}
    low = min(ref.min(), cur.min())
    high = max(ref.max(), cur.max())
    if low == high:
        high = low + 1.0
    ref_hist, edges = np.histogram(ref, bins=bins, range=(low, high))
    cur_hist, _ = np.histogram(cur, bins=edges)
    return histogram_drift_metrics(ref_hist, cur_hist)


DRIFT_DETECTED_THRESHOLD = 0.7   # defaults; overridable via config
POSSIBLE_DRIFT_THRESHOLD = 0.5   # model_monitoring.drift_thresholds


def drift_status(drift_score: float) -> str:
    """Classify a drift score (reference histogram_data_drift
    thresholds, overridable via
    config.model_monitoring.drift_thresholds.{detected,possible})."""
    detected, possible = (DRIFT_DETECTED_THRESHOLD,
                          POSSIBLE_DRIFT_THRESHOLD)
    try:
        from ..config import config

        thresholds = config.model_monitoring.drift_thresholds
        detected = float(thresholds.detected)
        possible = float(thresholds.possible)
    except Exception:
        pass
    if drift_score >= detected:
        return "drift_detected"
    if drift_score >= possible:
        return "possible_drift"
    return "no_drift"
