"""Autoscalers (reference: sky/serve/autoscalers.py — Autoscaler:138
with hysteresis :393, RequestRateAutoscaler:479)."""
from __future__ import annotations

import math
import time
from typing import Optional

from skypilot_amd.serve.service_spec import ReplicaPolicy

QPS_WINDOW_SECONDS = 60.0


class Autoscaler:
    def __init__(self, policy: ReplicaPolicy):
        self.policy = policy

    def target_replicas(self, qps: float, current: int) -> int:
        raise NotImplementedError


class FixedAutoscaler(Autoscaler):
    """No target_qps: keep min_replicas (== replicas for `replicas: N`)."""

    def target_replicas(self, qps: float, current: int) -> int:
        return self.policy.min_replicas


class RequestRateAutoscaler(Autoscaler):
    """Scale to ceil(qps / target_qps_per_replica) with upscale/downscale
    delay hysteresis (reference: autoscalers.py:479)."""

    def __init__(self, policy: ReplicaPolicy):
        super().__init__(policy)
        self._upscale_since: Optional[float] = None
        self._downscale_since: Optional[float] = None

    def _raw_target(self, qps: float) -> int:
        p = self.policy
        want = math.ceil(qps / p.target_qps_per_replica) if qps > 0 else 0
        want = max(want, p.min_replicas) + p.num_overprovision
        if p.max_replicas is not None:
            want = min(want, p.max_replicas)
        return want

    def target_replicas(self, qps: float, current: int) -> int:
        want = self._raw_target(qps)
        now = time.time()
        if want > current:
            self._downscale_since = None
            if self._upscale_since is None:
                self._upscale_since = now
            if now - self._upscale_since >= self.policy.upscale_delay_seconds:
                self._upscale_since = None
                return want
            return current
        if want < current:
            self._upscale_since = None
            if self._downscale_since is None:
                self._downscale_since = now
            if now - self._downscale_since >= \
                    self.policy.downscale_delay_seconds:
                self._downscale_since = None
                return want
            return current
        self._upscale_since = self._downscale_since = None
        return current


def make_autoscaler(policy: ReplicaPolicy) -> Autoscaler:
    if policy.target_qps_per_replica:
        return RequestRateAutoscaler(policy)
    return FixedAutoscaler(policy)
