"""Service spec — the `service:` section of a task YAML.

Reference: sky/serve/service_spec.py:24-48 (readiness_probe,
replica_policy {min/max_replicas, target_qps_per_replica,
upscale/downscale_delay_seconds}, replicas, load_balancing_policy).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from skypilot_amd.exceptions import TaskValidationError


@dataclass
class ReadinessProbe:
    path: str = "/health"
    initial_delay_seconds: int = 60
    timeout_seconds: int = 10
    post_data: Optional[Dict[str, Any]] = None

    @classmethod
    def from_config(cls, cfg) -> "ReadinessProbe":
        if cfg is None:
            return cls()
        if isinstance(cfg, str):
            return cls(path=cfg)
        return cls(
            path=cfg.get("path", "/health"),
            initial_delay_seconds=int(cfg.get("initial_delay_seconds", 60)),
            timeout_seconds=int(cfg.get("timeout_seconds", 10)),
            post_data=cfg.get("post_data"),
        )


@dataclass
class ReplicaPolicy:
    min_replicas: int = 1
    max_replicas: Optional[int] = None
    target_qps_per_replica: Optional[float] = None
    num_overprovision: int = 0  # extra replicas above the computed target
    upscale_delay_seconds: int = 300
    downscale_delay_seconds: int = 1200

    @classmethod
    def from_config(cls, cfg) -> "ReplicaPolicy":
        if cfg is None:
            return cls()
        mn = int(cfg.get("min_replicas", 1))
        mx = cfg.get("max_replicas")
        return cls(
            min_replicas=mn,
            max_replicas=int(mx) if mx is not None else None,
            target_qps_per_replica=cfg.get("target_qps_per_replica"),
            num_overprovision=int(cfg.get("num_overprovision", 0)),
            upscale_delay_seconds=int(cfg.get("upscale_delay_seconds", 300)),
            downscale_delay_seconds=int(
                cfg.get("downscale_delay_seconds", 1200)),
        )


@dataclass
class TLSConfig:
    """LB TLS termination (reference: sky/serve service schema `tls:`
    keyfile/certfile).  `tls: true` auto-generates a self-signed pair
    under the service dir (openssl), for pools without a CA."""
    certfile: Optional[str] = None
    keyfile: Optional[str] = None
    auto: bool = False

    @classmethod
    def from_config(cls, cfg) -> Optional["TLSConfig"]:
        if not cfg:
            return None
        if cfg is True:
            return cls(auto=True)
        if isinstance(cfg, dict):
            return cls(certfile=cfg.get("certfile"),
                       keyfile=cfg.get("keyfile"),
                       auto=bool(cfg.get("auto", False)))
        raise TaskValidationError(f"bad tls config: {cfg!r}")

    def ensure_materialized(self, out_dir) -> "TLSConfig":
        """Generate a self-signed pair if auto and none given."""
        import pathlib
        import subprocess
        if self.certfile and self.keyfile:
            return self
        d = pathlib.Path(out_dir)
        d.mkdir(parents=True, exist_ok=True)
        cert, key = d / "lb.crt", d / "lb.key"
        if not (cert.exists() and key.exists()):
            subprocess.run(
                ["openssl", "req", "-x509", "-newkey", "rsa:2048",
                 "-keyout", str(key), "-out", str(cert), "-days", "3650",
                 "-nodes", "-subj", "/CN=skypilot-amd-lb"],
                check=True, capture_output=True)
        return TLSConfig(certfile=str(cert), keyfile=str(key))


@dataclass
class ServiceSpec:
    readiness_probe: ReadinessProbe = field(default_factory=ReadinessProbe)
    policy: ReplicaPolicy = field(default_factory=ReplicaPolicy)
    load_balancing_policy: str = "least_load"
    port: Optional[int] = None  # replica port (task may also use $PORT)
    tls: Optional[TLSConfig] = None

    KNOWN_KEYS = {"readiness_probe", "replica_policy", "replicas",
                  "load_balancing_policy", "load_balancer", "ports", "tls",
                  "endpoint_probe_interval_seconds",
                  "stream_timeout_seconds"}

    @classmethod
    def from_config(cls, cfg: Dict[str, Any]) -> "ServiceSpec":
        if not isinstance(cfg, dict):
            raise TaskValidationError("service: must be a mapping")
        unknown = set(cfg) - cls.KNOWN_KEYS
        if unknown:
            raise TaskValidationError(f"unknown service keys: {unknown}")
        if "replicas" in cfg and "replica_policy" in cfg:
            raise TaskValidationError(
                "use either replicas or replica_policy, not both")
        if "replicas" in cfg:
            n = int(cfg["replicas"])
            policy = ReplicaPolicy(min_replicas=n, max_replicas=n)
        else:
            policy = ReplicaPolicy.from_config(cfg.get("replica_policy"))
        port = cfg.get("ports")
        return cls(
            readiness_probe=ReadinessProbe.from_config(
                cfg.get("readiness_probe")),
            policy=policy,
            load_balancing_policy=cfg.get("load_balancing_policy",
                                          "least_load"),
            port=int(port) if port is not None else None,
            tls=TLSConfig.from_config(cfg.get("tls")),
        )
