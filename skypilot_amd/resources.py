"""Resources — the task placement spec.

Keeps the reference's task-YAML `resources:` surface (reference:
sky/resources.py:142, sky/utils/schemas.py:293) for the fields that are
meaningful on an MI355X pool: accelerators ("MI355X:8"), cpus/memory
(`4+` minimums), infra/cloud (local | ssh | kubernetes), ports, labels,
autostop, job_recovery, image_id, disk_size.  Multi-cloud catalog fields
(region/zone/spot pricing across 20 clouds) collapse to the pool model.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from skypilot_amd.exceptions import TaskValidationError

ACCELERATOR_ALIASES = {
    "MI355X": "MI355X",
    "MI355": "MI355X",
    "AMD-MI355X": "MI355X",
}

MEMORY_PER_GPU_GB = 288  # HBM3E per MI355X


@dataclass
class AutostopConfig:
    """reference: sky/resources.py:64 (AutostopConfig)."""
    idle_minutes: int = -1
    down: bool = False

    @classmethod
    def from_yaml_config(cls, cfg) -> Optional["AutostopConfig"]:
        if cfg is None:
            return None
        if isinstance(cfg, bool):
            return cls(idle_minutes=5) if cfg else None
        if isinstance(cfg, (int, float)):
            return cls(idle_minutes=int(cfg))
        if isinstance(cfg, str):
            return cls(idle_minutes=int(cfg.rstrip("m")))
        if isinstance(cfg, dict):
            return cls(idle_minutes=int(cfg.get("idle_minutes", 5)),
                       down=bool(cfg.get("down", False)))
        raise TaskValidationError(f"bad autostop config: {cfg!r}")


@dataclass
class JobRecovery:
    """reference: sky/utils/schemas.py job_recovery subschema."""
    strategy: str = "FAILOVER"
    max_restarts_on_errors: int = 0
    # Exit codes that always recover (not billed to the restart budget)
    # — reference: job_recovery.recover_on_exit_codes.
    recover_on_exit_codes: tuple = ()

    @classmethod
    def from_yaml_config(cls, cfg) -> Optional["JobRecovery"]:
        if cfg is None:
            return None
        if isinstance(cfg, str):
            return cls(strategy=cfg.upper())
        if isinstance(cfg, dict):
            return cls(
                strategy=str(cfg.get("strategy", "FAILOVER") or "FAILOVER").upper(),
                max_restarts_on_errors=int(cfg.get("max_restarts_on_errors", 0)),
                recover_on_exit_codes=tuple(
                    int(c) for c in cfg.get("recover_on_exit_codes") or ()))
        raise TaskValidationError(f"bad job_recovery: {cfg!r}")


def _parse_plus(v) -> tuple[Optional[float], bool]:
    """'4+' -> (4.0, True); 4 -> (4.0, False); None -> (None, False)."""
    if v is None:
        return None, False
    if isinstance(v, (int, float)):
        return float(v), False
    s = str(v).strip()
    if s.endswith("+"):
        return float(s[:-1]), True
    return float(s), False


def _acc_count(v) -> float:
    """Counts may be fractional (reference schema: e.g. "A100:0.5" on
    k8s); fractions must be in (0, 1) — whole GPUs stay ints."""
    c = float(v)
    if c <= 0:
        raise TaskValidationError(f"accelerator count must be > 0: {v}")
    if c >= 1:
        if c != int(c):
            raise TaskValidationError(
                f"fractional accelerator counts must be < 1: {v}")
        return int(c)
    return c


def parse_accelerators(spec) -> tuple[Optional[str], float]:
    if spec is None:
        return None, 0
    if isinstance(spec, dict):
        (name, count), = spec.items()
        return canonical_accelerator(name), _acc_count(count)
    s = str(spec)
    if ":" in s:
        name, count = s.split(":", 1)
        return canonical_accelerator(name), _acc_count(count)
    return canonical_accelerator(s), 1


def canonical_accelerator(name: str) -> str:
    key = name.strip().upper().replace(" ", "")
    return ACCELERATOR_ALIASES.get(key, name.strip())


@dataclass
class Resources:
    """Immutable-ish resource filter + launchable spec
    (reference: sky/resources.py:142)."""
    infra: Optional[str] = None            # local | ssh | k8s pool name
    cloud: Optional[str] = None
    accelerators: Optional[str] = None     # canonical name
    accelerator_count: float = 0           # int, or a fraction in (0,1)
    cpus: Optional[float] = None
    cpus_is_min: bool = False
    memory: Optional[float] = None
    memory_is_min: bool = False
    use_spot: bool = False
    job_recovery: Optional[JobRecovery] = None
    image_id: Optional[str] = None
    disk_size: Optional[int] = None
    ports: tuple = ()
    labels: Dict[str, str] = field(default_factory=dict)
    autostop: Optional[AutostopConfig] = None
    priority: Optional[int] = None
    # True when the candidate list came from `ordered` (preference
    # order is the user's; the optimizer must not re-rank it).
    ordered: bool = False
    # Failover candidates (reference: any_of/ordered resources,
    # sky/resources.py multi-candidate sets).  candidates[0] is self's
    # config; execution tries each in order on
    # ResourcesUnavailableError.
    candidates: tuple = ()
    _raw: Dict[str, Any] = field(default_factory=dict, repr=False)

    KNOWN_KEYS = {
        "infra", "cloud", "region", "zone", "instance_type", "cpus",
        "memory", "accelerators", "accelerator_args", "use_spot",
        "job_recovery", "image_id", "disk_size", "disk_tier",
        "network_tier", "ports", "labels", "autostop", "priority",
        "any_of", "ordered",
    }

    @classmethod
    def from_yaml_config(cls, cfg: Optional[Dict[str, Any]]) -> "Resources":
        if cfg is None:
            return cls()
        if not isinstance(cfg, dict):
            raise TaskValidationError(f"resources must be a mapping: {cfg!r}")
        unknown = set(cfg) - cls.KNOWN_KEYS
        if unknown:
            raise TaskValidationError(
                f"unknown resources keys: {sorted(unknown)}")
        if "any_of" in cfg or "ordered" in cfg:
            # Multi-candidate resources (reference: sky/resources.py
            # any_of/ordered): `ordered` keeps the given preference
            # order; `any_of` is sorted cheapest-first (fewest
            # accelerators, then cpus).  execution.py retries down the
            # list on ResourcesUnavailableError.
            cand_cfgs = cfg.get("ordered") or cfg.get("any_of")
            base = {k: v for k, v in cfg.items() if k not in ("any_of",
                                                              "ordered")}
            cands = [cls.from_yaml_config({**base, **c})
                     for c in cand_cfgs]
            if "ordered" not in cfg:
                cands.sort(key=lambda r: (r.accelerator_count,
                                          r.cpus or 0))
            primary = cands[0]
            primary.candidates = tuple(cands)
            primary.ordered = "ordered" in cfg
            return primary
        acc, n = parse_accelerators(cfg.get("accelerators"))
        cpus, cpus_min = _parse_plus(cfg.get("cpus"))
        mem, mem_min = _parse_plus(cfg.get("memory"))
        ports = cfg.get("ports") or ()
        if isinstance(ports, (int, str)):
            ports = (str(ports),)
        else:
            ports = tuple(str(p) for p in ports)
        return cls(
            infra=cfg.get("infra") or cfg.get("cloud"),
            cloud=cfg.get("cloud"),
            accelerators=acc,
            accelerator_count=n,
            cpus=cpus, cpus_is_min=cpus_min,
            memory=mem, memory_is_min=mem_min,
            use_spot=bool(cfg.get("use_spot", False)),
            job_recovery=JobRecovery.from_yaml_config(cfg.get("job_recovery")),
            image_id=cfg.get("image_id"),
            disk_size=cfg.get("disk_size"),
            ports=ports,
            labels=dict(cfg.get("labels") or {}),
            autostop=AutostopConfig.from_yaml_config(cfg.get("autostop")),
            priority=cfg.get("priority"),
            _raw=dict(cfg),
        )

    def to_yaml_config(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        if self.infra:
            out["infra"] = self.infra
        if self.accelerators:
            out["accelerators"] = f"{self.accelerators}:{self.accelerator_count}"
        if self.cpus is not None:
            out["cpus"] = f"{self.cpus:g}+" if self.cpus_is_min else self.cpus
        if self.memory is not None:
            out["memory"] = (f"{self.memory:g}+" if self.memory_is_min
                             else self.memory)
        if self.use_spot:
            out["use_spot"] = True
        if self.job_recovery:
            out["job_recovery"] = {
                "strategy": self.job_recovery.strategy,
                "max_restarts_on_errors":
                    self.job_recovery.max_restarts_on_errors,
            }
        if self.ports:
            out["ports"] = list(self.ports)
        if self.labels:
            out["labels"] = dict(self.labels)
        return out

    def copy(self, **overrides) -> "Resources":
        import copy as _copy
        r = _copy.deepcopy(self)
        for k, v in overrides.items():
            setattr(r, k, v)
        return r
