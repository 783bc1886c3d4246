"""Task — one unit of work: setup + run + resources + mounts.

Keeps the reference task-YAML surface (reference: sky/task.py:318,
sky/utils/schemas.py:get_task_schema:1161): name, workdir, num_nodes,
resources, file_mounts, storage mounts, service, setup, run, envs,
secrets.  YAML round-trip preserved so reference examples run unmodified.
"""
from __future__ import annotations

import os
import re
from typing import Any, Callable, Dict, Optional, Union

import yaml

from skypilot_amd.exceptions import TaskValidationError
from skypilot_amd.resources import Resources

_VALID_NAME = re.compile(r"^[a-zA-Z0-9][a-zA-Z0-9._-]*$")

KNOWN_TASK_KEYS = {
    "name", "workdir", "event_callback", "num_nodes", "resources",
    "file_mounts", "service", "pool", "setup", "run", "envs", "secrets",
    "config", "volumes", "volume_mounts", "inputs", "outputs",
    "file_mounts_mapping", "managed_secrets", "api_server_access",
    "_metadata",
}



def _resolve_upload(value):
    """Resolve a client-upload marker {"upload": <id>} to the server's
    extracted blob directory (see server /api/upload)."""
    if isinstance(value, dict) and "upload" in value:
        import re as _re
        uid = str(value["upload"])
        if not _re.fullmatch(r"[a-f0-9]{32}", uid):
            raise TaskValidationError(f"bad upload id {uid!r}")
        from skypilot_amd import global_state
        path = global_state.root_dir() / "api" / "uploads" / uid
        if not path.exists():
            raise TaskValidationError(
                f"upload {uid} not found on the API server")
        return str(path)
    return value


class Task:
    def __init__(self,
                 name: Optional[str] = None,
                 *,
                 setup: Optional[str] = None,
                 run: Optional[Union[str, Callable]] = None,
                 envs: Optional[Dict[str, str]] = None,
                 secrets: Optional[Dict[str, str]] = None,
                 workdir: Optional[str] = None,
                 num_nodes: int = 1,
                 file_mounts: Optional[Dict[str, Any]] = None,
                 resources: Optional[Resources] = None,
                 service: Optional[Dict[str, Any]] = None,
                 event_callback: Optional[str] = None):
        if name is not None and not _VALID_NAME.match(name):
            raise TaskValidationError(f"invalid task name: {name!r}")
        self.name = name
        self.setup = setup
        self.run = run
        self.envs = dict(envs or {})
        self.secrets = dict(secrets or {})
        self.workdir = workdir
        self.num_nodes = int(num_nodes or 1)
        self.file_mounts = dict(file_mounts or {})
        self.resources = resources or Resources()
        self.service = service
        self.event_callback = event_callback
        self.validate()

    # -- fluent setters (reference: sky/task.py fluent API) -----------------
    def set_resources(self, resources: Resources) -> "Task":
        self.resources = resources
        return self

    def set_file_mounts(self, fm: Dict[str, Any]) -> "Task":
        self.file_mounts = dict(fm or {})
        return self

    def update_envs(self, envs: Dict[str, str]) -> "Task":
        self.envs.update(envs or {})
        return self

    def set_service(self, service) -> "Task":
        self.service = service
        return self

    # -- validation ---------------------------------------------------------
    def validate(self):
        self.validate_name()
        self.validate_run()
        if self.num_nodes < 1:
            raise TaskValidationError("num_nodes must be >= 1")
        if self.workdir is not None and not isinstance(self.workdir,
                                                       (str, dict)):
            raise TaskValidationError(
                "workdir must be a path or {url, ref} git source")
        if isinstance(self.workdir, dict):
            # reference: schemas.py workdir {url, ref} git-source form
            if "url" not in self.workdir:
                raise TaskValidationError("git workdir needs a url")
        for k in self.envs:
            if not isinstance(k, str):
                raise TaskValidationError(f"env name must be str: {k!r}")

    def validate_name(self):
        if self.name is not None and not _VALID_NAME.match(self.name):
            raise TaskValidationError(f"invalid task name: {self.name!r}")

    def validate_run(self):
        if self.run is not None and not isinstance(self.run, str) \
                and not callable(self.run):
            raise TaskValidationError("run must be a string or callable")

    # -- YAML round trip ----------------------------------------------------
    @classmethod
    def from_yaml_config(cls, config: Dict[str, Any],
                         env_overrides: Optional[Dict[str, str]] = None
                         ) -> "Task":
        if not isinstance(config, dict):
            raise TaskValidationError("task YAML must be a mapping")
        unknown = set(config) - KNOWN_TASK_KEYS
        if unknown:
            raise TaskValidationError(
                f"unknown task YAML keys: {sorted(unknown)}")
        envs = dict(config.get("envs") or {})
        if env_overrides:
            envs.update(env_overrides)
        # Client-uploaded blobs (reference: sky server /upload +
        # client/common.py:154-192): {"upload": <id>} markers in workdir
        # and file_mounts resolve to the server-side extraction dir.
        config = dict(config)
        config["workdir"] = _resolve_upload(config.get("workdir"))
        if config.get("file_mounts"):
            config["file_mounts"] = {
                k: _resolve_upload(v)
                for k, v in config["file_mounts"].items()}
        # ${VAR} substitution in run/setup from envs (reference behavior is
        # shell-level; we keep envs as env vars, no substitution needed).
        task = cls(
            name=config.get("name"),
            setup=config.get("setup"),
            run=config.get("run"),
            envs={k: "" if v is None else str(v) for k, v in envs.items()},
            secrets=dict(config.get("secrets") or {}),
            workdir=config.get("workdir"),
            num_nodes=config.get("num_nodes") or 1,
            file_mounts=dict(config.get("file_mounts") or {}),
            resources=Resources.from_yaml_config(config.get("resources")),
            service=config.get("service"),
            event_callback=config.get("event_callback"),
        )
        return task

    @classmethod
    def from_yaml(cls, path: str,
                  env_overrides: Optional[Dict[str, str]] = None) -> "Task":
        with open(os.path.expanduser(path)) as f:
            config = yaml.safe_load(f) or {}
        return cls.from_yaml_config(config, env_overrides)

    def to_yaml_config(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        for key, val in [
            ("name", self.name), ("workdir", self.workdir),
            ("num_nodes", self.num_nodes if self.num_nodes != 1 else None),
            ("setup", self.setup),
            ("run", self.run if isinstance(self.run, str) else None),
        ]:
            if val is not None:
                out[key] = val
        if self.envs:
            out["envs"] = dict(self.envs)
        if self.secrets:
            out["secrets"] = {k: "<redacted>" for k in self.secrets}
        if self.file_mounts:
            out["file_mounts"] = dict(self.file_mounts)
        res = self.resources.to_yaml_config()
        if res:
            out["resources"] = res
        if self.service:
            out["service"] = self.service
        return out

    def to_yaml(self) -> str:
        return yaml.safe_dump(self.to_yaml_config(), sort_keys=False)

    def __repr__(self):
        acc = self.resources.accelerators
        acc_s = f", {acc}:{self.resources.accelerator_count}" if acc else ""
        return f"Task({self.name or '<unnamed>'}{acc_s})"
