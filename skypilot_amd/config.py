"""Layered configuration (reference: sky/skypilot_config.py — server
config -> user ~/.sky_amd/config.yaml -> project ./sky_amd_config.yaml
-> task-YAML `config:` overrides, with get_nested access)."""
from __future__ import annotations

import copy
import os
import threading
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml

_lock = threading.Lock()
_cache: Optional[Dict[str, Any]] = None

# The user config lives under the state root so SKY_AMD_HOME relocates
# it together with the rest of the deployment (and tests isolate it).
def _user_config_path() -> str:
    import os as _os
    if USER_CONFIG_PATH != "~/.sky_amd/config.yaml":
        return USER_CONFIG_PATH  # explicitly overridden (tests/tools)
    return _os.path.join(
        _os.environ.get("SKY_AMD_HOME", "~/.sky_amd"), "config.yaml")


USER_CONFIG_PATH = "~/.sky_amd/config.yaml"  # legacy fallback
PROJECT_CONFIG_PATH = "./sky_amd_config.yaml"

DEFAULTS: Dict[str, Any] = {
    "pool": {
        "name": "local",
        "gpus_per_node": 8,
        "accelerator": "MI355X",
    },
    "api_server": {"port": 46580},
    "jobs": {"controller_poll_seconds": 2.0},
    "serve": {"controller_poll_seconds": 2.0},
    "train": {"bucket_mb": 64},
}


def _merge(base: Dict[str, Any], override: Dict[str, Any]) -> Dict[str, Any]:
    out = copy.deepcopy(base)
    for k, v in (override or {}).items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def _load_file(path: str) -> Dict[str, Any]:
    p = Path(os.path.expanduser(path))
    if not p.exists():
        return {}
    try:
        with open(p) as f:
            return yaml.safe_load(f) or {}
    except (OSError, yaml.YAMLError):
        return {}


def load(refresh: bool = False) -> Dict[str, Any]:
    global _cache
    with _lock:
        if _cache is None or refresh:
            cfg = copy.deepcopy(DEFAULTS)
            cfg = _merge(cfg, _load_file(_user_config_path()))
            cfg = _merge(cfg, _load_file(PROJECT_CONFIG_PATH))
            _cache = cfg
        return _cache


def get_nested(keys: List[str], default: Any = None,
               override_configs: Optional[Dict[str, Any]] = None) -> Any:
    """reference: skypilot_config.get_nested(keys, default, overrides)."""
    cfg = load()
    if override_configs:
        cfg = _merge(cfg, override_configs)
    cur: Any = cfg
    for k in keys:
        if not isinstance(cur, dict) or k not in cur:
            return default
        cur = cur[k]
    return cur
