"""Volumes — persistent named directories attachable to tasks
(reference: sky/volumes/ — k8s PVC / runpod volumes; on the local pool a
volume is a durable directory under ~/.sky_amd/volumes/<name>, mounted
into tasks via symlink like MOUNT-mode storage)."""
from __future__ import annotations

import shutil
import time
from pathlib import Path
from typing import Dict, List, Optional

from skypilot_amd import global_state


def volumes_root() -> Path:
    d = global_state.root_dir() / "volumes"
    d.mkdir(parents=True, exist_ok=True)
    return d


def create(name: str, size_gb: Optional[int] = None) -> Dict:
    d = volumes_root() / name
    d.mkdir(parents=True, exist_ok=True)
    meta = {"name": name, "path": str(d), "size_gb": size_gb,
            "created_at": time.time()}
    global_state.set_config(f"volume:{name}", meta)
    return meta


def get(name: str) -> Optional[Dict]:
    return global_state.get_config(f"volume:{name}")


def list_volumes() -> List[Dict]:
    out = []
    for d in sorted(volumes_root().iterdir()):
        if d.is_dir():
            meta = get(d.name) or {"name": d.name, "path": str(d)}
            out.append(meta)
    return out


def delete(name: str) -> bool:
    d = volumes_root() / name
    existed = d.exists()
    shutil.rmtree(d, ignore_errors=True)
    global_state.set_config(f"volume:{name}", None)
    return existed


def mount_path(name: str) -> Path:
    """Tasks mount volumes by symlinking dst -> this path (see
    data/storage.execute_file_mounts for the storage analog)."""
    create(name)
    return volumes_root() / name
