"""Storage + file mounts for the MI355X pool.

Reference: sky/data/storage.py (Storage/AbstractStore, modes
MOUNT/COPY/MOUNT_CACHED :336-430) and sky/data/mounting_utils.py.  On a
single-node pool the durable store is a local directory under
`~/.sky_amd/storage/<name>` (S3-compatible remotes can layer on via
rclone later); the three modes map to:

  COPY         — materialize a copy at the mount path
  MOUNT        — symlink the mount path to the store dir (write-through,
                 survives cluster teardown: this is the managed-jobs
                 checkpoint contract, reference SURVEY.md §2.7)
  MOUNT_CACHED — local store: same as MOUNT (writeback is a no-op on a
                 local FS).  REMOTE source (s3:// etc): a real rclone
                 FUSE mount with VFS write-back
                 (`rclone mount --vfs-cache-mode writes`), matching the
                 reference's MOUNT_CACHED semantics
                 (sky/data/mounting_utils.py:698 + storage.py:378 VFS
                 presets) — writes land locally and flush to the bucket
                 asynchronously; on recovery the same bucket re-mounts.
"""
from __future__ import annotations

import os
import shutil
import time
from pathlib import Path
from typing import Any, Dict

from skypilot_amd import global_state
from skypilot_amd.exceptions import TaskValidationError
from skypilot_amd.utils.command_runner import LocalProcessCommandRunner

MODE_COPY = "COPY"
MODE_MOUNT = "MOUNT"
MODE_MOUNT_CACHED = "MOUNT_CACHED"

# Remote (S3-compatible) stores layer on via rclone (reference:
# sky/data/storage.py S3Store/GcsStore/R2Store + data_transfer).  A
# source like s3://bucket/path pulls into the local store dir at create
# time and pushes back with sync_to_remote (sky storage sync) — the
# local dir stays the mount contract, so COPY/MOUNT modes are unchanged.
REMOTE_SCHEMES = ("s3://", "gs://", "r2://", "b2://", "minio://")


def _is_remote(source: str | None) -> bool:
    return bool(source) and source.startswith(REMOTE_SCHEMES)


def _rclone_target(source: str) -> str:
    """s3://bucket/path -> s3:bucket/path (rclone remote syntax; the
    remote must be configured in ~/.config/rclone/rclone.conf)."""
    scheme, rest = source.split("://", 1)
    return f"{scheme}:{rest}"


def _rclone(*args: str) -> None:
    import shutil as _sh
    import subprocess
    if _sh.which("rclone") is None:
        raise TaskValidationError(
            "remote storage requires rclone on PATH (not bundled in "
            "this offline image); configure ~/.config/rclone/rclone.conf")
    proc = subprocess.run(["rclone", *args], capture_output=True,
                          text=True)
    if proc.returncode != 0:
        raise TaskValidationError(
            f"rclone {' '.join(args[:2])} failed: {proc.stderr[:400]}")


def sync_to_remote(name: str) -> str:
    """Push a store's local contents back to its remote source."""
    rec = next((r for r in list_storage() if r["name"] == name), None)
    if rec is None:
        raise TaskValidationError(f"no such storage {name!r}")
    if not _is_remote(rec["source"]):
        raise TaskValidationError(
            f"storage {name!r} has no remote source to sync to")
    d = storage_root() / name
    _rclone("sync", str(d), _rclone_target(rec["source"]))
    return rec["source"]


def storage_root() -> Path:
    d = global_state.root_dir() / "storage"
    d.mkdir(parents=True, exist_ok=True)
    return d


def get_or_create_store(name: str, source: str | None = None) -> Path:
    d = storage_root() / name
    first = not d.exists()
    d.mkdir(parents=True, exist_ok=True)
    if first:
        store_type = ("s3" if _is_remote(source) else "local")
        with global_state._DB_LOCK, global_state._conn() as c:
            c.execute(
                "INSERT OR IGNORE INTO storage "
                "(name,source,store_type,created_at) VALUES (?,?,?,?)",
                (name, source or "", store_type, time.time()))
        if _is_remote(source):
            _rclone("sync", _rclone_target(source), str(d))
        elif source:
            src = os.path.expanduser(source)
            if os.path.isdir(src):
                LocalProcessCommandRunner().rsync(src.rstrip("/") + "/",
                                                  str(d))
            elif os.path.exists(src):
                shutil.copy2(src, d)
    return d


def list_storage():
    with global_state._DB_LOCK, global_state._conn() as c:
        rows = c.execute(
            "SELECT name,source,store_type,created_at FROM storage").fetchall()
    return [{"name": r[0], "source": r[1], "store_type": r[2],
             "created_at": r[3]} for r in rows]


def delete_storage(name: str) -> bool:
    d = storage_root() / name
    existed = d.exists()
    shutil.rmtree(d, ignore_errors=True)
    with global_state._DB_LOCK, global_state._conn() as c:
        c.execute("DELETE FROM storage WHERE name=?", (name,))
    return existed


def _mount_cached_remote(source: str, target: Path,
                         handle: Dict[str, Any]) -> None:
    """rclone FUSE mount with VFS write-back for a remote bucket
    (reference: sky/data/mounting_utils.py:698 rclone mount command,
    sky/data/storage.py:378 MOUNT_CACHED VFS presets)."""
    import shutil as _sh
    import subprocess
    if _sh.which("rclone") is None:
        raise TaskValidationError(
            "MOUNT_CACHED with a remote source requires rclone on PATH")
    target.mkdir(parents=True, exist_ok=True)
    if os.path.ismount(target):
        return
    cache_dir = Path(handle["cluster_dir"]) / ".rclone-vfs-cache"
    cache_dir.mkdir(parents=True, exist_ok=True)
    proc = subprocess.run(
        ["rclone", "mount", _rclone_target(source), str(target),
         "--daemon", "--allow-non-empty",
         "--vfs-cache-mode", "writes",
         "--cache-dir", str(cache_dir),
         "--dir-cache-time", "10s",
         "--vfs-write-back", "1s"],
        capture_output=True, text=True, timeout=60)
    if proc.returncode != 0:
        raise TaskValidationError(
            f"rclone mount failed: {proc.stderr[:400]}")
    # record for best-effort unmount at teardown
    mounts = Path(handle["cluster_dir"]) / "rclone-mounts"
    with open(mounts, "a") as f:
        f.write(str(target) + "\n")


def unmount_cluster_mounts(cluster_dir: str) -> None:
    """Best-effort fusermount -u of rclone mounts at teardown."""
    import subprocess
    mounts = Path(cluster_dir) / "rclone-mounts"
    if not mounts.exists():
        return
    for line in mounts.read_text().splitlines():
        if line.strip():
            subprocess.run(["fusermount", "-u", line.strip()],
                           capture_output=True, timeout=30)
    mounts.unlink(missing_ok=True)


def _resolve_dst(handle: Dict[str, Any], dst: str) -> Path:
    """Absolute mount paths are real paths (single-node pool shares the
    FS); relative paths land in the cluster workdir."""
    if os.path.isabs(dst):
        return Path(dst)
    return Path(handle["cluster_dir"]) / "workdir" / dst


def execute_file_mounts(handle: Dict[str, Any],
                        file_mounts: Dict[str, Any]) -> None:
    runner = LocalProcessCommandRunner()
    for dst, src in (file_mounts or {}).items():
        target = _resolve_dst(handle, dst)
        target.parent.mkdir(parents=True, exist_ok=True)
        if isinstance(src, str):
            path = os.path.expanduser(src)
            if not os.path.exists(path):
                raise TaskValidationError(
                    f"file_mount source not found: {src}")
            if os.path.isdir(path):
                runner.rsync(path.rstrip("/") + "/", str(target))
            else:
                target.parent.mkdir(parents=True, exist_ok=True)
                shutil.copy2(path, target)
        elif isinstance(src, dict):
            if "volume" in src:
                from skypilot_amd.data import volumes
                vpath = volumes.mount_path(src["volume"])
                if target.is_symlink():
                    target.unlink()
                if not target.exists():
                    target.symlink_to(vpath)
                continue
            name = src.get("name")
            if not name:
                raise TaskValidationError(
                    f"storage mount for {dst} needs a name")
            mode = str(src.get("mode", MODE_MOUNT)).upper()
            if mode == MODE_MOUNT_CACHED and _is_remote(src.get("source")):
                # register the store row (no local pull: the FUSE mount
                # IS the data path) and mount the bucket directly.
                get_or_create_store(name, None)
                with global_state._DB_LOCK, global_state._conn() as c:
                    c.execute("UPDATE storage SET source=?, store_type=? "
                              "WHERE name=?",
                              (src["source"], "s3-mount", name))
                _mount_cached_remote(src["source"], target, handle)
                continue
            store = get_or_create_store(name, src.get("source"))
            if mode == MODE_COPY:
                if target.is_symlink():
                    target.unlink()
                runner.rsync(str(store) + "/", str(target))
            elif mode in (MODE_MOUNT, MODE_MOUNT_CACHED):
                if target.is_symlink() or target.exists():
                    if target.is_symlink():
                        target.unlink()
                    elif target.is_dir() and not any(target.iterdir()):
                        target.rmdir()
                    else:
                        continue  # real data already at mount point
                target.symlink_to(store)
            else:
                raise TaskValidationError(f"unknown mount mode {mode}")
        else:
            raise TaskValidationError(f"bad file_mount value for {dst}")
