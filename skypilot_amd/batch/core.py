"""Batch processing — shard a dataset over pool jobs.

Reference: sky/batch/ (Dataset over JSONL, @remote_function,
BatchCoordinator:132 fanning rows out to a jobs pool).  Local-pool
version: the dataset is JSONL on the shared FS; `run_batch` shards it,
launches one cluster job per shard running the user's command with
SKY_BATCH_INPUT/SKY_BATCH_OUTPUT env, and concatenates shard outputs.
"""
from __future__ import annotations

import json
import math
import time
from pathlib import Path
from typing import Any, Dict, Iterable, List, Optional

from skypilot_amd import execution, global_state
from skypilot_amd.backends.pool_backend import PoolBackend
from skypilot_amd.task import Task


class Dataset:
    """A JSONL-backed dataset (reference: sky/batch Dataset)."""

    def __init__(self, rows: List[Dict[str, Any]]):
        self.rows = rows

    @classmethod
    def from_jsonl(cls, path: str) -> "Dataset":
        rows = []
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line:
                    rows.append(json.loads(line))
        return cls(rows)

    @classmethod
    def from_list(cls, rows: Iterable[Dict[str, Any]]) -> "Dataset":
        return cls(list(rows))

    def write_jsonl(self, path: str) -> None:
        Path(path).parent.mkdir(parents=True, exist_ok=True)
        with open(path, "w") as f:
            for r in self.rows:
                f.write(json.dumps(r) + "\n")

    def shard(self, n: int) -> List["Dataset"]:
        per = math.ceil(len(self.rows) / n) if self.rows else 1
        return [Dataset(self.rows[i * per:(i + 1) * per])
                for i in range(n) if self.rows[i * per:(i + 1) * per]]

    def __len__(self):
        return len(self.rows)


def run_batch(dataset: Dataset, run_command: str, *,
              num_workers: int = 2,
              gpus_per_worker: int = 0,
              cluster_prefix: str = "sky-batch",
              workdir: Optional[str] = None,
              timeout: float = 600.0) -> Dataset:
    """Fan shards out to pool jobs and gather results.

    ``run_command`` reads JSONL from $SKY_BATCH_INPUT and writes JSONL to
    $SKY_BATCH_OUTPUT (one output row per input row, any order across
    shards).
    """
    work_root = global_state.root_dir() / "batch" / \
        f"{cluster_prefix}-{int(time.time())}"
    work_root.mkdir(parents=True, exist_ok=True)
    shards = dataset.shard(num_workers)
    backend = PoolBackend()
    pending = []  # (handle, job_id, out_path)
    for i, shard in enumerate(shards):
        in_path = work_root / f"in-{i}.jsonl"
        out_path = work_root / f"out-{i}.jsonl"
        shard.write_jsonl(str(in_path))
        res = "MI355X:" + str(gpus_per_worker) if gpus_per_worker else None
        task = Task.from_yaml_config({
            "name": f"batch-shard-{i}",
            "workdir": workdir,
            "resources": {"accelerators": res} if res else {},
            "envs": {"SKY_BATCH_INPUT": str(in_path),
                     "SKY_BATCH_OUTPUT": str(out_path),
                     "SKY_BATCH_SHARD": str(i)},
            "run": run_command,
        })
        job_id, handle = execution.launch(
            task, f"{cluster_prefix}-{i}", detach_run=True)
        pending.append((handle, job_id, out_path, f"{cluster_prefix}-{i}"))

    rows: List[Dict[str, Any]] = []
    errors = []
    try:
        for handle, job_id, out_path, _ in pending:
            job = backend.wait_job(handle, job_id, timeout=timeout)
            if job["status"] != "SUCCEEDED":
                errors.append((job_id, job["status"]))
            elif out_path.exists():
                rows.extend(Dataset.from_jsonl(str(out_path)).rows)
    finally:
        for handle, _, _, name in pending:
            try:
                backend.teardown(handle, terminate=True)
            except Exception:  # noqa: BLE001
                pass
    if errors:
        raise RuntimeError(f"batch shards failed: {errors}")
    return Dataset(rows)
