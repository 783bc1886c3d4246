"""Abstract backend interface (reference: sky/backends/backend.py)."""
from __future__ import annotations

from typing import Any, Dict, Optional

from skypilot_amd.task import Task


class Backend:
    def provision(self, task: Task, cluster_name: str,
                  retry_until_up: bool = False) -> Dict[str, Any]:
        raise NotImplementedError

    def sync_workdir(self, handle: Dict[str, Any], workdir: str) -> None:
        raise NotImplementedError

    def sync_file_mounts(self, handle: Dict[str, Any],
                         file_mounts: Dict[str, Any]) -> None:
        raise NotImplementedError

    def setup(self, handle: Dict[str, Any], task: Task) -> None:
        raise NotImplementedError

    def execute(self, handle: Dict[str, Any], task: Task,
                detach_run: bool = False) -> int:
        raise NotImplementedError

    def post_execute(self, handle: Dict[str, Any]) -> None:
        pass

    def teardown(self, handle: Dict[str, Any], terminate: bool = True
                 ) -> None:
        raise NotImplementedError

    def tail_logs(self, handle: Dict[str, Any], job_id: Optional[int],
                  follow: bool = True):
        raise NotImplementedError

    def cancel_jobs(self, handle: Dict[str, Any],
                    job_ids: Optional[list] = None) -> int:
        raise NotImplementedError

    def set_autostop(self, handle: Dict[str, Any], idle_minutes: int,
                     down: bool = False) -> None:
        raise NotImplementedError
