"""Pluggable admin policy — server-side request mutation.

Reference: sky/admin_policy.py (UserRequest -> MutatedUserRequest,
applied at every launch, execution.py:303).  Configure with

    admin_policy: mypkg.mymodule.MyPolicy

in ~/.sky_amd/config.yaml; the class gets each task config before
execution and may mutate or reject it.
"""
from __future__ import annotations

import importlib
from dataclasses import dataclass
from typing import Any, Dict, Optional

from skypilot_amd import config
from skypilot_amd.exceptions import TaskValidationError


@dataclass
class UserRequest:
    task_config: Dict[str, Any]
    cluster_name: Optional[str] = None
    operation: str = "launch"


@dataclass
class MutatedUserRequest:
    task_config: Dict[str, Any]


class AdminPolicy:
    """Base policy: identity."""

    def validate_and_mutate(self, request: UserRequest
                            ) -> MutatedUserRequest:
        return MutatedUserRequest(task_config=request.task_config)


class RejectPolicy(AdminPolicy):
    """Example policy: reject everything (for tests)."""

    def validate_and_mutate(self, request):
        raise TaskValidationError("request rejected by admin policy")


def load_policy() -> Optional[AdminPolicy]:
    path = config.get_nested(["admin_policy"])
    if not path:
        return None
    mod_name, _, cls_name = str(path).rpartition(".")
    cls = getattr(importlib.import_module(mod_name), cls_name)
    return cls()


def apply(task_config: Dict[str, Any], cluster_name: Optional[str] = None,
          operation: str = "launch") -> Dict[str, Any]:
    policy = load_policy()
    if policy is None:
        return task_config
    mutated = policy.validate_and_mutate(
        UserRequest(task_config=task_config, cluster_name=cluster_name,
                    operation=operation))
    return mutated.task_config
