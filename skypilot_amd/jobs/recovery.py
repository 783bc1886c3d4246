"""Recovery strategies for managed jobs.

Reference: sky/jobs/recovery_strategy.py (StrategyExecutor:175,
FailoverStrategyExecutor:1376 — retry same placement then failover;
EagerFailoverStrategyExecutor:1497 — tear down and move immediately;
registry via utils/registry.py:132).  On the one-node MI355X pool
"failover" means re-allocating GPUs from the pool (possibly different
indices) rather than moving region; the strategy interface is kept so
SSH/K8s pools can failover across machines.
"""
from __future__ import annotations

import time
from typing import Dict, Optional, Type

STRATEGY_REGISTRY: Dict[str, Type["RecoveryStrategy"]] = {}


def register_strategy(name: str):
    def deco(cls):
        STRATEGY_REGISTRY[name] = cls
        cls.NAME = name
        return cls
    return deco


def make(name: Optional[str], max_restarts_on_errors: int = 0
         ) -> "RecoveryStrategy":
    cls = STRATEGY_REGISTRY.get((name or "FAILOVER").upper(),
                                STRATEGY_REGISTRY["FAILOVER"])
    return cls(max_restarts_on_errors)


class RecoveryStrategy:
    NAME = "base"
    RETRY_GAP_SECONDS = 2.0

    def __init__(self, max_restarts_on_errors: int = 0):
        self.max_restarts_on_errors = max_restarts_on_errors
        self.restarts_on_errors = 0

    def should_restart_on_failure(self) -> bool:
        """User-code failure (nonzero exit): restart only within budget
        (reference: recovery_strategy.py:1342)."""
        if self.restarts_on_errors >= self.max_restarts_on_errors:
            return False
        self.restarts_on_errors += 1
        return True

    def wait_before_retry(self) -> None:
        time.sleep(self.RETRY_GAP_SECONDS)

    # Preemption / infra failure recovery is unconditional for both
    # built-in strategies; they differ in placement policy.
    def keep_placement_first(self) -> bool:
        raise NotImplementedError


@register_strategy("FAILOVER")
class FailoverStrategy(RecoveryStrategy):
    """Retry the same placement first, then fail over."""

    def keep_placement_first(self) -> bool:
        return True


@register_strategy("EAGER_NEXT_CLUSTER")
class EagerFailoverStrategy(RecoveryStrategy):
    """Immediately tear down and take a different placement."""

    def keep_placement_first(self) -> bool:
        return False
