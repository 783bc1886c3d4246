"""Load-balancing policies (reference:
sky/serve/load_balancing_policies.py — RoundRobinPolicy:85,
LeastLoadPolicy:111 which is the default)."""
from __future__ import annotations

import itertools
import threading
from typing import Dict, List, Optional


class LBPolicy:
    def pick(self, replicas: List[str]) -> Optional[str]:
        raise NotImplementedError

    def on_start(self, replica: str) -> None:
        pass

    def on_finish(self, replica: str) -> None:
        pass


class RoundRobinPolicy(LBPolicy):
    def __init__(self):
        self._counter = itertools.count()

    def pick(self, replicas):
        if not replicas:
            return None
        return replicas[next(self._counter) % len(replicas)]


class LeastLoadPolicy(LBPolicy):
    """Route to the replica with the fewest outstanding requests."""

    def __init__(self):
        self._outstanding: Dict[str, int] = {}
        self._lock = threading.Lock()

    def pick(self, replicas):
        if not replicas:
            return None
        with self._lock:
            return min(replicas,
                       key=lambda r: self._outstanding.get(r, 0))

    def on_start(self, replica):
        with self._lock:
            self._outstanding[replica] = \
                self._outstanding.get(replica, 0) + 1

    def on_finish(self, replica):
        with self._lock:
            self._outstanding[replica] = \
                max(0, self._outstanding.get(replica, 0) - 1)


def make_policy(name: str) -> LBPolicy:
    return {"round_robin": RoundRobinPolicy,
            "least_load": LeastLoadPolicy}.get(name, LeastLoadPolicy)()
