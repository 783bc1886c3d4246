"""Dag — ordered collection of Tasks with a context-manager API
(reference: sky/dag.py:26, push/pop context :220, chain check :177)."""
from __future__ import annotations

import threading
from typing import List, Optional

from skypilot_amd.task import Task

_dag_stack = threading.local()


def _stack() -> List["Dag"]:
    if not hasattr(_dag_stack, "s"):
        _dag_stack.s = []
    return _dag_stack.s


class Dag:
    def __init__(self, name: Optional[str] = None):
        self.name = name
        self.tasks: List[Task] = []
        self._edges: List[tuple] = []

    def add(self, task: Task):
        self.tasks.append(task)

    def remove(self, task: Task):
        self.tasks.remove(task)
        self._edges = [(u, v) for u, v in self._edges
                       if u is not task and v is not task]

    def add_edge(self, op1: Task, op2: Task):
        self._edges.append((op1, op2))

    def is_chain(self) -> bool:
        n = len(self.tasks)
        if n <= 1:
            return True
        if len(self._edges) != n - 1:
            return False
        order = {t: i for i, t in enumerate(self.tasks)}
        return all(order[v] == order[u] + 1 for u, v in self._edges)

    def __len__(self):
        return len(self.tasks)

    def __enter__(self):
        _stack().append(self)
        return self

    def __exit__(self, *args):
        _stack().pop()

    def __repr__(self):
        return f"Dag({self.name or '<unnamed>'}, {len(self.tasks)} tasks)"


def get_current_dag() -> Optional[Dag]:
    s = _stack()
    return s[-1] if s else None


def to_dag(task_or_dag) -> Dag:
    if isinstance(task_or_dag, Dag):
        return task_or_dag
    d = Dag(name=getattr(task_or_dag, "name", None))
    d.add(task_or_dag)
    return d
