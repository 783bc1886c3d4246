"""Optimizer — feasibility + placement over the pool.

Reference: sky/optimizer.py:109 (DP over DAG with per-cloud catalog
enumeration).  On a one-pool deployment the search space is the pool
inventory; the interface (optimize(dag) -> annotated dag) is preserved
so a multi-pool catalog can slot in later.
"""
from __future__ import annotations

from skypilot_amd.dag import Dag, to_dag
from skypilot_amd.exceptions import ResourcesUnavailableError
from skypilot_amd.resources import MEMORY_PER_GPU_GB
from skypilot_amd.utils.gpu_topology import detect_gpus


class Optimizer:
    @classmethod
    def optimize(cls, dag: Dag, quiet: bool = True) -> Dag:
        dag = to_dag(dag)
        for task in dag.tasks:
            cands = task.resources.candidates or (task.resources,)
            feasible, errs = [], []
            for cand in cands:
                try:
                    cls._check_feasible(task, cand)
                    feasible.append(cand)
                except ResourcesUnavailableError as e:
                    errs.append(str(e))
            if not feasible:
                raise ResourcesUnavailableError("; ".join(errs))
            # keep the feasible candidates (in order) for provision-time
            # failover in execution.py
            task.resources = feasible[0]
            task.resources.candidates = tuple(feasible)
        return dag

    @staticmethod
    def _check_feasible(task, res=None) -> None:
        res = res if res is not None else task.resources
        need = task.num_nodes * res.accelerator_count
        if need == 0:
            return
        if 0 < res.accelerator_count < 1:
            # fractional share of one GPU (reference: fractional
            # accelerators on k8s) — single node only; feasible iff the
            # pool has any GPU at all (load is checked at provision).
            if task.num_nodes != 1:
                raise ResourcesUnavailableError(
                    "fractional accelerators require num_nodes=1")
            if not detect_gpus():
                raise ResourcesUnavailableError("pool has no GPUs")
            return
        if res.accelerators not in (None, "MI355X"):
            raise ResourcesUnavailableError(
                f"pool has MI355X only, requested {res.accelerators}")
        pool = detect_gpus()
        if need > len(pool):
            raise ResourcesUnavailableError(
                f"requested {need} GPUs ({task.num_nodes} nodes x "
                f"{res.accelerator_count}), pool has {len(pool)}")
        if res.memory and res.memory > MEMORY_PER_GPU_GB * need:
            raise ResourcesUnavailableError(
                f"requested {res.memory} GB accelerator memory; "
                f"{need} GPUs provide {MEMORY_PER_GPU_GB * need} GB")
