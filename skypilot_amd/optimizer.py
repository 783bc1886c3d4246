"""Optimizer — feasibility + cost ranking over the pool.

Reference: sky/optimizer.py:109 (DP over DAG with per-cloud catalog
enumeration and a price model).  On a one-pool deployment the search
space is the pool inventory; `any_of` candidates are ranked by a
config-driven hourly-cost estimate (pool.prices; spot discounted),
`ordered` keeps the user's preference order, and provision-time
failover walks the ranked list.  The interface (optimize(dag) ->
annotated dag) is preserved so a multi-pool catalog can slot in later.
"""
from __future__ import annotations

from skypilot_amd.dag import Dag, to_dag
from skypilot_amd.exceptions import ResourcesUnavailableError
from skypilot_amd.resources import MEMORY_PER_GPU_GB
from skypilot_amd.utils.gpu_topology import detect_gpus


class Optimizer:
    @classmethod
    def estimate_hourly_cost(cls, task, res) -> float:
        """Config-driven price model (config `pool.prices`: per-GPU and
        per-CPU $/hr plus a spot discount factor; defaults keep the
        round-1 cheapest-first heuristic ordering)."""
        from skypilot_amd import config as sky_config
        gpu_price = float(sky_config.get_nested(
            ["pool", "prices", "MI355X"], 2.0))
        cpu_price = float(sky_config.get_nested(
            ["pool", "prices", "cpu"], 0.05))
        spot_mult = float(sky_config.get_nested(
            ["pool", "prices", "spot_discount"], 0.3))
        acc = res.accelerator_count * task.num_nodes
        cost = acc * gpu_price + (res.cpus or 1) * cpu_price
        if res.use_spot:
            cost *= spot_mult
        return cost

    @classmethod
    def optimize(cls, dag: Dag, quiet: bool = True) -> Dag:
        dag = to_dag(dag)
        for task in dag.tasks:
            cands = task.resources.candidates or (task.resources,)
            ordered = getattr(task.resources, "ordered", False)
            feasible, errs = [], []
            for cand in cands:
                try:
                    cls._check_feasible(task, cand)
                    feasible.append(cand)
                except ResourcesUnavailableError as e:
                    errs.append(str(e))
            if not feasible:
                raise ResourcesUnavailableError("; ".join(errs))
            if not ordered and len(feasible) > 1:
                # rank any_of candidates cheapest-first by the price
                # model (reference: optimizer cost ranking)
                feasible.sort(
                    key=lambda r: cls.estimate_hourly_cost(task, r))
            # keep the feasible candidates (in order) for provision-time
            # failover in execution.py
            task.resources = feasible[0]
            task.resources.candidates = tuple(feasible)
            task.estimated_hourly_cost = cls.estimate_hourly_cost(
                task, feasible[0])
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
