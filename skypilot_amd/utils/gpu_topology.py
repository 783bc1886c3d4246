"""MI355X pool inventory + xGMI/NUMA topology probe.

Reference parity: SkyPilot reads accelerator inventories from cloud
catalogs (sky/catalog/); on a local pool we probe the actual node via
amd-smi/rocm-smi and sysfs.  The rank->GPU/NUMA plan feeds the gang
launcher so each RCCL rank is pinned to its GPU's NUMA domain
(xGMI is point-to-point, affinity matters).
"""
from __future__ import annotations

import functools
import json
import os
import subprocess
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class GpuInfo:
    index: int
    name: str = "MI355X"
    memory_gb: int = 288
    numa_node: int = -1
    pci_bus: str = ""


def _sysfs_numa(card_idx: int) -> int:
    p = f"/sys/class/drm/card{card_idx}/device/numa_node"
    try:
        with open(p) as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return -1


@functools.lru_cache(maxsize=1)
def detect_gpus() -> List[GpuInfo]:
    """Enumerate visible AMD GPUs.  SKY_AMD_FAKE_GPUS=<n> fakes an n-GPU
    node for CPU-only tests."""
    fake = os.environ.get("SKY_AMD_FAKE_GPUS")
    if fake:
        return [GpuInfo(index=i) for i in range(int(fake))]
    gpus: List[GpuInfo] = []
    try:
        out = subprocess.run(
            ["rocm-smi", "--showid", "--showbus", "--json"],
            capture_output=True, text=True, timeout=20)
        if out.returncode == 0 and out.stdout.strip():
            data = json.loads(out.stdout)
            for key in sorted(k for k in data if k.startswith("card")):
                idx = int(key[4:])
                gpus.append(GpuInfo(
                    index=idx,
                    pci_bus=data[key].get("PCI Bus", ""),
                    numa_node=_sysfs_numa(idx)))
            if gpus:
                return gpus
    except (OSError, subprocess.TimeoutExpired, json.JSONDecodeError,
            ValueError):
        pass
    try:
        import torch
        if torch.cuda.is_available():
            return [GpuInfo(index=i, name=torch.cuda.get_device_name(i))
                    for i in range(torch.cuda.device_count())]
    except Exception:  # noqa: BLE001
        pass
    return gpus


def xgmi_topology() -> Optional[Dict]:
    """Link map from `rocm-smi --showtopo` (best effort; informational)."""
    try:
        out = subprocess.run(["rocm-smi", "--showtopo"], capture_output=True,
                             text=True, timeout=20)
        if out.returncode == 0:
            return {"raw": out.stdout}
    except (OSError, subprocess.TimeoutExpired):
        pass
    return None


@dataclass
class RankPlan:
    rank: int
    gpu: int
    numa_node: int = -1

    def env(self) -> Dict[str, str]:
        e = {"HIP_VISIBLE_DEVICES": str(self.gpu)}
        return e

    def numactl_prefix(self) -> List[str]:
        if self.numa_node >= 0 and os.path.exists("/usr/bin/numactl"):
            return ["numactl", f"--cpunodebind={self.numa_node}",
                    f"--membind={self.numa_node}"]
        return []


def plan_ranks(gpu_ids: List[int]) -> List[RankPlan]:
    infos = {g.index: g for g in detect_gpus()}
    return [RankPlan(rank=i, gpu=g,
                     numa_node=infos.get(g, GpuInfo(g)).numa_node)
            for i, g in enumerate(gpu_ids)]
