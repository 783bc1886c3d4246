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
from dataclasses import dataclass
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


def parse_showtopo(text: str) -> Dict:
    """Parse `rocm-smi --showtopo` output into a structured link map.

    Returns {n_gpus, link_type (NxN, '0' on the diagonal), hops (NxN
    ints), numa {gpu: node}, fully_connected_xgmi}.  The MI355X node is
    expected to be fully connected over xGMI (7 p2p links per GPU,
    SURVEY.md §2.12); `fully_connected_xgmi` asserts exactly that so
    rank planning can rely on any-to-any placement.
    """
    import re

    def matrix(section: str) -> List[List[str]]:
        m = re.search(section + r"[^\n]*\n(.*?)(?:\n\s*\n|\n=|\Z)", text,
                      re.S)
        if not m:
            return []
        rows = []
        for line in m.group(1).splitlines():
            toks = line.split()
            # data rows are "GPU<i> <val> <val> ..."; the header row is
            # all GPU<i> tokens and is skipped.
            if (len(toks) >= 2 and re.fullmatch(r"GPU\d+", toks[0])
                    and not toks[1].startswith("GPU")):
                rows.append(toks[1:])
        return rows

    link = matrix(r"Link Type between two GPUs")
    hops_raw = matrix(r"Hops between two GPUs")
    numa: Dict[int, int] = {}
    for m in re.finditer(
            r"GPU\[(\d+)\]\s*:\s*\(Topology\) Numa Node:\s*(\d+)", text):
        numa[int(m.group(1))] = int(m.group(2))
    n = len(link)
    hops = [[int(x) for x in row] for row in hops_raw] if hops_raw else []
    fully = n >= 2 and all(
        link[i][j].upper() == "XGMI"
        for i in range(n) for j in range(n) if i != j)
    return {"n_gpus": n, "link_type": link, "hops": hops, "numa": numa,
            "fully_connected_xgmi": fully}


def xgmi_topology() -> Optional[Dict]:
    """Structured link map from `rocm-smi --showtopo` (xGMI link types,
    hop counts, NUMA affinity) for rank planning and `sky check`."""
    try:
        out = subprocess.run(["rocm-smi", "--showtopo"], capture_output=True,
                             text=True, timeout=20)
        if out.returncode == 0:
            topo = parse_showtopo(out.stdout)
            topo["raw"] = out.stdout
            return topo
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
