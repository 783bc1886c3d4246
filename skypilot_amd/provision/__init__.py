"""Provisioner dispatch — name-based routing to per-pool modules
(reference: sky/provision/__init__.py:147 _route_to_cloud_impl)."""
from __future__ import annotations

import importlib
from typing import Any, Dict

_IMPLS = {
    "local": "skypilot_amd.provision.local",
    "ssh": "skypilot_amd.provision.ssh_pool",
    "kubernetes": "skypilot_amd.provision.k8s",
    "k8s": "skypilot_amd.provision.k8s",
}

DEFAULT_CLOUD = "local"


def _impl(cloud: str | None):
    cloud = (cloud or DEFAULT_CLOUD).lower()
    if cloud not in _IMPLS:
        from skypilot_amd.exceptions import ResourcesUnavailableError
        raise ResourcesUnavailableError(
            f"unknown infra {cloud!r}; available: {sorted(_IMPLS)}")
    return importlib.import_module(_IMPLS[cloud])


def run_instances(cloud, *args, **kwargs):
    return _impl(cloud).run_instances(*args, **kwargs)


def stop_instances(cloud, *args, **kwargs):
    return _impl(cloud).stop_instances(*args, **kwargs)


def terminate_instances(cloud, *args, **kwargs):
    return _impl(cloud).terminate_instances(*args, **kwargs)


def query_instances(cloud, *args, **kwargs):
    return _impl(cloud).query_instances(*args, **kwargs)


def get_cluster_info(cloud, *args, **kwargs):
    return _impl(cloud).get_cluster_info(*args, **kwargs)
