"""skypilot_amd — an MI355X-native Sky-style orchestrator + training stack.

A from-scratch framework with SkyPilot's capabilities (`sky launch/exec/
jobs/serve`, task YAML, managed jobs, serving) re-targeted at an
8xMI355X pool, plus the MI355X-native compute stack the reference lacks:
hand-written CDNA4 HIP kernels, RCCL-over-xGMI launch backend, bundled
Llama train/serve entrypoints.

Public API mirrors the reference's `sky/__init__.py` re-exports
(reference: sky/__init__.py:95-140).
"""
__version__ = "0.1.0"

_LAZY = {
    "Task": ("skypilot_amd.task", "Task"),
    "Resources": ("skypilot_amd.resources", "Resources"),
    "Dag": ("skypilot_amd.dag", "Dag"),
    "launch": ("skypilot_amd.execution", "launch"),
    "exec": ("skypilot_amd.execution", "exec_"),
}


def __getattr__(name):
    if name in _LAZY:
        import importlib
        mod, attr = _LAZY[name]
        return getattr(importlib.import_module(mod), attr)
    raise AttributeError(f"module 'skypilot_amd' has no attribute {name!r}")
