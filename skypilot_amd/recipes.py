"""Recipe hub — named, versioned task YAMLs served by the framework
(reference: sky/recipes/core.py).  Recipes live in the repo's examples/
plus ~/.sky_amd/recipes/; `sky launch recipe:<name>` resolves them."""
from __future__ import annotations

import os
from pathlib import Path
from typing import Dict, List, Optional

import yaml

BUILTIN_DIR = Path(__file__).resolve().parent.parent / "examples"
USER_DIR = "~/.sky_amd/recipes"


def _dirs() -> List[Path]:
    out = [BUILTIN_DIR]
    u = Path(os.path.expanduser(USER_DIR))
    if u.exists():
        out.append(u)
    return out


def list_recipes() -> List[Dict[str, str]]:
    seen = {}
    for d in _dirs():
        for f in sorted(d.glob("*.yaml")):
            name = f.stem
            try:
                cfg = yaml.safe_load(f.read_text()) or {}
            except yaml.YAMLError:
                continue
            seen[name] = {"name": name, "path": str(f),
                          "task_name": cfg.get("name", name),
                          "accelerators": str((cfg.get("resources") or {})
                                              .get("accelerators", "-"))}
    return list(seen.values())


def get_recipe_path(name: str) -> Optional[str]:
    for d in reversed(_dirs()):  # user recipes shadow builtins
        p = d / f"{name}.yaml"
        if p.exists():
            return str(p)
    return None
