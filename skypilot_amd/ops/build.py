"""In-tree hipcc build for the skypilot_amd native extension (gfx950).

We drive hipcc directly (not torch's Extension machinery) so the HIP
sources compile exactly as written — no hipify pass, no CUDA shims.  The
resulting ``_C.so`` lands next to this file and travels to GPU boxes with
the repo snapshot.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

HERE = Path(__file__).resolve().parent
CSRC = HERE / "csrc"
BUILD = HERE / "_build"
SO_PATH = HERE / "_C.so"

HIP_SOURCES = [
    "rmsnorm.hip",
    "rope.hip",
    "adamw.hip",
    "cross_entropy.hip",
    "attention_fwd.hip",
    "attention_fwd_v3.hip",
    "attention_bwd.hip",
    "attention_bwd_v3.hip",
    "attention_decode.hip",
    "swiglu.hip",
    "skinny_gemm.hip",
    "wgrad_gemm.hip",
    "decode_fused.hip",
    "mfma_probe.hip",
]
CPP_SOURCES = ["bindings.cpp"]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    return {
        "includes": ce.include_paths() + [sysconfig.get_paths()["include"]],
        "lib_dir": str(Path(torch.__file__).parent / "lib"),
        "cxx11_abi": "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0",
    }


def _run(cmd: list[str]) -> None:
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        sys.stderr.write(" ".join(cmd) + "\n")
        sys.stderr.write(proc.stdout[-4000:] + "\n" + proc.stderr[-8000:] + "\n")
        raise RuntimeError(f"build command failed ({proc.returncode})")


def _needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    for f in CSRC.iterdir():
        if f.suffix in (".hip", ".cpp", ".h") and f.stat().st_mtime > so_mtime:
            return True
    return False


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not _needs_rebuild():
        return SO_PATH
    tp = _torch_paths()
    BUILD.mkdir(exist_ok=True)
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

    common_flags = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"-D_GLIBCXX_USE_CXX11_ABI={tp['cxx11_abi']}",
    ]
    # SKY_AMD_SANITIZE=1: host AddressSanitizer on the binding/runtime
    # code (SURVEY.md §5.2 — the reference is pure Python and needs
    # none; the native layer here does).  Device code stays unsanitized
    # (use SKY_AMD_SANITIZE=xnack + HSA_XNACK=1 on a box that supports
    # it for amdgpu ASAN).
    san = os.environ.get("SKY_AMD_SANITIZE")
    if san == "1":
        common_flags += ["-fsanitize=address", "-shared-libasan",
                         "-g", "-fno-omit-frame-pointer"]
    elif san == "xnack":
        common_flags += ["-fsanitize=address", "-shared-libasan", "-g",
                         f"--offload-arch={ARCH}:xnack+"]
    objs = []
    # Device TUs: pure HIP, no torch headers.
    for src in HIP_SOURCES:
        obj = BUILD / (src.replace(".hip", ".o"))
        if verbose:
            print(f"[ops.build] hipcc {src}")
        _run([hipcc, "-x", "hip", str(CSRC / src), "-c", "-o", str(obj)]
             + common_flags)
        objs.append(str(obj))
    # Binding TU: torch headers under hipcc host-compilation.
    torch_flags = [
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DHIPBLAS_V2",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
    ] + [f"-I{p}" for p in tp["includes"]]
    for src in CPP_SOURCES:
        obj = BUILD / (src.replace(".cpp", ".o"))
        if verbose:
            print(f"[ops.build] hipcc {src}")
        _run([hipcc, "-x", "hip", str(CSRC / src), "-c", "-o", str(obj)]
             + common_flags + torch_flags)
        objs.append(str(obj))

    link = [hipcc, "-shared", "-fPIC", "-o", str(SO_PATH)] + objs
    if os.environ.get("SKY_AMD_SANITIZE"):
        link += ["-fsanitize=address", "-shared-libasan"]
    link += [
        f"-L{tp['lib_dir']}",
        "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
        "-ltorch_python",
        f"-Wl,-rpath,{tp['lib_dir']}",
        "-L/opt/rocm/lib", "-lamdhip64",
    ]
    if verbose:
        print("[ops.build] linking _C.so")
    _run(link)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {SO_PATH}")
