"""In-tree build of the ``_hipcore`` HIP extension for gfx950.

Usage: ``python -m optuna_amd._hip.build``. Produces
``optuna_amd/_hip/_hipcore.<abi>.so`` next to the sources, so the built artifact
travels with the repo snapshot to GPU boxes (no JIT cache dependence).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path


HIP_ARCH = os.environ.get("OPTUNA_AMD_HIP_ARCH", "gfx950")
_HIP_DIR = Path(__file__).resolve().parent
_SRC = _HIP_DIR / "csrc" / "tpe_core.hip"


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def output_path() -> Path:
    return _HIP_DIR / f"_hipcore{_ext_suffix()}"


def _sources_mtime() -> float:
    return max(p.stat().st_mtime for p in (_HIP_DIR / "csrc").glob("*"))


def build(force: bool = False, verbose: bool = True) -> Path:
    out = output_path()
    if out.exists() and not force and out.stat().st_mtime >= _sources_mtime():
        if verbose:
            print(f"[optuna_amd._hip] up to date: {out}")
        return out

    import pybind11

    hipcc = os.environ.get("HIPCC", "hipcc")
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        hipcc,
        f"--offload-arch={HIP_ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        f"-I{pybind11.get_include()}",
        f"-I{py_include}",
        str(_SRC),
        "-lrocblas",
        "-o",
        str(out),
    ]
    if verbose:
        print("[optuna_amd._hip] " + " ".join(cmd))
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
