"""In-tree build of the native extensions (no JIT cache — the built .so must
live next to the sources so it travels to GPU boxes with the repo snapshot).

Extensions:
  * gpuhealth  — HIP/gfx950 on-device health probes (gpuhealth.hip)
  * engine     — C++ control-plane core: object cache + indices (engine.cpp)

Build: ``python -m kuberay_amd._native.build`` or ``__graft_entry__.build()``.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

HERE = Path(__file__).resolve().parent
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def _pybind11_includes() -> list:
    import pybind11
    return [f"-I{pybind11.get_include()}",
            f"-I{sysconfig.get_paths()['include']}"]


def _needs_rebuild(src: Path, out: Path) -> bool:
    return not out.exists() or src.stat().st_mtime > out.stat().st_mtime


def build_gpuhealth(force: bool = False) -> Path:
    src = HERE / "gpuhealth.hip"
    out = HERE / f"gpuhealth{_ext_suffix()}"
    if not force and not _needs_rebuild(src, out):
        return out
    cmd = [
        HIPCC, f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC", *_pybind11_includes(),
        str(src), "-o", str(out),
    ]
    subprocess.run(cmd, check=True)
    return out


def build_engine(force: bool = False) -> Path:
    src = HERE / "engine.cpp"
    out = HERE / f"engine{_ext_suffix()}"
    if not src.exists():
        return out
    if not force and not _needs_rebuild(src, out):
        return out
    cxx = os.environ.get("CXX", "g++")
    cmd = [
        cxx, "-O3", "-std=c++17", "-shared", "-fPIC",
        *_pybind11_includes(), str(src), "-o", str(out),
    ]
    subprocess.run(cmd, check=True)
    return out


def build_all(force: bool = False) -> list:
    outs = [build_gpuhealth(force)]
    if (HERE / "engine.cpp").exists():
        outs.append(build_engine(force))
    return outs


if __name__ == "__main__":
    force = "--force" in sys.argv
    for out in build_all(force):
        print(out)
