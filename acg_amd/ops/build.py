"""In-tree build of the gfx950 HIP kernel extension.

Reference analog: the CMake HIP build (CMakeLists.txt `--offload-arch`,
ACG_WITH_HIP) reduced to two direct compiler invocations.

The built ``.so`` lives next to the sources (inside the package) so the
repo snapshot carries it to GPU boxes; a JIT cache under ~/.cache would
not travel.  Compilation is a direct ``hipcc --offload-arch=gfx950`` of
kernels.hip with pybind11 -- no hipify, no torch C++ ABI dependency.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

HERE = Path(__file__).resolve().parent
SRC = HERE / "kernels.hip"
HOST_SRC = HERE.parent / "host" / "host.cpp"
ARCH = os.environ.get("ACG_AMD_ARCH", "gfx950")


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def so_path() -> Path:
    return HERE / f"_acg_kernels{_ext_suffix()}"


def needs_build() -> bool:
    so = so_path()
    return (not so.exists()) or so.stat().st_mtime < SRC.stat().st_mtime


def host_so_path() -> Path:
    return HOST_SRC.parent / f"_acg_host{_ext_suffix()}"


def _run(cmd, verbose):
    if verbose:
        print("[acg_amd.ops.build]", " ".join(cmd), file=sys.stderr)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"{cmd[0]} failed ({r.returncode}):\n{r.stdout}\n{r.stderr}")


def build(verbose: bool = True, force: bool = False) -> Path:
    """Compile kernels.hip -> _acg_kernels.so (gfx950) and the C++/OpenMP
    host extension _acg_host.so."""
    import pybind11

    inc = [f"-I{pybind11.get_include()}", f"-I{sysconfig.get_paths()['include']}"]
    so = so_path()
    if force or needs_build():
        _run(["hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17",
              "-shared", "-fPIC", *inc, str(SRC), "-o", str(so)], verbose)
    hso = host_so_path()
    if force or (not hso.exists()) or hso.stat().st_mtime < HOST_SRC.stat().st_mtime:
        # -march=x86-64-v3 (not native): the built .so travels to GPU boxes
        # with possibly different host CPUs
        _run(["g++", "-O3", "-std=c++17", "-fopenmp", "-shared", "-fPIC",
              "-march=x86-64-v3", *inc, str(HOST_SRC), "-o", str(hso)], verbose)
    return so


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(so_path())
