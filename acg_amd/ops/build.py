"""In-tree build of the gfx950 HIP kernel extension.

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
ARCH = os.environ.get("ACG_AMD_ARCH", "gfx950")


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def so_path() -> Path:
    return HERE / f"_acg_kernels{_ext_suffix()}"


def needs_build() -> bool:
    so = so_path()
    return (not so.exists()) or so.stat().st_mtime < SRC.stat().st_mtime


def build(verbose: bool = True, force: bool = False) -> Path:
    """Compile kernels.hip -> _acg_kernels.so (gfx950)."""
    so = so_path()
    if not force and not needs_build():
        return so
    import pybind11

    cmd = [
        "hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        str(SRC), "-o", str(so),
    ]
    if verbose:
        print("[acg_amd.ops.build]", " ".join(cmd), file=sys.stderr)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"hipcc failed ({r.returncode}):\n{r.stdout}\n{r.stderr}")
    return so


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(so_path())
