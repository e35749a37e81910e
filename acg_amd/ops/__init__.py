"""GPU/CPU op dispatch.

``gpu_ops`` wraps the hand-written gfx950 HIP kernels (kernels.hip).  On a
GPU box the extension MUST be present -- ops raise loudly rather than fall
back to eager torch, so a silently-slow path can never pass for the native
one.  ``torch_ref`` provides the plain-PyTorch fp64 reference used by
numerics tests and by the CPU solver.
"""

from . import torch_ref  # noqa: F401


class _LazyGpuOps:
    """Import-on-first-use proxy so CPU-only work never needs the .so."""

    _mod = None

    def _load(self):
        if self._mod is None:
            import importlib

            self._mod = importlib.import_module("acg_amd.ops.gpu_ops")
        return self._mod

    def __getattr__(self, name):
        return getattr(self._load(), name)


gpu_ops = _LazyGpuOps()
