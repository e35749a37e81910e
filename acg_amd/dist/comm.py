"""Communicator wrapper (reference: acg/comm.{c,h} struct acgcomm).

The reference wraps {null, mpi, nccl/rccl, nvshmem, rocshmem} behind one
tagged union (comm.h:84-117).  The MI355X-native build keeps exactly two
live backends behind torch.distributed:

- ``rccl``  (torch.distributed backend "nccl", which IS RCCL on ROCm):
  the production path -- grouped send/recv halo + 1-2 double allreduce on
  HIP streams over xGMI.
- ``gloo``: CPU backend used for multi-process tests without GPUs.
- ``none``: serial, no process group.

Bootstrap is torchrun/env:// (RANK/WORLD_SIZE/MASTER_ADDR), replacing the
reference's MPI_Comm_split_type + ncclCommInitRank dance
(hip/acg-hip.c:971-1081).
"""

from __future__ import annotations

import datetime
import os

import numpy as np
import torch


class Comm:
    """Process-group communicator; one process per GPU."""

    def __init__(self, kind: str = "none", device: torch.device | None = None,
                 timeout_s: float = 600.0):
        kind = {"nccl": "rccl"}.get(kind, kind)
        if kind not in ("none", "rccl", "gloo"):
            raise ValueError(f"unsupported comm kind {kind!r} (none|rccl|gloo)")
        self.kind = kind
        self.device = device
        if kind == "none":
            self.rank, self.size = 0, 1
            self.group = None
            return
        import torch.distributed as dist

        self._dist = dist
        if not dist.is_initialized():
            backend = "nccl" if kind == "rccl" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            dist.init_process_group(
                backend=backend,
                rank=int(os.environ.get("RANK", 0)),
                world_size=int(os.environ.get("WORLD_SIZE", 1)),
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        self.rank = dist.get_rank()
        self.size = dist.get_world_size()
        self.group = dist.group.WORLD
        if kind == "rccl" and device is None:
            self.device = torch.device("cuda", self.rank % max(torch.cuda.device_count(), 1))

    @property
    def can_capture(self) -> bool:
        """True when this backend's collectives are hipGraph-capturable
        (RCCL on-stream).  The gloo test backend stages CUDA tensors
        through blocking host copies -- attempting capture there not only
        fails but can invalidate cross-stream event state mid-body, so
        callers must not even TRY (solvers consult this before capture)."""
        return self.kind == "rccl"

    # -- collectives ------------------------------------------------------

    def allreduce_(self, t: torch.Tensor) -> torch.Tensor:
        """In-place sum-allreduce on the *current stream* (RCCL) or host (gloo).

        Reference analog: acgcomm_allreduce_hip (comm.c:451-471).
        """
        if self.kind == "none" or self.size == 1:
            return t
        if self.kind == "gloo" and t.is_cuda:
            # CPU-staged path so the GPU solver's multi-rank code can be
            # exercised end-to-end on a single GPU (gloo cannot carry CUDA
            # tensors).  Test harness only -- production multi-GPU is RCCL.
            h = t.detach().cpu()
            self._dist.all_reduce(h, op=self._dist.ReduceOp.SUM)
            t.copy_(h)
            return t
        self._dist.all_reduce(t, op=self._dist.ReduceOp.SUM)
        return t

    def allreduce_max_int(self, v: int) -> int:
        if self.kind == "none" or self.size == 1:
            return v
        dev = self.device if self.kind == "rccl" else "cpu"
        t = torch.tensor([v], dtype=torch.int64, device=dev)
        self._dist.all_reduce(t, op=self._dist.ReduceOp.MAX)
        return int(t.item())

    def barrier(self):
        """Reference analog: acgcomm_barrier (0-count allreduce trick, comm.c:418)."""
        if self.kind == "none" or self.size == 1:
            return
        if self.kind == "rccl":
            self._dist.barrier(device_ids=[self.device.index])
        else:
            self._dist.barrier()

    # -- setup-phase object transport (root-centric scatter) --------------

    def scatter_object(self, objs: list | None, src: int = 0):
        """Scatter a python object per rank (reference acgsymcsrmatrix_scatter
        / acggraph_scatter field-by-field MPI_Send/Recv, graph.c:1529-1893 --
        here a single pickled object per rank over the store/gloo channel).
        """
        if self.kind == "none" or self.size == 1:
            return objs[0] if objs is not None else None
        out = [None]
        self._dist.scatter_object_list(out, objs if self.rank == src else None, src=src)
        return out[0]

    def gather_object(self, obj, dst: int = 0):
        if self.kind == "none" or self.size == 1:
            return [obj]
        out = [None] * self.size if self.rank == dst else None
        self._dist.gather_object(obj, out, dst=dst)
        return out

    def bcast_object(self, obj, src: int = 0):
        if self.kind == "none" or self.size == 1:
            return obj
        lst = [obj]
        self._dist.broadcast_object_list(lst, src=src)
        return lst[0]

    # -- distributed vector IO (reference mtxfile_fwrite_mpi_double) ------

    def gather_vector(self, x_local: np.ndarray, owned_global: np.ndarray,
                      n_global: int, dst: int = 0) -> np.ndarray | None:
        """Gather a distributed vector to root in global row order
        (reference mtxfile_fwrite_mpi_double, mtxfile.h:1087)."""
        pieces = self.gather_object((owned_global, x_local), dst=dst)
        if pieces is None:
            return None
        out = np.empty(n_global, dtype=np.float64)
        for og, xl in pieces:
            out[og] = xl
        return out

    def finalize(self):
        if self.kind != "none" and self._dist.is_initialized():
            self._dist.destroy_process_group()
