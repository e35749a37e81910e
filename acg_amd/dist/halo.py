"""Halo exchange engine (reference: acg/halo.{c,h}, halo-kernels-hip.hip).

The reference's RCCL engine (halo.c:1828-1883) packs border values with a
gather kernel, does ncclGroupStart / ncclRecv x nsenders / ncclSend x
nrecipients / ncclGroupEnd on a side stream, then scatters with an unpack
kernel.

MI355X-native differences:

- Ghost values land *directly* in the vector's ghost tail: the subdomain
  builder (acg_amd.part.subdomain) sorts ghosts by (owner, global id), so
  each sender's block is a contiguous slice of the tail and
  ``dist.batch_isend_irecv`` receives into views of the vector itself --
  no unpack kernel, no receive staging buffer.
- Only the send side needs a pack (gather) kernel
  (ops.pack_gather on GPU, index_select on CPU).
- Begin/end split: ``begin`` launches pack + grouped send/recv (RCCL:
  ncclGroupStart/End under torch.distributed.batch_isend_irecv) on the
  caller's current stream; ``end`` waits the requests -- with NCCL this is
  a stream-ordered event wait, not a host block, so SpMV(matA) overlaps
  the exchange exactly as in the reference's split (cghip.c:887-931).
"""

from __future__ import annotations

import torch

from ..part.subdomain import HaloPattern


class HaloExchange:
    """One vector's neighbour exchange (reference struct acghaloexchange)."""

    def __init__(self, halo: HaloPattern, nowned: int, device, comm,
                 dtype=torch.float64):
        self.halo = halo
        self.nowned = int(nowned)
        self.comm = comm
        self.device = torch.device(device)
        self.dtype = dtype
        idx = torch.from_numpy(halo.sendidx.astype("int64")
                               if halo.sendidx.dtype.kind != "i" else halo.sendidx)
        self.sendidx = idx.to(self.device).long()
        self.sendbuf = torch.empty(halo.sendsize, dtype=dtype, device=self.device)
        self._reqs: list = []
        self._staged_recv: list = []
        self._x = None
        # per-iteration traffic counters (reference cghip.h:109-118 stats)
        self.nexchanges = 0
        self.bytes_sent = 0
        self.bytes_recv = 0
        self.nmsgs_sent = 0

    def begin(self, x: torch.Tensor) -> None:
        """Pack + post grouped send/recv.  ``x`` is the full local vector
        (owned + ghost tail); ghosts are received in place."""
        h = self.halo
        if h.nrecipients == 0 and h.nsenders == 0:
            return
        import torch.distributed as dist

        if x.is_cuda:
            from ..ops import gpu_ops

            gpu_ops.pack_gather(self.sendbuf, x, self.sendidx)
        else:
            torch.index_select(x, 0, self.sendidx, out=self.sendbuf)
        # gloo cannot carry CUDA tensors: stage through host buffers so the
        # GPU solver's multi-rank path can be tested end-to-end on one GPU.
        # Test harness only -- production multi-GPU runs RCCL (device direct).
        staged = x.is_cuda and getattr(self.comm, "kind", "rccl") == "gloo"
        sendsrc = self.sendbuf.cpu() if staged else self.sendbuf
        self._staged_recv = []
        ops = []
        for i in range(h.nsenders):
            lo = self.nowned + int(h.rdispls[i])
            hi = lo + int(h.recvcounts[i])
            if staged:
                buf = torch.empty(hi - lo, dtype=self.dtype)
                self._staged_recv.append((buf, lo, hi))
                ops.append(dist.P2POp(dist.irecv, buf, int(h.senders[i])))
            else:
                ops.append(dist.P2POp(dist.irecv, x[lo:hi], int(h.senders[i])))
        for i in range(h.nrecipients):
            lo = int(h.sdispls[i])
            hi = lo + int(h.sendcounts[i])
            ops.append(dist.P2POp(dist.isend, sendsrc[lo:hi], int(h.recipients[i])))
        self._x = x if staged else None
        self._reqs = dist.batch_isend_irecv(ops) if ops else []
        self.nexchanges += 1
        esize = x.element_size()
        self.bytes_sent += h.sendsize * esize
        self.bytes_recv += h.recvsize * esize
        self.nmsgs_sent += h.nrecipients

    def end(self) -> None:
        """Complete the exchange.  NCCL: stream-ordered wait (the current
        stream waits on the comm kernels -- host does not block).  Gloo:
        blocking wait."""
        for r in self._reqs:
            r.wait()
        self._reqs = []
        if self._staged_recv:
            for buf, lo, hi in self._staged_recv:
                self._x[lo:hi].copy_(buf)
            self._staged_recv = []
            self._x = None

    def exchange(self, x: torch.Tensor) -> None:
        self.begin(x)
        self.end()
