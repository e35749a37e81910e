"""Device-side slab generation: build the rank-local operator directly in
HBM as SELL arrays (no host CSR assembly, no PCIe copy).

The operator is analytic (stencil offsets + dof x dof blocks), so a 2048^3
7-pt Poisson slab (~90 GB SELL per GPU) generates at HBM write speed --
this is what makes BASELINE config 5 (8.6e9 rows, 288 GB HBM3E sizing)
feasible, and it cuts the Queen-shaped bench startup from ~80 s of numpy
to milliseconds.  Layouts are identical to gen.stencil.stencil_local_slab
except column order within a row (enumeration order; SpMV is
order-independent -- verified against the host generator in tests).
"""

from __future__ import annotations

import numpy as np
import torch

from ..part.subdomain import HaloPattern
from .stencil import _blocks, _slab_bounds


class DeviceSlabSystem:
    """Rank-local system with device-resident SELL operators.

    Duck-types the LocalSystem surface CGSolverHIP needs (sizes + halo),
    plus ``A_sell``/``O_sell`` tensor triples.
    """

    def __init__(self, **kw):
        self.__dict__.update(kw)

    @property
    def nlocal(self):
        return self.nowned + self.nghost

    @property
    def nnzA(self):
        return self._nnzA

    @property
    def nnzO(self):
        return self._nnzO


def estimate_slab_gib(gx: int, gy: int, gz: int, spec: dict, rank: int,
                      nranks: int, matfree: bool = False) -> float:
    """Pre-flight per-rank GPU memory estimate (GiB) for a device-generated
    stencil slab + pipelined solver workspace.

    Calibrated against measured peaks on MI355X (tools/slab_probe.py,
    profiles/RESULTS.md): 2048^3 7-pt rank 0/8 = 1.074B rows measured
    157.2 GiB assembled / 80.9 GiB matrix-free.  Model: assembled
    operator ~ 12 B/nnz (8 B vals + 4 B int32 cols incl. SELL padding)
    + 73 B/row of vectors/transients; matrix-free 81 B/row flat."""
    dof = spec["dof"]
    pts = len(spec["offsets"]) + 1  # neighbours + self
    z0, z1 = _slab_bounds(gz, rank, nranks)
    nghost_planes = (1 if rank > 0 else 0) + (1 if rank < nranks - 1 else 0)
    nlocal = (z1 - z0 + nghost_planes) * gx * gy * dof
    bytes_per_row = 81.0 if matfree else 12.0 * pts * dof + 73.0
    return nlocal * bytes_per_row / 2**30


def preflight_slab(gx: int, gy: int, gz: int, spec: dict, rank: int,
                   nranks: int, matfree: bool, total_bytes: int,
                   headroom: float = 0.94) -> float:
    """Assert the slab + solver fits this GPU BEFORE any allocation
    (BASELINE config-5 rehearsal: fail in seconds, not after a 40-GiB
    partial generation OOMs the box).  Returns the estimate in GiB."""
    est = estimate_slab_gib(gx, gy, gz, spec, rank, nranks, matfree)
    budget = headroom * total_bytes / 2**30
    if est > budget:
        raise MemoryError(
            f"rank {rank}/{nranks} of {gx}x{gy}x{gz} needs ~{est:.0f} GiB "
            f"({'matfree' if matfree else 'assembled'}) but this GPU has "
            f"{total_bytes / 2**30:.0f} GiB ({budget:.0f} usable); use more "
            f"ranks, --matfree, or a smaller --grid")
    return est


def device_stencil_slab(gx: int, gy: int, gz: int, spec: dict,
                        rank: int, nranks: int, device,
                        operator: bool = True) -> DeviceSlabSystem:
    """``operator=False`` skips assembling the SELL/BSELL arrays (row
    lengths are still counted for nnz stats): matrix-free solves (dof=1)
    need no memory-resident operator at all, freeing the ~12 B/nnz for
    larger grids per GPU."""
    from ..ops import gpu_ops

    K = gpu_ops.K
    device = torch.device(device)
    dof = spec["dof"]
    M, D = _blocks(spec)
    z0, z1 = _slab_bounds(gz, rank, nranks)
    nz_own = z1 - z0
    if nz_own <= 0:
        raise ValueError(f"rank {rank} owns no z-planes")
    plane_nodes = gx * gy
    has_lo, has_hi = rank > 0, rank < nranks - 1
    border_planes = sorted(set(([z0] if has_lo else []) +
                               ([z1 - 1] if has_hi else [])))
    interior_planes = [z for z in range(z0, z1) if z not in border_planes]
    ghost_planes = ([z0 - 1] if has_lo else []) + ([z1] if has_hi else [])
    plane_seq = interior_planes + border_planes + ghost_planes

    nown_nodes = nz_own * plane_nodes
    ninterior_nodes = len(interior_planes) * plane_nodes
    nborder_nodes = len(border_planes) * plane_nodes
    nghost_nodes = len(ghost_planes) * plane_nodes
    nowned = nown_nodes * dof
    nghost = nghost_nodes * dof
    assert (nowned + nghost) * 1 < 2**31, "device generator needs int32 local cols"

    # plane tables
    pb = torch.full((gz + 2,), -1, dtype=torch.int64)
    for k, z in enumerate(plane_seq):
        pb[z + 1] = k * plane_nodes
    zs_own = torch.tensor([z for z in interior_planes + border_planes],
                          dtype=torch.int32)
    pb = pb.to(device)
    zs_own = zs_own.to(device)
    offs_np = np.asarray([[dx, dy, dz, w] for (dx, dy, dz, w) in spec["offsets"]],
                         dtype=np.float64)
    offs = torch.from_numpy(offs_np).reshape(-1).to(device)
    ksten = len(spec["offsets"])
    blocks = torch.from_numpy(
        np.concatenate([M.reshape(-1), D.reshape(-1)])).to(device)
    stream = torch.cuda.current_stream(device).cuda_stream

    def build(filter_ghost: int, nodes: int, row0_node: int):
        if nodes == 0:
            empty = torch.zeros(1, dtype=torch.int64, device=device)
            return (torch.zeros(1, dtype=torch.int64, device=device),
                    torch.zeros(0, dtype=torch.int32, device=device),
                    torch.zeros(0, dtype=torch.float64, device=device), 0)
        rowlen_nodes = torch.empty(nodes, dtype=torch.int64, device=device)
        K.stencil_rowlen(nodes, row0_node, gx, gy, gz, dof, nown_nodes,
                         zs_own.data_ptr(), pb.data_ptr(), offs.data_ptr(),
                         ksten, filter_ghost, rowlen_nodes.data_ptr(), stream)
        # avoid an extra multi-GB copy for dof=1 (the 8.6e9-row Poisson)
        rowlen = rowlen_nodes if dof == 1 else rowlen_nodes.repeat_interleave(dof)
        nnz = int(rowlen.sum())
        if not operator:
            return None, None, None, nnz
        nrows = nodes * dof
        nslices = (nrows + 63) // 64
        padded = torch.zeros(nslices * 64, dtype=torch.int64, device=device)
        padded[:nrows] = rowlen
        slice_len = padded.view(nslices, 64).max(dim=1).values
        sellptr = torch.zeros(nslices + 1, dtype=torch.int64, device=device)
        torch.cumsum(slice_len * 64, dim=0, out=sellptr[1:])
        total = int(sellptr[-1])
        cols = torch.zeros(total, dtype=torch.int32, device=device)
        vals = torch.zeros(total, dtype=torch.float64, device=device)
        K.stencil_fill(nodes, row0_node, gx, gy, gz, dof, nown_nodes,
                       zs_own.data_ptr(), pb.data_ptr(), offs.data_ptr(),
                       ksten, blocks.data_ptr(), filter_ghost,
                       sellptr.data_ptr(), cols.data_ptr(), vals.data_ptr(),
                       stream)
        return sellptr, cols, vals, nnz

    A_sellptr, A_cols, A_vals, nnzA = build(0, nown_nodes, 0)
    O_sellptr, O_cols, O_vals, nnzO = build(1, nborder_nodes, ninterior_nodes)

    # matrix-free operator tables (dof=1 constant-coefficient stencils):
    # everything k_stencil_spmv / k_stencil_pipe need to apply the operator
    # without any assembled matrix (ops.gpu_ops.stencil_spmv).
    mf_tables = None
    if dof == 1:
        # 7-pt fast path: the z-column-walk kernels need the per-axis
        # weights and the slab bounds (w_old is then read exactly once,
        # with the z +- 1 neighbours carried in registers).
        w7 = None
        if ksten == 6:
            wmap = {(int(dx), int(dy), int(dz)): float(w)
                    for (dx, dy, dz, w) in spec["offsets"]}
            keys = [(-1, 0, 0), (1, 0, 0), (0, -1, 0),
                    (0, 1, 0), (0, 0, -1), (0, 0, 1)]
            if set(wmap) == set(keys):
                w7 = tuple(wmap[k] for k in keys)
        mf_tables = dict(zs=zs_own, pb=pb, offs=offs, ksten=ksten,
                         diag=float(D[0, 0]), gx=gx, gy=gy, gz=gz,
                         nown_nodes=nown_nodes, z0=z0, z1=z1, w7=w7)

    # Block-SELL for matA when the operator has dense dof x dof blocks:
    # one int32 index per block instead of per entry (4 -> 4/dof^2 B/nnz).
    A_bsell = None
    if operator and 2 <= dof <= 4 and nown_nodes > 0:
        blocklen = torch.empty(nown_nodes, dtype=torch.int64, device=device)
        K.stencil_blocklen(nown_nodes, gx, gy, gz, nown_nodes,
                           zs_own.data_ptr(), pb.data_ptr(), offs.data_ptr(),
                           ksten, blocklen.data_ptr(), stream)
        nbslices = (nown_nodes + 63) // 64
        bp = torch.zeros(nbslices * 64, dtype=torch.int64, device=device)
        bp[:nown_nodes] = blocklen
        blk_slice = bp.view(nbslices, 64).max(dim=1).values
        bptr = torch.zeros(nbslices + 1, dtype=torch.int64, device=device)
        torch.cumsum(blk_slice * 64, dim=0, out=bptr[1:])
        btotal = int(bptr[-1])
        bcol = torch.zeros(btotal, dtype=torch.int32, device=device)
        bvals = torch.zeros(btotal * dof * dof, dtype=torch.float64, device=device)
        K.stencil_bfill(nown_nodes, gx, gy, gz, dof, nown_nodes,
                        zs_own.data_ptr(), pb.data_ptr(), offs.data_ptr(),
                        ksten, blocks.data_ptr(), bptr.data_ptr(),
                        bcol.data_ptr(), bvals.data_ptr(), stream)
        A_bsell = (bptr, bcol, bvals, dof)

    # halo pattern (analytic, host-side; identical to stencil_local_slab)
    senders, recvcounts, rdispls = [], [], []
    recipients, sendidx_parts = [], []
    off = 0
    if has_lo:
        senders.append(rank - 1)
        recvcounts.append(plane_nodes * dof)
        rdispls.append(off)
        off += plane_nodes * dof
    if has_hi:
        senders.append(rank + 1)
        recvcounts.append(plane_nodes * dof)
        rdispls.append(off)
        off += plane_nodes * dof

    plane_local = {z: k for k, z in enumerate(plane_seq)}

    def plane_rows(z):
        b = plane_local[z] * plane_nodes
        node_loc = b + np.arange(plane_nodes, dtype=np.int64)
        return (node_loc[:, None] * dof + np.arange(dof)[None, :]).ravel()

    if has_lo:
        recipients.append(rank - 1)
        sendidx_parts.append(plane_rows(z0))
    if has_hi:
        recipients.append(rank + 1)
        sendidx_parts.append(plane_rows(z1 - 1))
    sendcounts = [len(s) for s in sendidx_parts]
    sdispls = list(np.concatenate([[0], np.cumsum(sendcounts)[:-1]])) if sendcounts else []
    sendidx = (np.concatenate(sendidx_parts) if sendidx_parts
               else np.zeros(0, np.int64))
    halo = HaloPattern(
        senders=np.asarray(senders, dtype=np.int32),
        recvcounts=np.asarray(recvcounts, dtype=np.int64),
        rdispls=np.asarray(rdispls, dtype=np.int64),
        recipients=np.asarray(recipients, dtype=np.int32),
        sendcounts=np.asarray(sendcounts, dtype=np.int64),
        sdispls=np.asarray(sdispls, dtype=np.int64),
        sendidx=sendidx.astype(np.int32),
    )
    return DeviceSlabSystem(
        rank=rank, nparts=nranks, n_global=gx * gy * gz * dof,
        nowned=nowned, ninterior=ninterior_nodes * dof,
        nborder=nborder_nodes * dof, nghost=nghost,
        A_sell=(A_sellptr, A_cols, A_vals) if operator else None,
        A_bsell=A_bsell,
        O_sell=(O_sellptr, O_cols, O_vals) if operator else None,
        mf_tables=mf_tables,
        _nnzA=nnzA, _nnzO=nnzO, halo=halo, device=device,
    )
