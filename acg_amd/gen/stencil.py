"""Synthetic SPD stencil operators, global and distributed-slab.

Benchmark matrices (BASELINE.json) are SuiteSparse downloads in the
reference's world; there is no network here, so the benchmarks use
synthetic SPD matrices of the same shape:

- 5-pt 2D / 7-pt 3D Poisson (configs 1, 5),
- a "Queen_4147-like" operator (config 2-4): 3-D 27-point stencil with a
  ``dof``-vector block per grid node (Queen_4147 is a 3-D structural
  problem: 4.15M rows, ~79 nnz/row).  dof=3 gives 81 nnz/row.

Two construction paths:

- :func:`stencil_global`: a SymCSRMatrix (packed upper triangle) for the
  file-driven pipeline and for small tests (goes through the generic
  partitioner/extractor like any read matrix).
- :func:`stencil_local_slab`: builds rank-local LocalSystems *directly*
  (contiguous z-slab partition), never materialising the global matrix --
  required for the 2048^3 (~8.6e9 rows) configuration and used by
  bench.py.  The layouts (interior|border|ghost ordering, ghost sorted by
  (owner, global id)) are identical to the generic extractor's.
"""

from __future__ import annotations

import numpy as np

from ..core.symcsr import SymCSRMatrix
from ..part.subdomain import HaloPattern, LocalSystem

# (dx, dy, dz, weight) — weight is the off-diagonal value; diag handled below
def _offsets_27():
    offs = [(dx, dy, dz, -1.0)
            for dz in (-1, 0, 1) for dy in (-1, 0, 1) for dx in (-1, 0, 1)
            if not (dx == 0 and dy == 0 and dz == 0)]
    return offs


STENCIL_5PT_2D = {
    "offsets": [(-1, 0, 0, -1.0), (1, 0, 0, -1.0), (0, -1, 0, -1.0), (0, 1, 0, -1.0)],
    "diag": 4.0,
    "dof": 1,
}
STENCIL_7PT_3D = {
    "offsets": [(-1, 0, 0, -1.0), (1, 0, 0, -1.0), (0, -1, 0, -1.0),
                (0, 1, 0, -1.0), (0, 0, -1, -1.0), (0, 0, 1, -1.0)],
    "diag": 6.0,
    "dof": 1,
}
STENCIL_27PT_3D = {
    "offsets": _offsets_27(),
    "diag": 27.0,  # > sum|w| = 26 → strictly diagonally dominant, SPD
    "dof": 1,
}


def _blocks(spec: dict):
    """Per-neighbour block matrix M (scaled by the offset weight) and the
    self/diagonal block D; dof=1 reduces to scalars."""
    dof = spec["dof"]
    M = np.asarray(spec.get("offblock", np.eye(dof)), dtype=np.float64)
    D = np.asarray(spec.get("diagblock", spec.get("diag", 1.0) * np.eye(dof)),
                   dtype=np.float64)
    assert M.shape == (dof, dof) and D.shape == (dof, dof)
    assert np.allclose(M, M.T) and np.allclose(D, D.T), "blocks must be symmetric"
    return M, D


def queen_like_spec(dof: int = 3) -> dict:
    """27-point stencil with dense dof x dof blocks per neighbour:
    27*dof nnz/row (81 for dof=3) — the shape class of Queen_4147
    (3-D structural mechanics, 4.15M rows, ~79 nnz/row).

    SPD by strict diagonal dominance (Gershgorin): per row, off-diagonal
    magnitude sums to 26*|M|_row + |D|_offdiag < diag."""
    s = dict(STENCIL_27PT_3D)
    s["dof"] = dof
    if dof == 3:
        M = np.array([[1.0, 0.3, 0.2],
                      [0.3, 1.0, 0.3],
                      [0.2, 0.3, 1.0]])
        D = 45.0 * np.eye(3) + 0.5 * (M - np.eye(3))
        s["offblock"] = M
        s["diagblock"] = D
    else:
        s["offblock"] = np.eye(dof)
        s["diagblock"] = 27.0 * np.eye(dof)
    return s


def queen_grid_for_rows(target_rows: int, dof: int = 3) -> int:
    """Grid edge G such that 3 G^3 ~= target_rows."""
    return int(round((target_rows / dof) ** (1.0 / 3.0)))


def _node_id(x, y, z, gx, gy):
    return x + gx * (y + gy * z)


def stencil_global(gx: int, gy: int, gz: int, spec: dict) -> SymCSRMatrix:
    """Assemble the global operator as packed-upper SymCSRMatrix."""
    dof = spec["dof"]
    M, D = _blocks(spec)
    nnodes = gx * gy * gz
    n = nnodes * dof
    X, Y, Z = np.meshgrid(np.arange(gx), np.arange(gy), np.arange(gz), indexing="ij")
    X, Y, Z = X.ravel(), Y.ravel(), Z.ravel()
    gid = _node_id(X, Y, Z, gx, gy)
    ii, jj, vv = [], [], []
    # self block: upper triangle of D only (from_coo sums duplicates, so the
    # mirrored half must not be emitted twice)
    for a in range(dof):
        for b in range(a, dof):
            if D[a, b] == 0.0:
                continue
            ii.append(gid * dof + a)
            jj.append(gid * dof + b)
            vv.append(np.full(nnodes, D[a, b]))
    for (dx, dy, dz, w) in spec["offsets"]:
        # keep only upper-triangle node pairs (i < j): offset lexicographically > 0
        if (dz, dy, dx) <= (0, 0, 0):
            continue
        nx, ny, nz = X + dx, Y + dy, Z + dz
        ok = (nx >= 0) & (nx < gx) & (ny >= 0) & (ny < gy) & (nz >= 0) & (nz < gz)
        gi = gid[ok]
        gj = _node_id(nx[ok], ny[ok], nz[ok], gx, gy)
        for a in range(dof):
            for b in range(dof):
                if M[a, b] == 0.0:
                    continue
                ii.append(gi * dof + a)
                jj.append(gj * dof + b)
                vv.append(np.full(len(gi), w * M[a, b]))
    i = np.concatenate(ii)
    j = np.concatenate(jj)
    v = np.concatenate(vv)
    return SymCSRMatrix.from_coo(n, i, j, v)


def _slab_bounds(gz: int, rank: int, nranks: int) -> tuple[int, int]:
    z0 = (gz * rank) // nranks
    z1 = (gz * (rank + 1)) // nranks
    return z0, z1


def stencil_local_slab(gx: int, gy: int, gz: int, spec: dict,
                       rank: int, nranks: int) -> LocalSystem:
    """Build rank ``rank``'s LocalSystem directly (z-slab partition).

    Layout matches acg_amd.part.subdomain.extract_subdomains: local node
    order is [interior planes | border planes | ghost planes], each group
    ascending in global id; ghosts sorted by (owner rank, global id).
    """
    dof = spec["dof"]
    z0, z1 = _slab_bounds(gz, rank, nranks)
    nz_own = z1 - z0
    if nz_own <= 0:
        raise ValueError(f"rank {rank} has no z-planes ({gz} planes / {nranks} ranks)")
    plane_nodes = gx * gy
    has_lo = rank > 0
    has_hi = rank < nranks - 1
    # which owned planes are border?
    maxreach = max(abs(o[2]) for o in spec["offsets"])
    if maxreach > 1:
        raise ValueError("slab generator supports |dz| <= 1 stencils")
    border_planes = set()
    if has_lo:
        border_planes.add(z0)
    if has_hi:
        border_planes.add(z1 - 1)
    interior_planes = [z for z in range(z0, z1) if z not in border_planes]
    border_list = sorted(border_planes)
    ghost_planes = ([z0 - 1] if has_lo else []) + ([z1] if has_hi else [])

    # local plane ordering and per-plane local node base
    plane_seq = interior_planes + border_list + ghost_planes
    plane_base = {z: k * plane_nodes for k, z in enumerate(plane_seq)}
    nown_nodes = nz_own * plane_nodes
    nghost_nodes = len(ghost_planes) * plane_nodes
    ninterior = len(interior_planes) * plane_nodes * dof
    nborder = len(border_list) * plane_nodes * dof
    nowned = nown_nodes * dof
    nghost = nghost_nodes * dof

    # per-owned-node coordinates in local order
    zs = np.repeat(np.asarray(interior_planes + border_list, dtype=np.int64), plane_nodes)
    xy = np.tile(np.arange(plane_nodes, dtype=np.int64), nz_own)
    xs = xy % gx
    ys = xy // gx

    # stencil columns: dense [nown_nodes, k] col/val with valid mask
    offs = spec["offsets"]
    k = len(offs) + 1
    coln = np.empty((nown_nodes, k), dtype=np.int64)
    valn = np.empty((nown_nodes, k), dtype=np.float64)
    okn = np.ones((nown_nodes, k), dtype=bool)
    # self
    pb = np.empty(gz + 2, dtype=np.int64)  # plane z -> local node base (+1 shift)
    pb.fill(-1)
    for z, b in plane_base.items():
        pb[z + 1] = b
    coln[:, 0] = pb[zs + 1] + xy
    valn[:, 0] = 1.0  # self entries take the D block (see expand_block)
    for c, (dx, dy, dz, w) in enumerate(offs, start=1):
        nx, ny, nzp = xs + dx, ys + dy, zs + dz
        ok = (nx >= 0) & (nx < gx) & (ny >= 0) & (ny < gy) & (nzp >= 0) & (nzp < gz)
        nzc = np.clip(nzp, -1, gz)
        base = pb[np.clip(nzc, 0, gz - 1) + 1]
        ok &= base >= 0
        coln[:, c] = np.where(ok, base + nx + gx * ny, 0)
        valn[:, c] = w
        okn[:, c] = ok

    selfn = np.zeros((nown_nodes, k), dtype=bool)
    selfn[:, 0] = True

    # sort columns within each row for ascending local index
    order = np.argsort(np.where(okn, coln, np.iinfo(np.int64).max), axis=1, kind="stable")
    coln = np.take_along_axis(coln, order, axis=1)
    valn = np.take_along_axis(valn, order, axis=1)
    selfn = np.take_along_axis(selfn, order, axis=1)
    okn = np.take_along_axis(okn, order, axis=1)

    # expand node-level structure to dof-level CSR rows with dof x dof blocks
    M, D = _blocks(spec)
    counts_node = okn.sum(axis=1).astype(np.int64)
    flat_cols = coln[okn]
    flat_vals = valn[okn]
    flat_self = selfn[okn]
    # split: ghost columns are local node idx >= nown_nodes
    ghost_mask = flat_cols >= nown_nodes
    node_of_entry = np.repeat(np.arange(nown_nodes, dtype=np.int64), counts_node)
    cntA_node = np.bincount(node_of_entry[~ghost_mask], minlength=nown_nodes)
    cntO_node = np.bincount(node_of_entry[ghost_mask], minlength=nown_nodes)

    colsA_node = flat_cols[~ghost_mask]
    valsA_node = flat_vals[~ghost_mask]
    selfA_node = flat_self[~ghost_mask]
    colsO_node = flat_cols[ghost_mask]
    valsO_node = flat_vals[ghost_mask]
    selfO_node = flat_self[ghost_mask]

    def expand_block(cnt_node, cols_node, vals_node, self_node):
        """Expand node-level entries to dof-level rows: node entry (i,j,w)
        becomes the dof x dof block w*M (or D for the self entry)."""
        nnode = len(cnt_node)
        nnzn = len(cols_node)
        rowptr = np.zeros(nnode * dof + 1, dtype=np.int64)
        cnt_rows = np.repeat(cnt_node * dof, dof)
        np.cumsum(cnt_rows, out=rowptr[1:])
        cols = np.empty(nnzn * dof * dof, dtype=np.int64)
        vals = np.empty(nnzn * dof * dof, dtype=np.float64)
        ends_node = np.cumsum(cnt_node)
        starts_node = ends_node - cnt_node
        entry_node = np.repeat(np.arange(nnode, dtype=np.int64), cnt_node)
        within = np.arange(nnzn, dtype=np.int64) - starts_node[entry_node]
        for a in range(dof):
            rbase = rowptr[entry_node * dof + a]
            for b in range(dof):
                dst = rbase + within * dof + b
                cols[dst] = cols_node * dof + b
                vals[dst] = np.where(self_node, D[a, b], vals_node * M[a, b])
        return rowptr, cols, vals

    A_rowptr, A_cols, A_vals = expand_block(cntA_node, colsA_node, valsA_node, selfA_node)
    # matO rows: only border rows carry ghost couplings; rows base = ninterior
    nborder_nodes = len(border_list) * plane_nodes
    bsel = slice(nown_nodes - nborder_nodes, nown_nodes)
    cntO_border = cntO_node[bsel]
    # colsO/valsO already only contain entries of border nodes (interior
    # nodes have no ghost neighbours) and are ordered by node
    O_rowptr, O_cols, O_vals = expand_block(cntO_border, colsO_node, valsO_node, selfO_node)

    cdt = np.int32 if (nowned + nghost) < 2**31 else np.int64
    # global ids of local rows (for IO / manufactured solutions)
    owned_node_gid = _node_id(xs, ys, zs, gx, gy)
    owned_global = (owned_node_gid[:, None] * dof + np.arange(dof)[None, :]).ravel()
    gzs = np.asarray(ghost_planes, dtype=np.int64)
    ghost_node_gid = (np.tile(np.arange(plane_nodes, dtype=np.int64), len(ghost_planes))
                      + np.repeat(gzs * plane_nodes, plane_nodes))
    ghost_global = (ghost_node_gid[:, None] * dof + np.arange(dof)[None, :]).ravel()

    # halo pattern: neighbours are rank-1 (plane z0-1 <-> z0) and rank+1
    senders, recvcounts, rdispls = [], [], []
    recipients, sendcounts, sdispls, sendidx_parts = [], [], [], []
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
    # send my border planes: plane z0 to rank-1, plane z1-1 to rank+1
    def plane_rows(z):
        b = plane_base[z]
        node_loc = b + np.arange(plane_nodes, dtype=np.int64)
        return (node_loc[:, None] * dof + np.arange(dof)[None, :]).ravel()

    if has_lo:
        recipients.append(rank - 1)
        sendidx_parts.append(plane_rows(z0))
    if has_hi:
        recipients.append(rank + 1)
        sendidx_parts.append(plane_rows(z1 - 1))
    # recipients must be ascending by rank for deterministic order
    if has_lo and has_hi:
        pass  # already ascending: rank-1 < rank+1
    sendcounts = [len(s) for s in sendidx_parts]
    sdispls = list(np.concatenate([[0], np.cumsum(sendcounts)[:-1]])) if sendcounts else []
    sendidx = np.concatenate(sendidx_parts) if sendidx_parts else np.zeros(0, np.int64)

    halo = HaloPattern(
        senders=np.asarray(senders, dtype=np.int32),
        recvcounts=np.asarray(recvcounts, dtype=np.int64),
        rdispls=np.asarray(rdispls, dtype=np.int64),
        recipients=np.asarray(recipients, dtype=np.int32),
        sendcounts=np.asarray(sendcounts, dtype=np.int64),
        sdispls=np.asarray(sdispls, dtype=np.int64),
        sendidx=sendidx.astype(cdt),
    )
    return LocalSystem(
        rank=rank, nparts=nranks, n_global=gx * gy * gz * dof,
        nowned=nowned, ninterior=ninterior, nborder=nborder, nghost=nghost,
        A_rowptr=A_rowptr, A_colidx=A_cols.astype(cdt), A_vals=A_vals,
        O_rowptr=O_rowptr, O_colidx=O_cols.astype(cdt), O_vals=O_vals,
        owned_global=owned_global, ghost_global=ghost_global, halo=halo,
    )
