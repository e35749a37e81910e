"""Tensor-level wrappers around the gfx950 HIP kernels (_acg_kernels.so).

Reference analog: the BLAS/SpMV call layer of acg/cghip.c (hipsparse
SpMV setup :463-585, hipblas dot/axpy wrappers) -- here every hot op is
one of our own kernels (ops/kernels.hip cites the per-kernel analogs).

Raw pointers + the caller's current HIP stream are passed to the
extension; all type/shape checks happen here.  If the extension is
missing on a machine with a GPU, import fails loudly (no eager fallback).

Reductions are two-phase (per-block partials + finalize kernel): callers
pass a ``partials`` scratch tensor of size 2*MAXG allocated once via
:func:`alloc_partials`.
"""

from __future__ import annotations

import torch

try:
    from . import _acg_kernels as K
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "acg_amd HIP kernel extension (_acg_kernels) is not built. "
        "Run `python -m acg_amd.ops.build` (hipcc --offload-arch=gfx950). "
        f"Original error: {e}"
    ) from e

# scalar-slab slot indices (shared with kernels.hip)
S_RR = K.S_RR
S_PT = K.S_PT
S_RR_PREV = K.S_RR_PREV
S_BNRM2 = K.S_BNRM2
S_GAMMA = K.S_GAMMA
S_DELTA = K.S_DELTA
S_GAMMA_PREV = K.S_GAMMA_PREV
S_ALPHA_PREV = K.S_ALPHA_PREV
S_NSLOTS = K.S_NSLOTS
MAXG = K.MAXG


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def alloc_scalars(device) -> torch.Tensor:
    return torch.zeros(S_NSLOTS, dtype=torch.float64, device=device)


def alloc_partials(device) -> torch.Tensor:
    return torch.zeros(2 * MAXG, dtype=torch.float64, device=device)


def pick_lanes(mean_nnz_per_row: float) -> int:
    """Lanes-per-row heuristic for the CSR vector kernel (CDNA4: wave=64).

    MEASURED mapping (tools/lanes_sweep.py, fixed-length rows, 30M nnz):
    the optimum sits near 1-2 nnz/lane, NOT the round-1 ~6 rule-of-thumb
    (len-24 rows: 32 lanes 372 us vs 4 lanes 557; len-16: 16 lanes 432
    vs 4 lanes 588).  len<=8 -> 8, <=16 -> 16, <=96 -> 32, else 64."""
    if mean_nnz_per_row <= 4.0:
        return 4
    if mean_nnz_per_row <= 8.0:
        return 8
    if mean_nnz_per_row <= 16.0:
        return 16
    if mean_nnz_per_row <= 96.0:
        return 32
    return 64


def spmv(rowptr: torch.Tensor, colidx: torch.Tensor, vals: torch.Tensor,
         x: torch.Tensor, y: torch.Tensor, *, rowbase: int = 0,
         lanes: int = 16, accum: bool = False,
         partials: torch.Tensor | None = None,
         scal: torch.Tensor | None = None, dotslot: int = -1,
         dot_accum: bool = True) -> None:
    """CSR-vector SpMV: y[rowbase:+nrows] (=|+=) A x; optional fused
    dot(x,y) finalized into scal[dotslot]."""
    nrows = rowptr.numel() - 1
    if nrows <= 0:
        return
    fuse = scal is not None and dotslot >= 0
    K.spmv(nrows, rowbase, rowptr.data_ptr(), colidx.data_ptr(),
           1 if colidx.dtype == torch.int64 else 0,
           vals.data_ptr(), x.data_ptr(), y.data_ptr(), lanes, accum,
           partials.data_ptr() if fuse else 0,
           scal.data_ptr() if fuse else 0, dotslot, dot_accum, _stream())


def build_row_bins(rowptr, max_lanes: int = 64):
    """Sort rows into lane-count bins by length for the hybrid binned SpMV.

    Returns (rowlist int32 numpy, bins [(start, count, lanes)]).  Bin
    thresholds follow the pick_lanes rule (~6 nnz per lane); rows inside
    a bin are ordered longest-first so grid-stride waves retire the
    heavy tail early."""
    import numpy as np

    rowptr = np.asarray(rowptr)
    lens = np.diff(rowptr)
    order = np.argsort(-lens, kind="stable").astype(np.int32)
    slens = lens[order.astype(np.int64)]
    bins = []
    lo = 0
    # descending order: long rows first; thresholds from the high end
    # (measured mapping, see pick_lanes: >96 -> 64, >16 -> 32, >8 -> 16,
    # >4 -> 8, rest 4)
    for lanes, thresh in ((64, 96), (32, 16), (16, 8), (8, 4), (4, None)):
        if lanes > max_lanes:
            continue
        if thresh is None:
            hi = len(slens)
        else:
            hi = int(np.searchsorted(-slens, -thresh - 1, side="right"))
        if hi > lo:
            bins.append((lo, hi - lo, lanes))
            lo = hi
    return order, bins


def build_sellcsr_hybrid(rowptr, colidx, vals, cut: int = 192,
                         window: int = 0, bucket: int = 8):
    """Host prep for the SELL+CSR hybrid: short rows (len <= cut) as
    sigma-sorted SELL, long rows as a longest-first 64/32-lane binned CSR
    list.  Returns (sellptr, cols, svals, perm, rowlist_long, bins) numpy
    arrays (sellptr None when every row is long)."""
    import numpy as np

    from .torch_ref import sellcsr_split

    sellptr, cols, svals, perm, long_rows, nshort = sellcsr_split(
        rowptr, colidx, vals, cut=cut, window=window, bucket=bucket)
    lens = np.diff(np.asarray(rowptr))
    longlens = lens[long_rows.astype(np.int64)]  # descending
    bins = []
    lo = 0
    for lanes, thresh in ((64, 96), (32, 16), (16, 8), (8, 4), (4, None)):
        if thresh is None:
            hi = len(longlens)
        else:
            hi = int(np.searchsorted(-longlens, -thresh - 1, side="right"))
        if hi > lo:
            bins.append((lo, hi - lo, lanes))
            lo = hi
        if lo >= len(longlens):
            break
    if nshort == 0:
        sellptr = None
    return sellptr, cols, svals, perm, long_rows, bins


def spmv_binned(rowptr: torch.Tensor, colidx: torch.Tensor, vals: torch.Tensor,
                rowlist: torch.Tensor, bins, x: torch.Tensor, y: torch.Tensor,
                *, rowbase: int = 0, accum: bool = False,
                partials: torch.Tensor | None = None,
                scal: torch.Tensor | None = None, dotslot: int = -1,
                dot_accum: bool = True) -> None:
    """Row-binned hybrid CSR SpMV (load balance for power-law rows).
    ``rowlist`` int32 rows sorted longest-first; ``bins`` from
    :func:`build_row_bins`."""
    if rowlist.numel() == 0:
        return
    assert rowlist.dtype == torch.int32
    fuse = scal is not None and dotslot >= 0
    K.spmv_binned(rowbase, rowptr.data_ptr(), colidx.data_ptr(),
                  1 if colidx.dtype == torch.int64 else 0,
                  vals.data_ptr(), x.data_ptr(), y.data_ptr(),
                  rowlist.data_ptr(), [list(b) for b in bins], accum,
                  partials.data_ptr() if fuse else 0,
                  scal.data_ptr() if fuse else 0, dotslot, dot_accum,
                  _stream())


# SELL kernel variant bits (see kernels.hip): +1 non-temporal vals/cols,
# +2 XCD-aware block swizzle, +4 unroll-8.
SELL_NT, SELL_SWZ, SELL_U8 = 1, 2, 4
# A/B on MI355X (tools/spmv_bench.py, Queen-shaped 326M nnz): base 5.53,
# NT 5.66, NT+U8 5.78 TB/s (92% of the ~6.3 TB/s achievable); SWZ slightly
# negative (x is L3-resident, natural slice order already L2-local).
DEFAULT_SELL_VARIANT = SELL_NT | SELL_U8


def spmv_sell(sellptr: torch.Tensor, cols: torch.Tensor, vals: torch.Tensor,
              nrows: int, x: torch.Tensor, y: torch.Tensor, *,
              rowbase: int = 0, accum: bool = False,
              partials: torch.Tensor | None = None,
              scal: torch.Tensor | None = None, dotslot: int = -1,
              dot_accum: bool = True, variant: int | None = None,
              perm: torch.Tensor | None = None) -> None:
    """SELL-C-64(-sigma) SpMV.  ``perm`` (int32) maps SELL row -> matrix
    row for sigma-sorted irregular matrices."""
    nslices = sellptr.numel() - 1
    if nslices <= 0:
        return
    if variant is None:
        variant = DEFAULT_SELL_VARIANT
    fuse = scal is not None and dotslot >= 0
    K.spmv_sell(nslices, nrows, rowbase, sellptr.data_ptr(), cols.data_ptr(),
                1 if cols.dtype == torch.int64 else 0,
                vals.data_ptr(), x.data_ptr(), y.data_ptr(), accum,
                partials.data_ptr() if fuse else 0,
                scal.data_ptr() if fuse else 0, dotslot, dot_accum,
                variant, perm.data_ptr() if perm is not None else 0, _stream())


def spmv_bsell(bptr: torch.Tensor, bcol: torch.Tensor, bvals: torch.Tensor,
               nnodes: int, dof: int, x: torch.Tensor, y: torch.Tensor, *,
               partials: torch.Tensor | None = None,
               scal: torch.Tensor | None = None, dotslot: int = -1,
               dot_accum: bool = True) -> None:
    """Block-SELL SpMV (dense dof x dof blocks; one int32 index per block)."""
    nslices = bptr.numel() - 1
    if nslices <= 0:
        return
    fuse = scal is not None and dotslot >= 0
    K.spmv_bsell(nslices, nnodes, dof, bptr.data_ptr(), bcol.data_ptr(),
                 bvals.data_ptr(), x.data_ptr(), y.data_ptr(),
                 partials.data_ptr() if fuse else 0,
                 scal.data_ptr() if fuse else 0, dotslot, dot_accum, _stream())


def stencil_spmv(mf, nrows_nodes: int, row0_node: int, x: torch.Tensor,
                 y: torch.Tensor, *, mato: bool = False,
                 partials: torch.Tensor | None = None,
                 scal: torch.Tensor | None = None, dotslot: int = -1,
                 dot_accum: bool = True) -> None:
    """Matrix-free stencil SpMV (dof=1 constant-coefficient operators).
    ``mf`` = gen.device_slab mf_tables dict.  7-pt matA passes take the
    z-column-walk kernel (w_old read exactly once)."""
    if nrows_nodes <= 0:
        return
    fuse = scal is not None and dotslot >= 0
    pp = partials.data_ptr() if fuse else 0
    sp = scal.data_ptr() if fuse else 0
    if not mato and mf["w7"] is not None:
        K.stencil_spmv7(mf["gx"] * mf["gy"], mf["gx"], mf["z0"], mf["z1"],
                        mf["gz"], mf["nown_nodes"], mf["pb"].data_ptr(),
                        mf["diag"], *mf["w7"], x.data_ptr(), y.data_ptr(),
                        pp, sp, dotslot, dot_accum, _stream())
        return
    K.stencil_spmv(nrows_nodes, row0_node, mf["gx"], mf["gy"], mf["gz"],
                   mf["nown_nodes"], mf["zs"].data_ptr(),
                   mf["pb"].data_ptr(), mf["offs"].data_ptr(), mf["ksten"],
                   mf["diag"], x.data_ptr(), y.data_ptr(), mato,
                   pp, sp, dotslot, dot_accum, _stream())


def stencil_pipe(mf, nrows_nodes: int, row0_node: int, border_base: int,
                 w_old, qpart, z, t, p, x, r, w_new, scal, first: bool,
                 partials, partials_off: int, mato: bool) -> int:
    """Megafused matrix-free pipelined iteration pass (stencil SpMV +
    6-vector update + dots); same protocol as sell_pipe."""
    if nrows_nodes <= 0:
        return 0
    qp = qpart.data_ptr() if qpart is not None else 0
    if not mato and mf["w7"] is not None:
        return K.stencil_pipe7(mf["gx"] * mf["gy"], mf["gx"], mf["z0"],
                               mf["z1"], mf["gz"], border_base,
                               mf["nown_nodes"], mf["pb"].data_ptr(),
                               mf["diag"], *mf["w7"], w_old.data_ptr(), qp,
                               z.data_ptr(), t.data_ptr(), p.data_ptr(),
                               x.data_ptr(), r.data_ptr(), w_new.data_ptr(),
                               scal.data_ptr(), 1 if first else 0,
                               partials.data_ptr(), partials_off, _stream())
    return K.stencil_pipe(nrows_nodes, row0_node, border_base, mf["gx"],
                          mf["gy"], mf["gz"], mf["nown_nodes"],
                          mf["zs"].data_ptr(), mf["pb"].data_ptr(),
                          mf["offs"].data_ptr(), mf["ksten"], mf["diag"],
                          w_old.data_ptr(), qp,
                          z.data_ptr(), t.data_ptr(), p.data_ptr(),
                          x.data_ptr(), r.data_ptr(), w_new.data_ptr(),
                          scal.data_ptr(), 1 if first else 0,
                          partials.data_ptr(), partials_off, mato, _stream())


def spmv_bsell_daypx(bptr, bcol, bvals, nnodes: int, dof: int,
                     pold: torch.Tensor, rvec: torch.Tensor,
                     pnew: torch.Tensor, y: torch.Tensor,
                     scal: torch.Tensor, partials: torch.Tensor,
                     dotslot: int) -> None:
    """Classic-CG fold: p_new = beta*p_old + r materialised inside the
    BSELL SpMV (gather computes it on the fly), (p_new, t) dot fused.
    beta = scal[RR]/scal[RR_PREV]; serial matA-only (no matO pass)."""
    nslices = bptr.numel() - 1
    if nslices <= 0:
        return
    K.spmv_bsell_daypx(nslices, nnodes, dof, bptr.data_ptr(), bcol.data_ptr(),
                       bvals.data_ptr(), pold.data_ptr(), rvec.data_ptr(),
                       pnew.data_ptr(), y.data_ptr(), scal.data_ptr(),
                       partials.data_ptr(), dotslot, _stream())


def zero_scalars(scal: torch.Tensor, i0: int = 0, count: int | None = None) -> None:
    K.zero_scalars(scal.data_ptr(), i0, count if count is not None else scal.numel() - i0,
                   _stream())


def cg_prep_pt(scal: torch.Tensor) -> None:
    K.cg_prep_pt(scal.data_ptr(), _stream())


def cg_prep_rr(scal: torch.Tensor) -> None:
    K.cg_prep_rr(scal.data_ptr(), _stream())


def dot(x: torch.Tensor, y: torch.Tensor, partials: torch.Tensor,
        scal: torch.Tensor, slot: int, n: int | None = None,
        accumulate: bool = False) -> None:
    n = x.numel() if n is None else n
    K.dot(x.data_ptr(), y.data_ptr(), n, partials.data_ptr(), scal.data_ptr(),
          slot, accumulate, _stream())


def dot2(r: torch.Tensor, w: torch.Tensor, partials: torch.Tensor,
         scal: torch.Tensor, n: int, accumulate: bool = False) -> None:
    """scal[GAMMA] (=|+=) (r,r); scal[DELTA] (=|+=) (w,r): one pass."""
    K.dot2(r.data_ptr(), w.data_ptr(), n, partials.data_ptr(), scal.data_ptr(),
           accumulate, _stream())


def axpy_ratio(y: torch.Tensor, x: torch.Tensor, scal: torch.Tensor,
               num: int, den: int, sign: float = 1.0, n: int | None = None) -> None:
    K.axpy_ratio(y.data_ptr(), x.data_ptr(), y.numel() if n is None else n,
                 scal.data_ptr(), num, den, sign, _stream())


def daypx_ratio(y: torch.Tensor, x: torch.Tensor, scal: torch.Tensor,
                num: int, den: int, n: int | None = None) -> None:
    K.daypx_ratio(y.data_ptr(), x.data_ptr(), y.numel() if n is None else n,
                  scal.data_ptr(), num, den, _stream())


def cg_fused_update(r: torch.Tensor, x: torch.Tensor, p: torch.Tensor,
                    t: torch.Tensor, scal: torch.Tensor,
                    partials: torch.Tensor, n: int) -> None:
    """alpha = rr_prev/pt (device); r -= alpha t; x += alpha p;
    scal[RR] = (r,r) (finalized)."""
    K.cg_fused_update(r.data_ptr(), x.data_ptr(), p.data_ptr(), t.data_ptr(),
                      n, scal.data_ptr(), partials.data_ptr(), _stream())


def pcg_fused_update(r, x, p, t, z, dinv, scal, partials, n: int) -> None:
    """Jacobi-PCG fused epilogue: alpha = rz/pt (device); r -= alpha t;
    x += alpha p; z = dinv*r; rotates rz -> RR_PREV and publishes the new
    rz in S_RR plus the TRUE (r,r) in S_GAMMA."""
    K.pcg_fused_update(r.data_ptr(), x.data_ptr(), p.data_ptr(), t.data_ptr(),
                       z.data_ptr(), dinv.data_ptr(), n, scal.data_ptr(),
                       partials.data_ptr(), _stream())


def sell_pipe(sellptr, cols, vals, nrows_pass: int, rowbase: int,
              border_base: int, w_old, qpart, z, t, p, x, r, w_new,
              scal, first: bool, partials, partials_off: int,
              mato: bool) -> int:
    """Megafused pipelined iteration pass (SpMV + 6-vector update + dots).
    Returns the number of partial blocks written."""
    nslices = sellptr.numel() - 1
    assert cols.dtype == torch.int32
    return K.sell_pipe(nslices, nrows_pass, rowbase, border_base,
                       sellptr.data_ptr(), cols.data_ptr(), vals.data_ptr(),
                       w_old.data_ptr(),
                       qpart.data_ptr() if qpart is not None else 0,
                       z.data_ptr(), t.data_ptr(), p.data_ptr(), x.data_ptr(),
                       r.data_ptr(), w_new.data_ptr(), scal.data_ptr(),
                       1 if first else 0, partials.data_ptr(), partials_off,
                       mato, _stream())


def pipelined_finalize(partials, nblocks: int, scal, first: bool) -> None:
    K.pipelined_finalize(partials.data_ptr(), nblocks, scal.data_ptr(),
                         1 if first else 0, _stream())


def pipelined_fused(z, t, p, x, r, w, q, scal: torch.Tensor,
                    partials: torch.Tensor, n: int, first: bool,
                    nt_update: bool = True) -> None:
    """Fused pipelined update + next gamma/delta + scalar rotation."""
    K.pipelined_fused(z.data_ptr(), t.data_ptr(), p.data_ptr(), x.data_ptr(),
                      r.data_ptr(), w.data_ptr(), q.data_ptr(), n,
                      scal.data_ptr(), 1 if first else 0, partials.data_ptr(),
                      nt_update, _stream())


BAR_STATE_WORDS = 400  # shared with kernels.hip (flat + hierarchical v3)


def cg_device(sellptr: torch.Tensor, cols: torch.Tensor, vals: torch.Tensor,
              nrows: int, b: torch.Tensor, x: torch.Tensor, r: torch.Tensor,
              p: torch.Tensor, t: torch.Tensor, scal: torch.Tensor,
              partials: torch.Tensor, out2: torch.Tensor,
              barrier_state: torch.Tensor, maxits: int,
              res_atol: float, res_rtol: float,
              hier: bool | None = None) -> int:
    """Monolithic device-side CG: one cooperative launch runs the whole
    solve.  ``barrier_state``: BAR_STATE_WORDS zeroed uint32 words (must
    be zeroed before EVERY launch).  ``hier`` selects the hierarchical
    (per-XCD then global) grid barrier; default = env ACG_DEVCG_HIER
    ("auto": hierarchical from 64 blocks up -- measured crossover,
    tools/devcg_barrier_ab.py).  Returns the grid size used."""
    nslices = sellptr.numel() - 1
    assert barrier_state.numel() >= BAR_STATE_WORDS
    if hier is None:
        import os

        v = os.environ.get("ACG_DEVCG_HIER", "auto")
        ih = -1 if v == "auto" else (0 if v == "0" else 1)
    else:
        ih = 1 if hier else 0
    return K.cg_device(nslices, nrows, sellptr.data_ptr(), cols.data_ptr(),
                       1 if cols.dtype == torch.int64 else 0, vals.data_ptr(),
                       b.data_ptr(), x.data_ptr(), r.data_ptr(), p.data_ptr(),
                       t.data_ptr(), scal.data_ptr(), partials.data_ptr(),
                       out2.data_ptr(), barrier_state.data_ptr(),
                       maxits, res_atol, res_rtol, _stream(), ih)


def pack_gather(sendbuf: torch.Tensor, x: torch.Tensor, idx: torch.Tensor) -> None:
    K.pack_gather(sendbuf.data_ptr(), x.data_ptr(), idx.data_ptr(),
                  1 if idx.dtype == torch.int64 else 0,
                  sendbuf.numel(), _stream())
