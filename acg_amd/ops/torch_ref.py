"""Plain-PyTorch fp64 reference implementations of every HIP op.

Used (a) as the oracle in kernel numerics tests (GPU kernel vs torch fp64,
same semantics), (b) as the compute engine of the CPU solver path
(reference analog: acg/cg.c host solver).  Mirrors the device-scalar
convention: coefficients live in a small fp64 "scal" tensor; ``partials``
arguments are accepted for API parity but unused (torch reduces directly).
"""

from __future__ import annotations

import torch

# keep slot numbering identical to kernels.hip
S_RR, S_PT, S_RR_PREV, S_BNRM2 = 0, 1, 2, 3
S_GAMMA, S_DELTA, S_GAMMA_PREV, S_ALPHA_PREV = 4, 5, 6, 7
S_NSLOTS = 8


def alloc_scalars(device="cpu") -> torch.Tensor:
    return torch.zeros(S_NSLOTS, dtype=torch.float64, device=device)


def alloc_partials(device="cpu") -> torch.Tensor:
    return torch.zeros(1, dtype=torch.float64, device=device)


def spmv(rowptr, colidx, vals, x, y, *, rowbase: int = 0, accum: bool = False,
         partials=None, scal=None, dotslot: int = -1, dot_accum: bool = True) -> None:
    nrows = rowptr.numel() - 1
    if nrows <= 0:
        return
    counts = rowptr[1:] - rowptr[:-1]
    rows = torch.repeat_interleave(torch.arange(nrows, dtype=torch.int64,
                                                device=rowptr.device), counts)
    prod = vals * x[colidx.long()]
    contrib = torch.zeros(nrows, dtype=torch.float64, device=x.device)
    contrib.index_add_(0, rows, prod)
    sl = slice(rowbase, rowbase + nrows)
    if accum:
        y[sl] += contrib
    else:
        y[sl] = contrib
    if scal is not None and dotslot >= 0:
        d = torch.dot(x[sl], contrib)
        scal[dotslot] = scal[dotslot] + d if dot_accum else d


def sell_from_csr(rowptr, colidx, vals, C: int = 64, sigma: int = 1):
    """Convert CSR -> SELL-C-sigma (vectorized numpy host prep).

    Element j of SELL row (s*C+lane) lives at sellptr[s] + j*C + lane.
    ``sigma`` > 1 sorts rows by descending length within windows of
    sigma*C rows, shrinking padding for irregular matrices; the returned
    ``perm`` (or None for sigma=1) maps SELL row -> matrix row (pad lanes
    hold nrows as a sentinel).  Padding entries point at a valid row with
    value 0.  Returns (sellptr int64, cols, vals f64[, perm int32])."""
    import numpy as np

    rowptr = np.asarray(rowptr)
    colidx = np.asarray(colidx)
    vals = np.asarray(vals)
    nrows = len(rowptr) - 1
    nslices = (nrows + C - 1) // C
    counts = np.diff(rowptr)
    perm = None
    rowof = np.arange(nslices * C, dtype=np.int64)  # SELL row -> matrix row
    if sigma > 1 and nrows:
        win = sigma * C
        order = np.empty(nrows, dtype=np.int64)
        for w0 in range(0, nrows, win):
            w1 = min(w0 + win, nrows)
            sub = np.argsort(-counts[w0:w1], kind="stable")
            order[w0:w1] = w0 + sub
        rowof = np.full(nslices * C, nrows, dtype=np.int64)
        rowof[:nrows] = order
        perm = rowof.astype(np.int32)
    cpad = np.zeros(nslices * C, dtype=np.int64)
    valid = rowof < nrows
    cpad[valid] = counts[rowof[valid]]
    slice_len = cpad.reshape(nslices, C).max(axis=1)
    sellptr = np.zeros(nslices + 1, dtype=np.int64)
    np.cumsum(slice_len * C, out=sellptr[1:])
    total = int(sellptr[-1])
    # defaults: col = first valid row's own col target (clipped), val = 0
    slice_of_p = np.repeat(np.arange(nslices, dtype=np.int64), slice_len * C)
    lane = (np.arange(total, dtype=np.int64) - sellptr[slice_of_p]) % C
    defrow = np.minimum(rowof[np.minimum(slice_of_p * C + lane,
                                         nslices * C - 1)], nrows - 1)
    cols = defrow.astype(colidx.dtype)
    svals = np.zeros(total, dtype=np.float64)
    # scatter real entries: matrix row r sits at SELL row invperm[r]
    nnz = len(colidx)
    if perm is not None:
        invperm = np.empty(nslices * C, dtype=np.int64)
        invperm[rowof] = np.arange(nslices * C, dtype=np.int64)
        sellrow = invperm[:nrows]
    else:
        sellrow = np.arange(nrows, dtype=np.int64)
    rows = np.repeat(sellrow, counts)
    within = np.arange(nnz, dtype=np.int64) - rowptr[np.repeat(
        np.arange(nrows, dtype=np.int64), counts)]
    dst = sellptr[rows // C] + within * C + rows % C
    cols[dst] = colidx
    svals[dst] = vals
    if perm is not None:
        return sellptr, cols, svals, perm
    return sellptr, cols, svals


def sellcsr_split(rowptr, colidx, vals, cut: int = 192, C: int = 64,
                  window: int = 0, bucket: int = 8):
    """Split a CSR matrix by row length for the SELL+CSR hybrid format.

    Rows with len <= ``cut`` (the regular majority) go into a sigma-SELL
    structure; rows with len > cut (the power-law tail) stay CSR and are
    listed longest-first for the 64-lane vector kernel.  MEASURED motive
    (MI355X, 1M-row power-law): the pure binned hybrid spends 241 us/it
    in its 4-lane short-row bin -- 4-lane groups issue 16 scattered row
    streams per wave, while SELL's 64-lane slices load one contiguous
    512 B line set per step.

    ``window``: 0 = short rows globally length-descending (minimal
    padding, longest slices first).  W > 0 = sort by length only WITHIN
    windows of W consecutive short rows (original index order preserved
    across windows): slightly more padding, but the x gather keeps the
    source ordering's column locality -- the lever when the gather, not
    the vals/cols stream, bounds the SELL part.

    Returns (sellptr i64, cols, svals f64, perm int32 [sentinel n],
    rowlist_long int32 desc, nshort).  SELL part empty => sellptr len 1.
    """
    import numpy as np

    rowptr = np.asarray(rowptr)
    colidx = np.asarray(colidx)
    vals = np.asarray(vals)
    n = len(rowptr) - 1
    lens = np.diff(rowptr)
    order = np.argsort(-lens, kind="stable")
    slens = lens[order]
    nlong = int(np.searchsorted(-slens, -int(cut) - 1, side="right"))
    long_rows = order[:nlong].astype(np.int32)
    short_rows = order[nlong:].astype(np.int64)  # descending lengths
    if bucket and nlong and cut >= 96:
        # same line-sharing ordering for the tail.  Only when every long
        # row gets 64 lanes (cut >= 96): the bin builder derives bin
        # boundaries by searchsorted on descending lengths, which this
        # reorder would break for multi-bin tails.
        ll = lens[long_rows.astype(np.int64)]
        fc = colidx[rowptr[long_rows.astype(np.int64)]]
        bw = max(bucket * 8, 64)
        long_rows = long_rows[np.lexsort((fc, -((ll + bw - 1) // bw)))]
    if window and len(short_rows):
        sr = np.sort(short_rows)  # original index order
        out = np.empty_like(sr)
        for w0 in range(0, len(sr), window):
            w1 = min(w0 + window, len(sr))
            sub = np.argsort(-lens[sr[w0:w1]], kind="stable")
            out[w0:w1] = sr[w0:w1][sub]
        short_rows = out
    elif bucket and len(short_rows):
        # (length-bucket desc, first column asc): padding bounded by the
        # bucket width while rows inside a slice share nearby columns --
        # their 128 B x-line fetches overlap (the gather is the measured
        # bound: 50% L2 miss with 8-of-128-byte line use, see
        # profiles/irr_pmc_tcc_r02.txt)
        sl = lens[short_rows]
        firstcol = np.where(sl > 0, colidx[rowptr[short_rows]], 0)
        bkt = (sl + bucket - 1) // bucket
        order2 = np.lexsort((firstcol, -bkt))
        short_rows = short_rows[order2]
    nshort = len(short_rows)
    nslices = (nshort + C - 1) // C
    # SELL arrays over the short subset, slice lengths from the sorted order
    padlens = np.zeros(nslices * C, dtype=np.int64)
    padlens[:nshort] = lens[short_rows]
    slice_len = padlens.reshape(nslices, C).max(axis=1) if nslices else \
        np.zeros(0, np.int64)
    sellptr = np.zeros(nslices + 1, dtype=np.int64)
    np.cumsum(slice_len * C, out=sellptr[1:])
    total = int(sellptr[-1])
    perm = np.full(nslices * C, n, dtype=np.int32)
    perm[:nshort] = short_rows
    # defaults: pad cols point at a valid x slot, pad vals 0
    slice_of_p = np.repeat(np.arange(nslices, dtype=np.int64), slice_len * C)
    lane = (np.arange(total, dtype=np.int64) - sellptr[slice_of_p]) % C
    defrow = np.minimum(perm[np.minimum(slice_of_p * C + lane,
                                        max(nslices * C - 1, 0))], n - 1)
    cols = defrow.astype(colidx.dtype)
    svals = np.zeros(total, dtype=np.float64)
    # scatter the short rows' entries (vectorised: no per-row python loop)
    counts = lens[short_rows]
    rows_rep = np.repeat(np.arange(nshort, dtype=np.int64), counts)
    excl = np.zeros(nshort, dtype=np.int64)
    if nshort:
        np.cumsum(counts[:-1], out=excl[1:])
    within = np.arange(int(counts.sum()), dtype=np.int64) - excl[rows_rep]
    src = np.repeat(rowptr[short_rows], counts) + within
    dst = sellptr[rows_rep // C] + within * C + rows_rep % C
    cols[dst] = colidx[src]
    svals[dst] = vals[src]
    return sellptr, cols, svals, perm, long_rows, nshort


def bsell_from_csr(rowptr, colidx, vals, dof: int, C: int = 64):
    """Convert CSR -> Block-SELL (dense dof x dof blocks per node pair).

    Returns (bptr int64[nslices+1] in block units, bcol int32, bvals f64
    laid out [slice][j][k][64], density) or None when the matrix shape is
    not divisible by dof.  Blocks missing entries are zero-filled; the
    caller decides (via density) whether the 4->4/dof^2 B/nnz index saving
    beats the zero-fill cost."""
    import numpy as np

    rowptr = np.asarray(rowptr)
    colidx = np.asarray(colidx, dtype=np.int64)
    vals = np.asarray(vals)
    nrows = len(rowptr) - 1
    if dof < 2 or nrows % dof:
        return None
    nnodes = nrows // dof
    nnz = len(colidx)
    rows = np.repeat(np.arange(nrows, dtype=np.int64), np.diff(rowptr))
    try:
        from ..host import _acg_host as H

        blocks_per_node, bcols, inv = (np.asarray(a) for a in
                                       H.bsell_blocks(rowptr, colidx, dof))
        nblocks = len(bcols)
        bnode = np.repeat(np.arange(nnodes, dtype=np.int64), blocks_per_node)
    except ImportError:  # numpy fallback: O(nnz log nnz) unique
        node = rows // dof
        bcol_of_entry = colidx // dof
        key = node * (np.int64(1) << 32) | bcol_of_entry.astype(np.int64)
        uk, inv = np.unique(key, return_inverse=True)
        nblocks = len(uk)
        bnode = (uk >> 32).astype(np.int64)
        bcols = (uk & 0xFFFFFFFF).astype(np.int64)
        blocks_per_node = np.bincount(bnode, minlength=nnodes)
    density = nnz / (nblocks * dof * dof)
    nslices = (nnodes + C - 1) // C
    padlen = np.zeros(nslices * C, dtype=np.int64)
    padlen[:nnodes] = blocks_per_node
    slice_len = padlen.reshape(nslices, C).max(axis=1)
    bptr = np.zeros(nslices + 1, dtype=np.int64)
    np.cumsum(slice_len * C, out=bptr[1:])
    btotal = int(bptr[-1])
    # default: self block col (in-bounds), zero values
    slice_of_p = np.repeat(np.arange(nslices, dtype=np.int64), slice_len * C)
    lane = (np.arange(btotal, dtype=np.int64) - bptr[slice_of_p]) % C
    bcol = np.minimum(slice_of_p * C + lane, nnodes - 1).astype(np.int32)
    bvals = np.zeros(btotal * dof * dof, dtype=np.float64)
    # position of block b within its node's list (blocks sorted by key =>
    # grouped by node, ascending bcol)
    starts = np.zeros(nnodes + 1, dtype=np.int64)
    np.cumsum(blocks_per_node, out=starts[1:])
    within = np.arange(nblocks, dtype=np.int64) - starts[bnode]
    bdst = bptr[bnode // C] + within * C + bnode % C
    bcol[bdst] = bcols.astype(np.int32)
    # scatter entries into their block slots
    entry_block_dst = bdst[inv]
    k = (rows % dof) * dof + colidx % dof
    # bvals index, PAIR-major (see kernels.hip bval_off): element (j,k,lane)
    # at base_p*dof^2 + (k//2)*2C + lane*2 + k%2 for paired k, odd tail at
    # (dof^2-1)*C + lane
    lane_p = entry_block_dst % C
    base_p = entry_block_dst - lane_p
    D2 = dof * dof
    even = D2 & ~1
    off = np.where(k < even, (k >> 1) * (2 * C) + lane_p * 2 + (k & 1),
                   even * C + lane_p)
    vdst = base_p * D2 + off
    bvals[vdst] = vals
    return bptr, bcol, bvals, density


def spmv_sell(sellptr, cols, vals, nrows, x, y, *, rowbase: int = 0,
              accum: bool = False, partials=None, scal=None,
              dotslot: int = -1, dot_accum: bool = True, C: int = 64) -> None:
    nslices = sellptr.numel() - 1
    assert (nrows + C - 1) // C == nslices, \
        f"SELL structure was built with a different chunk size than C={C}"
    contrib = torch.zeros(nrows, dtype=torch.float64, device=x.device)
    for s in range(nslices):
        base = int(sellptr[s])
        L = (int(sellptr[s + 1]) - base) // C
        block = (vals[base:base + L * C].view(L, C)
                 * x[cols[base:base + L * C].long()].view(L, C)).sum(dim=0)
        hi = min(C, nrows - s * C)
        contrib[s * C:s * C + hi] = block[:hi]
    sl = slice(rowbase, rowbase + nrows)
    if accum:
        y[sl] += contrib
    else:
        y[sl] = contrib
    if scal is not None and dotslot >= 0:
        d = torch.dot(x[sl], contrib)
        scal[dotslot] = scal[dotslot] + d if dot_accum else d


def cg_prep_pt(scal) -> None:
    scal[S_PT] = 0.0


def cg_prep_rr(scal) -> None:
    scal[S_RR_PREV] = scal[S_RR].clone()


def dot(x, y, partials, scal, slot, n=None, accumulate=False) -> None:
    n = x.numel() if n is None else n
    d = torch.dot(x[:n], y[:n])
    scal[slot] = scal[slot] + d if accumulate else d


def dot2(r, w, partials, scal, n, accumulate=False) -> None:
    g = torch.dot(r[:n], r[:n])
    d = torch.dot(w[:n], r[:n])
    if accumulate:
        scal[S_GAMMA] += g
        scal[S_DELTA] += d
    else:
        scal[S_GAMMA] = g
        scal[S_DELTA] = d


def _sdiv(a, b):
    # 0/0-safe coefficient division (see kernels.hip safe_div): a converged
    # solve driven past convergence underflows the recursion residual to
    # exact 0; dividing to 0 freezes the iterate instead of NaN-poisoning it
    return a / b if float(b) != 0.0 else torch.zeros_like(a) if torch.is_tensor(a) else 0.0


def axpy_ratio(y, x, scal, num, den, sign=1.0, n=None) -> None:
    n = y.numel() if n is None else n
    a = sign * _sdiv(float(scal[num]), float(scal[den]))
    y[:n] += a * x[:n]


def daypx_ratio(y, x, scal, num, den, n=None) -> None:
    n = y.numel() if n is None else n
    b = _sdiv(float(scal[num]), float(scal[den]))
    y[:n] = b * y[:n] + x[:n]


def cg_fused_update(r, x, p, t, scal, partials, n) -> None:
    """alpha = rr/pt; update r,x; then rotate rr->rr_prev and publish the
    new (r,r) (matches k_cg_fused_update + k_cg_finalize)."""
    alpha = _sdiv(float(scal[S_RR]), float(scal[S_PT]))
    r[:n] -= alpha * t[:n]
    x[:n] += alpha * p[:n]
    scal[S_RR_PREV] = scal[S_RR].clone()
    scal[S_RR] = torch.dot(r[:n], r[:n])


def _pipelined_coeffs(scal, first: bool):
    gamma = float(scal[S_GAMMA])
    delta = float(scal[S_DELTA])
    if first:
        return 0.0, _sdiv(gamma, delta)
    beta = _sdiv(gamma, float(scal[S_GAMMA_PREV]))
    alpha = _sdiv(gamma, delta - beta * _sdiv(gamma, float(scal[S_ALPHA_PREV])))
    return beta, alpha


def pipelined_fused(z, t, p, x, r, w, q, scal, partials, n, first: bool) -> None:
    """Update the 6 vectors AND produce the next gamma/delta + rotate the
    scalar history (matches k_pipelined_fused + k_pipelined_finalize)."""
    beta, alpha = _pipelined_coeffs(scal, first)
    z[:n] = q[:n] + beta * z[:n]
    t[:n] = w[:n] + beta * t[:n]
    p[:n] = r[:n] + beta * p[:n]
    x[:n] += alpha * p[:n]
    r[:n] -= alpha * t[:n]
    w[:n] -= alpha * z[:n]
    scal[S_GAMMA_PREV] = scal[S_GAMMA].clone()
    scal[S_ALPHA_PREV] = alpha
    scal[S_GAMMA] = torch.dot(r[:n], r[:n])
    scal[S_DELTA] = torch.dot(w[:n], r[:n])


def pack_gather(sendbuf, x, idx) -> None:
    torch.index_select(x, 0, idx.long(), out=sendbuf)
