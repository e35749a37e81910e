"""Plain-PyTorch fp64 reference implementations of every HIP op.

Used (a) as the oracle in kernel numerics tests (GPU kernel vs torch fp64,
same semantics), (b) as the compute engine of the CPU solver path
(reference analog: acg/cg.c host solver).  Mirrors the device-scalar
convention: coefficients live in a small fp64 "scal" tensor.
"""

from __future__ import annotations

import torch

# keep slot numbering identical to kernels.hip
S_RR, S_PT, S_RR_PREV, S_BNRM2 = 0, 1, 2, 3
S_GAMMA, S_DELTA, S_GAMMA_PREV, S_ALPHA_PREV = 4, 5, 6, 7
S_NSLOTS = 8


def alloc_scalars(device="cpu") -> torch.Tensor:
    return torch.zeros(S_NSLOTS, dtype=torch.float64, device=device)


def spmv(rowptr, colidx, vals, x, y, *, rowbase: int = 0, accum: bool = False,
         scal=None, dotslot: int = -1) -> None:
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
        scal[dotslot] += torch.dot(x[sl], contrib)


def cg_prep_pt(scal) -> None:
    scal[S_PT] = 0.0


def cg_prep_rr(scal) -> None:
    scal[S_RR_PREV] = scal[S_RR].clone()
    scal[S_RR] = 0.0


def dot(x, y, scal, slot, n=None, zero_first=True) -> None:
    n = x.numel() if n is None else n
    if zero_first:
        scal[slot] = 0.0
    scal[slot] += torch.dot(x[:n], y[:n])


def dot2(r, w, scal, n) -> None:
    scal[S_GAMMA] += torch.dot(r[:n], r[:n])
    scal[S_DELTA] += torch.dot(w[:n], r[:n])


def axpy_ratio(y, x, scal, num, den, sign=1.0, n=None) -> None:
    n = y.numel() if n is None else n
    a = sign * float(scal[num]) / float(scal[den])
    y[:n] += a * x[:n]


def daypx_ratio(y, x, scal, num, den, n=None) -> None:
    n = y.numel() if n is None else n
    b = float(scal[num]) / float(scal[den])
    y[:n] = b * y[:n] + x[:n]


def cg_fused_update(r, x, p, t, scal, n) -> None:
    alpha = float(scal[S_RR_PREV]) / float(scal[S_PT])
    r[:n] -= alpha * t[:n]
    x[:n] += alpha * p[:n]
    scal[S_RR] += torch.dot(r[:n], r[:n])


def _pipelined_coeffs(scal, first: bool):
    gamma = float(scal[S_GAMMA])
    delta = float(scal[S_DELTA])
    if first:
        return 0.0, gamma / delta
    beta = gamma / float(scal[S_GAMMA_PREV])
    alpha = gamma / (delta - beta * gamma / float(scal[S_ALPHA_PREV]))
    return beta, alpha


def pipelined_fused(z, t, p, x, r, w, q, scal, n, first: bool) -> None:
    beta, alpha = _pipelined_coeffs(scal, first)
    z[:n] = q[:n] + beta * z[:n]
    t[:n] = w[:n] + beta * t[:n]
    p[:n] = r[:n] + beta * p[:n]
    x[:n] += alpha * p[:n]
    r[:n] -= alpha * t[:n]
    w[:n] -= alpha * z[:n]


def pipelined_reset(scal, first: bool) -> None:
    _, alpha = _pipelined_coeffs(scal, first)
    scal[S_GAMMA_PREV] = scal[S_GAMMA].clone()
    scal[S_ALPHA_PREV] = alpha
    scal[S_GAMMA] = 0.0
    scal[S_DELTA] = 0.0


def pack_gather(sendbuf, x, idx) -> None:
    torch.index_select(x, 0, idx.long(), out=sendbuf)
