"""Tensor-level wrappers around the gfx950 HIP kernels (_acg_kernels.so).

Raw pointers + the caller's current HIP stream are passed to the
extension; all type/shape checks happen here.  If the extension is
missing on a machine with a GPU, import fails loudly (no eager fallback).
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


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _chk(t: torch.Tensor, dtype, name: str):
    assert t.is_cuda and t.dtype == dtype and t.is_contiguous(), \
        f"{name}: need contiguous cuda {dtype}, got {t.device} {t.dtype} contig={t.is_contiguous()}"


def alloc_scalars(device) -> torch.Tensor:
    return torch.zeros(S_NSLOTS, dtype=torch.float64, device=device)


def pick_lanes(mean_nnz_per_row: float) -> int:
    """Lanes-per-row heuristic for the CSR vector kernel (CDNA4: wave=64).

    Measured on MI355X (Queen-shaped, ~80 nnz/row): 16 lanes (≈5 nnz/lane)
    beats 64 lanes (≈1.2 nnz/lane) by ~25% — deep-enough per-lane runs
    amortize the shuffle reduction and keep loads contiguous."""
    for lanes in (4, 8, 16, 32):
        if mean_nnz_per_row <= lanes * 6.0:
            return lanes
    return 64


def spmv(rowptr: torch.Tensor, colidx: torch.Tensor, vals: torch.Tensor,
         x: torch.Tensor, y: torch.Tensor, *, rowbase: int = 0,
         lanes: int = 16, accum: bool = False,
         scal: torch.Tensor | None = None, dotslot: int = -1) -> None:
    """y[rowbase:rowbase+nrows] (=|+=) A x, optionally fusing dot(x,y)."""
    nrows = rowptr.numel() - 1
    if nrows <= 0:
        return
    _chk(vals, torch.float64, "vals")
    _chk(x, torch.float64, "x")
    _chk(y, torch.float64, "y")
    assert rowptr.dtype == torch.int64
    col64 = 1 if colidx.dtype == torch.int64 else 0
    fuse = scal is not None and dotslot >= 0
    K.spmv(nrows, rowbase, rowptr.data_ptr(), colidx.data_ptr(), col64,
           vals.data_ptr(), x.data_ptr(), y.data_ptr(), lanes, accum, fuse,
           scal.data_ptr() if fuse else 0, dotslot, _stream())


def zero_scalars(scal: torch.Tensor, i0: int = 0, count: int | None = None) -> None:
    K.zero_scalars(scal.data_ptr(), i0, count if count is not None else scal.numel() - i0,
                   _stream())


def cg_prep_pt(scal: torch.Tensor) -> None:
    K.cg_prep_pt(scal.data_ptr(), _stream())


def cg_prep_rr(scal: torch.Tensor) -> None:
    K.cg_prep_rr(scal.data_ptr(), _stream())


def dot(x: torch.Tensor, y: torch.Tensor, scal: torch.Tensor, slot: int,
        n: int | None = None, zero_first: bool = True) -> None:
    n = x.numel() if n is None else n
    if zero_first:
        K.zero_scalars(scal.data_ptr(), slot, 1, _stream())
    K.dot(x.data_ptr(), y.data_ptr(), n, scal.data_ptr(), slot, _stream())


def dot2(r: torch.Tensor, w: torch.Tensor, scal: torch.Tensor, n: int) -> None:
    """gamma += (r,r), delta += (w,r); slots must be pre-zeroed."""
    K.dot2(r.data_ptr(), w.data_ptr(), n, scal.data_ptr(), _stream())


def axpy_ratio(y: torch.Tensor, x: torch.Tensor, scal: torch.Tensor,
               num: int, den: int, sign: float = 1.0, n: int | None = None) -> None:
    K.axpy_ratio(y.data_ptr(), x.data_ptr(), y.numel() if n is None else n,
                 scal.data_ptr(), num, den, sign, _stream())


def daypx_ratio(y: torch.Tensor, x: torch.Tensor, scal: torch.Tensor,
                num: int, den: int, n: int | None = None) -> None:
    K.daypx_ratio(y.data_ptr(), x.data_ptr(), y.numel() if n is None else n,
                  scal.data_ptr(), num, den, _stream())


def cg_fused_update(r: torch.Tensor, x: torch.Tensor, p: torch.Tensor,
                    t: torch.Tensor, scal: torch.Tensor, n: int) -> None:
    K.cg_fused_update(r.data_ptr(), x.data_ptr(), p.data_ptr(), t.data_ptr(),
                      n, scal.data_ptr(), _stream())


def pipelined_fused(z, t, p, x, r, w, q, scal: torch.Tensor, n: int, first: bool) -> None:
    K.pipelined_fused(z.data_ptr(), t.data_ptr(), p.data_ptr(), x.data_ptr(),
                      r.data_ptr(), w.data_ptr(), q.data_ptr(), n,
                      scal.data_ptr(), 1 if first else 0, _stream())


def pipelined_reset(scal: torch.Tensor, first: bool) -> None:
    K.pipelined_reset(scal.data_ptr(), 1 if first else 0, _stream())


def pack_gather(sendbuf: torch.Tensor, x: torch.Tensor, idx: torch.Tensor) -> None:
    idx64 = 1 if idx.dtype == torch.int64 else 0
    K.pack_gather(sendbuf.data_ptr(), x.data_ptr(), idx.data_ptr(), idx64,
                  sendbuf.numel(), _stream())
