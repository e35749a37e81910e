"""Synthetic irregular SPD matrices: power-law degrees + geometric locality.

The reference's merge-path SpMV exists to load-balance power-law rows
(/root/reference/acg/cg-kernels-hip.hip:348-1175); its benchmark inputs
(SuiteSparse Queen_4147 etc.) are irregular FEM/graph matrices.  This
module generates matrices of that *shape class* without network access:

- per-row off-diagonal degree ~ Pareto(alpha-1) clipped to ``clip``
  (heavy tail: a few rows with thousands of nonzeros, median far below
  the mean -- the regime where fixed-lane kernels lose),
- neighbour offsets ~ Laplace(0, locality): mostly-banded structure with
  long-range tails, like meshes with irregular refinement,
- symmetric by construction (entries canonicalised to the upper triangle,
  duplicates summed),
- STRICTLY diagonally dominant with positive diagonal
  (diag_i = 1 + sum_j |a_ij|), hence symmetric positive definite --
  a rigorous SPD guarantee, not a heuristic shift.
"""

from __future__ import annotations

import numpy as np

from ..core.symcsr import SymCSRMatrix


def powerlaw_spd(n: int, mean_nnz: float = 32.0, alpha: float = 2.2,
                 locality: float | None = None, clip: int = 8192,
                 seed: int = 0) -> SymCSRMatrix:
    """Power-law-degree SPD matrix with ~``mean_nnz`` nonzeros/row (full).

    ``alpha``: degree-distribution tail exponent (smaller = heavier tail;
    2.1-2.5 matches social/FEM-refinement graphs).  ``locality``: Laplace
    scale of the column offsets (default n/64).  ``clip``: max per-row
    generated degree (keeps the worst row finite, like real matrices).
    """
    if locality is None:
        locality = max(n / 64.0, 8.0)
    rng = np.random.default_rng(seed)
    # directed draw count per row; symmetrisation roughly doubles the
    # off-diagonal count, the diagonal adds 1
    half = max((mean_nnz - 1.0) / 2.0, 1.0)
    # Pareto(a) with scale m has mean m*a/(a-1); we draw m*(1+pareto(a))
    a = alpha - 1.0
    dmin = half * (a - 1.0) / a if a > 1.0 else half
    deg = np.minimum(dmin * (1.0 + rng.pareto(a, n)), float(clip))
    deg = np.maximum(deg, 1.0).astype(np.int64)
    M = int(deg.sum())
    i = np.repeat(np.arange(n, dtype=np.int64), deg)
    offs = np.trunc(rng.laplace(0.0, locality, M)).astype(np.int64)
    offs[offs == 0] = 1
    j = (i + offs) % n
    keep = i != j  # offsets that wrapped a full period
    i, j = i[keep], j[keep]
    v = -np.abs(rng.standard_normal(len(i))) - 0.05
    A0 = SymCSRMatrix.from_coo(n, i, j, v)
    # full-row |offdiag| sums from the packed upper triangle
    rows_u = np.repeat(np.arange(n, dtype=np.int64), np.diff(A0.rowptr))
    s = (np.bincount(rows_u, weights=np.abs(A0.vals), minlength=n)
         + np.bincount(A0.colidx, weights=np.abs(A0.vals), minlength=n))
    return SymCSRMatrix.from_coo(
        n,
        np.concatenate([rows_u, np.arange(n, dtype=np.int64)]),
        np.concatenate([A0.colidx, np.arange(n, dtype=np.int64)]),
        np.concatenate([A0.vals, 1.0 + s]))


def degree_stats(A: SymCSRMatrix) -> dict:
    """Full-row nonzero-count stats (tail diagnostics for reports)."""
    rows_u = np.repeat(np.arange(A.n, dtype=np.int64), np.diff(A.rowptr))
    offd = rows_u != A.colidx
    deg = (np.bincount(rows_u[offd], minlength=A.n)
           + np.bincount(A.colidx[offd], minlength=A.n) + 1)
    return {
        "n": A.n,
        "nnz_full": int(deg.sum()),
        "mean": float(deg.mean()),
        "median": float(np.median(deg)),
        "p99": float(np.percentile(deg, 99)),
        "max": int(deg.max()),
    }
