"""Independent oracle solver (reference role: acg/cgpetsc.c PETSc KSPCG).

PETSc is not available in this image; scipy.sparse.linalg.cg plays the
same role -- an independently implemented CG on the identical system, used
to cross-check both correctness (solution/iterations) and convergence
behaviour.  Distributed runs gather the system to root, solve, and
scatter the solution back (the oracle is a correctness baseline, not a
performance path)."""

from __future__ import annotations

import time

import numpy as np
import scipy.sparse as sp

from .base import SolveResult


def _local_to_coo(S):
    """Local rows (global numbering) as COO triplets."""
    rows_l = np.repeat(np.arange(S.nowned, dtype=np.int64),
                       np.diff(S.A_rowptr))
    gi = S.owned_global[rows_l]
    local_cols = np.concatenate([S.owned_global, S.ghost_global]) \
        if S.nghost else S.owned_global
    gj = local_cols[S.A_colidx.astype(np.int64)]
    vals = S.A_vals
    if S.nnzO:
        rows_o = np.repeat(np.arange(S.nborder, dtype=np.int64) + S.ninterior,
                           np.diff(S.O_rowptr))
        gi = np.concatenate([gi, S.owned_global[rows_o]])
        gj = np.concatenate([gj, local_cols[S.O_colidx.astype(np.int64)]])
        vals = np.concatenate([vals, S.O_vals])
    return gi, gj, vals


def solve_scipy(S, comm, b_local: np.ndarray, x0_local: np.ndarray,
                maxits: int = 100, res_rtol: float = 1e-9,
                res_atol: float = 0.0, pipelined: bool = False):
    """Solve with scipy CG; returns (SolveResult, x_local)."""
    pieces = comm.gather_object((_local_to_coo(S), S.owned_global, b_local,
                                 x0_local)) if comm else \
        [((_local_to_coo(S)), S.owned_global, b_local, x0_local)]
    res = SolveResult(solver="scipy-cg", maxits=maxits, res_rtol=res_rtol,
                      res_atol=res_atol, nranks=comm.size if comm else 1)
    x_global = None
    if pieces is not None:  # root
        n = S.n_global
        gi = np.concatenate([p[0][0] for p in pieces])
        gj = np.concatenate([p[0][1] for p in pieces])
        vv = np.concatenate([p[0][2] for p in pieces])
        A = sp.csr_matrix((vv, (gi, gj)), shape=(n, n))
        b = np.empty(n)
        x0 = np.empty(n)
        for (_, og, bl, xl) in pieces:
            b[og] = bl
            x0[og] = xl
        import scipy.sparse.linalg as spla

        it_count = [0]

        def cb(_xk):
            it_count[0] += 1

        t0 = time.perf_counter()
        bn = float(np.linalg.norm(b))
        rtol = max(res_rtol, res_atol / bn if bn > 0 else 0.0)
        x_global, info = spla.cg(A, b, x0=x0, rtol=rtol, atol=res_atol,
                                 maxiter=maxits, callback=cb)
        res.tsolve = time.perf_counter() - t0
        res.niterations = it_count[0]
        res.converged = info == 0
        res.bnrm2 = bn
        r = b - A @ x_global
        res.rnrm2 = float(np.linalg.norm(r))
        r0 = b - A @ x0
        res.r0nrm2 = float(np.linalg.norm(r0))
    if comm:
        res = comm.bcast_object(res)
        xl = comm.scatter_object(
            [x_global[p[1]] for p in pieces] if pieces is not None else None)
    else:
        xl = x_global[S.owned_global]
    return res, np.ascontiguousarray(xl)
