"""Host (CPU) CG solvers — the reference oracle path.

Reference: acg/cg.c — acgsolver_solve / acgsolver_solvempi: textbook CG
with a halo exchange before each SpMV and an all-reduce per dot
(cg.c:408-649).  Unlike the GPU solvers it also supports the
diff_atol/diff_rtol stopping criteria.  solve_pipelined mirrors the
Ghysels–Vanroose recurrences used by the GPU pipelined solver so the
algorithm is testable without a GPU.

Runs on torch CPU tensors (gloo for multi-rank tests); the same code
drives the ops in acg_amd.ops.torch_ref that the HIP kernels are tested
against.
"""

from __future__ import annotations

import math
import time

import numpy as np
import torch

from ..dist.halo import HaloExchange
from ..ops import torch_ref as ops
from ..part.subdomain import LocalSystem
from .base import SolveResult, cg_flops_per_iter


class CGSolverCPU:
    """Distributed classic / pipelined CG on host tensors."""

    def __init__(self, local: LocalSystem, comm=None, device="cpu"):
        self.local = local
        self.comm = comm
        self.device = torch.device(device)
        L = local
        self.A_rowptr = torch.from_numpy(np.ascontiguousarray(L.A_rowptr)).to(self.device)
        self.A_colidx = torch.from_numpy(np.ascontiguousarray(L.A_colidx)).to(self.device)
        self.A_vals = torch.from_numpy(np.ascontiguousarray(L.A_vals)).to(self.device)
        self.O_rowptr = torch.from_numpy(np.ascontiguousarray(L.O_rowptr)).to(self.device)
        self.O_colidx = torch.from_numpy(np.ascontiguousarray(L.O_colidx)).to(self.device)
        self.O_vals = torch.from_numpy(np.ascontiguousarray(L.O_vals)).to(self.device)
        self.halo = HaloExchange(L.halo, L.nowned, self.device, comm)
        self.n = L.nowned
        self.nlocal = L.nowned + L.nghost

    # -- helpers ----------------------------------------------------------

    def _allreduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.comm is not None:
            self.comm.allreduce_(t)
        return t

    def _dot(self, a: torch.Tensor, b: torch.Tensor) -> float:
        v = torch.dot(a[: self.n], b[: self.n]).reshape(1)
        return float(self._allreduce(v)[0])

    def _spmv(self, xfull: torch.Tensor, y: torch.Tensor) -> None:
        """y = A_local x (halo exchange + split matA/matO SpMV)."""
        self.halo.exchange(xfull)
        ops.spmv(self.A_rowptr, self.A_colidx, self.A_vals, xfull, y)
        ops.spmv(self.O_rowptr, self.O_colidx, self.O_vals, xfull, y,
                 rowbase=self.local.ninterior, accum=True)

    def _vec(self, nghost: bool = False) -> torch.Tensor:
        return torch.zeros(self.nlocal if nghost else self.n,
                           dtype=torch.float64, device=self.device)

    # -- classic CG (reference acgsolver_solvempi, cg.c:408-649) ----------

    def solve(self, b: torch.Tensor, x: torch.Tensor, maxits: int = 100,
              res_atol: float = 0.0, res_rtol: float = 1e-9,
              diff_atol: float = 0.0, diff_rtol: float = 0.0) -> SolveResult:
        res = SolveResult(solver="cg-cpu", maxits=maxits, res_atol=res_atol,
                          res_rtol=res_rtol,
                          nranks=self.comm.size if self.comm else 1)
        n = self.n
        t0 = time.perf_counter()
        bnrm2 = math.sqrt(self._dot(b, b))
        res.bnrm2 = bnrm2
        r = self._vec()
        t = self._vec()
        p = self._vec(nghost=True)
        # r0 = b - A x0
        self._spmv(x, t)
        r[:] = b[:n] - t
        p[:n] = r
        rr = self._dot(r, r)
        res.r0nrm2 = math.sqrt(rr)
        rtol2 = max(res_atol, res_rtol * bnrm2) ** 2
        dtol = max(diff_atol, diff_rtol * bnrm2)
        if rr <= rtol2 and (dtol == 0.0):
            res.converged = True
            res.rnrm2 = math.sqrt(rr)
            res.tsolve = time.perf_counter() - t0
            return res
        for k in range(maxits):
            self._spmv(p, t)
            pt = self._dot(p, t)
            alpha = rr / pt if pt != 0.0 else 0.0  # 0/0 at underflow: freeze
            r -= alpha * t
            x[:n] += alpha * p[:n]
            rr_new = self._dot(r, r)
            res.niterations = k + 1
            converged = False
            if rtol2 > 0 and rr_new <= rtol2:
                converged = True
            if dtol > 0:
                dx = abs(alpha) * math.sqrt(self._dot(p, p))
                if dx <= dtol:
                    converged = True
            if converged:
                res.converged = True
                res.rnrm2 = math.sqrt(rr_new)
                break
            beta = rr_new / rr if rr != 0.0 else 0.0
            p[:n] = r + beta * p[:n]
            rr = rr_new
            res.rnrm2 = math.sqrt(rr_new)
        res.tsolve = time.perf_counter() - t0
        nnz_full = self.local.nnzA + self.local.nnzO
        res.nflops = res.niterations * cg_flops_per_iter(nnz_full, n)
        res.halo_bytes_sent = self.halo.bytes_sent
        res.halo_msgs_sent = self.halo.nmsgs_sent
        return res

    # -- Jacobi-preconditioned CG (beyond reference: aCG runs
    # unpreconditioned CG only, PCNONE even in its PETSc oracle,
    # cgpetsc.c:181-193.  Diagonal preconditioning is the standard
    # production lever for time-to-solution on ill-conditioned SPD
    # systems; opt-in so every parity bench stays unpreconditioned) -----

    def _diag_inv(self) -> torch.Tensor:
        d = torch.zeros(self.n, dtype=torch.float64, device=self.device)
        rowptr = self.A_rowptr
        rows = torch.repeat_interleave(
            torch.arange(self.n, dtype=torch.int64, device=self.device),
            rowptr[1:] - rowptr[:-1])
        mask = rows == self.A_colidx.long()
        d[rows[mask]] = self.A_vals[mask]
        if (d == 0).any():
            from ..utils.errors import AcgError, ErrCode

            raise AcgError(ErrCode.INVALID_VALUE,
                           "jacobi preconditioner needs a full diagonal")
        return 1.0 / d

    def solve_jacobi(self, b: torch.Tensor, x: torch.Tensor,
                     maxits: int = 100, res_atol: float = 0.0,
                     res_rtol: float = 1e-9) -> SolveResult:
        """Jacobi-PCG: z = D^-1 r, alpha = (r,z)/(p,t), beta = rz'/rz.
        Convergence is tested on the TRUE residual 2-norm (same semantics
        as the unpreconditioned solvers)."""
        res = SolveResult(solver="cg-jacobi-cpu", maxits=maxits,
                          res_atol=res_atol, res_rtol=res_rtol,
                          nranks=self.comm.size if self.comm else 1)
        n = self.n
        t0 = time.perf_counter()
        dinv = self._diag_inv()
        bnrm2 = math.sqrt(self._dot(b, b))
        res.bnrm2 = bnrm2
        r = self._vec()
        t = self._vec()
        p = self._vec(nghost=True)
        self._spmv(x, t)
        r[:] = b[:n] - t
        z = dinv * r
        p[:n] = z
        rz = self._dot(r, z)
        rr = self._dot(r, r)
        res.r0nrm2 = math.sqrt(rr)
        rtol2 = max(res_atol, res_rtol * bnrm2) ** 2
        if rtol2 > 0 and rr <= rtol2:
            res.converged = True
            res.rnrm2 = math.sqrt(rr)
            res.tsolve = time.perf_counter() - t0
            return res
        for k in range(maxits):
            self._spmv(p, t)
            pt = self._dot(p, t)
            alpha = rz / pt if pt != 0.0 else 0.0
            r -= alpha * t
            x[:n] += alpha * p[:n]
            z = dinv * r
            rz_new = self._dot(r, z)
            rr = self._dot(r, r)
            res.niterations = k + 1
            res.rnrm2 = math.sqrt(max(rr, 0.0))
            if rtol2 > 0 and rr <= rtol2:
                res.converged = True
                break
            beta = rz_new / rz if rz != 0.0 else 0.0
            p[:n] = z + beta * p[:n]
            rz = rz_new
        res.tsolve = time.perf_counter() - t0
        nnz_full = self.local.nnzA + self.local.nnzO
        res.nflops = res.niterations * (cg_flops_per_iter(nnz_full, n) + 3.0 * n)
        res.halo_bytes_sent = self.halo.bytes_sent
        res.halo_msgs_sent = self.halo.nmsgs_sent
        return res

    # -- pipelined CG (reference acgsolverhip_solve_pipelined, §3.3) ------

    def solve_pipelined(self, b: torch.Tensor, x: torch.Tensor, maxits: int = 100,
                        res_atol: float = 0.0, res_rtol: float = 1e-9) -> SolveResult:
        res = SolveResult(solver="cg-pipelined-cpu", maxits=maxits,
                          res_atol=res_atol, res_rtol=res_rtol,
                          nranks=self.comm.size if self.comm else 1)
        n = self.n
        t0 = time.perf_counter()
        bnrm2 = math.sqrt(self._dot(b, b))
        res.bnrm2 = bnrm2
        tmp = self._vec()
        r = self._vec(nghost=True)
        w = self._vec(nghost=True)
        q = self._vec()
        z = self._vec()
        t = self._vec()
        p = self._vec()
        self._spmv(x, tmp)
        r[:n] = b[:n] - tmp
        self._spmv(r, w)  # w = A r  (writes w[:n])
        rtol2 = max(res_atol, res_rtol * bnrm2) ** 2
        gamma_prev = alpha_prev = None
        for k in range(maxits):
            gd = torch.stack([torch.dot(r[:n], r[:n]), torch.dot(w[:n], r[:n])])
            self._allreduce(gd)
            gamma, delta = float(gd[0]), float(gd[1])
            if k == 0:
                res.r0nrm2 = math.sqrt(gamma)
            res.rnrm2 = math.sqrt(gamma)
            if rtol2 > 0 and gamma <= rtol2:
                res.converged = True
                res.niterations = k
                break
            self._spmv(w, q)  # q = A w (halo on w inside)
            if k == 0:
                beta = 0.0
                alpha = gamma / delta if delta != 0.0 else 0.0
            else:
                beta = gamma / gamma_prev if gamma_prev != 0.0 else 0.0
                den = delta - beta * (gamma / alpha_prev if alpha_prev != 0.0 else 0.0)
                alpha = gamma / den if den != 0.0 else 0.0
            z[:] = q + beta * z
            t[:] = w[:n] + beta * t
            p[:] = r[:n] + beta * p
            x[:n] += alpha * p
            r[:n] -= alpha * t
            w[:n] -= alpha * z
            gamma_prev, alpha_prev = gamma, alpha
            res.niterations = k + 1
        res.tsolve = time.perf_counter() - t0
        nnz_full = self.local.nnzA + self.local.nnzO
        res.nflops = res.niterations * (cg_flops_per_iter(nnz_full, n) + 8.0 * n)
        res.halo_bytes_sent = self.halo.bytes_sent
        res.halo_msgs_sent = self.halo.nmsgs_sent
        return res
