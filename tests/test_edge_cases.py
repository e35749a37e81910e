"""Edge cases and robustness (CPU)."""

import numpy as np
import pytest
import torch

from acg_amd.core.symcsr import SymCSRMatrix
from acg_amd.gen import STENCIL_5PT_2D, stencil_global
from acg_amd.part import extract_subdomains, partition_rows
from acg_amd.solvers.cpu import CGSolverCPU
from acg_amd.utils.errors import AcgError


def test_tiny_1x1():
    A = SymCSRMatrix.from_coo(1, [0], [0], [4.0])
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    x = torch.zeros(1, dtype=torch.float64)
    res = solver.solve(torch.tensor([8.0], dtype=torch.float64), x,
                       maxits=5, res_rtol=1e-12)
    assert res.converged and abs(float(x[0]) - 2.0) < 1e-12


def test_identity_converges_one_iter():
    n = 50
    A = SymCSRMatrix.from_coo(n, np.arange(n), np.arange(n), np.ones(n))
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    b = torch.randn(n, dtype=torch.float64)
    x = torch.zeros(n, dtype=torch.float64)
    res = solver.solve(b.clone(), x, maxits=5, res_rtol=1e-12)
    assert res.converged and res.niterations <= 1
    torch.testing.assert_close(x, b)


def test_x0_initial_guess():
    A = stencil_global(10, 10, 1, STENCIL_5PT_2D)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    rng = np.random.default_rng(0)
    xsol = rng.standard_normal(A.n)
    b = torch.from_numpy(A.dsymv(xsol))
    # exact initial guess -> converged with 0 iterations
    x = torch.from_numpy(xsol.copy())
    res = solver.solve(b.clone(), x, maxits=10, res_rtol=1e-10)
    assert res.converged and res.niterations == 0


def test_diff_stopping_criteria():
    A = stencil_global(12, 12, 1, STENCIL_5PT_2D)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    b = torch.ones(A.n, dtype=torch.float64)
    x = torch.zeros(A.n, dtype=torch.float64)
    res = solver.solve(b, x, maxits=5000, res_rtol=0.0, diff_rtol=1e-12)
    assert res.converged
    assert res.niterations < 5000


def test_epsilon_shift():
    A = stencil_global(8, 8, 1, STENCIL_5PT_2D)
    part = partition_rows(A, 1)
    S0 = extract_subdomains(A, part, 1, eps=0.0)[0]
    S1 = extract_subdomains(A, part, 1, eps=2.5)[0]
    # diagonal entries shifted by eps
    import scipy.sparse as sp

    A0 = sp.csr_matrix((S0.A_vals, S0.A_colidx.astype(np.int64), S0.A_rowptr))
    A1 = sp.csr_matrix((S1.A_vals, S1.A_colidx.astype(np.int64), S1.A_rowptr))
    d = (A1 - A0).diagonal()
    np.testing.assert_allclose(d, 2.5)


def test_partition_more_parts_than_structure():
    A = stencil_global(4, 4, 1, STENCIL_5PT_2D)  # 16 rows
    part = partition_rows(A, 8)
    systems = extract_subdomains(A, part, 8)
    assert sum(s.nowned for s in systems) == A.n


def test_nonsquare_rejected():
    from acg_amd.io.mtx import MtxFile

    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=3, ncols=4, nnz=1,
                rowidx=np.array([0]), colidx=np.array([1]), a=np.array([1.0]))
    with pytest.raises(AcgError):
        SymCSRMatrix.from_mtxfile(m)


def test_general_symmetry_rejected():
    from acg_amd.io.mtx import MtxFile

    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="general", nrows=3, ncols=3, nnz=1,
                rowidx=np.array([0]), colidx=np.array([1]), a=np.array([1.0]))
    with pytest.raises(AcgError):
        SymCSRMatrix.from_mtxfile(m)


def test_cpu_solver_survives_residual_underflow():
    """Driven far past convergence with rtol=0 (fixed-iteration benches),
    the recursion residual underflows to exact 0: coefficients must
    freeze (0) instead of becoming 0/0 = NaN."""
    import torch

    from acg_amd.gen import STENCIL_5PT_2D, stencil_global
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.cpu import CGSolverCPU

    A = stencil_global(12, 12, 1, STENCIL_5PT_2D)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(S.nowned))
    for method in ("solve", "solve_pipelined"):
        solver = CGSolverCPU(S)
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
        res = getattr(solver, method)(b, x, maxits=3000, res_rtol=0.0)
        assert res.niterations == 3000, (method, res.niterations)
        assert torch.isfinite(x).all(), method
