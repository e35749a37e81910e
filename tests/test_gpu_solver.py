"""GPU CG solver end-to-end tests (single GPU): convergence, parity with
the CPU oracle solver, manufactured solutions."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def problem():
    from acg_amd.gen import queen_like_spec, stencil_global
    from acg_amd.part import extract_subdomains, partition_rows

    A = stencil_global(10, 10, 10, queen_like_spec(3))  # 3000 rows
    part = partition_rows(A, 1)
    S = extract_subdomains(A, part, 1)[0]
    return A, S


def test_hip_cg_matches_cpu_iterations(problem):
    from acg_amd.solvers.cpu import CGSolverCPU
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(0)
    b_np = rng.standard_normal(A.n)
    b = torch.from_numpy(b_np[S.owned_global])
    cpu = CGSolverCPU(S)
    xc = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    rc = cpu.solve(b.clone(), xc, maxits=400, res_rtol=1e-9)
    gpu = CGSolverHIP(S, device="cuda:0")
    xg = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    rg = gpu.solve(b.to("cuda"), xg, maxits=400, res_rtol=1e-9)
    assert rc.converged and rg.converged
    # same algorithm, same fp64: iteration counts must agree to +-2
    assert abs(rc.niterations - rg.niterations) <= 2, (rc.niterations, rg.niterations)
    np.testing.assert_allclose(xg[:S.nowned].cpu().numpy(),
                               xc[:S.nowned].numpy(), rtol=1e-7, atol=1e-9)


def test_hip_manufactured_solution(problem):
    """b := A x_sol on CPU, solve on GPU, error norm must be tiny
    (reference --manufactured-solution, acg-hip.c:1940-2087)."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(1)
    xsol = rng.standard_normal(A.n)
    xsol /= np.linalg.norm(xsol)
    b_np = A.dsymv(xsol)
    b = torch.from_numpy(b_np[S.owned_global]).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = gpu.solve(b, x, maxits=600, res_rtol=1e-11)
    assert res.converged
    err = np.linalg.norm(x[:S.nowned].cpu().numpy() - xsol[S.owned_global])
    assert err < 1e-8, err


def test_hip_pipelined_matches_classic(problem):
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(2)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    x1 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    r1 = gpu.solve(b, x1, maxits=400, res_rtol=1e-10)
    x2 = torch.zeros_like(x1)
    r2 = gpu.solve_pipelined(b, x2, maxits=400, res_rtol=1e-10)
    assert r1.converged and r2.converged
    assert abs(r1.niterations - r2.niterations) <= 3
    torch.testing.assert_close(x1[:S.nowned], x2[:S.nowned], rtol=1e-6, atol=1e-8)


def test_hip_device_monolithic_cg(problem):
    """Whole-solve cooperative kernel vs the host-driven classic solver."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(4)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    assert gpu.sell is not None
    x1 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    r1 = gpu.solve(b, x1, maxits=400, res_rtol=1e-10)
    x2 = torch.zeros_like(x1)
    r2 = gpu.solve_device(b, x2, maxits=400, res_rtol=1e-10)
    assert r1.converged and r2.converged, (r1.summary(), r2.summary())
    assert abs(r1.niterations - r2.niterations) <= 2
    torch.testing.assert_close(x1[:S.nowned], x2[:S.nowned], rtol=1e-7, atol=1e-9)


def test_slab_generated_gpu_solve():
    """Flagship path: slab-generated Queen-like system, single GPU."""
    from acg_amd.gen import queen_like_spec, stencil_local_slab
    from acg_amd.solvers.hip import CGSolverHIP

    S = stencil_local_slab(16, 16, 16, queen_like_spec(3), 0, 1)
    rng = np.random.default_rng(3)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    solver = CGSolverHIP(S, device="cuda:0")
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = solver.solve(b, x, maxits=300, res_rtol=1e-9)
    assert res.converged, res.summary()
