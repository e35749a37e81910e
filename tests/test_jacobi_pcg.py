"""Jacobi-preconditioned CG (beyond reference, opt-in): correctness vs
scipy and the iteration-count win on ill-conditioned SPD systems."""

import numpy as np
import pytest
import torch

from acg_amd.gen.irregular import powerlaw_spd
from acg_amd.part import extract_subdomains, partition_rows
from acg_amd.solvers.cpu import CGSolverCPU


def _ill_conditioned(n=6000, seed=3):
    # heavy clip => Gershgorin diag spread => kappa ~ max row sum
    return powerlaw_spd(n, mean_nnz=20, clip=2000, seed=seed)


def test_cpu_jacobi_matches_scipy_and_cuts_iterations():
    import scipy.sparse.linalg as spla

    A = _ill_conditioned()
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    rng = np.random.default_rng(0)
    b_np = rng.standard_normal(S.nowned)
    b = torch.from_numpy(b_np)
    xj = torch.zeros(S.nowned, dtype=torch.float64)
    rj = solver.solve_jacobi(b, xj, maxits=3000, res_rtol=1e-10)
    assert rj.converged
    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_np)
    np.testing.assert_allclose(xj.numpy(), x_ref, rtol=1e-6, atol=1e-8)
    # plain CG on the same system needs SEVERAL TIMES more iterations
    xc = torch.zeros(S.nowned, dtype=torch.float64)
    rc = solver.solve(b, xc, maxits=3000, res_rtol=1e-10)
    assert rj.niterations * 3 < rc.niterations, \
        (rj.niterations, rc.niterations)


def test_cpu_jacobi_distributed_matches_serial():
    # structure-level: jacobi on a 1-part extraction equals the serial
    # result (multi-rank transport is the same halo/allreduce machinery
    # every other solver uses)
    A = _ill_conditioned(3000, seed=5)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    rng = np.random.default_rng(1)
    b = torch.from_numpy(rng.standard_normal(S.nowned))
    x1 = torch.zeros(S.nowned, dtype=torch.float64)
    r1 = solver.solve_jacobi(b, x1, maxits=2000, res_rtol=1e-9)
    assert r1.converged
    # true residual honours the reported tolerance semantics
    rtrue = np.linalg.norm(b.numpy() - A.to_scipy_full() @ x1.numpy())
    assert rtrue <= 1.05e-9 * r1.bnrm2


def test_cli_cpu_jacobi(tmp_path, monkeypatch, capsys):
    from acg_amd import cli
    from acg_amd.io.mtx import MtxFile, write_mtx

    A = _ill_conditioned(1200, seed=7)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    p = tmp_path / "A.mtx"
    write_mtx(p, m)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(p), "--solver", "cpu-jacobi",
                   "--manufactured-solution", "--max-iterations", "2000",
                   "--residual-rtol", "1e-9", "-q"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    assert "manufactured solution" in out.err


@pytest.mark.gpu
def test_gpu_jacobi_matches_cpu_and_cuts_iterations():
    from acg_amd.solvers.hip import CGSolverHIP

    A = _ill_conditioned()
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b_np = rng.standard_normal(S.nowned)
    gpu = CGSolverHIP(S, device="cuda:0")
    b = torch.from_numpy(b_np).cuda()
    xg = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    rg = gpu.solve_jacobi(b, xg, maxits=3000, res_rtol=1e-10)
    assert rg.converged
    cpu = CGSolverCPU(S)
    xc = torch.zeros(S.nowned, dtype=torch.float64)
    rc_ = cpu.solve_jacobi(torch.from_numpy(b_np), xc, maxits=3000,
                           res_rtol=1e-10)
    # same algorithm: iteration counts within a couple of each other
    assert abs(rg.niterations - rc_.niterations) <= 3, \
        (rg.niterations, rc_.niterations)
    np.testing.assert_allclose(xg[:S.nowned].cpu().numpy(), xc.numpy(),
                               rtol=1e-6, atol=1e-8)
    # and beats plain classic on iterations
    xp = torch.zeros_like(xg)
    rp = gpu.solve(b, xp, maxits=3000, res_rtol=1e-10)
    assert rg.niterations * 3 < rp.niterations


@pytest.mark.gpu
def test_cli_gpu_jacobi(tmp_path, monkeypatch, capsys):
    from acg_amd import cli
    from acg_amd.io.mtx import MtxFile, write_mtx

    A = _ill_conditioned(2000, seed=9)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    p = tmp_path / "A.mtx"
    write_mtx(p, m)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(p), "--solver", "acg-jacobi",
                   "--manufactured-solution", "--max-iterations", "3000",
                   "--residual-rtol", "1e-9", "-q"])
    out = capsys.readouterr()
    assert rc == 0, out.err


# ---- jacobi SCALING (composes preconditioning with every solver) ----

def test_jacobi_scale_pipelined_cpu_matches_scipy():
    """Scaled system + plain PIPELINED CG == preconditioned solve: same
    iteration-count class as explicit PCG, back-transformed solution
    matches scipy."""
    import scipy.sparse.linalg as spla

    from acg_amd.solvers.precond import jacobi_scale_system

    A = _ill_conditioned()
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b_np = rng.standard_normal(S.nowned)
    Ss, s = jacobi_scale_system(S)
    # scaled diagonal is exactly 1
    rows = np.repeat(np.arange(Ss.nowned), np.diff(Ss.A_rowptr))
    dm = rows == Ss.A_colidx
    np.testing.assert_allclose(Ss.A_vals[dm], 1.0, rtol=1e-14)
    solver = CGSolverCPU(Ss)
    bs = torch.from_numpy(b_np * s)
    xs = torch.zeros(Ss.nowned, dtype=torch.float64)
    r = solver.solve_pipelined(bs, xs, maxits=3000, res_rtol=1e-10)
    assert r.converged
    # iteration-count class of explicit PCG (same spectrum)
    plain = CGSolverCPU(S)
    xj = torch.zeros(S.nowned, dtype=torch.float64)
    rj = plain.solve_jacobi(torch.from_numpy(b_np), xj, maxits=3000,
                            res_rtol=1e-10)
    assert r.niterations <= rj.niterations * 2
    x = xs.numpy() * s
    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_np)
    np.testing.assert_allclose(x, x_ref, rtol=1e-6, atol=1e-8)


def test_cli_jacobi_scale_pipelined(tmp_path, monkeypatch, capsys):
    from acg_amd import cli
    from acg_amd.io.mtx import MtxFile, write_mtx

    A = _ill_conditioned(1500, seed=11)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    p = tmp_path / "A.mtx"
    write_mtx(p, m)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(p), "--solver", "cpu-pipelined", "--jacobi-scale",
                   "--max-iterations", "2000", "--residual-rtol", "1e-10"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    # output is the UNSCALED solution
    got = np.array([float(v) for v in out.out.strip().splitlines()[2:]])
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), np.ones(A.n))
    np.testing.assert_allclose(got, x_ref, rtol=1e-5, atol=1e-7)


@pytest.mark.gpu
def test_gpu_jacobi_scale_pipelined_matches_explicit_pcg():
    from acg_amd.solvers.hip import CGSolverHIP
    from acg_amd.solvers.precond import jacobi_scale_system

    A = _ill_conditioned()
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b_np = rng.standard_normal(S.nowned)
    Ss, s = jacobi_scale_system(S)
    gpu = CGSolverHIP(Ss, device="cuda:0")
    bs = torch.from_numpy(b_np * s).cuda()
    xs = torch.zeros(Ss.nowned + Ss.nghost, dtype=torch.float64,
                     device="cuda")
    r = gpu.solve_pipelined(bs, xs, maxits=3000, res_rtol=1e-10)
    assert r.converged
    x = xs[:S.nowned].cpu().numpy() * s
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_np)
    np.testing.assert_allclose(x, x_ref, rtol=1e-6, atol=1e-8)
    # explicit-M PCG iteration class
    plain = CGSolverHIP(S, device="cuda:0")
    xj = torch.zeros_like(xs)
    rj = plain.solve_jacobi(torch.from_numpy(b_np).cuda(), xj, maxits=3000,
                            res_rtol=1e-10)
    assert r.niterations <= rj.niterations * 2


def test_cli_jacobi_scale_with_manufactured(tmp_path, monkeypatch, capsys):
    """--jacobi-scale + --manufactured-solution: b is built from the
    UNSCALED A, scaling happens after scatter, the reported error norm is
    against the back-transformed solution."""
    from acg_amd import cli
    from acg_amd.io.mtx import MtxFile, write_mtx

    A = _ill_conditioned(1000, seed=13)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    p = tmp_path / "A.mtx"
    write_mtx(p, m)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(p), "--solver", "cpu", "--jacobi-scale",
                   "--manufactured-solution", "--max-iterations", "2000",
                   "--residual-rtol", "1e-10", "-q"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    line = next(l for l in out.err.splitlines() if "manufactured" in l)
    enorm = float(line.split("=")[1].split("(")[0])
    assert enorm < 1e-6, line


def test_jacobi_error_paths():
    from acg_amd.solvers.precond import jacobi_scale_system
    from acg_amd.core.symcsr import SymCSRMatrix
    from acg_amd.utils.errors import AcgError

    # zero diagonal -> clean error (both transform and explicit PCG)
    A = SymCSRMatrix.from_coo(3, [0, 1, 0], [0, 1, 2], [1.0, 1.0, 0.5])
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    with pytest.raises(AcgError):
        jacobi_scale_system(S)
    solver = CGSolverCPU(S)
    with pytest.raises(AcgError):
        solver.solve_jacobi(torch.ones(3, dtype=torch.float64),
                            torch.zeros(3, dtype=torch.float64), maxits=5)
    # ghosts without a communicator -> clean error
    B = powerlaw_spd(600, mean_nnz=10, seed=1)
    part = partition_rows(B, 2, method="block")
    S2 = extract_subdomains(B, part, 2)[0]
    assert S2.nghost > 0
    with pytest.raises(AcgError):
        jacobi_scale_system(S2, comm=None)
