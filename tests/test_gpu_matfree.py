"""Matrix-free stencil operator (beyond reference): k_stencil_spmv /
k_stencil_pipe must reproduce the assembled SELL operator and solver
exactly (same matA/matO split, same fused-dot protocol), with zero
matrix bytes read.  Oracle = the assembled path, itself validated
against the host generator and scipy."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _spec(name):
    from acg_amd.gen import STENCIL_7PT_3D, STENCIL_27PT_3D

    return dict(STENCIL_7PT_3D if name == "7pt" else STENCIL_27PT_3D)


@pytest.mark.parametrize("stencil", ["7pt", "27pt"])
@pytest.mark.parametrize("nranks", [1, 2])
def test_stencil_spmv_matches_assembled(stencil, nranks):
    from acg_amd.gen import stencil_global, stencil_local_slab
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.ops import gpu_ops

    spec = _spec(stencil)
    gx, gy, gz = 7, 6, 5 * nranks
    A = stencil_global(gx, gy, gz, spec)
    rng = np.random.default_rng(0)
    xg = rng.standard_normal(A.n)
    yg = A.dsymv(xg)
    xg_t = torch.from_numpy(xg).cuda()
    for rank in range(nranks):
        H = stencil_local_slab(gx, gy, gz, spec, rank, nranks)
        S = device_stencil_slab(gx, gy, gz, spec, rank, nranks, "cuda:0")
        assert S.mf_tables is not None
        xl = torch.empty(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
        xl[:S.nowned] = xg_t[torch.from_numpy(H.owned_global).cuda()]
        if S.nghost:
            xl[S.nowned:] = xg_t[torch.from_numpy(H.ghost_global).cuda()]
        y = torch.zeros(S.nowned, dtype=torch.float64, device="cuda")
        scal = gpu_ops.alloc_scalars("cuda:0")
        partials = gpu_ops.alloc_partials("cuda:0")
        fuse = dict(partials=partials, scal=scal, dotslot=gpu_ops.S_PT)
        gpu_ops.stencil_spmv(S.mf_tables, S.nowned, 0, xl, y,
                             mato=False, dot_accum=False, **fuse)
        if S.nnzO:
            gpu_ops.stencil_spmv(S.mf_tables, S.nborder, S.ninterior, xl, y,
                                 mato=True, **fuse)
        np.testing.assert_allclose(y.cpu().numpy(), yg[H.owned_global],
                                   rtol=1e-13, atol=1e-12)
        pt = float(scal[gpu_ops.S_PT])
        pt_want = float(np.dot(xg[H.owned_global], yg[H.owned_global]))
        assert abs(pt - pt_want) < 1e-9 * max(abs(pt_want), 1.0)


def _solve(matfree, method, operator=True, megafuse=None, G=20):
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    S = device_stencil_slab(G, G, G, _spec("7pt"), 0, 1, "cuda:0",
                            operator=operator)
    solver = CGSolverHIP(S, device="cuda:0", matfree=matfree)
    rng = np.random.default_rng(5)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    kw = {} if method == "solve" else {"megafuse": megafuse}
    res = getattr(solver, method)(b, x, maxits=400, res_rtol=1e-10, **kw)
    assert res.converged, res.summary()
    return x[:S.nowned].cpu().numpy(), res


def test_matfree_pipelined_matches_assembled():
    x_mf, r_mf = _solve(True, "solve_pipelined")
    x_as, r_as = _solve(False, "solve_pipelined")
    assert abs(r_mf.niterations - r_as.niterations) <= 2
    np.testing.assert_allclose(x_mf, x_as, rtol=1e-8, atol=1e-10)


def test_matfree_classic_matches_assembled():
    x_mf, r_mf = _solve(True, "solve")
    x_as, r_as = _solve(False, "solve")
    assert abs(r_mf.niterations - r_as.niterations) <= 2
    np.testing.assert_allclose(x_mf, x_as, rtol=1e-8, atol=1e-10)


def test_matfree_27pt_pipelined_matches_assembled():
    """27-pt has no column-walk fast path: this exercises the GENERIC
    megafused k_stencil_pipe end-to-end (w7 is None)."""
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    S = device_stencil_slab(14, 14, 14, _spec("27pt"), 0, 1, "cuda:0")
    assert S.mf_tables["w7"] is None
    rng = np.random.default_rng(9)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()

    def run(matfree):
        solver = CGSolverHIP(S, device="cuda:0", matfree=matfree)
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64,
                        device="cuda")
        res = solver.solve_pipelined(b, x, maxits=400, res_rtol=1e-10)
        assert res.converged, res.summary()
        return x[:S.nowned].cpu().numpy(), res

    x_mf, r_mf = run(True)
    x_as, r_as = run(False)
    assert abs(r_mf.niterations - r_as.niterations) <= 2
    np.testing.assert_allclose(x_mf, x_as, rtol=1e-8, atol=1e-10)


def test_matfree_megafused_matches_nomega():
    """Default matfree pipelined runs the megafused k_stencil_pipe (+ graph
    replay); it must agree with the non-megafused matfree path."""
    x_mega, r_mega = _solve(True, "solve_pipelined", megafuse=True)
    x_no, r_no = _solve(True, "solve_pipelined", megafuse=False)
    assert abs(r_mega.niterations - r_no.niterations) <= 2
    np.testing.assert_allclose(x_mega, x_no, rtol=1e-8, atol=1e-10)


def test_matfree_without_assembled_operator():
    """operator=False: no SELL arrays exist at all; nnz stats still there."""
    from acg_amd.gen.device_slab import device_stencil_slab

    S = device_stencil_slab(16, 16, 16, _spec("7pt"), 0, 1, "cuda:0",
                            operator=False)
    assert S.A_sell is None and S.nnzA > 0
    x_mf, res = _solve(True, "solve_pipelined", operator=False, G=16)
    x_as, _ = _solve(False, "solve_pipelined", G=16)
    np.testing.assert_allclose(x_mf, x_as, rtol=1e-8, atol=1e-10)


@pytest.mark.parametrize("nranks", [1, 2])
def test_column_walk_matches_generic(nranks):
    """7-pt matA dispatches to the z-column-walk kernel; forcing the
    generic kernel (w7=None) must give bitwise-comparable results."""
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.ops import gpu_ops

    for rank in range(nranks):
        S = device_stencil_slab(9, 7, 6 * nranks, _spec("7pt"), rank, nranks,
                                "cuda:0")
        mf = S.mf_tables
        assert mf["w7"] is not None
        rng = np.random.default_rng(rank)
        xl = torch.from_numpy(
            rng.standard_normal(S.nowned + S.nghost)).cuda()
        y7 = torch.zeros(S.nowned, dtype=torch.float64, device="cuda")
        yg = torch.zeros(S.nowned, dtype=torch.float64, device="cuda")
        gpu_ops.stencil_spmv(mf, S.nowned, 0, xl, y7, mato=False)
        gpu_ops.stencil_spmv({**mf, "w7": None}, S.nowned, 0, xl, yg,
                             mato=False)
        np.testing.assert_allclose(y7.cpu().numpy(), yg.cpu().numpy(),
                                   rtol=1e-14, atol=1e-13)


@pytest.mark.parametrize("matfree", [False, True])
def test_megafused_large_matA_leaves_matO_blocks(matfree):
    """Regression: at >=3.9M owned rows the matA pass used to claim all
    MAXG partials blocks, so a matO pass (any multi-rank system) launched
    with grid 0 -> 'invalid configuration argument'.  Rank 0 of 2 at
    256^3 crosses the threshold (8.4M rows) with a live matO pass."""
    from acg_amd.gen import STENCIL_7PT_3D
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    S = device_stencil_slab(256, 256, 256, dict(STENCIL_7PT_3D), 0, 2,
                            "cuda:0", operator=not matfree)
    assert S.nnzO > 0
    rng = np.random.default_rng(3)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()

    def run(megafuse):
        solver = CGSolverHIP(S, device="cuda:0", matfree=matfree)
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64,
                        device="cuda")
        solver.solve_pipelined(b, x, maxits=25, res_rtol=0.0,
                               megafuse=megafuse)
        return x[:S.nowned].cpu().numpy()

    if matfree:
        np.testing.assert_allclose(run(True), run(False), rtol=1e-8,
                                   atol=1e-10)
    else:
        run(True)  # operator=True build is the expensive part; mega only


def test_matfree_requires_stencil_system():
    from acg_amd.gen import queen_like_spec
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    S = device_stencil_slab(5, 5, 5, queen_like_spec(3), 0, 1, "cuda:0")
    with pytest.raises(ValueError):
        CGSolverHIP(S, device="cuda:0", matfree=True)  # dof=3: no mf tables
