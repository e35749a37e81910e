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


@pytest.mark.parametrize("specname,g,nranks_sim", [
    ("queen", (7, 7, 9), 1),
    ("queen", (6, 6, 10), 3),
    ("poisson7", (12, 12, 12), 2),
])
def test_device_generation_matches_host(specname, g, nranks_sim):
    """On-GPU SELL generation must produce the same operator as the host
    slab generator (SpMV results equal on every simulated rank)."""
    from acg_amd.gen import STENCIL_7PT_3D, queen_like_spec, stencil_local_slab
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.ops import gpu_ops

    spec = queen_like_spec(3) if specname == "queen" else dict(STENCIL_7PT_3D)
    gx, gy, gz = g
    for rank in range(nranks_sim):
        H = stencil_local_slab(gx, gy, gz, spec, rank, nranks_sim)
        D = device_stencil_slab(gx, gy, gz, spec, rank, nranks_sim, "cuda:0")
        assert (D.nowned, D.ninterior, D.nborder, D.nghost) == \
               (H.nowned, H.ninterior, H.nborder, H.nghost)
        assert D.nnzA == H.nnzA and D.nnzO == H.nnzO
        nlocal = H.nowned + H.nghost
        x = torch.randn(nlocal, dtype=torch.float64, device="cuda")
        # host operator result
        import acg_amd.ops.torch_ref as tr

        yh = torch.zeros(H.nowned, dtype=torch.float64)
        xh = x.cpu()
        tr.spmv(torch.from_numpy(H.A_rowptr), torch.from_numpy(H.A_colidx.astype(np.int64)),
                torch.from_numpy(H.A_vals), xh, yh)
        tr.spmv(torch.from_numpy(H.O_rowptr), torch.from_numpy(H.O_colidx.astype(np.int64)),
                torch.from_numpy(H.O_vals), xh, yh, rowbase=H.ninterior, accum=True)
        yd = torch.zeros(H.nowned, dtype=torch.float64, device="cuda")
        sp, sc, sv = D.A_sell
        gpu_ops.spmv_sell(sp, sc, sv, D.nowned, x, yd)
        op, oc, ov = D.O_sell
        if D.nnzO:
            gpu_ops.spmv_sell(op, oc, ov, D.nborder, x, yd,
                              rowbase=D.ninterior, accum=True)
        torch.testing.assert_close(yd.cpu(), yh, rtol=1e-12, atol=1e-10)
        # halo patterns identical
        np.testing.assert_array_equal(D.halo.senders, H.halo.senders)
        np.testing.assert_array_equal(D.halo.recvcounts, H.halo.recvcounts)
        np.testing.assert_array_equal(np.asarray(D.halo.sendidx, dtype=np.int64),
                                      np.asarray(H.halo.sendidx, dtype=np.int64))


def test_csr_vector_fallback_path(problem):
    """Forced CSR-vector operator (use_sell=False): both solvers still
    converge and agree with the SELL/BSELL path."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(31)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    fast = CGSolverHIP(S, device="cuda:0")
    csr = CGSolverHIP(S, device="cuda:0", use_sell=False)
    assert csr.sell is None and csr.A_rowptr is not None
    x1 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    r1 = fast.solve(b, x1, maxits=400, res_rtol=1e-10)
    x2 = torch.zeros_like(x1)
    r2 = csr.solve(b, x2, maxits=400, res_rtol=1e-10)
    assert r1.converged and r2.converged
    assert abs(r1.niterations - r2.niterations) <= 2
    torch.testing.assert_close(x1[:S.nowned], x2[:S.nowned], rtol=1e-6, atol=1e-8)


def test_pipelined_graph_matches_eager(problem):
    """hipGraph-replayed pipelined CG == eager pipelined CG."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(7)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    x1 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    r1 = gpu.solve_pipelined(b, x1, maxits=300, res_rtol=1e-10, use_graph=False)
    x2 = torch.zeros_like(x1)
    r2 = gpu.solve_pipelined(b, x2, maxits=300, res_rtol=1e-10, use_graph=True)
    assert r1.converged and r2.converged
    assert r1.niterations == r2.niterations
    # graph path applies one extra (converged) update to x; both solve Ax=b
    torch.testing.assert_close(x1[:S.nowned], x2[:S.nowned], rtol=1e-6, atol=1e-8)


def test_pipelined_megafused_matches_fallback(problem):
    """Megafused single-kernel iteration == separate SpMV + update path."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(11)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    assert gpu.can_megafuse
    x1 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    r1 = gpu.solve_pipelined(b, x1, maxits=300, res_rtol=1e-10, use_graph=False,
                             megafuse=False)
    x2 = torch.zeros_like(x1)
    r2 = gpu.solve_pipelined(b, x2, maxits=300, res_rtol=1e-10, use_graph=False,
                             megafuse=True)
    assert r1.converged and r2.converged
    assert r1.niterations == r2.niterations, (r1.niterations, r2.niterations)
    torch.testing.assert_close(x1[:S.nowned], x2[:S.nowned], rtol=1e-6, atol=1e-8)


def test_no_memory_growth_across_solves(problem):
    """Repeated solves (each capturing fresh hipGraphs) must not leak."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(23)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    for _ in range(3):  # warm the pools
        gpu.solve_pipelined(b, x.clone(), maxits=10, res_rtol=0.0)
    torch.cuda.synchronize()
    base = torch.cuda.memory_allocated()
    for _ in range(10):
        gpu.solve_pipelined(b, x.clone(), maxits=10, res_rtol=0.0)
        gpu.solve(b, x.clone(), maxits=10, res_rtol=0.0)
    torch.cuda.synchronize()
    grown = torch.cuda.memory_allocated() - base
    assert grown < 32 << 20, f"leaked {grown} bytes across 20 solves"


def test_profile_mode(problem):
    """hipEvent per-op profiling produces sane per-op stats on GPU."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(21)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0", profile=True)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = gpu.solve_pipelined(b, x, maxits=50, res_rtol=0.0)
    assert "update" in res.ops
    up = res.ops["update"]
    assert up.count >= 50 and up.seconds > 0
    assert up.bytes > 0  # analytic annotation attached
    res2 = gpu.solve(b, torch.zeros_like(x), maxits=50, res_rtol=0.0)
    assert "spmvA" in res2.ops and res2.ops["spmvA"].seconds > 0


def test_bsell_matches_scalar_sell():
    """Block-SELL SpMV == scalar SELL SpMV on the device-generated system."""
    from acg_amd.gen import queen_like_spec
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.ops import gpu_ops

    for g, nranks, rank in ((13, 1, 0), (9, 2, 1)):
        S = device_stencil_slab(g, g, g, queen_like_spec(3), rank, nranks,
                                "cuda:0")
        assert S.A_bsell is not None
        x = torch.randn(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
        y1 = torch.zeros(S.nowned, dtype=torch.float64, device="cuda")
        sp, sc, sv = S.A_sell
        gpu_ops.spmv_sell(sp, sc, sv, S.nowned, x, y1)
        y2 = torch.zeros(S.nowned, dtype=torch.float64, device="cuda")
        bptr, bcol, bvals, dof = S.A_bsell
        gpu_ops.spmv_bsell(bptr, bcol, bvals, S.nowned // dof, dof, x, y2)
        torch.testing.assert_close(y1, y2, rtol=1e-13, atol=1e-11)
        # fused dot agrees too
        scal1 = gpu_ops.alloc_scalars("cuda:0")
        scal2 = gpu_ops.alloc_scalars("cuda:0")
        part = gpu_ops.alloc_partials("cuda:0")
        gpu_ops.spmv_sell(sp, sc, sv, S.nowned, x, y1, partials=part,
                          scal=scal1, dotslot=0, dot_accum=False)
        gpu_ops.spmv_bsell(bptr, bcol, bvals, S.nowned // dof, dof, x, y2,
                           partials=part, scal=scal2, dotslot=0,
                           dot_accum=False)
        torch.testing.assert_close(scal1[0], scal2[0], rtol=1e-12, atol=1e-9)


def test_device_generated_solver():
    from acg_amd.gen import queen_like_spec
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    S = device_stencil_slab(14, 14, 14, queen_like_spec(3), 0, 1, "cuda:0")
    rng = np.random.default_rng(6)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    solver = CGSolverHIP(S, device="cuda:0")
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = solver.solve_pipelined(b, x, maxits=300, res_rtol=1e-9)
    assert res.converged, res.summary()


@pytest.mark.parametrize("maxits", [1, 2, 3, 5])
def test_pipelined_small_maxits(problem, maxits):
    """Graph/lag machinery must be exact at tiny iteration counts."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(13)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    gpu = CGSolverHIP(S, device="cuda:0")
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = gpu.solve_pipelined(b, x, maxits=maxits, res_rtol=0.0)
    assert res.niterations == maxits
    res2 = gpu.solve(b, torch.zeros_like(x), maxits=maxits, res_rtol=0.0)
    assert res2.niterations == maxits


def test_convergence_iteration_counts_match_cpu(problem):
    """Detected convergence iteration must equal the CPU oracle's for
    classic and pipelined, graphs on and off."""
    from acg_amd.solvers.cpu import CGSolverCPU
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(17)
    b_np = rng.standard_normal(S.nowned)
    cpu = CGSolverCPU(S)
    xc = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    rc = cpu.solve(torch.from_numpy(b_np.copy()), xc, maxits=500, res_rtol=1e-8)
    gpu = CGSolverHIP(S, device="cuda:0")
    b = torch.from_numpy(b_np).cuda()
    for name, fn in (("classic", gpu.solve),
                     ("pipelined", gpu.solve_pipelined)):
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
        rg = fn(b, x, maxits=500, res_rtol=1e-8)
        assert rg.converged
        assert abs(rg.niterations - rc.niterations) <= 2, \
            (name, rg.niterations, rc.niterations)


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


def test_repeated_solves_deterministic_and_isolated(problem):
    """Workspace vectors and captured graphs persist across solve calls:
    repeated identical solves must be bitwise identical (deterministic
    reductions + fixed kernel geometry), interleaving classic/pipelined/
    graph variants must not leak state, and the caller's x must receive
    the solution (it lives in an internal staggered buffer)."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    rng = np.random.default_rng(23)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    solver = CGSolverHIP(S, device="cuda:0")

    def run(fn, **kw):
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64,
                        device="cuda")
        r = fn(b, x, maxits=60, res_rtol=0.0, **kw)
        assert r.niterations == 60
        return x[:S.nowned].cpu().numpy()

    p1 = run(solver.solve_pipelined)
    c1 = run(solver.solve)
    pg = run(solver.solve_pipelined, use_graph=True)
    p2 = run(solver.solve_pipelined)
    c2 = run(solver.solve)
    pg2 = run(solver.solve_pipelined, use_graph=True)
    np.testing.assert_array_equal(p1, p2)     # bitwise repeatable
    np.testing.assert_array_equal(pg, pg2)    # cached-graph replays too
    np.testing.assert_array_equal(c1, c2)
    np.testing.assert_allclose(p1, pg, rtol=1e-12, atol=1e-12)
    np.testing.assert_allclose(p1, c1, rtol=1e-5, atol=1e-7)
    assert np.all(np.isfinite(p1)) and np.abs(p1).max() > 0


def test_classic_daypx_fold_matches_unfolded(problem):
    """Serial BSELL classic CG folds daypx into the SpMV (p ping-pong);
    must match the unfolded path in iterations and solution."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    solver = CGSolverHIP(S, device="cuda:0")
    if solver.bsell is None or S.nnzO > 0:
        pytest.skip("fold needs serial BSELL matA-only")
    rng = np.random.default_rng(31)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()

    def run(**kw):
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64,
                        device="cuda")
        res = solver.solve(b, x, maxits=500, res_rtol=1e-9, **kw)
        assert res.converged, res.summary()
        return x[:S.nowned].cpu().numpy(), res

    xf, rf = run(fold_daypx=True)     # measured-negative, opt-in
    xu, ru = run()
    assert abs(rf.niterations - ru.niterations) <= 2
    np.testing.assert_allclose(xf, xu, rtol=1e-8, atol=1e-10)
    xf2, _ = run(fold_daypx=True)
    np.testing.assert_array_equal(xf, xf2)  # fold path deterministic


@pytest.mark.parametrize("method", ["solve", "solve_pipelined"])
def test_gpu_solver_survives_residual_underflow(problem, method):
    """rtol=0 driven ~1600 iterations past convergence: the recursion
    residual underflows to exact 0 (measured ~0.5x/iteration decay) and
    the device coefficients must freeze via safe_div instead of going
    0/0 = NaN (which would raise FloatingPointError and poison x)."""
    from acg_amd.solvers.hip import CGSolverHIP

    A, S = problem
    solver = CGSolverHIP(S, device="cuda:0")
    rng = np.random.default_rng(41)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = getattr(solver, method)(b, x, maxits=1600, res_rtol=0.0)
    assert res.niterations == 1600
    assert torch.isfinite(x).all()
    # the frozen iterate is the converged solution: true residual tiny
    t = torch.zeros(S.nowned, dtype=torch.float64, device="cuda")
    solver._spmv_overlapped(x, t)
    rel = float(torch.linalg.norm(b - t) / torch.linalg.norm(b))
    # classic floors at ~1e-15; pipelined's recursion drift leaves the
    # frozen iterate at ~1e-12 true residual (textbook behaviour)
    assert rel < 1e-10, rel


@pytest.mark.parametrize("solver_name", ["acg", "acg-pipelined", "acg-device", "acg-jacobi"])
def test_cli_gpu_end_to_end(tmp_path, capsys, monkeypatch, solver_name):
    """Full CLI pipeline on a real GPU: mtx file -> assembly -> extraction
    -> CGSolverHIP (all three GPU solver types) -> solution to stdout."""
    from acg_amd import cli
    from acg_amd.gen import STENCIL_5PT_2D, stencil_global
    from acg_amd.io.mtx import MtxFile, write_mtx

    A = stencil_global(24, 24, 1, STENCIL_5PT_2D)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=A.colidx.astype(np.int64),
                colidx=rows.astype(np.int64), a=A.vals)
    path = tmp_path / "p.mtx"
    write_mtx(path, m)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(path), "--solver", solver_name,
                   "--manufactured-solution", "--max-iterations", "2000",
                   "--residual-rtol", "1e-9", "-v"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    assert out.out.startswith("%%MatrixMarket matrix array real general")
    vals = np.array([float(v) for v in out.out.strip().splitlines()[2:]])
    assert len(vals) == A.n and np.isfinite(vals).all()


class _FakeCaptureComm:
    """comm.size>1 with no-op collectives and an empty halo: drives the
    DISTRIBUTED code path (graph capture + replay loop) on one GPU with
    serial semantics -- the capture/replay machinery the first real N=8
    run relies on is otherwise unexercised before that run."""

    kind = "rccl"
    size = 2
    rank = 0
    can_capture = True

    def __init__(self, device):
        self.device = device

    def allreduce_(self, t):
        return t

    def barrier(self):
        pass


@pytest.mark.parametrize("method", ["solve", "solve_pipelined"])
def test_dist_graph_capture_replay_fakecomm(method):
    """Classic + pipelined multi-GPU graph capture: the whole iteration
    (halo no-op, allreduce no-op, SpMV, fused update) is captured at k==2
    and REPLAYED; the result must match the serial eager solve."""
    import numpy as np
    import torch

    from acg_amd.gen import queen_like_spec, stencil_global
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    A = stencil_global(8, 8, 8, queen_like_spec(3))
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()

    ser = CGSolverHIP(S, comm=None, device="cuda:0")
    xs = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    rs = getattr(ser, method)(b, xs, maxits=60, res_rtol=1e-10)
    assert rs.converged

    comm = _FakeCaptureComm(torch.device("cuda", 0))
    dist = CGSolverHIP(S, comm=comm, device="cuda:0")
    xd = torch.zeros_like(xs)
    rd = getattr(dist, method)(b, xd, maxits=60, res_rtol=1e-10)
    assert rd.converged
    key = ("classic:dist" if method == "solve"
           else "pipelined:dist")
    assert not dist._graphs.get(key.split(":")[0] + ":capture_failed", False)
    assert dist._graphs.get(key) is not None, \
        f"{method}: distributed graph was not captured"
    # same Krylov process (identical arithmetic): tight agreement
    np.testing.assert_allclose(xd[:S.nowned].cpu().numpy(),
                               xs[:S.nowned].cpu().numpy(),
                               rtol=1e-9, atol=1e-11)
    # replay path must also be numerically sane standalone
    rel = rd.rnrm2 / rd.bnrm2
    assert rel < 1e-9


def test_cli_gpu_irregular_file_hybrid(tmp_path, capsys, monkeypatch):
    """File-driven irregular path end-to-end on a real GPU: power-law .mtx
    -> assembly -> auto partition -> SELL+CSR hybrid format -> solve ->
    solution matches scipy."""
    import scipy.sparse.linalg as spla

    from acg_amd import cli
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.io.mtx import MtxFile, write_mtx
    from acg_amd.solvers.hip import CGSolverHIP
    from acg_amd.part import extract_subdomains, partition_rows

    A = powerlaw_spd(20_000, mean_nnz=24, clip=48, seed=8)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows.astype(np.int64),
                colidx=A.colidx.astype(np.int64), a=A.vals)
    path = tmp_path / "pl.mtx"
    write_mtx(path, m)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(path), "--solver", "acg", "--max-iterations", "3000",
                   "--residual-rtol", "1e-11", "-v"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    # the irregular matrix must have gone through a non-plain-SELL format
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    probe = CGSolverHIP(S, device="cuda:0")
    assert (probe.hybrid is not None or probe.sell_perm is not None)
    vals = np.array([float(v) for v in out.out.strip().splitlines()[2:]])
    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), np.ones(A.n))
    np.testing.assert_allclose(vals, x_ref, rtol=1e-6, atol=1e-8)


def test_dist_graph_capture_replay_megafused_fakecomm():
    """The MEGAFUSED pipelined distributed path (what config-5 poisson
    runs at N=8: narrow rows => megafuse auto-on) with ping-pong graphs:
    capture + replay must match the serial solve."""
    import numpy as np
    import torch

    from acg_amd.gen import STENCIL_7PT_3D, stencil_global
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    A = stencil_global(16, 16, 16, STENCIL_7PT_3D)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()

    ser = CGSolverHIP(S, comm=None, device="cuda:0")
    assert ser.megafuse_auto  # 7-pt rows are narrow: mega path engages
    xs = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    rs = ser.solve_pipelined(b, xs, maxits=200, res_rtol=1e-10)
    assert rs.converged

    comm = _FakeCaptureComm(torch.device("cuda", 0))
    dist = CGSolverHIP(S, comm=comm, device="cuda:0")
    xd = torch.zeros_like(xs)
    rd = dist.solve_pipelined(b, xd, maxits=200, res_rtol=1e-10)
    assert rd.converged
    assert not dist._graphs.get("pipelined_mega:capture_failed", False)
    graphs = dist._graphs.get("pipelined_mega:dist")
    assert graphs is not None and all(g is not None for g in graphs), \
        "megafused distributed ping-pong graphs were not captured"
    np.testing.assert_allclose(xd[:S.nowned].cpu().numpy(),
                               xs[:S.nowned].cpu().numpy(),
                               rtol=1e-9, atol=1e-11)


def test_bsell_dof4_detected_and_correct():
    """dof=4 dense-block matrices take the Block-SELL path (dispatch
    covers 2/3/4; auto-detect previously only tried 3 and 2)."""
    import numpy as np
    import scipy.sparse.linalg as spla
    import torch

    from acg_amd.gen import STENCIL_27PT_3D, stencil_global
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    spec = dict(STENCIL_27PT_3D)
    spec["dof"] = 4
    M = np.array([[1.0, .3, .2, .1], [.3, 1.0, .3, .2],
                  [.2, .3, 1.0, .3], [.1, .2, .3, 1.0]])
    spec["offblock"] = M
    spec["diagblock"] = 60.0 * np.eye(4) + 0.5 * (M - np.eye(4))
    A = stencil_global(8, 8, 8, spec)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverHIP(S, device="cuda:0")
    assert solver.bsell is not None and solver.bsell[3] == 4
    rng = np.random.default_rng(0)
    b_np = rng.standard_normal(S.nowned)
    b = torch.from_numpy(b_np).cuda()
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = solver.solve(b, x, maxits=400, res_rtol=1e-11)
    assert res.converged
    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_np)
    np.testing.assert_allclose(x[:S.nowned].cpu().numpy(), x_ref,
                               rtol=1e-7, atol=1e-9)
