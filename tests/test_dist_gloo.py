"""Multi-process distributed tests on CPU (gloo, world_size=2): the halo
exchange + distributed CG must reproduce the serial solution.  This covers
the construction-correctness of the RCCL path (identical code, different
backend) without GPUs."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from acg_amd.gen import STENCIL_27PT_3D, queen_like_spec, stencil_global


def _worker(rank, world, port, fn_name, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        from acg_amd.dist.comm import Comm

        comm = Comm("gloo")
        result = globals()[fn_name](comm)
        q.put((rank, "ok", result))
        comm.finalize()
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "err", traceback.format_exc()))
        raise


def _run_dist(fn_name, world=2, port=29600):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn_name, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return results


# -- worker bodies (module-level for spawn picklability) -------------------

def _body_halo(comm):
    from acg_amd.dist.halo import HaloExchange
    from acg_amd.gen import stencil_local_slab

    S = stencil_local_slab(6, 6, 8, STENCIL_27PT_3D, comm.rank, comm.size)
    # fill owned with global ids, exchange, check ghosts
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    x[:S.nowned] = torch.from_numpy(S.owned_global.astype(np.float64))
    hx = HaloExchange(S.halo, S.nowned, "cpu", comm)
    hx.exchange(x)
    got = x[S.nowned:].numpy()
    want = S.ghost_global.astype(np.float64)
    np.testing.assert_array_equal(got, want)
    return True


def _body_cg(comm):
    from acg_amd.gen import stencil_local_slab
    from acg_amd.solvers.cpu import CGSolverCPU

    spec = queen_like_spec(3)
    S = stencil_local_slab(5, 5, 8, spec, comm.rank, comm.size)
    # global RHS so every rank agrees
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(S.n_global)
    b = torch.from_numpy(b_global[S.owned_global])
    solver = CGSolverCPU(S, comm=comm)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    res = solver.solve(b, x, maxits=500, res_rtol=1e-10)
    assert res.converged, res.summary()
    return (S.owned_global, x[:S.nowned].numpy(), res.niterations)


def _body_cg_pipelined(comm):
    from acg_amd.gen import stencil_local_slab
    from acg_amd.solvers.cpu import CGSolverCPU

    spec = queen_like_spec(3)
    S = stencil_local_slab(5, 5, 8, spec, comm.rank, comm.size)
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(S.n_global)
    b = torch.from_numpy(b_global[S.owned_global])
    solver = CGSolverCPU(S, comm=comm)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    res = solver.solve_pipelined(b, x, maxits=500, res_rtol=1e-10)
    assert res.converged, res.summary()
    return (S.owned_global, x[:S.nowned].numpy(), res.niterations)


def _body_cg_ws4(comm):
    """4-rank pipelined CG on a generic (non-slab) partition via the full
    extract_subdomains path — covers multi-neighbour halos."""
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.cpu import CGSolverCPU

    A = stencil_global(6, 6, 6, STENCIL_27PT_3D)
    part = partition_rows(A, comm.size, method="rgb", seed=1)
    S = extract_subdomains(A, part, comm.size)[comm.rank]
    rng = np.random.default_rng(7)
    b_global = rng.standard_normal(A.n)
    b = torch.from_numpy(b_global[S.owned_global])
    solver = CGSolverCPU(S, comm=comm)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    res = solver.solve_pipelined(b, x, maxits=800, res_rtol=1e-10)
    assert res.converged, res.summary()
    return (S.owned_global, x[:S.nowned].numpy(), res.niterations)


def test_distributed_cg_ws4_rgb():
    results = _run_dist("_body_cg_ws4", world=4, port=29604)
    A = stencil_global(6, 6, 6, STENCIL_27PT_3D)
    rng = np.random.default_rng(7)
    b_global = rng.standard_normal(A.n)
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_global)
    for rank, (owned_global, xloc, nit) in results.items():
        np.testing.assert_allclose(xloc, x_ref[owned_global], rtol=1e-6, atol=1e-8)


def _body_cg_jacobi(comm):
    """2-rank Jacobi-PCG must reproduce the serial solution."""
    from acg_amd.gen import stencil_local_slab
    from acg_amd.solvers.cpu import CGSolverCPU

    spec = queen_like_spec(3)
    S = stencil_local_slab(5, 5, 8, spec, comm.rank, comm.size)
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(S.n_global)
    b = torch.from_numpy(b_global[S.owned_global])
    solver = CGSolverCPU(S, comm=comm)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    res = solver.solve_jacobi(b, x, maxits=800, res_rtol=1e-10)
    assert res.converged, res.summary()
    return (S.owned_global, x[:S.nowned].numpy(), res.niterations)


def test_distributed_jacobi_matches_serial():
    results = _run_dist("_body_cg_jacobi", world=2, port=29608)
    spec = queen_like_spec(3)
    A = stencil_global(5, 5, 8, spec)
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(A.n)
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_global)
    for rank, (owned_global, xloc, nit) in results.items():
        np.testing.assert_allclose(xloc, x_ref[owned_global], rtol=1e-6,
                                   atol=1e-8)


def _body_jacobi_scale(comm):
    """Distributed jacobi scaling: the setup halo exchange of s plus a
    plain pipelined solve on the scaled system reproduces the serial
    solution after back-transform."""
    from acg_amd.gen import stencil_local_slab
    from acg_amd.solvers.cpu import CGSolverCPU
    from acg_amd.solvers.precond import jacobi_scale_system

    spec = queen_like_spec(3)
    S = stencil_local_slab(5, 5, 8, spec, comm.rank, comm.size)
    Ss, s = jacobi_scale_system(S, comm)
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(S.n_global)
    b = torch.from_numpy(b_global[S.owned_global] * s)
    solver = CGSolverCPU(Ss, comm=comm)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    res = solver.solve_pipelined(b, x, maxits=800, res_rtol=1e-10)
    assert res.converged, res.summary()
    return (S.owned_global, x[:S.nowned].numpy() * s, res.niterations)


def test_distributed_jacobi_scale_matches_serial():
    results = _run_dist("_body_jacobi_scale", world=2, port=29609)
    spec = queen_like_spec(3)
    A = stencil_global(5, 5, 8, spec)
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(A.n)
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_global)
    for rank, (owned_global, xloc, nit) in results.items():
        np.testing.assert_allclose(xloc, x_ref[owned_global], rtol=1e-6,
                                   atol=1e-8)


def _body_collective_error(comm):
    """Reference acgerrmpi semantics (error.c:149): one rank's failure must
    raise on ALL ranks instead of deadlocking."""
    from acg_amd.utils.errors import AcgError, ErrCode, collective_raise

    exc = None
    if comm.rank == 1:
        exc = AcgError(ErrCode.INVALID_VALUE, "rank 1 exploded")
    try:
        collective_raise(comm, exc)
    except AcgError as e:
        return ("raised", int(e.code))
    return ("no-raise", 0)


def test_collective_error_agreement():
    results = _run_dist("_body_collective_error", world=2, port=29605)
    for rank, (status, code) in results.items():
        assert status == "raised", f"rank {rank} did not raise"
        assert code == int(__import__("acg_amd.utils.errors",
                                      fromlist=["ErrCode"]).ErrCode.INVALID_VALUE)


def test_halo_exchange_gloo_ws2():
    _run_dist("_body_halo", world=2, port=29601)


def _body_halo_audit(comm):
    """Collective halo dry-run audit (dist.verify) on a 4-way rgb
    partition: passes clean, and a corrupted pattern raises on ALL ranks."""
    from acg_amd.dist.verify import verify_halo
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.utils.errors import AcgError

    A = stencil_global(6, 6, 6, STENCIL_27PT_3D)
    part = partition_rows(A, comm.size, method="rgb", seed=1)
    S = extract_subdomains(A, part, comm.size)[comm.rank]
    verify_halo(S, comm)  # must pass
    # corrupt ONE rank's ghost ordering: every rank must raise
    if comm.rank == 1 and S.nghost >= 2:
        S.ghost_global[:2] = S.ghost_global[:2][::-1].copy()
    try:
        verify_halo(S, comm)
        return ("no-raise",)
    except AcgError:
        return ("raised",)


def _body_chunked_scatter(comm):
    """Streaming field-by-field scatter (comm.scatter_systems) must equal
    direct local extraction; tiny CHUNK_BYTES exercises the chunk loop
    (reference MPI_Send64 chunking, graph.c:1529-1893)."""
    from acg_amd.dist.comm import Comm
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.part.subdomain import SubdomainExtractor

    Comm.CHUNK_BYTES = 64  # force many chunks per array
    A = stencil_global(6, 6, 6, STENCIL_27PT_3D)
    part = partition_rows(A, comm.size, method="rgb", seed=1)
    if comm.rank == 0:
        ex = SubdomainExtractor(A, part, comm.size)
        S = comm.scatter_systems(lambda p: ex.build(p))
        rng = np.random.default_rng(3)
        bg = rng.standard_normal(A.n)
        b = comm.scatter_rows(lambda p: bg[ex.owned_globals[p]])
    else:
        S = comm.scatter_systems(None)
        b = comm.scatter_rows(None)
    # oracle: local extraction of the same part
    ref = extract_subdomains(A, part, comm.size,
                             only_parts=[comm.rank])[comm.rank]
    ok = True
    for fld in ("nowned", "ninterior", "nborder", "nghost", "n_global"):
        ok &= getattr(S, fld) == getattr(ref, fld)
    import numpy as _np

    for fld in ("A_rowptr", "A_colidx", "A_vals", "O_rowptr", "O_colidx",
                "O_vals", "owned_global", "ghost_global"):
        a, r = getattr(S, fld), getattr(ref, fld)
        ok &= a.dtype == r.dtype and _np.array_equal(a, r)
    for fld in ("senders", "recvcounts", "rdispls", "recipients",
                "sendcounts", "sdispls", "sendidx"):
        ok &= _np.array_equal(getattr(S.halo, fld), getattr(ref.halo, fld))
    rng = np.random.default_rng(3)
    bg = rng.standard_normal(A.n)
    ok &= _np.array_equal(b, bg[ref.owned_global])
    return ("ok" if ok else "mismatch",)


def test_chunked_scatter_ws4():
    results = _run_dist("_body_chunked_scatter", world=4, port=29607)
    for rank, (status,) in results.items():
        assert status == "ok", f"rank {rank}: {status}"


def test_halo_audit_ws4():
    results = _run_dist("_body_halo_audit", world=4, port=29606)
    for rank, (status,) in results.items():
        assert status == "raised", f"rank {rank}: corrupted halo not detected"


@pytest.mark.parametrize("body,port", [("_body_cg", 29602),
                                       ("_body_cg_pipelined", 29603)])
def test_distributed_cg_matches_serial(body, port):
    results = _run_dist(body, world=2, port=port)
    # serial oracle
    spec = queen_like_spec(3)
    A = stencil_global(5, 5, 8, spec)
    rng = np.random.default_rng(42)
    b_global = rng.standard_normal(A.n)
    import scipy.sparse.linalg as spla

    X = A.to_scipy_full()
    x_ref = spla.spsolve(X.tocsc(), b_global)
    for rank, (owned_global, xloc, nit) in results.items():
        np.testing.assert_allclose(xloc, x_ref[owned_global], rtol=1e-6, atol=1e-8)


def _body_scatter_poison(comm):
    """Root-side factory failure mid-scatter must raise on EVERY rank
    (matching-size poison headers), not hang the unsent ranks."""
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.part.subdomain import SubdomainExtractor
    from acg_amd.utils.errors import AcgError

    A = stencil_global(6, 6, 6, STENCIL_27PT_3D)
    part = partition_rows(A, comm.size, method="rgb", seed=1)
    if comm.rank == 0:
        ex = SubdomainExtractor(A, part, comm.size)

        def factory(p):
            if p == 1:
                raise RuntimeError("boom at part 1")
            return ex.build(p)

        try:
            comm.scatter_systems(factory)
            return ("no-raise",)
        except RuntimeError:
            return ("raised",)
    try:
        comm.scatter_systems(None)
        return ("no-raise",)
    except AcgError:
        return ("raised",)


def test_scatter_poison_ws4():
    results = _run_dist("_body_scatter_poison", world=4, port=29610)
    for rank, (status,) in results.items():
        assert status == "raised", f"rank {rank}: {status}"
