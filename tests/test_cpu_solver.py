"""End-to-end CPU CG tests (reference test strategy: SURVEY.md §4 --
manufactured solutions + independent scipy oracle on 5-pt Poisson)."""

import numpy as np
import pytest
import torch

from acg_amd.gen import STENCIL_5PT_2D, STENCIL_27PT_3D, queen_like_spec, stencil_global
from acg_amd.part import extract_subdomains, partition_rows
from acg_amd.solvers import CGSolverCPU


def _build(nx=32, ny=32, nparts=1, spec=STENCIL_5PT_2D, gz=1, method="block"):
    A = stencil_global(nx, ny, gz, spec)
    part = partition_rows(A, nparts, method=method)
    systems = extract_subdomains(A, part, nparts)
    return A, systems


def test_baseline_config1_poisson128():
    """BASELINE.json config 1: 5-pt 2D Poisson 128x128, classic CG, CPU,
    1 rank (the plumbing check, verbatim)."""
    A, systems = _build(128, 128, 1)
    S = systems[0]
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(A.n))
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    solver = CGSolverCPU(S)
    res = solver.solve(b, x, maxits=2000, res_rtol=1e-9)
    assert res.converged, res.summary()
    assert res.rnrm2 <= 1e-9 * res.bnrm2 * 1.01


def test_poisson_cg_serial_vs_scipy():
    A, systems = _build(32, 32, 1)
    S = systems[0]
    n = S.nowned
    rng = np.random.default_rng(0)
    xsol = rng.standard_normal(A.n)
    b_global = A.dsymv(xsol)
    b = torch.from_numpy(b_global[S.owned_global])
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    solver = CGSolverCPU(S, comm=None)
    res = solver.solve(b, x, maxits=2000, res_rtol=1e-10)
    assert res.converged, res.summary()
    err = np.linalg.norm(x[:n].numpy() - xsol[S.owned_global]) / np.linalg.norm(xsol)
    assert err < 1e-7, err


def test_poisson_cg_pipelined_matches_classic():
    A, systems = _build(24, 24, 1)
    S = systems[0]
    rng = np.random.default_rng(1)
    b_np = rng.standard_normal(A.n)
    b = torch.from_numpy(b_np[S.owned_global])
    solver = CGSolverCPU(S, comm=None)
    x1 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    r1 = solver.solve(b.clone(), x1, maxits=500, res_rtol=1e-10)
    x2 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    r2 = solver.solve_pipelined(b.clone(), x2, maxits=500, res_rtol=1e-10)
    assert r1.converged and r2.converged
    # pipelined CG is algebraically equivalent; allow fp jitter
    assert abs(r1.niterations - r2.niterations) <= 3
    np.testing.assert_allclose(x1[:S.nowned].numpy(), x2[:S.nowned].numpy(),
                               rtol=1e-6, atol=1e-8)


@pytest.mark.parametrize("method", ["block", "rgb"])
def test_partitioned_serial_consistency(method):
    """Multi-part extraction run serially (all parts in one process) must
    reproduce the global SpMV exactly."""
    A, systems = _build(16, 16, 4, method=method)
    n = A.n
    rng = np.random.default_rng(2)
    xg = rng.standard_normal(n)
    yg = A.dsymv(xg)
    # emulate the halo by hand: fill ghost tails from the global vector
    for S in systems:
        xl = np.concatenate([xg[S.owned_global], xg[S.ghost_global]])
        import acg_amd.ops.torch_ref as ops

        xt = torch.from_numpy(xl)
        yt = torch.zeros(S.nowned, dtype=torch.float64)
        ops.spmv(torch.from_numpy(S.A_rowptr), torch.from_numpy(S.A_colidx),
                 torch.from_numpy(S.A_vals), xt, yt)
        ops.spmv(torch.from_numpy(S.O_rowptr), torch.from_numpy(S.O_colidx),
                 torch.from_numpy(S.O_vals), xt, yt,
                 rowbase=S.ninterior, accum=True)
        np.testing.assert_allclose(yt.numpy(), yg[S.owned_global], rtol=1e-12, atol=1e-12)


def test_subdomain_send_recv_pairing():
    """Sender block ordering must match receiver ghost ordering."""
    A, systems = _build(16, 16, 4)
    for S in systems:
        h = S.halo
        for qi, q in enumerate(h.recipients):
            R = systems[q]
            # R's ghosts owned by S.rank
            gmask = np.isin(R.ghost_global, S.owned_global)
            gset = R.ghost_global[gmask]
            owners = np.array([systems[S.rank].owned_global[i]
                               for i in h.sendidx[h.sdispls[qi]:h.sdispls[qi] + h.sendcounts[qi]]])
            # what S sends to q == q's ghosts from S, in q's tail order
            si = np.where(np.asarray(R.halo.senders) == S.rank)[0][0]
            lo = int(R.halo.rdispls[si])
            hi = lo + int(R.halo.recvcounts[si])
            np.testing.assert_array_equal(owners, R.ghost_global[lo:hi])


def test_queen_like_spec_shape():
    spec = queen_like_spec(dof=3)
    A = stencil_global(6, 6, 6, spec)
    n = 6 * 6 * 6 * 3
    assert A.n == n
    F = A.to_full_csr()
    # interior rows have 27 neighbours x 3 dof = 81 nonzeros (Queen_4147: ~79)
    counts = np.diff(F.rowptr)
    assert counts.max() == 27 * 3
    # SPD sanity: diagonally dominant
    X = A.to_scipy_full()
    d = X.diagonal()
    offsum = np.abs(X).sum(axis=1).A1 - np.abs(d)
    assert np.all(d > offsum * 0.99)


@pytest.mark.parametrize("spec,g,nranks", [
    (STENCIL_5PT_2D, (12, 12, 1), 1),
    (STENCIL_27PT_3D, (6, 6, 8), 3),
    (queen_like_spec(3), (5, 5, 7), 2),
])
def test_slab_generator_matches_global(spec, g, nranks):
    """stencil_local_slab must produce the same operator as the global
    assembly + hand-filled halo."""
    from acg_amd.gen import stencil_local_slab
    import acg_amd.ops.torch_ref as ops

    gx, gy, gz = g
    A = stencil_global(gx, gy, gz, spec)
    rng = np.random.default_rng(3)
    xg = rng.standard_normal(A.n)
    yg = A.dsymv(xg)
    covered = np.zeros(A.n, dtype=bool)
    for rank in range(nranks):
        S = stencil_local_slab(gx, gy, gz, spec, rank, nranks)
        xl = np.concatenate([xg[S.owned_global], xg[S.ghost_global]])
        xt = torch.from_numpy(xl)
        yt = torch.zeros(S.nowned, dtype=torch.float64)
        ops.spmv(torch.from_numpy(S.A_rowptr), torch.from_numpy(S.A_colidx.astype(np.int64)),
                 torch.from_numpy(S.A_vals), xt, yt)
        ops.spmv(torch.from_numpy(S.O_rowptr), torch.from_numpy(S.O_colidx.astype(np.int64)),
                 torch.from_numpy(S.O_vals), xt, yt, rowbase=S.ninterior, accum=True)
        np.testing.assert_allclose(yt.numpy(), yg[S.owned_global], rtol=1e-12, atol=1e-10)
        assert not covered[S.owned_global].any()
        covered[S.owned_global] = True
    assert covered.all()


def test_bsell_from_csr_layout():
    """Host Block-SELL conversion: emulate the kernel's indexing in numpy
    and compare against the CSR SpMV."""
    from acg_amd.gen import queen_like_spec
    from acg_amd.ops.torch_ref import bsell_from_csr
    from acg_amd.part import extract_subdomains, partition_rows

    A = stencil_global(5, 5, 5, queen_like_spec(3))
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    out = bsell_from_csr(S.A_rowptr, S.A_colidx, S.A_vals, 3)
    assert out is not None
    bptr, bcol, bvals, density = out
    assert density > 0.95, density
    rng = np.random.default_rng(0)
    x = rng.standard_normal(S.nowned)
    dof, C = 3, 64
    nnodes = S.nowned // dof
    y = np.zeros(S.nowned)
    nslices = len(bptr) - 1
    for s in range(nslices):
        b0 = int(bptr[s])
        blen = (int(bptr[s + 1]) - b0) // C
        for lane in range(C):
            node = s * C + lane
            if node >= nnodes:
                continue
            acc = np.zeros(dof)
            D2 = dof * dof
            even = D2 & ~1
            for j in range(blen):
                cb = int(bcol[b0 + j * C + lane])
                xv = x[cb * dof:(cb + 1) * dof]
                for k in range(D2):
                    off = ((k >> 1) * 2 * C + lane * 2 + (k & 1)
                           if k < even else even * C + lane)
                    a = bvals[b0 * D2 + j * D2 * C + off]
                    acc[k // dof] += a * xv[k % dof]
            y[node * dof:(node + 1) * dof] = acc
    import scipy.sparse as sp

    Acsr = sp.csr_matrix((S.A_vals, S.A_colidx.astype(np.int64), S.A_rowptr))
    np.testing.assert_allclose(y, Acsr @ x, rtol=1e-12, atol=1e-12)


def test_slab_halo_pairing():
    from acg_amd.gen import stencil_local_slab

    spec = queen_like_spec(3)
    systems = [stencil_local_slab(5, 5, 9, spec, r, 3) for r in range(3)]
    for S in systems:
        h = S.halo
        for qi, q in enumerate(h.recipients):
            R = systems[q]
            sl = h.sendidx[int(h.sdispls[qi]):int(h.sdispls[qi]) + int(h.sendcounts[qi])]
            sent_globals = S.owned_global[np.asarray(sl, dtype=np.int64)]
            si = np.where(np.asarray(R.halo.senders) == S.rank)[0][0]
            lo = int(R.halo.rdispls[si])
            hi = lo + int(R.halo.recvcounts[si])
            np.testing.assert_array_equal(sent_globals, R.ghost_global[lo:hi])


def test_slab_generator_equals_generic_extractor():
    """The analytic slab generator and the generic extractor are two
    independent constructions of the same partition: layouts, halos and
    operators must agree EXACTLY on slab-aligned partitions."""
    import scipy.sparse as sp

    from acg_amd.gen import stencil_local_slab
    from acg_amd.part import extract_subdomains

    spec = queen_like_spec(3)
    gx, gy, gz, R, dof = 5, 5, 8, 2, 3
    A = stencil_global(gx, gy, gz, spec)
    plane = gx * gy
    part = np.zeros(A.n, dtype=np.int32)
    for r in range(R):
        z0, z1 = gz * r // R, gz * (r + 1) // R
        part[z0 * plane * dof:z1 * plane * dof] = r
    gen = [stencil_local_slab(gx, gy, gz, spec, r, R) for r in range(R)]
    ext = extract_subdomains(A, part, R)
    for g, e in zip(gen, ext):
        assert (g.nowned, g.ninterior, g.nborder, g.nghost) == \
               (e.nowned, e.ninterior, e.nborder, e.nghost)
        np.testing.assert_array_equal(g.owned_global, e.owned_global)
        np.testing.assert_array_equal(g.ghost_global, e.ghost_global)
        np.testing.assert_array_equal(g.halo.senders, e.halo.senders)
        np.testing.assert_array_equal(g.halo.sendidx, e.halo.sendidx)

        def toop(S):
            a = sp.csr_matrix((S.A_vals, S.A_colidx.astype(np.int64),
                               S.A_rowptr),
                              shape=(S.nowned, S.nowned + S.nghost))
            if S.nnzO:
                o = sp.csr_matrix((S.O_vals, S.O_colidx.astype(np.int64),
                                   S.O_rowptr),
                                  shape=(S.nborder, S.nowned + S.nghost))
                return sp.vstack([a[:S.ninterior], a[S.ninterior:] + o]).tocsr()
            return a

        assert abs(toop(g) - toop(e)).max() < 1e-14
