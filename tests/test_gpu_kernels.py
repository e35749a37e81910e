"""GPU kernel numerics: every gfx950 HIP kernel vs the plain-torch fp64
reference (ops.torch_ref), per the test strategy in SURVEY.md §4."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _rand_csr(nrows, ncols, nnz_per_row, seed=0, col64=False):
    rng = np.random.default_rng(seed)
    counts = rng.integers(1, 2 * nnz_per_row, size=nrows)
    rowptr = np.zeros(nrows + 1, dtype=np.int64)
    np.cumsum(counts, out=rowptr[1:])
    nnz = int(rowptr[-1])
    cols = rng.integers(0, ncols, size=nnz)
    rows = np.repeat(np.arange(nrows), counts)
    order = np.lexsort((cols, rows))
    cols = cols[order]
    vals = rng.standard_normal(nnz)
    cdt = np.int64 if col64 else np.int32
    return (torch.from_numpy(rowptr), torch.from_numpy(cols.astype(cdt)),
            torch.from_numpy(vals))


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


@pytest.mark.parametrize("lanes", [4, 8, 16, 32, 64])
@pytest.mark.parametrize("col64", [False, True])
def test_spmv_vector(dev, lanes, col64):
    from acg_amd.ops import gpu_ops, torch_ref

    rowptr, colidx, vals = _rand_csr(5000, 6000, 40, seed=lanes, col64=col64)
    x = torch.randn(6000, dtype=torch.float64)
    y_ref = torch.zeros(5000, dtype=torch.float64)
    torch_ref.spmv(rowptr, colidx, vals, x, y_ref)
    yg = torch.zeros(5000, dtype=torch.float64, device=dev)
    gpu_ops.spmv(rowptr.to(dev), colidx.to(dev), vals.to(dev), x.to(dev), yg,
                 lanes=lanes)
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)


@pytest.mark.parametrize("col64", [False, True])
@pytest.mark.parametrize("nrows", [64, 100, 5000])
def test_spmv_sell(dev, col64, nrows):
    from acg_amd.ops import gpu_ops, torch_ref

    rowptr, colidx, vals = _rand_csr(nrows, nrows + 37, 30, seed=nrows, col64=col64)
    sellptr, scols, svals = torch_ref.sell_from_csr(
        rowptr.numpy(), colidx.numpy(), vals.numpy())
    x = torch.randn(nrows + 37, dtype=torch.float64)
    y_ref = torch.zeros(nrows, dtype=torch.float64)
    torch_ref.spmv(rowptr, colidx, vals, x, y_ref)
    yg = torch.zeros(nrows, dtype=torch.float64, device=dev)
    gpu_ops.spmv_sell(torch.from_numpy(sellptr).to(dev),
                      torch.from_numpy(scols).to(dev),
                      torch.from_numpy(svals).to(dev),
                      nrows, x.to(dev), yg)
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)


def test_spmv_sell_sigma_sorted(dev):
    """Sigma-sorted SELL (row permutation) on a power-law irregular matrix."""
    from acg_amd.ops import gpu_ops, torch_ref

    rng = np.random.default_rng(5)
    nrows = 8000
    # power-law row lengths: a few hubs, many short rows
    counts = np.minimum((rng.pareto(1.2, nrows) * 4 + 1).astype(np.int64), 800)
    rowptr = np.zeros(nrows + 1, dtype=np.int64)
    np.cumsum(counts, out=rowptr[1:])
    nnz = int(rowptr[-1])
    cols = rng.integers(0, nrows, nnz).astype(np.int32)
    vals = rng.standard_normal(nnz)
    plain = torch_ref.sell_from_csr(rowptr, cols, vals)
    sp1, sc1, sv1 = plain
    out = torch_ref.sell_from_csr(rowptr, cols, vals, sigma=16)
    sp2, sc2, sv2, perm = out
    waste1 = (int(sp1[-1]) - nnz) / nnz
    waste2 = (int(sp2[-1]) - nnz) / nnz
    assert waste2 < waste1 * 0.7, (waste1, waste2)  # sorting shrinks padding
    x = torch.randn(nrows, dtype=torch.float64)
    y_ref = torch.zeros(nrows, dtype=torch.float64)
    torch_ref.spmv(torch.from_numpy(rowptr), torch.from_numpy(cols),
                   torch.from_numpy(vals), x, y_ref)
    yg = torch.zeros(nrows, dtype=torch.float64, device=dev)
    gpu_ops.spmv_sell(torch.from_numpy(sp2).to(dev), torch.from_numpy(sc2).to(dev),
                      torch.from_numpy(sv2).to(dev), nrows, x.to(dev), yg,
                      perm=torch.from_numpy(perm).to(dev))
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-9)


def test_irregular_matrix_end_to_end(dev):
    """CG on an irregular random SPD matrix goes through the sigma-SELL or
    CSR-vector path and still converges."""
    import scipy.sparse as sp

    from acg_amd.core.symcsr import SymCSRMatrix
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    rng = np.random.default_rng(9)
    n = 3000
    # random sparse SPD: B B^T + diag
    B = sp.random(n, n, density=0.004, random_state=1, format="csr")
    M = (B @ B.T).tocoo()
    A = SymCSRMatrix.from_coo(n, M.row, M.col, M.data)
    # strengthen diagonal
    d = np.zeros(n)
    rows_u = np.repeat(np.arange(n), np.diff(A.rowptr))
    np.add.at(d, rows_u, np.abs(A.vals))
    np.add.at(d, A.colidx, np.abs(A.vals))
    diag_add = d + 1.0
    A2 = SymCSRMatrix.from_coo(
        n, np.concatenate([rows_u, np.arange(n)]),
        np.concatenate([A.colidx, np.arange(n)]),
        np.concatenate([A.vals, diag_add]))
    S = extract_subdomains(A2, partition_rows(A2, 1), 1)[0]
    solver = CGSolverHIP(S, device="cuda:0")
    b = torch.from_numpy(rng.standard_normal(n)).cuda()
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    res = solver.solve(b, x, maxits=600, res_rtol=1e-9)
    assert res.converged, res.summary()


def test_spmv_sell_fused_dot(dev):
    from acg_amd.ops import gpu_ops, torch_ref

    nrows = 3000
    rowptr, colidx, vals = _rand_csr(nrows, nrows, 25, seed=3)
    sellptr, scols, svals = torch_ref.sell_from_csr(
        rowptr.numpy(), colidx.numpy(), vals.numpy())
    x = torch.randn(nrows, dtype=torch.float64)
    y_ref = torch.zeros(nrows, dtype=torch.float64)
    sr = torch_ref.alloc_scalars()
    torch_ref.spmv(rowptr, colidx, vals, x, y_ref, scal=sr,
                   dotslot=torch_ref.S_PT, dot_accum=False)
    yg = torch.zeros(nrows, dtype=torch.float64, device=dev)
    sg = gpu_ops.alloc_scalars(dev)
    pg = gpu_ops.alloc_partials(dev)
    gpu_ops.spmv_sell(torch.from_numpy(sellptr).to(dev),
                      torch.from_numpy(scols).to(dev),
                      torch.from_numpy(svals).to(dev),
                      nrows, x.to(dev), yg, partials=pg, scal=sg,
                      dotslot=gpu_ops.S_PT, dot_accum=False)
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)
    torch.testing.assert_close(sg.cpu()[gpu_ops.S_PT], sr[torch_ref.S_PT],
                               rtol=1e-10, atol=1e-8)


def test_spmv_accum_rowbase_fused_dot(dev):
    from acg_amd.ops import gpu_ops, torch_ref

    n = 4000
    rowptr, colidx, vals = _rand_csr(1000, n, 30, seed=7)
    x = torch.randn(n, dtype=torch.float64)
    y0 = torch.randn(n, dtype=torch.float64)
    scal_ref = torch_ref.alloc_scalars()
    y_ref = y0.clone()
    torch_ref.spmv(rowptr, colidx, vals, x, y_ref, rowbase=500, accum=True,
                   scal=scal_ref, dotslot=torch_ref.S_PT, dot_accum=False)
    scal = gpu_ops.alloc_scalars(dev)
    part = gpu_ops.alloc_partials(dev)
    yg = y0.to(dev)
    gpu_ops.spmv(rowptr.to(dev), colidx.to(dev), vals.to(dev), x.to(dev), yg,
                 rowbase=500, accum=True, lanes=8, partials=part, scal=scal,
                 dotslot=gpu_ops.S_PT, dot_accum=False)
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)
    torch.testing.assert_close(scal.cpu()[gpu_ops.S_PT], scal_ref[torch_ref.S_PT],
                               rtol=1e-10, atol=1e-10)


def test_dot_and_dot2(dev):
    from acg_amd.ops import gpu_ops, torch_ref

    n = 100_003
    r = torch.randn(n, dtype=torch.float64)
    w = torch.randn(n, dtype=torch.float64)
    sr = torch_ref.alloc_scalars()
    torch_ref.dot(r, w, None, sr, torch_ref.S_PT)
    torch_ref.dot2(r, w, None, sr, n)
    sg = gpu_ops.alloc_scalars(dev)
    pg = gpu_ops.alloc_partials(dev)
    gpu_ops.dot(r.to(dev), w.to(dev), pg, sg, gpu_ops.S_PT)
    gpu_ops.dot2(r.to(dev), w.to(dev), pg, sg, n)
    torch.testing.assert_close(sg.cpu(), sr, rtol=1e-10, atol=1e-8)


def test_dot_deterministic(dev):
    """Partials-based reduction must be bitwise deterministic across runs
    (the reference's atomicAdd dots are not)."""
    from acg_amd.ops import gpu_ops

    n = 1_000_001
    r = torch.randn(n, dtype=torch.float64).to(dev)
    w = torch.randn(n, dtype=torch.float64).to(dev)
    sg = gpu_ops.alloc_scalars(dev)
    pg = gpu_ops.alloc_partials(dev)
    vals = set()
    for _ in range(5):
        gpu_ops.dot(r, w, pg, sg, 0)
        vals.add(float(sg.cpu()[0]))
    assert len(vals) == 1, vals


def test_fused_update_and_daypx(dev):
    from acg_amd.ops import gpu_ops, torch_ref

    n = 50_001
    g = torch.Generator().manual_seed(5)
    r = torch.randn(n, dtype=torch.float64, generator=g)
    x = torch.randn(n, dtype=torch.float64, generator=g)
    p = torch.randn(n, dtype=torch.float64, generator=g)
    t = torch.randn(n, dtype=torch.float64, generator=g)
    sr = torch_ref.alloc_scalars()
    sr[torch_ref.S_RR] = 3.7
    sr[torch_ref.S_PT] = 1.9
    sg = sr.clone().to(dev)
    pg = gpu_ops.alloc_partials(dev)
    rr, xr, pr = r.clone(), x.clone(), p.clone()
    torch_ref.cg_prep_rr(sr)
    torch_ref.cg_fused_update(rr, xr, pr, t, sr, None, n)
    torch_ref.daypx_ratio(pr, rr, sr, torch_ref.S_RR, torch_ref.S_RR_PREV)
    rg, xg, pg_, tg = r.to(dev), x.to(dev), p.to(dev), t.to(dev)
    gpu_ops.cg_prep_rr(sg)
    gpu_ops.cg_fused_update(rg, xg, pg_, tg, sg, pg, n)
    gpu_ops.daypx_ratio(pg_, rg, sg, gpu_ops.S_RR, gpu_ops.S_RR_PREV)
    torch.testing.assert_close(rg.cpu(), rr, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(xg.cpu(), xr, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(pg_.cpu(), pr, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(sg.cpu()[gpu_ops.S_RR], sr[torch_ref.S_RR],
                               rtol=1e-10, atol=1e-8)


@pytest.mark.parametrize("first", [True, False])
def test_pipelined_fused(dev, first):
    from acg_amd.ops import gpu_ops, torch_ref

    n = 30_000
    g = torch.Generator().manual_seed(9)
    vecs = {k: torch.randn(n, dtype=torch.float64, generator=g)
            for k in "ztpxrwq"}
    sr = torch_ref.alloc_scalars()
    sr[torch_ref.S_GAMMA] = 2.1
    sr[torch_ref.S_DELTA] = 4.3
    sr[torch_ref.S_GAMMA_PREV] = 1.7
    sr[torch_ref.S_ALPHA_PREV] = 0.9
    sg = sr.clone().to(dev)
    pg = gpu_ops.alloc_partials(dev)
    ref = {k: v.clone() for k, v in vecs.items()}
    torch_ref.pipelined_fused(ref["z"], ref["t"], ref["p"], ref["x"], ref["r"],
                              ref["w"], ref["q"], sr, None, n, first)
    gv = {k: v.to(dev) for k, v in vecs.items()}
    gpu_ops.pipelined_fused(gv["z"], gv["t"], gv["p"], gv["x"], gv["r"],
                            gv["w"], gv["q"], sg, pg, n, first)
    for k in "ztpxrw":
        torch.testing.assert_close(gv[k].cpu(), ref[k], rtol=1e-12, atol=1e-12,
                                   msg=f"vec {k}")
    torch.testing.assert_close(sg.cpu(), sr, rtol=1e-10, atol=1e-8)


def test_pack_gather(dev):
    from acg_amd.ops import gpu_ops, torch_ref

    n = 10000
    x = torch.randn(n, dtype=torch.float64)
    idx = torch.randint(0, n, (777,), dtype=torch.int32)
    ref = torch.empty(777, dtype=torch.float64)
    torch_ref.pack_gather(ref, x, idx)
    out = torch.empty(777, dtype=torch.float64, device=dev)
    gpu_ops.pack_gather(out, x.to(dev), idx.to(dev))
    torch.testing.assert_close(out.cpu(), ref)


def test_axpy_ratio(dev):
    from acg_amd.ops import gpu_ops, torch_ref

    n = 12345
    y = torch.randn(n, dtype=torch.float64)
    x = torch.randn(n, dtype=torch.float64)
    s = torch_ref.alloc_scalars()
    s[0], s[1] = 2.5, 0.5
    yr = y.clone()
    torch_ref.axpy_ratio(yr, x, s, 0, 1, sign=-1.0)
    yg = y.to(dev)
    gpu_ops.axpy_ratio(yg, x.to(dev), s.to(dev), 0, 1, sign=-1.0)
    torch.testing.assert_close(yg.cpu(), yr, rtol=1e-14, atol=1e-14)


@pytest.mark.parametrize("col64", [False, True])
def test_spmv_binned_hybrid(dev, col64):
    """Row-binned hybrid CSR vs plain torch SpMV on power-law rows."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.ops import gpu_ops, torch_ref
    from acg_amd.part import extract_subdomains, partition_rows

    A = powerlaw_spd(20_000, mean_nnz=35, seed=11)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rowptr = torch.from_numpy(S.A_rowptr)
    cdt = np.int64 if col64 else np.int32
    colidx = torch.from_numpy(S.A_colidx.astype(cdt))
    vals = torch.from_numpy(S.A_vals)
    rowlist_np, bins = gpu_ops.build_row_bins(S.A_rowptr)
    assert len(bins) >= 3  # heavy tail must populate several lane bins
    x = torch.randn(S.nowned, dtype=torch.float64)
    y_ref = torch.zeros(S.nowned, dtype=torch.float64)
    torch_ref.spmv(rowptr, colidx, vals, x, y_ref)
    yg = torch.zeros(S.nowned, dtype=torch.float64, device=dev)
    rowlist = torch.from_numpy(rowlist_np).to(dev)
    gpu_ops.spmv_binned(rowptr.to(dev), colidx.to(dev), vals.to(dev),
                        rowlist, bins, x.to(dev), yg)
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)
    # fused dot + accumulate path
    scal = gpu_ops.alloc_scalars(dev)
    partials = gpu_ops.alloc_partials(dev)
    yg2 = torch.ones(S.nowned, dtype=torch.float64, device=dev)
    gpu_ops.spmv_binned(rowptr.to(dev), colidx.to(dev), vals.to(dev),
                        rowlist, bins, x.to(dev), yg2, accum=True,
                        partials=partials, scal=scal,
                        dotslot=gpu_ops.S_PT, dot_accum=False)
    torch.testing.assert_close(yg2.cpu(), y_ref + 1.0, rtol=1e-12, atol=1e-10)
    want = float(torch.dot(x, y_ref))
    got = float(scal[gpu_ops.S_PT])
    assert abs(got - want) < 1e-8 * max(1.0, abs(want))


def test_spmv_sell_sigma_wide(dev):
    """Wide-window sigma-SELL (4096) exactness on power-law rows."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.ops import gpu_ops, torch_ref
    from acg_amd.part import extract_subdomains, partition_rows

    A = powerlaw_spd(20_000, mean_nnz=35, seed=11)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    sellptr, scols, svals, perm = torch_ref.sell_from_csr(
        S.A_rowptr, S.A_colidx, S.A_vals, sigma=4096)
    x = torch.randn(S.nowned, dtype=torch.float64)
    y_ref = torch.zeros(S.nowned, dtype=torch.float64)
    torch_ref.spmv(torch.from_numpy(S.A_rowptr),
                   torch.from_numpy(S.A_colidx.astype(np.int64)),
                   torch.from_numpy(S.A_vals), x, y_ref)
    yg = torch.zeros(S.nowned, dtype=torch.float64, device=dev)
    gpu_ops.spmv_sell(torch.from_numpy(sellptr).to(dev),
                      torch.from_numpy(scols).to(dev),
                      torch.from_numpy(svals).to(dev), S.nowned,
                      x.to(dev), yg, perm=torch.from_numpy(perm).to(dev))
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)


def test_powerlaw_solver_formats_agree(dev):
    """Solver-level: hybrid, sigma-SELL and CSR-vector formats all solve
    the same irregular system to the same answer (vs scipy direct).
    Mild tail (clip=48) keeps kappa ~ O(100) so rtol 1e-11 converges;
    exactness of each kernel on the HEAVY tail is covered above."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP
    import scipy.sparse.linalg as spla

    A = powerlaw_spd(30_000, mean_nnz=24, clip=48, seed=4)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(1)
    b_np = rng.standard_normal(S.nowned)
    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b_np)
    for fmt in ("hybrid", "sigma", "csr"):
        solver = CGSolverHIP(S, device=dev, force_format=fmt)
        if fmt == "hybrid":
            assert solver.hybrid is not None and solver.sell is None
        elif fmt == "sigma":
            assert solver.sell_perm is not None
        else:
            assert solver.sell is None and solver.hybrid is None
        b = torch.from_numpy(b_np).to(dev)
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device=dev)
        res = solver.solve(b, x, maxits=2000, res_rtol=1e-11)
        assert res.converged, (fmt, res.niterations, res.rnrm2, res.bnrm2)
        np.testing.assert_allclose(x[:S.nowned].cpu().numpy(), x_ref,
                                   rtol=1e-6, atol=1e-8, err_msg=fmt)


def test_powerlaw_auto_format_is_hybrid_or_sigma(dev):
    """The auto ladder must NOT pick plain SELL for power-law rows (padding
    explodes); it lands on wide-sigma or hybrid."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    A = powerlaw_spd(30_000, mean_nnz=30, seed=4)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverHIP(S, device=dev)
    assert solver.hybrid is not None or solver.sell_perm is not None


def test_sellcsr_split_spmv_exact(dev):
    """SELL+CSR split SpMV (sell part + binned long part, fused dot) is
    EXACT vs the torch reference on the heavy tail -- the per-iteration
    building block, free of CG rounding amplification."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.ops import gpu_ops, torch_ref
    from acg_amd.part import extract_subdomains, partition_rows

    A = powerlaw_spd(50_000, mean_nnz=35, seed=11)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    n = S.nowned
    sellptr, cols, svals, perm, rowlist, bins = gpu_ops.build_sellcsr_hybrid(
        S.A_rowptr, S.A_colidx, S.A_vals)
    assert sellptr is not None and len(rowlist)
    rowptr_t = torch.from_numpy(S.A_rowptr)
    colidx_t = torch.from_numpy(S.A_colidx.astype(np.int32))
    vals_t = torch.from_numpy(S.A_vals)
    x = torch.randn(n, dtype=torch.float64)
    y_ref = torch.zeros(n, dtype=torch.float64)
    torch_ref.spmv(rowptr_t, colidx_t, vals_t, x, y_ref)
    xg = x.to(dev)
    yg = torch.zeros(n, dtype=torch.float64, device=dev)
    scal = gpu_ops.alloc_scalars(dev)
    partials = gpu_ops.alloc_partials(dev)
    gpu_ops.spmv_sell(torch.from_numpy(sellptr).to(dev),
                      torch.from_numpy(cols).to(dev),
                      torch.from_numpy(svals).to(dev), n, xg, yg,
                      perm=torch.from_numpy(perm).to(dev),
                      partials=partials, scal=scal,
                      dotslot=gpu_ops.S_PT, dot_accum=False)
    gpu_ops.spmv_binned(rowptr_t.to(dev), colidx_t.to(dev), vals_t.to(dev),
                        torch.from_numpy(rowlist).to(dev), bins, xg, yg,
                        partials=partials, scal=scal,
                        dotslot=gpu_ops.S_PT, dot_accum=True)
    torch.testing.assert_close(yg.cpu(), y_ref, rtol=1e-12, atol=1e-10)
    want = float(torch.dot(x, y_ref))
    assert abs(float(scal[gpu_ops.S_PT]) - want) < 1e-8 * max(1.0, abs(want))


def test_sellcsr_hybrid_solver_heavy_tail(dev):
    """Solver-level split-format run on the HEAVY tail: converges like the
    pure-binned format (cross-format iterates only agree to rounding-
    amplified tolerance at kappa~5e3, so compare residual QUALITY; exact
    SpMV equality is test_sellcsr_split_spmv_exact)."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    A = powerlaw_spd(50_000, mean_nnz=35, seed=11)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(1)
    b_np = rng.standard_normal(S.nowned)
    rnorm = {}
    for fmt in ("hybrid", "binned"):
        solver = CGSolverHIP(S, device=dev, force_format=fmt)
        if fmt == "hybrid":
            assert solver.hybrid["sellptr"] is not None
            assert solver.hybrid["rowlist"] is not None
        b = torch.from_numpy(b_np).to(dev)
        x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device=dev)
        res = solver.solve(b, x, maxits=120, res_rtol=0.0)
        assert res.niterations == 120
        xh = x[:S.nowned].cpu().numpy()
        rnorm[fmt] = np.linalg.norm(b_np - (A.to_scipy_full() @ xh))
        assert rnorm[fmt] < 1e-2 * np.linalg.norm(b_np), fmt
    # equivalent convergence quality (same algorithm, different rounding)
    ratio = rnorm["hybrid"] / rnorm["binned"]
    assert 0.3 < ratio < 3.0, rnorm


def test_hybrid_mato_split_spmv(dev):
    """Hybrid format on a PARTITIONED irregular system: the matA SELL+CSR
    split plus the binned matO pass must equal the global SpMV restricted
    to this rank's rows (ghost tail filled by hand, comm=None)."""
    from acg_amd.gen.irregular import powerlaw_spd
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    A = powerlaw_spd(30_000, mean_nnz=30, seed=6)
    part = partition_rows(A, 2, method="ml", seed=1)
    S = extract_subdomains(A, part, 2, only_parts=[0])[0]
    assert S.nnzO > 0
    solver = CGSolverHIP(S, device=dev, force_format="hybrid")
    assert "rowlistO" in solver.hybrid
    rng = np.random.default_rng(2)
    xg_np = rng.standard_normal(A.n)
    xfull = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device=dev)
    xfull[:S.nowned] = torch.from_numpy(xg_np[S.owned_global]).to(dev)
    xfull[S.nowned:] = torch.from_numpy(xg_np[S.ghost_global]).to(dev)
    y = torch.zeros(S.nowned, dtype=torch.float64, device=dev)
    solver._spmv_overlapped(xfull, y)
    y_ref = (A.to_scipy_full() @ xg_np)[S.owned_global]
    np.testing.assert_allclose(y.cpu().numpy(), y_ref, rtol=1e-11, atol=1e-9)
