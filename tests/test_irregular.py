"""Irregular (power-law) SPD family: generator, streaming extraction,
row-binned hybrid prep (CPU parts)."""

import numpy as np
import pytest
import torch

from acg_amd.gen.irregular import degree_stats, powerlaw_spd
from acg_amd.part import extract_subdomains, partition_rows


def test_powerlaw_spd_is_spd():
    A = powerlaw_spd(400, mean_nnz=24, seed=7)
    F = A.to_scipy_full().toarray()
    assert np.allclose(F, F.T)
    np.linalg.cholesky(F)  # raises if not SPD


def test_powerlaw_heavy_tail():
    A = powerlaw_spd(50_000, mean_nnz=32, seed=0)
    st = degree_stats(A)
    # heavy tail: max far above the mean, median below it
    assert st["max"] > 8 * st["mean"]
    assert st["median"] < st["mean"]


def test_powerlaw_cpu_cg_matches_scipy():
    from acg_amd.solvers.cpu import CGSolverCPU

    A = powerlaw_spd(2000, mean_nnz=16, seed=3)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    solver = CGSolverCPU(S)
    rng = np.random.default_rng(0)
    b = rng.standard_normal(S.nowned)
    x = torch.zeros(S.nowned, dtype=torch.float64)
    res = solver.solve(torch.from_numpy(b), x, maxits=500, res_rtol=1e-12)
    assert res.converged
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), b)
    np.testing.assert_allclose(x.numpy(), x_ref, rtol=1e-8, atol=1e-10)


def test_only_parts_matches_full_extraction():
    A = powerlaw_spd(3000, mean_nnz=20, seed=1)
    part = partition_rows(A, 4, method="rgb", seed=2)
    full = extract_subdomains(A, part, 4)
    for p in (0, 2, 3):
        lone = extract_subdomains(A, part, 4, only_parts=[p])
        assert lone[(p + 1) % 4] is None
        Sp, Sf = lone[p], full[p]
        for fld in ("nowned", "ninterior", "nborder", "nghost"):
            assert getattr(Sp, fld) == getattr(Sf, fld), fld
        for fld in ("A_rowptr", "A_colidx", "A_vals", "O_rowptr", "O_colidx",
                    "O_vals", "owned_global", "ghost_global"):
            np.testing.assert_array_equal(getattr(Sp, fld), getattr(Sf, fld),
                                          err_msg=fld)
        for fld in ("senders", "recvcounts", "rdispls", "recipients",
                    "sendcounts", "sdispls", "sendidx"):
            np.testing.assert_array_equal(getattr(Sp.halo, fld),
                                          getattr(Sf.halo, fld), err_msg=fld)


def test_build_row_bins_partitions_all_rows():
    from acg_amd.ops.gpu_ops import build_row_bins

    A = powerlaw_spd(5000, mean_nnz=30, seed=5)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rowlist, bins = build_row_bins(S.A_rowptr)
    n = S.nowned
    # every row exactly once
    assert sorted(rowlist.tolist()) == list(range(n))
    total = sum(c for _, c, _ in bins)
    assert total == n
    lens = np.diff(S.A_rowptr)
    # bin lane assignment follows the MEASURED mapping (lanes_sweep:
    # >96 -> 64, >16 -> 32, >8 -> 16, >4 -> 8, rest 4); rows are
    # longest-first within the global order
    slens = lens[rowlist.astype(np.int64)]
    assert np.all(np.diff(slens) <= 0)
    upper = {64: None, 32: 96, 16: 16, 8: 8, 4: 4}
    lower = {64: 96, 32: 16, 16: 8, 8: 4, 4: None}
    for start, count, lanes in bins:
        seg = slens[start:start + count]
        if upper[lanes] is not None:
            assert seg.max() <= upper[lanes]
        if lower[lanes] is not None:
            assert seg.min() > lower[lanes]


def test_hybrid_torch_fallback_matches_spmv():
    # CPU check of the binning semantics: summing per-bin contributions
    # equals the plain SpMV (the GPU kernel equality test is in
    # test_gpu_kernels.py)
    from acg_amd.ops.gpu_ops import build_row_bins

    A = powerlaw_spd(1200, mean_nnz=25, seed=9)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rowlist, bins = build_row_bins(S.A_rowptr)
    import scipy.sparse as sp

    M = sp.csr_matrix((S.A_vals, S.A_colidx.astype(np.int64), S.A_rowptr),
                      shape=(S.nowned, S.nowned))
    x = np.random.default_rng(0).standard_normal(S.nowned)
    y = np.zeros(S.nowned)
    for start, count, lanes in bins:
        rows = rowlist[start:start + count].astype(np.int64)
        y[rows] = M[rows] @ x
    np.testing.assert_allclose(y, M @ x, rtol=1e-12)


def test_sellcsr_split_structure():
    """Every entry lands exactly once in SELL or stays CSR-long; perm
    sentinel and bin thresholds consistent."""
    from acg_amd.ops.gpu_ops import build_sellcsr_hybrid

    A = powerlaw_spd(8000, mean_nnz=30, seed=2)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    cut = 96
    sellptr, cols, svals, perm, rowlist, bins = build_sellcsr_hybrid(
        S.A_rowptr, S.A_colidx, S.A_vals, cut=cut)
    n = S.nowned
    lens = np.diff(S.A_rowptr)
    short = set(np.where(lens <= cut)[0].tolist())
    long_ = set(np.where(lens > cut)[0].tolist())
    assert set(rowlist.tolist()) == long_
    sell_rows = set(int(r) for r in perm if r < n)
    assert sell_rows == short
    # SELL holds exactly the short rows' values
    assert float(np.abs(svals).sum()) == pytest.approx(
        float(np.abs(S.A_vals[np.isin(np.repeat(np.arange(n), lens),
                                      list(short))]).sum()), rel=1e-12)
    # reconstruct y = A x from the two halves on CPU
    import scipy.sparse as sp

    M = sp.csr_matrix((S.A_vals, S.A_colidx.astype(np.int64), S.A_rowptr),
                      shape=(n, n))
    x = np.random.default_rng(0).standard_normal(n)
    y = np.zeros(n)
    C = 64
    nslices = len(sellptr) - 1
    for s in range(nslices):
        base = int(sellptr[s])
        L = (int(sellptr[s + 1]) - base) // C
        blk = (svals[base:base + L * C].reshape(L, C)
               * x[cols[base:base + L * C].astype(np.int64)].reshape(L, C)
               ).sum(axis=0)
        for lane in range(min(C, nslices * C - s * C)):
            r = int(perm[s * C + lane])
            if r < n:
                y[r] = blk[lane]
    lr = rowlist.astype(np.int64)
    y[lr] = M[lr] @ x
    np.testing.assert_allclose(y, M @ x, rtol=1e-12)


@pytest.mark.parametrize("seed,cut", [(2, 48), (5, 96), (9, 192)])
def test_sellcsr_split_fuzz(seed, cut):
    """Split reconstruction equals the plain SpMV across seeds and cuts."""
    from acg_amd.ops.gpu_ops import build_sellcsr_hybrid

    A = powerlaw_spd(4000, mean_nnz=26, seed=seed)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    n = S.nowned
    sellptr, cols, svals, perm, rowlist, bins = build_sellcsr_hybrid(
        S.A_rowptr, S.A_colidx, S.A_vals, cut=cut)
    import scipy.sparse as sp

    M = sp.csr_matrix((S.A_vals, S.A_colidx.astype(np.int64), S.A_rowptr),
                      shape=(n, n))
    x = np.random.default_rng(0).standard_normal(n)
    y = np.zeros(n)
    if sellptr is not None:
        C = 64
        for s in range(len(sellptr) - 1):
            base = int(sellptr[s])
            L = (int(sellptr[s + 1]) - base) // C
            blk = (svals[base:base + L * C].reshape(L, C)
                   * x[cols[base:base + L * C].astype(np.int64)].reshape(L, C)
                   ).sum(axis=0)
            for lane in range(C):
                r = int(perm[s * C + lane])
                if r < n:
                    y[r] = blk[lane]
    lr = rowlist.astype(np.int64)
    if len(lr):
        y[lr] = M[lr] @ x
    np.testing.assert_allclose(y, M @ x, rtol=1e-12)
