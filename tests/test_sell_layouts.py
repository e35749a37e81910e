"""CPU layout-contract tests: emulate the GPU kernels' SELL / sigma-SELL
indexing in numpy and check the conversion against scipy SpMV.  The GPU
numerics tests cover the kernels themselves; these lock the host-side
conversion format (element j of SELL row s*64+lane at sellptr[s]+j*64+lane,
pad entries value-0 pointing at a valid row, sigma windows sorted by
descending row length with perm sentinel = nrows)."""

import numpy as np
import pytest
import scipy.sparse as sp

from acg_amd.ops.torch_ref import sell_from_csr

C = 64


def _random_csr(nrows, density_rows, seed):
    """Irregular matrix: row lengths drawn from a skewed distribution."""
    rng = np.random.default_rng(seed)
    counts = np.minimum(rng.geometric(1.0 / density_rows, nrows), nrows)
    rowptr = np.zeros(nrows + 1, dtype=np.int64)
    np.cumsum(counts, out=rowptr[1:])
    nnz = int(rowptr[-1])
    colidx = np.empty(nnz, dtype=np.int64)
    for r in range(nrows):
        k0, k1 = rowptr[r], rowptr[r + 1]
        colidx[k0:k1] = np.sort(rng.choice(nrows, size=k1 - k0, replace=False))
    vals = rng.standard_normal(nnz)
    return rowptr, colidx, vals


def _emulate_sell_spmv(sellptr, cols, vals, nrows, x, perm=None):
    """Exactly the k_spmv_sell index arithmetic, scalar numpy."""
    y = np.zeros(nrows)
    nslices = len(sellptr) - 1
    for s in range(nslices):
        base = int(sellptr[s])
        length = (int(sellptr[s + 1]) - base) // C
        for lane in range(C):
            acc = 0.0
            for j in range(length):
                e = base + j * C + lane
                acc += vals[e] * x[int(cols[e])]
            row = int(perm[s * C + lane]) if perm is not None else s * C + lane
            if row < nrows:
                y[row] = acc
    return y


@pytest.mark.parametrize("nrows,dens,seed", [(130, 6, 0), (257, 15, 1)])
def test_sell_plain_layout(nrows, dens, seed):
    rowptr, colidx, vals = _random_csr(nrows, dens, seed)
    sellptr, scols, svals = sell_from_csr(rowptr, colidx, vals)
    rng = np.random.default_rng(2)
    x = rng.standard_normal(nrows)
    y = _emulate_sell_spmv(sellptr, scols, svals, nrows, x)
    want = sp.csr_matrix((vals, colidx, rowptr), shape=(nrows, nrows)) @ x
    np.testing.assert_allclose(y, want, rtol=1e-13, atol=1e-13)


@pytest.mark.parametrize("nrows,dens,seed", [(300, 8, 3), (1000, 20, 4)])
def test_sell_sigma_layout(nrows, dens, seed):
    rowptr, colidx, vals = _random_csr(nrows, dens, seed)
    out = sell_from_csr(rowptr, colidx, vals, sigma=16)
    sellptr, scols, svals, perm = out
    # sentinel rows must be exactly the pad lanes
    assert (perm == nrows).sum() == len(perm) - nrows
    # sigma sorting must not lose or duplicate any row
    assert sorted(p for p in perm if p < nrows) == list(range(nrows))
    rng = np.random.default_rng(5)
    x = rng.standard_normal(nrows)
    y = _emulate_sell_spmv(sellptr, scols, svals, nrows, x, perm=perm)
    want = sp.csr_matrix((vals, colidx, rowptr), shape=(nrows, nrows)) @ x
    np.testing.assert_allclose(y, want, rtol=1e-13, atol=1e-13)
    # and the padding must actually shrink vs sigma=1 for skewed rows
    plain = sell_from_csr(rowptr, colidx, vals)[0]
    assert int(sellptr[-1]) <= int(plain[-1])


def test_pick_lanes_policy():
    """Measured lanes-per-row policy (tools/lanes_sweep.py round 2: the
    optimum is ~1-2 nnz/lane, e.g. len-24 rows 50% faster at 32 lanes
    than at 4)."""
    from acg_amd.ops.gpu_ops import pick_lanes

    assert pick_lanes(3) == 4
    assert pick_lanes(7) == 8        # 7-pt Poisson
    assert pick_lanes(16) == 16
    assert pick_lanes(24) == 32
    assert pick_lanes(80) == 32      # Queen-shaped
    assert pick_lanes(96) == 32
    assert pick_lanes(150) == 64
    assert pick_lanes(500) == 64
