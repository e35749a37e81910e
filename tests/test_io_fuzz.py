"""Property-based Matrix-Market IO fuzz: random matrices and vectors
through every (format x field x container) roundtrip the reference
supports (mtxfile.c text/gzip/binary, 32/64-bit binary indices)."""

import numpy as np
import pytest

from acg_amd.io.mtx import MtxFile, read_mtx, write_mtx


def _rand_coo(rng, nrows, ncols, nnz, field_, symmetry):
    if symmetry == "symmetric":
        i = rng.integers(0, nrows, nnz)
        j = rng.integers(0, ncols, nnz)
        lo, hi = np.minimum(i, j), np.maximum(i, j)
        i, j = hi, lo  # lower triangle, aCG packed convention
    else:
        i = rng.integers(0, nrows, nnz)
        j = rng.integers(0, ncols, nnz)
    if field_ == "real":
        a = np.round(rng.standard_normal(nnz), 12)
    elif field_ == "integer":
        a = rng.integers(-1000, 1000, nnz).astype(np.int64)
    else:  # pattern
        a = None
    return i, j, a


@pytest.mark.parametrize("field_", ["real", "integer", "pattern"])
@pytest.mark.parametrize("container", ["text", "gz", "bin32", "bin64"])
def test_coordinate_roundtrip_fuzz(tmp_path, field_, container):
    rng = np.random.default_rng(hash((field_, container)) % 2**31)
    for trial in range(3):
        nrows = int(rng.integers(1, 40))
        ncols = int(rng.integers(1, 40))
        nnz = int(rng.integers(0, nrows * ncols // 2 + 1))
        symmetry = "symmetric" if (nrows == ncols and trial % 2) else "general"
        i, j, a = _rand_coo(rng, nrows, ncols, nnz, field_, symmetry)
        m = MtxFile(object="matrix", format="coordinate", field_=field_,
                    symmetry=symmetry, nrows=nrows, ncols=ncols, nnz=nnz,
                    rowidx=i.astype(np.int64), colidx=j.astype(np.int64),
                    a=a)
        path = tmp_path / f"m_{field_}_{container}_{trial}.mtx"
        if field_ == "pattern" and container.startswith("bin"):
            # the reference binary layout stores a value array: pattern is
            # text-only (mtx2bin offers --double/--integer); must raise
            # cleanly, not crash
            from acg_amd.utils.errors import AcgError

            with pytest.raises(AcgError):
                write_mtx(path, m, binary=True, idxsize=int(container[3:]))
            return
        kw = {}
        if container == "gz":
            path = tmp_path / (path.name + ".gz")
            write_mtx(path, m, gzipped=True)
            kw = dict(gzipped=True)
        elif container.startswith("bin"):
            write_mtx(path, m, binary=True, idxsize=int(container[3:]))
            kw = dict(binary=True, idxsize=int(container[3:]))
        else:
            write_mtx(path, m)
        r = read_mtx(path, **kw)
        assert (r.nrows, r.ncols, r.nnz) == (nrows, ncols, nnz)
        assert r.field_ == field_ and r.symmetry == symmetry
        np.testing.assert_array_equal(np.asarray(r.rowidx), i)
        np.testing.assert_array_equal(np.asarray(r.colidx), j)
        if field_ == "real":
            np.testing.assert_allclose(np.asarray(r.a), a, rtol=0, atol=1e-14)
        elif field_ == "integer":
            np.testing.assert_array_equal(np.asarray(r.a, dtype=np.int64), a)


@pytest.mark.parametrize("container", ["text", "bin64"])
def test_array_vector_roundtrip_fuzz(tmp_path, container):
    rng = np.random.default_rng(7)
    for trial in range(3):
        n = int(rng.integers(1, 200))
        a = np.round(rng.standard_normal(n), 12)
        m = MtxFile(object="matrix", format="array", field_="real",
                    symmetry="general", nrows=n, ncols=1, nnz=n, a=a)
        path = tmp_path / f"v{trial}.mtx"
        if container == "bin64":
            write_mtx(path, m, binary=True, idxsize=64)
            r = read_mtx(path, binary=True, idxsize=64)
        else:
            write_mtx(path, m)
            r = read_mtx(path)
        np.testing.assert_allclose(np.asarray(r.a), a, rtol=0, atol=1e-14)


def test_numfmt_grammar_fuzz():
    """Random printf specifiers over the reference grammar (fmtspec.c):
    every accepted spec must format like C/python %-formatting; malformed
    ones must be rejected, never crash."""

    from acg_amd.utils.numfmt import FmtSpec, parse_numfmt

    rng = np.random.default_rng(11)
    convs = "eEfFgG"
    vals = [0.0, 1.0, -1.5, 3.141592653589793e-8, 2.718e120, -7e-300]
    for _ in range(200):
        flags = "".join(rng.choice(list("-+ #0"),
                                   size=rng.integers(0, 3), replace=False))
        width = str(rng.integers(0, 25)) if rng.random() < 0.6 else ""
        prec = f".{rng.integers(0, 18)}" if rng.random() < 0.7 else ""
        conv = convs[rng.integers(len(convs))]
        spec = f"%{flags}{width}{prec}{conv}"
        f = parse_numfmt(spec)
        for v in vals:
            got = f.format(v)
            want = spec.replace("'", "") % v
            assert got == want, (spec, v, got, want)
    # hex-float conversions format without crashing and round-trip
    fa = parse_numfmt("%a")
    for v in vals:
        assert float.fromhex(fa.format(v)) == v
    # malformed specs are rejected
    for bad in ("%d", "%.5", "%", "abc", "%5.2x", "%ld", "%*g", "%.*e", "%%g"):
        try:
            FmtSpec(bad)
            rejected = False
        except ValueError:
            rejected = True
        assert rejected, bad


def test_symcsr_assembly_fuzz():
    """Random COO (duplicates, unsorted, mixed triangles) through both the
    native C++ assembly and the numpy fallback must equal the scipy
    symmetrisation, and the packed-upper -> full expansion must be exact."""
    import scipy.sparse as sp

    from acg_amd.core.symcsr import SymCSRMatrix

    rng = np.random.default_rng(23)
    for trial in range(6):
        n = int(rng.integers(2, 60))
        nnz = int(rng.integers(1, 4 * n))
        i = rng.integers(0, n, nnz)
        j = rng.integers(0, n, nnz)
        # guarantee a full diagonal so the operator is well-formed
        i = np.concatenate([i, np.arange(n)])
        j = np.concatenate([j, np.arange(n)])
        v = rng.standard_normal(len(i))
        A = SymCSRMatrix.from_coo(n, i, j, v)
        # scipy oracle: canonicalise each entry to (min,max), sum dups,
        # then expand symmetrically
        lo, hi = np.minimum(i, j), np.maximum(i, j)
        U = sp.coo_matrix((v, (lo, hi)), shape=(n, n)).tocsr()
        U.sum_duplicates()
        full = U + sp.triu(U, k=1).T
        X = A.to_scipy_full()
        assert abs(X - full).max() < 1e-12
        # dsymv agrees too
        x = rng.standard_normal(n)
        np.testing.assert_allclose(A.dsymv(x), full @ x, rtol=1e-12, atol=1e-12)


def test_bsell_native_matches_numpy_fallback(monkeypatch):
    """bsell_from_csr has a native C++ block-merge path and a numpy unique
    fallback: identical outputs on random dof-blocked matrices."""
    import sys

    from acg_amd.gen import queen_like_spec, stencil_global
    from acg_amd.ops import torch_ref
    from acg_amd.part import extract_subdomains, partition_rows

    A = stencil_global(5, 4, 6, queen_like_spec(3))
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    native = torch_ref.bsell_from_csr(S.A_rowptr, S.A_colidx, S.A_vals, 3)
    monkeypatch.setitem(sys.modules, "acg_amd.host._acg_host", None)
    fallback = torch_ref.bsell_from_csr(S.A_rowptr, S.A_colidx, S.A_vals, 3)
    assert native is not None and fallback is not None
    for a, b in zip(native[:3], fallback[:3]):
        np.testing.assert_array_equal(np.asarray(a), np.asarray(b))
    assert native[3] == fallback[3]
