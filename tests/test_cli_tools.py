"""CLI driver, IO formats, tools, oracle solver (CPU)."""

import gzip
import sys

import numpy as np
import pytest
import torch

from acg_amd.gen import STENCIL_5PT_2D, stencil_global
from acg_amd.io.mtx import MtxFile, read_mtx, write_mtx
from acg_amd.utils.numfmt import parse_numfmt


def _poisson_mtx(tmp_path, nx=16, ny=16, name="A.mtx"):
    A = stencil_global(nx, ny, 1, STENCIL_5PT_2D)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    path = tmp_path / name
    write_mtx(path, m)
    return A, path


def test_mtx_roundtrip_text(tmp_path):
    A, path = _poisson_mtx(tmp_path)
    m = read_mtx(path)
    assert m.symmetry == "symmetric" and m.nnz == A.nnz_stored
    np.testing.assert_allclose(m.a, A.vals)


def test_mtx_roundtrip_binary(tmp_path):
    A, path = _poisson_mtx(tmp_path)
    m = read_mtx(path)
    for idxsize in (32, 64):
        bpath = tmp_path / f"A{idxsize}.bin"
        write_mtx(bpath, m, binary=True, idxsize=idxsize)
        mb = read_mtx(bpath, binary=True, idxsize=idxsize)
        np.testing.assert_array_equal(mb.rowidx, m.rowidx)
        np.testing.assert_array_equal(mb.colidx, m.colidx)
        np.testing.assert_allclose(mb.a, m.a)


def test_mtx_gzip(tmp_path):
    A, path = _poisson_mtx(tmp_path)
    gzpath = tmp_path / "A.mtx.gz"
    with open(path, "rb") as f, gzip.open(gzpath, "wb") as g:
        g.write(f.read())
    m = read_mtx(gzpath)
    assert m.nnz == A.nnz_stored


def test_mtx2bin_tool(tmp_path):
    sys.path.insert(0, "tools")
    import importlib

    mtx2bin = importlib.import_module("mtx2bin")
    A, path = _poisson_mtx(tmp_path)
    out = tmp_path / "A.bin"
    assert mtx2bin.main([str(path), str(out)]) == 0
    mb = read_mtx(out, binary=True, idxsize=64)
    assert mb.nnz == A.nnz_stored


def test_mtxpartition_tool(tmp_path):
    sys.path.insert(0, "tools")
    import importlib

    mtxpartition = importlib.import_module("mtxpartition")
    A, path = _poisson_mtx(tmp_path)
    out = tmp_path / "part.mtx"
    assert mtxpartition.main([str(path), "--parts", "4", "--method", "rgb",
                              "--output", str(out)]) == 0
    from acg_amd.part import read_partition_file

    part = read_partition_file(out, A.n)
    assert part.min() == 0 and part.max() == 3
    # balanced within 2x
    counts = np.bincount(part)
    assert counts.max() <= 2 * counts.min()


def test_numfmt():
    f = parse_numfmt("%.3e")
    assert f(1.5) == "1.500e+00"
    f2 = parse_numfmt("%12.4f")
    assert f2(2.0) == "      2.0000"
    with pytest.raises(ValueError):
        parse_numfmt("%d")
    with pytest.raises(ValueError):
        parse_numfmt("nope")


def test_cli_serial_manufactured(tmp_path, capsys, monkeypatch):
    from acg_amd import cli

    A, path = _poisson_mtx(tmp_path)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    rc = cli.main([str(path), "--solver", "cpu", "--manufactured-solution",
                   "--max-iterations", "2000", "--residual-rtol", "1e-10",
                   "-v"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    assert "manufactured solution" in out.err
    # solution written as mtx array
    assert out.out.startswith("%%MatrixMarket matrix array real general")
    lines = out.out.strip().splitlines()
    assert int(lines[1].split()[0]) == A.n


def test_cli_partition_file_and_numfmt(tmp_path, capsys, monkeypatch):
    from acg_amd import cli
    from acg_amd.part import write_partition_file

    A, path = _poisson_mtx(tmp_path)
    ppath = tmp_path / "part.mtx"
    write_partition_file(ppath, np.zeros(A.n, dtype=np.int32))
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    rc = cli.main([str(path), "--solver", "cpu", "--partition", str(ppath),
                   "--max-iterations", "500", "--numfmt", "%.6e"])
    out = capsys.readouterr()
    assert rc == 0
    val_line = out.out.strip().splitlines()[2]
    assert "e" in val_line  # %.6e formatting applied


def test_oracle_matches_cpu(tmp_path):
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.cpu import CGSolverCPU
    from acg_amd.solvers.oracle import solve_scipy

    A = stencil_global(12, 12, 1, STENCIL_5PT_2D)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rng = np.random.default_rng(0)
    b = rng.standard_normal(A.n)
    res_o, x_o = solve_scipy(S, None, b, np.zeros(A.n), maxits=2000,
                             res_rtol=1e-10)
    solver = CGSolverCPU(S)
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64)
    res_c = solver.solve(torch.from_numpy(b.copy()), x, maxits=2000,
                         res_rtol=1e-10)
    assert res_o.converged and res_c.converged
    np.testing.assert_allclose(x[:A.n].numpy(), x_o, rtol=1e-6, atol=1e-8)


def test_cli_scipy_solver(tmp_path, capsys, monkeypatch):
    from acg_amd import cli

    A, path = _poisson_mtx(tmp_path)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    rc = cli.main([str(path), "--solver", "scipy", "--max-iterations", "2000",
                   "--residual-rtol", "1e-9", "-q"])
    out = capsys.readouterr()
    assert rc == 0, out.err
    assert "scipy-cg" in out.err


def test_mtx_edge_formats(tmp_path):
    """Empty matrices, vector-coordinate, pattern, complex, comments."""
    import numpy as np

    from acg_amd.io.mtx import MtxFile, read_mtx, write_mtx

    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=3, ncols=3, nnz=0,
                rowidx=np.zeros(0, np.int64), colidx=np.zeros(0, np.int64),
                a=np.zeros(0))
    p = tmp_path / "empty.mtx"
    write_mtx(p, m)
    assert read_mtx(p).nnz == 0
    (tmp_path / "v.mtx").write_text(
        "%%MatrixMarket vector coordinate real general\n5 2\n1 3.5\n4 -1.25\n")
    mv = read_mtx(tmp_path / "v.mtx")
    np.testing.assert_array_equal(mv.rowidx, [0, 3])
    (tmp_path / "p.mtx").write_text(
        "%%MatrixMarket matrix coordinate pattern symmetric\n3 3 2\n1 1\n3 2\n")
    mp_ = read_mtx(tmp_path / "p.mtx")
    np.testing.assert_array_equal(mp_.a, [1.0, 1.0])
    (tmp_path / "c.mtx").write_text(
        "%%MatrixMarket matrix coordinate complex general\n2 2 1\n1 2 1.5 -0.5\n")
    mc = read_mtx(tmp_path / "c.mtx")
    assert mc.a.dtype == np.complex128 and mc.a[0] == 1.5 - 0.5j
    (tmp_path / "h.mtx").write_text(
        "%%MatrixMarket matrix coordinate real general\n% c1\n%c2\n2 2 1\n1 1 2.0\n")
    mh = read_mtx(tmp_path / "h.mtx")
    assert len(mh.comments) == 2 and mh.a[0] == 2.0
