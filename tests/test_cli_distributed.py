"""End-to-end distributed CLI run: torchrun x2 CPU ranks (gloo), full
read -> partition -> scatter -> solve -> gather -> write pipeline
(the exact launch mode the driver uses for multi-GPU, minus the GPUs)."""

import subprocess
import sys
from pathlib import Path

import numpy as np

from acg_amd.gen import STENCIL_5PT_2D, stencil_global
from acg_amd.io.mtx import MtxFile, read_mtx, write_mtx

REPO = Path(__file__).resolve().parent.parent


def test_cli_torchrun_2ranks(tmp_path):
    A = stencil_global(16, 16, 1, STENCIL_5PT_2D)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    apath = tmp_path / "A.mtx"
    write_mtx(apath, m)
    out = tmp_path / "x.mtx"
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29731", "-m", "acg_amd.cli", str(apath),
           "--solver", "cpu", "--comm", "gloo", "--manufactured-solution",
           "--max-iterations", "2000", "--residual-rtol", "1e-10"]
    r = subprocess.run(cmd, capture_output=True, text=True, cwd=REPO,
                       timeout=300)
    assert r.returncode == 0, r.stderr[-3000:]
    assert "manufactured solution" in r.stderr
    assert "per-rank halo traffic" in r.stderr
    # stdout: the gathered solution as mtx array, written once by rank 0
    # (gloo chatter may precede it)
    lines = [ln for ln in r.stdout.splitlines() if ln.strip()]
    hdr = next(i for i, ln in enumerate(lines)
               if ln.startswith("%%MatrixMarket matrix array real general"))
    assert int(lines[hdr + 1].split()[0]) == A.n
    xs = np.array([float(v) for v in lines[hdr + 2:hdr + 2 + A.n]])
    # solution solves the manufactured system: || x - x* || reported small
    err_line = [ln for ln in r.stderr.splitlines() if "manufactured" in ln][0]
    err = float(err_line.split("=")[1].split("(")[0])
    assert err < 1e-7, err_line
    assert len(xs) == A.n


def test_cli_torchrun_2ranks_b_x0_jacobi_scale(tmp_path):
    """Chunked b/x0 row scatter + distributed jacobi scaling + pipelined
    solve through the REAL torchrun launch path."""
    A = stencil_global(14, 14, 1, STENCIL_5PT_2D)
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    apath = tmp_path / "A.mtx"
    write_mtx(apath, m)
    rng = np.random.default_rng(5)
    bvals = rng.standard_normal(A.n)
    write_mtx(tmp_path / "b.mtx",
              MtxFile(object="matrix", format="array", field_="real",
                      symmetry="general", nrows=A.n, ncols=1, nnz=A.n,
                      a=bvals))
    write_mtx(tmp_path / "x0.mtx",
              MtxFile(object="matrix", format="array", field_="real",
                      symmetry="general", nrows=A.n, ncols=1, nnz=A.n,
                      a=rng.standard_normal(A.n) * 0.1))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29732", "-m", "acg_amd.cli", str(apath),
           str(tmp_path / "b.mtx"), str(tmp_path / "x0.mtx"),
           "--solver", "cpu-pipelined", "--comm", "gloo", "--jacobi-scale",
           "--max-iterations", "3000", "--residual-rtol", "1e-10"]
    r = subprocess.run(cmd, capture_output=True, text=True, cwd=REPO,
                       timeout=300)
    assert r.returncode == 0, r.stderr[-3000:]
    lines = [ln for ln in r.stdout.splitlines() if ln.strip()]
    hdr = next(i for i, ln in enumerate(lines)
               if ln.startswith("%%MatrixMarket matrix array real general"))
    xs = np.array([float(v) for v in lines[hdr + 2:hdr + 2 + A.n]])
    import scipy.sparse.linalg as spla

    x_ref = spla.spsolve(A.to_scipy_full().tocsc(), bvals)
    np.testing.assert_allclose(xs, x_ref, rtol=1e-5, atol=1e-7)
