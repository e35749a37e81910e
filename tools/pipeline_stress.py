"""Full-pipeline stress: random irregular families through file IO ->
assembly -> partition (all methods) -> formats -> GPU solve -> oracle."""
import subprocess, sys, tempfile
from pathlib import Path
import numpy as np
from pathlib import Path as _P
sys.path.insert(0, str(_P(__file__).resolve().parent.parent))
from acg_amd.gen.irregular import powerlaw_spd
from acg_amd.io.mtx import MtxFile, write_mtx

rng = np.random.default_rng(99)
tmp = Path(tempfile.mkdtemp())
for t in range(6):
    n = int(rng.integers(2000, 30000))
    A = powerlaw_spd(n, mean_nnz=float(rng.uniform(8, 50)),
                     alpha=float(rng.uniform(2.05, 3.0)),
                     clip=int(rng.integers(16, 2000)),
                     locality=float(rng.uniform(20, n)),
                     seed=int(rng.integers(1 << 30)))
    rows = np.repeat(np.arange(A.n), np.diff(A.rowptr))
    m = MtxFile(object="matrix", format="coordinate", field_="real",
                symmetry="symmetric", nrows=A.n, ncols=A.n,
                nnz=A.nnz_stored, rowidx=rows, colidx=A.colidx, a=A.vals)
    p = tmp / f"A{t}.mtx"
    write_mtx(p, m)
    solver = ["acg", "acg-pipelined"][t % 2]
    method = ["auto", "ml", "rgb", "block"][t % 4]
    r = subprocess.run([sys.executable, "-m", "acg_amd.cli", str(p),
                        "--solver", solver, "--partition-method", method,
                        "--manufactured-solution", "--max-iterations", "4000",
                        "--residual-rtol", "1e-8", "-q"],
                       capture_output=True, text=True, cwd="/root/repo",
                       timeout=300)
    tailerr = r.stderr.strip().splitlines()[-1] if r.stderr.strip() else ""
    print(f"t{t}: n={n} {solver}/{method} rc={r.returncode} | {tailerr[:90]}",
          flush=True)
    assert r.returncode == 0, r.stderr[-1500:]
print("PIPELINE STRESS PASS")
