import sys, time
from pathlib import Path
import numpy as np, torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from acg_amd.gen.irregular import powerlaw_spd
from acg_amd.part import extract_subdomains, partition_rows
from acg_amd.solvers.hip import CGSolverHIP

dev = torch.device("cuda", 0)
A = powerlaw_spd(1_000_000, mean_nnz=40, seed=12345)
S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
solver = CGSolverHIP(S, device=dev)
rng = np.random.default_rng(1)
b = torch.from_numpy(rng.standard_normal(S.nowned)).to(dev)
for name, fn in (("classic", solver.solve),
                 ("pipelined", solver.solve_pipelined),
                 ("jacobi-pcg", solver.solve_jacobi)):
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device=dev)
    fn(b, x.clone(), maxits=10, res_rtol=0.0)  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    r = fn(b, x, maxits=20000, res_rtol=1e-8)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{name:11s} its={r.niterations:5d} converged={r.converged} "
          f"tts={dt:7.3f}s ({dt/max(r.niterations,1)*1e6:6.1f} us/it)",
          flush=True)
