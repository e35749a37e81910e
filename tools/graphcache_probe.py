"""Interleaved A/B: classic daypx-fold vs unfolded (queen G=111)."""
import sys
from pathlib import Path
import numpy as np, torch, statistics
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from acg_amd.gen import queen_like_spec
from acg_amd.gen.device_slab import device_stencil_slab
from acg_amd.solvers.hip import CGSolverHIP

S = device_stencil_slab(111, 111, 111, queen_like_spec(3), 0, 1, "cuda:0")
rng = np.random.default_rng(0)
b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
x0 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
sol = CGSolverHIP(S, device="cuda:0")
STEPS = 300

def t(**kw):
    r = sol.solve(b, x0.clone(), maxits=STEPS, res_rtol=0.0, **kw)
    return r.tsolve / STEPS * 1e6

arms = {"fold": lambda: t(), "nofold": lambda: t(fold_daypx=False)}
for f in arms.values(): f()
res = {k: [] for k in arms}
for _ in range(5):
    for k, f in arms.items():
        res[k].append(f())
for k, v in res.items():
    print(f"{k:8s} med {statistics.median(v):7.1f}  min {min(v):7.1f}")
