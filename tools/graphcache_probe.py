"""Graph-vs-eager defaults: classic queen, mega assembled 384, mega matfree 512."""
import sys
from pathlib import Path
import numpy as np, torch, statistics
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from acg_amd.gen import STENCIL_7PT_3D, queen_like_spec
from acg_amd.gen.device_slab import device_stencil_slab
from acg_amd.solvers.hip import CGSolverHIP

def arms_for(tag, S, steps, matfree=False, classic=False):
    sol = CGSolverHIP(S, device="cuda:0", matfree=matfree)
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    x0 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    def t(g):
        fn = sol.solve if classic else sol.solve_pipelined
        r = fn(b, x0.clone(), maxits=steps, res_rtol=0.0, use_graph=g)
        return r.tsolve / steps * 1e6
    return {f"{tag}-graph": lambda: t(True), f"{tag}-eager": lambda: t(False)}

arms = {}
Sq = device_stencil_slab(111, 111, 111, queen_like_spec(3), 0, 1, "cuda:0")
arms.update(arms_for("classicQ", Sq, 300, classic=True))
Sp = device_stencil_slab(384, 384, 384, dict(STENCIL_7PT_3D), 0, 1, "cuda:0")
arms.update(arms_for("megaP384", Sp, 150))
arms.update(arms_for("mfP384", Sp, 150, matfree=True))
for f in arms.values(): f()
res = {k: [] for k in arms}
for _ in range(5):
    for k, f in arms.items():
        res[k].append(f())
for k, v in res.items():
    print(f"{k:16s} med {statistics.median(v):8.1f}  min {min(v):8.1f}")
