import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from acg_amd.gen.irregular import powerlaw_spd
from acg_amd.ops import gpu_ops
from acg_amd.part import extract_subdomains, partition_rows
from acg_amd.solvers.hip import CGSolverHIP

dev = torch.device("cuda", 0)
A = powerlaw_spd(50_000, mean_nnz=35, seed=11)
S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
rng = np.random.default_rng(1)
b_np = rng.standard_normal(S.nowned)
b = torch.from_numpy(b_np).to(dev)
Afull = A.to_scipy_full()
for fmt in ("hybrid", "binned", "csr"):
    solver = CGSolverHIP(S, device=dev, force_format=fmt)
    x = torch.randn(S.nowned + S.nghost, dtype=torch.float64, device=dev)
    y0 = torch.zeros(S.nowned, dtype=torch.float64, device=dev)
    solver._spmv_overlapped(x, y0, fuse_dotslot=gpu_ops.S_PT)
    torch.cuda.synchronize()
    ref = y0.clone()
    d0 = float(solver.scal[gpu_ops.S_PT])
    spmv_det = True
    dot_det = True
    for _ in range(5):
        y = torch.zeros_like(y0)
        solver._spmv_overlapped(x, y, fuse_dotslot=gpu_ops.S_PT)
        torch.cuda.synchronize()
        spmv_det &= bool(torch.equal(y, ref))
        dot_det &= float(solver.scal[gpu_ops.S_PT]) == d0
    rn = []
    for rep in range(4):
        s2 = CGSolverHIP(S, device=dev, force_format=fmt)
        xx = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device=dev)
        s2.solve(b, xx, maxits=120, res_rtol=0.0)
        rn.append(float(np.linalg.norm(
            b_np - Afull @ xx[:S.nowned].cpu().numpy())))
    vals = ", ".join(f"{v:.6g}" for v in rn)
    print(f"{fmt:7s} spmv_det={spmv_det} dot_det={dot_det} rnorms=[{vals}]",
          flush=True)
