#!/usr/bin/env python3
"""Single-GPU probe of ONE rank's share of a multi-GPU configuration:
generate rank R of N's slab on this GPU and time solver iterations on it
(comm=None -> the halo is a no-op, so this measures per-rank memory
footprint + kernel throughput for capacity planning, NOT a distributed
solve).  Used to validate the BASELINE config-5 sizing (2048^3 / 8 GPUs)
on the 1-GPU pool.

    python tools/slab_probe.py --grid 2048 --ranks 8 [--matfree] [--steps 10]
"""

import argparse
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--grid", type=int, default=2048)
    ap.add_argument("--ranks", type=int, default=8)
    ap.add_argument("--rank", type=int, default=0)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--matfree", action="store_true")
    args = ap.parse_args()

    from acg_amd.gen import STENCIL_7PT_3D
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    G = args.grid
    t0 = time.perf_counter()
    S = device_stencil_slab(G, G, G, dict(STENCIL_7PT_3D), args.rank,
                            args.ranks, "cuda:0",
                            operator=not args.matfree)
    torch.cuda.synchronize()
    tgen = time.perf_counter() - t0
    free, total = torch.cuda.mem_get_info()
    print(f"slab rank {args.rank}/{args.ranks} of {G}^3: "
          f"{S.nowned:,} owned rows, {S.nnzA + S.nnzO:,} nnz, "
          f"generated in {tgen:.2f}s, "
          f"GPU mem used {(total - free) / 2**30:.1f} GiB", flush=True)
    solver = CGSolverHIP(S, device="cuda:0", matfree=args.matfree)
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")
    solver.solve_pipelined(b, x.clone(), maxits=3, res_rtol=0.0)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    solver.solve_pipelined(b, x, maxits=args.steps, res_rtol=0.0)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    free, total = torch.cuda.mem_get_info()
    print(f"per-rank iteration: {dt * 1e3:.2f} ms "
          f"({'matfree' if args.matfree else 'assembled'}; "
          f"peak GPU mem {(total - free) / 2**30:.1f} GiB)", flush=True)


if __name__ == "__main__":
    main()
