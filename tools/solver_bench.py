#!/usr/bin/env python3
"""Same-process solver A/B harness (box-to-box variance on the gpurun pool
is ~±8%, so solver comparisons are only valid interleaved in ONE process).

    python tools/solver_bench.py [--grid 111] [--config queen|poisson7]
                                 [--steps 200] [--reps 5]

Interleaves classic / pipelined (/ device for small systems) solves and
prints per-solver medians.
"""

import argparse
import statistics
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", choices=["queen", "poisson7"], default="queen")
    ap.add_argument("--grid", type=int, default=None)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--reps", type=int, default=5)
    args = ap.parse_args()

    from acg_amd.gen import STENCIL_7PT_3D, queen_like_spec
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.solvers.hip import CGSolverHIP

    if args.config == "queen":
        spec, G = queen_like_spec(3), args.grid or 111
    else:
        spec, G = dict(STENCIL_7PT_3D), args.grid or 512
    S = device_stencil_slab(G, G, G, spec, 0, 1, "cuda:0")
    solver = CGSolverHIP(S, device="cuda:0")
    rng = np.random.default_rng(0)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).cuda()
    x = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device="cuda")

    arms = {
        "classic": lambda: solver.solve(b, x.clone(), maxits=args.steps,
                                        res_rtol=0.0),
        "pipelined": lambda: solver.solve_pipelined(b, x.clone(),
                                                    maxits=args.steps,
                                                    res_rtol=0.0),
    }
    if getattr(S, "mf_tables", None) is not None:
        mfsolver = CGSolverHIP(S, device="cuda:0", matfree=True)
        arms["matfree-pipelined"] = lambda: mfsolver.solve_pipelined(
            b, x.clone(), maxits=args.steps, res_rtol=0.0)
        arms["matfree-classic"] = lambda: mfsolver.solve(
            b, x.clone(), maxits=args.steps, res_rtol=0.0)
    if solver.can_megafuse:
        arms["pipelined-nomega"] = lambda: solver.solve_pipelined(
            b, x.clone(), maxits=args.steps, res_rtol=0.0, megafuse=False)
        arms["pipelined-mega"] = lambda: solver.solve_pipelined(
            b, x.clone(), maxits=args.steps, res_rtol=0.0, megafuse=True)
    if S.nowned <= 1_100_000 and solver.sell is not None:
        arms["device"] = lambda: solver.solve_device(b, x.clone(),
                                                     maxits=args.steps,
                                                     res_rtol=0.0)
    for fn in arms.values():  # warmup
        fn()
    results = {k: [] for k in arms}
    for _ in range(args.reps):
        for name, fn in arms.items():
            r = fn()
            results[name].append(r.tsolve / args.steps * 1e6)
    print(f"config={args.config} G={G} rows={S.nowned} steps={args.steps} "
          f"reps={args.reps} (interleaved)")
    for name, v in results.items():
        med = statistics.median(v)
        print(f"  {name:18s} {med:9.2f} us/it  (min {min(v):.2f}, max {max(v):.2f})")


if __name__ == "__main__":
    main()
