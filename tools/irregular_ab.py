#!/usr/bin/env python3
"""A/B the matA operator formats on a power-law SPD system (one process,
interleaved reps -- box-to-box variance is ±8%, intra-process ±0.1%).

Usage: python tools/irregular_ab.py [--rows N] [--mean-nnz M] [--reps R]
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--mean-nnz", type=float, default=40.0)
    ap.add_argument("--alpha", type=float, default=2.2)
    ap.add_argument("--clip", type=int, default=8192)
    ap.add_argument("--reps", type=int, default=5)
    ap.add_argument("--its", type=int, default=30)
    ap.add_argument("--formats", default="hybrid,binned,csr")
    args = ap.parse_args()

    from acg_amd.gen.irregular import degree_stats, powerlaw_spd
    from acg_amd.part import extract_subdomains, partition_rows
    from acg_amd.solvers.hip import CGSolverHIP

    dev = torch.device("cuda", 0)
    t0 = time.time()
    A = powerlaw_spd(args.rows, mean_nnz=args.mean_nnz, alpha=args.alpha,
                     clip=args.clip, seed=12345)
    print(f"# generated in {time.time() - t0:.1f}s: {degree_stats(A)}",
          flush=True)
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    del A

    fmts = args.formats.split(",")
    solvers = {}
    for f in fmts:
        t0 = time.time()
        solvers[f] = CGSolverHIP(S, device=dev, force_format=f)
        print(f"# {f}: setup {time.time() - t0:.1f}s", flush=True)

    rng = np.random.default_rng(1)
    b = torch.from_numpy(rng.standard_normal(S.nowned)).to(dev)
    x0 = torch.zeros(S.nowned + S.nghost, dtype=torch.float64, device=dev)

    # warmup each arm
    for f in fmts:
        solvers[f].solve(b, x0.clone(), maxits=5, res_rtol=0.0)
    torch.cuda.synchronize(dev)

    times: dict[str, list] = {f: [] for f in fmts}
    for rep in range(args.reps):
        for f in fmts:
            x = x0.clone()
            torch.cuda.synchronize(dev)
            t0 = time.perf_counter()
            solvers[f].solve(b, x, maxits=args.its, res_rtol=0.0)
            torch.cuda.synchronize(dev)
            times[f].append((time.perf_counter() - t0) / args.its * 1e6)
    nnz = S.nnzA
    for f in fmts:
        med = float(np.median(times[f]))
        # effective traffic: vals 8B + idx 4B per nnz + 16B/row vectors
        gbs = (nnz * 12.0 + 16.0 * S.nowned) / (med * 1e-6) / 1e9
        print(f"{f:8s} median {med:8.1f} us/it  (reps "
              f"{['%.1f' % t for t in times[f]]})  ~{gbs:.0f} GB/s eff",
              flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
