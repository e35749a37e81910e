#!/usr/bin/env python3
"""A/B the device-CG grid barrier: flat parity counters vs hierarchical
(per-XCD then global).  Interleaved in one process; several system sizes
so both the latency-bound (small grid) and full-grid (1024-block) regimes
are covered.

Usage: python tools/devcg_barrier_ab.py [--reps 5] [--its 200]
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
    ap.add_argument("--reps", type=int, default=5)
    ap.add_argument("--its", type=int, default=200)
    ap.add_argument("--grids", default="16,35,64,100")
    args = ap.parse_args()

    from acg_amd.gen import STENCIL_27PT_3D
    from acg_amd.gen.device_slab import device_stencil_slab
    from acg_amd.ops import gpu_ops as ops

    dev = torch.device("cuda", 0)
    for G in (int(g) for g in args.grids.split(",")):
        S = device_stencil_slab(G, G, G, dict(STENCIL_27PT_3D), 0, 1, dev)
        n = S.nowned
        sellptr, cols, vals = S.A_sell
        b = torch.randn(n, dtype=torch.float64, device=dev)
        x0 = torch.zeros(n + S.nghost, dtype=torch.float64, device=dev)
        scal = ops.alloc_scalars(dev)
        partials = ops.alloc_partials(dev)
        out2 = torch.zeros(2, dtype=torch.int32, device=dev)
        bar = torch.zeros(ops.BAR_STATE_WORDS, dtype=torch.int32, device=dev)
        r = torch.zeros(n, dtype=torch.float64, device=dev)
        t = torch.zeros(n, dtype=torch.float64, device=dev)
        p = torch.zeros(n + S.nghost, dtype=torch.float64, device=dev)

        def run(hier):
            x = x0.clone()
            bar.zero_()
            torch.cuda.synchronize(dev)
            t0 = time.perf_counter()
            grid = ops.cg_device(sellptr, cols, vals, n, b, x, r, p, t,
                                 scal, partials, out2, bar, args.its,
                                 0.0, 0.0, hier=hier)
            torch.cuda.synchronize(dev)
            el = (time.perf_counter() - t0) / args.its * 1e6
            assert int(out2[1]) >= 0, "barrier timeout"
            return el, grid

        # warmup both
        run(False), run(True)
        tf, th = [], []
        grid = 0
        for _ in range(args.reps):
            e, grid = run(False)
            tf.append(e)
            e, grid = run(True)
            th.append(e)
        mf, mh = float(np.median(tf)), float(np.median(th))
        print(f"G={G:4d} n={n:9,} grid={grid:5d} blocks: "
              f"flat {mf:7.2f} us/it, hier {mh:7.2f} us/it "
              f"({(mf - mh):+.2f})", flush=True)
        del S, sellptr, cols, vals
        torch.cuda.empty_cache()
    return 0


if __name__ == "__main__":
    sys.exit(main())
