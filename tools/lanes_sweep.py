#!/usr/bin/env python3
"""CSR-vector lane-count sweep over fixed row lengths: grounds the
pick_lanes / bin thresholds (~6 nnz per lane rule) in measurement."""

import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from acg_amd.ops import gpu_ops  # noqa: E402

dev = torch.device("cuda", 0)
rng = np.random.default_rng(0)
target_nnz = 30_000_000
print(f"{'len':>5} " + " ".join(f"L={l:<3d}" for l in (4, 8, 16, 32, 64))
      + "  best")
for L in (4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256, 512):
    n = max(target_nnz // L, 1024)
    rowptr = np.arange(0, (n + 1) * L, L, dtype=np.int64)
    cols = rng.integers(0, n, n * L).astype(np.int32)
    # sort within rows for realism
    cols = np.sort(cols.reshape(n, L), axis=1).ravel()
    vals = rng.standard_normal(n * L)
    rp = torch.from_numpy(rowptr).to(dev)
    ci = torch.from_numpy(cols).to(dev)
    vv = torch.from_numpy(vals).to(dev)
    x = torch.randn(n, dtype=torch.float64, device=dev)
    y = torch.zeros(n, dtype=torch.float64, device=dev)
    times = {}
    for lanes in (4, 8, 16, 32, 64):
        gpu_ops.spmv(rp, ci, vv, x, y, lanes=lanes)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            gpu_ops.spmv(rp, ci, vv, x, y, lanes=lanes)
        torch.cuda.synchronize()
        times[lanes] = (time.perf_counter() - t0) / 5 * 1e6
    best = min(times, key=times.get)
    print(f"{L:5d} " + " ".join(f"{times[l]:5.0f}" for l in (4, 8, 16, 32, 64))
          + f"  -> {best}", flush=True)
