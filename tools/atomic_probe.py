#!/usr/bin/env python3
"""Measure sustained global_atomic_add_f64 throughput at SpMV-like target
distributions -- the go/no-go number for a symmetric-storage SpMV (half
the value bytes, transpose half scattered with HW atomics).

Patterns: banded (Queen-like transpose scatter: targets within +-band of
a moving row window, ~40 hits/row) and uniform-random (worst case).
"""

from __future__ import annotations

import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def run(K, y, idx, v, mode, reps=5):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        K.atomic_probe(y.data_ptr(), idx.data_ptr(), v.data_ptr(),
                       idx.numel(), mode,
                       torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main() -> int:
    from acg_amd.ops import gpu_ops

    K = gpu_ops.K
    dev = torch.device("cuda", 0)
    n = 4_100_000
    nnz = 160_000_000
    rng = np.random.default_rng(0)
    y = torch.zeros(n, dtype=torch.float64, device=dev)
    v = torch.randn(nnz, dtype=torch.float64, device=dev)
    for name, gen in (
        ("banded", lambda: (np.repeat(np.arange(nnz // 40, dtype=np.int64),
                                      40) * n // (nnz // 40)
                            + rng.integers(-2000, 2000, nnz)) % n),
        ("uniform", lambda: rng.integers(0, n, nnz)),
    ):
        idx = torch.from_numpy(gen().astype(np.int32)).to(dev)
        for mode, label in ((2, "gather-read"), (1, "plain-store"),
                            (0, "atomic-add")):
            run(K, y, idx, v, mode, reps=1)  # warmup
            dt = run(K, y, idx, v, mode)
            print(f"{name:8s} {label:12s} {nnz / dt / 1e9:7.2f} Gop/s "
                  f"({dt * 1e3:7.2f} ms for {nnz / 1e6:.0f}M)", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
