#!/usr/bin/env python3
"""SpMV microbenchmark: sweep SELL kernel variants and CSR lane counts on a
Queen-shaped operator.  Run on an MI355X box:

    python tools/spmv_bench.py [--grid 111] [--iters 50]

Prints effective GB/s per variant (bytes = vals + cols + x + y once each).
"""

import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--grid", type=int, default=111)
    ap.add_argument("--dof", type=int, default=3)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()

    from acg_amd.gen import queen_like_spec, stencil_local_slab
    from acg_amd.ops import gpu_ops
    from acg_amd.ops.torch_ref import sell_from_csr

    G = args.grid
    S = stencil_local_slab(G, G, G, queen_like_spec(args.dof), 0, 1)
    dev = torch.device("cuda:0")
    n = S.nowned
    nnz = S.nnzA
    print(f"rows={n} nnz={nnz} mean nnz/row={nnz/n:.1f}", flush=True)
    rowptr = torch.from_numpy(S.A_rowptr).to(dev)
    colidx = torch.from_numpy(S.A_colidx).to(dev)
    vals = torch.from_numpy(S.A_vals).to(dev)
    sellptr_np, scols_np, svals_np = sell_from_csr(S.A_rowptr, S.A_colidx, S.A_vals)
    sellptr = torch.from_numpy(sellptr_np).to(dev)
    scols = torch.from_numpy(scols_np).to(dev)
    svals = torch.from_numpy(svals_np).to(dev)
    pad = (int(sellptr_np[-1]) - nnz) / nnz
    print(f"SELL padding waste: {pad*100:.2f}%", flush=True)

    x = torch.randn(n, dtype=torch.float64, device=dev)
    y = torch.zeros(n, dtype=torch.float64, device=dev)
    yref = torch.zeros(n, dtype=torch.float64, device=dev)

    bytes_eff = nnz * (8 + S.A_colidx.dtype.itemsize) + 8 * n * 2

    def timeit(fn, label, check=True):
        fn()  # warmup + correctness snapshot
        torch.cuda.synchronize()
        if check:
            err = (y - yref).abs().max().item()
            denom = yref.abs().max().item()
            assert err <= 1e-9 * max(denom, 1.0), f"{label}: wrong result ({err})"
        t0 = time.perf_counter()
        for _ in range(args.iters):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(f"{label:34s} {dt*1e6:9.1f} us   {bytes_eff/dt/1e9:8.1f} GB/s", flush=True)

    # reference result from CSR lanes=16
    gpu_ops.spmv(rowptr, colidx, vals, x, yref, lanes=16)
    torch.cuda.synchronize()

    for lanes in (8, 16, 32):
        timeit(lambda l=lanes: gpu_ops.spmv(rowptr, colidx, vals, x, y, lanes=l),
               f"csr-vector lanes={lanes}")
    for variant in range(8):
        tags = [t for b, t in ((1, "NT"), (2, "SWZ"), (4, "U8")) if variant & b]
        timeit(lambda v=variant: gpu_ops.spmv_sell(sellptr, scols, svals, n, x, y,
                                                   variant=v),
               f"sell variant={variant} [{'+'.join(tags) or 'base'}]")

    # Block-SELL (if the operator has dof-blocks)
    from acg_amd.ops.torch_ref import bsell_from_csr

    out = bsell_from_csr(S.A_rowptr, S.A_colidx, S.A_vals, args.dof)
    if out is not None and out[3] >= 0.75:
        bptr = torch.from_numpy(out[0]).to(dev)
        bcol = torch.from_numpy(out[1]).to(dev)
        bvals = torch.from_numpy(out[2]).to(dev)
        print(f"BSELL dof={args.dof} density={out[3]:.3f}")
        timeit(lambda: gpu_ops.spmv_bsell(bptr, bcol, bvals, n // args.dof,
                                          args.dof, x, y),
               f"bsell dof={args.dof} (pair-major)")


if __name__ == "__main__":
    main()
