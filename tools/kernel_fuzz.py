"""One-off randomized GPU kernel fuzz: spmv/spmv_sell/spmv_binned/split
vs torch_ref over random shapes (beyond the fixed-seed suite)."""
import sys, numpy as np, torch
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from acg_amd.ops import gpu_ops, torch_ref
from acg_amd.gen.irregular import powerlaw_spd
from acg_amd.part import extract_subdomains, partition_rows

dev = torch.device("cuda", 0)
rng = np.random.default_rng(12345)
fails = 0
for trial in range(20):
    n = int(rng.integers(50, 30000))
    mean = float(rng.uniform(4, 60))
    A = powerlaw_spd(n, mean_nnz=mean, clip=int(rng.integers(32, 4096)),
                     seed=int(rng.integers(1 << 30)))
    S = extract_subdomains(A, partition_rows(A, 1), 1)[0]
    rowptr = torch.from_numpy(S.A_rowptr)
    colidx = torch.from_numpy(S.A_colidx.astype(
        np.int64 if trial % 3 == 0 else np.int32))
    vals = torch.from_numpy(S.A_vals)
    x = torch.randn(n, dtype=torch.float64)
    y_ref = torch.zeros(n, dtype=torch.float64)
    torch_ref.spmv(rowptr, colidx, vals, x, y_ref)
    xg = x.to(dev)
    # 1. CSR-vector, random lanes
    lanes = int(rng.choice([4, 8, 16, 32, 64]))
    yg = torch.zeros(n, dtype=torch.float64, device=dev)
    gpu_ops.spmv(rowptr.to(dev), colidx.to(dev), vals.to(dev), xg, yg, lanes=lanes)
    e1 = float((yg.cpu() - y_ref).abs().max())
    # 2. split hybrid with random cut
    cut = int(rng.integers(16, 400))
    sp_, cols, svals, perm, rowlist, bins = gpu_ops.build_sellcsr_hybrid(
        S.A_rowptr, S.A_colidx, S.A_vals, cut=cut)
    yh = torch.zeros(n, dtype=torch.float64, device=dev)
    if sp_ is not None:
        gpu_ops.spmv_sell(torch.from_numpy(sp_).to(dev),
                          torch.from_numpy(cols).to(dev),
                          torch.from_numpy(svals).to(dev), n, xg, yh,
                          perm=torch.from_numpy(perm).to(dev))
    if len(rowlist):
        gpu_ops.spmv_binned(rowptr.to(dev), colidx.to(dev), vals.to(dev),
                            torch.from_numpy(rowlist).to(dev), bins, xg, yh)
    e2 = float((yh.cpu() - y_ref).abs().max())
    tol = 1e-9 * max(1.0, float(y_ref.abs().max()))
    ok = e1 < tol and e2 < tol
    fails += not ok
    print(f"trial {trial:2d}: n={n:6d} mean={mean:5.1f} cut={cut:3d} "
          f"lanes={lanes:2d} e_csr={e1:.2e} e_split={e2:.2e} "
          f"{'OK' if ok else 'FAIL'}", flush=True)
print("FUZZ", "PASS" if fails == 0 else f"{fails} FAILURES")
