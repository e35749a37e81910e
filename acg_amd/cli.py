"""acg-compatible CLI driver (reference: hip/acg-hip.c).

Usage:  python -m acg_amd.cli [OPTION..] A.mtx [b.mtx] [x0.mtx]

Mirrors the reference option surface (acg-hip.c:312-374) and its pipeline
(§3.1 of SURVEY.md): read -> partition -> scatter -> solve -> report ->
write solution to stdout.  Multi-rank runs come from torchrun (one process
per GPU, RCCL) instead of mpirun; --comm gloo supports CPU multi-process.

Solvers: acg (GPU classic CG), acg-pipelined (GPU pipelined),
cpu / cpu-pipelined (host torch), scipy / scipy-pipelined (independent
oracle, the role PETSc KSPCG plays in the reference -- PETSc is not in
this image).
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import numpy as np

from . import __version__
from .utils.errors import AcgError, ErrCode, collective_raise
from .utils.numfmt import parse_numfmt


def make_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="acg-amd",
        description="MI355X-native distributed CG solver (aCG-compatible driver)")
    p.add_argument("A", help="matrix Market file for the SPD matrix A")
    p.add_argument("b", nargs="?", default=None, help="optional right-hand side")
    p.add_argument("x0", nargs="?", default=None, help="optional initial guess")
    p.add_argument("-z", "--gzip", action="store_true", help="gzip-compressed input")
    p.add_argument("--binary", action="store_true",
                   help="matrix file is in binary Matrix Market format (mtx2bin)")
    p.add_argument("--idxsize", type=int, choices=(32, 64), default=64,
                   help="index width of binary files (reference acgidx_t)")
    p.add_argument("--partition", metavar="FILE", default=None,
                   help="precomputed partition vector (mtx integer array)")
    p.add_argument("--binary-partition", action="store_true",
                   help="partition file is in binary Matrix Market format")
    p.add_argument("--partition-method",
                   choices=("block", "rgb", "ml", "auto"), default="auto")
    p.add_argument("--seed", type=int, default=0, help="partitioner seed")
    p.add_argument("--solver", default=None,
                   choices=("acg", "acg-pipelined", "acg-device",
                            "acg-jacobi", "cpu", "cpu-pipelined",
                            "cpu-jacobi", "scipy", "scipy-pipelined",
                            "petsc", "petsc-pipelined"),
                   help="default: acg on GPU, cpu otherwise.  petsc[-pipelined] "
                        "are accepted for aCG compatibility and run the scipy "
                        "oracle (PETSc is not available in this image)")
    p.add_argument("--max-iterations", type=int, default=100)
    p.add_argument("--diff-atol", type=float, default=0.0)
    p.add_argument("--diff-rtol", type=float, default=0.0)
    p.add_argument("--residual-atol", type=float, default=0.0)
    p.add_argument("--residual-rtol", type=float, default=1e-9)
    p.add_argument("--epsilon", type=float, default=0.0,
                   help="diagonal shift added to A")
    p.add_argument("--warmup", type=int, default=10,
                   help="untimed warmup iterations before the timed solve "
                        "(reference default 10, acg-hip.c:480-486)")
    p.add_argument("--comm", choices=("none", "rccl", "gloo"), default=None,
                   help="default: rccl when WORLD_SIZE>1 and GPUs exist")
    p.add_argument("--jacobi-scale", action="store_true",
                   help="diagonal (Jacobi) preconditioning via symmetric "
                        "system scaling -- composes with EVERY solver "
                        "(beyond reference; tolerances then apply to the "
                        "preconditioned residual norm)")
    p.add_argument("--manufactured-solution", action="store_true",
                   help="b := A x* for random x*; report error norms")
    p.add_argument("--numfmt", default=None, help="printf format for output values")
    p.add_argument("--output-comm-matrix", action="store_true",
                   help="print the rank x rank halo send-count matrix to "
                        "standard output (reference acg-hip.c:1683-1742)")
    p.add_argument("--no-output-comm-matrix", dest="output_comm_matrix",
                   action="store_false", help=argparse.SUPPRESS)
    p.add_argument("--profile", action="store_true",
                   help="per-op hipEvent timing (reference ACG_ENABLE_PROFILING)")
    p.add_argument("-q", "--quiet", action="store_true",
                   help="do not write the solution vector to stdout")
    p.add_argument("-v", "--verbose", action="count", default=0)
    p.add_argument("--version", action="version",
                   version=f"acg-amd {__version__} (gfx950 HIP kernels, RCCL)")
    return p


def _vector_from_mtx(m, n: int, what: str) -> np.ndarray:
    """Dense fp64 vector from an MtxFile: array format positionally, or
    coordinate (sparse) format scattered through rowidx -- never assign
    coordinate values positionally (a permuted file would silently produce
    a wrong b; reference validates at acg-hip.c:1786-2021)."""
    vals = np.asarray(m.a, dtype=np.float64)
    if m.format == "array":
        if m.ncols not in (0, 1) and m.nrows * m.ncols != n:
            raise AcgError(ErrCode.INVALID_VALUE,
                           f"{what}: array shape {m.nrows}x{m.ncols} != {n}")
        if len(vals) != n:
            raise AcgError(ErrCode.INVALID_VALUE,
                           f"{what}: length {len(vals)} != matrix rows {n}")
        return vals
    if m.format == "coordinate":
        if m.ncols != 1:
            raise AcgError(ErrCode.INVALID_VALUE,
                           f"{what}: coordinate vector must have 1 column")
        if m.nrows != n:
            raise AcgError(ErrCode.INVALID_VALUE,
                           f"{what}: length {m.nrows} != matrix rows {n}")
        out = np.zeros(n, dtype=np.float64)
        ri = np.asarray(m.rowidx, dtype=np.int64)
        if len(ri) and (ri.min() < 0 or ri.max() >= n):
            raise AcgError(ErrCode.INVALID_VALUE, f"{what}: row index out of range")
        out[ri] = vals
        return out
    raise AcgError(ErrCode.INVALID_FORMAT, f"{what}: unsupported format {m.format!r}")


def main(argv=None) -> int:
    args = make_parser().parse_args(argv)
    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    has_gpu = torch.cuda.is_available()
    commkind = args.comm
    if commkind is None:
        commkind = ("rccl" if has_gpu else "gloo") if world > 1 else "none"
    solver_name = args.solver or ("acg" if has_gpu else "cpu")
    # aCG compatibility: the PETSc baseline role is played by scipy here
    solver_name = {"petsc": "scipy",
                   "petsc-pipelined": "scipy-pipelined"}.get(solver_name,
                                                             solver_name)
    gpu_solver = solver_name.startswith("acg")
    if gpu_solver and not has_gpu:
        print("error: GPU solver requested but no GPU is available", file=sys.stderr)
        return 1

    from .dist.comm import Comm

    device = None
    if has_gpu:
        local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
        device = torch.device("cuda", local_rank % max(torch.cuda.device_count(), 1))
        torch.cuda.set_device(device)
    comm = Comm(commkind, device=device if commkind == "rccl" else None) \
        if commkind != "none" else None
    nparts = comm.size if comm else 1
    verbose = args.verbose and rank == 0

    def log(msg):
        if verbose:
            print(msg, file=sys.stderr, flush=True)

    numfmt = parse_numfmt(args.numfmt) if args.numfmt else None

    # ---- root: read + partition + structure pass (acg-hip.c:1267-1652)
    ex = None
    b_global = None
    x0_global = None
    xsol = None
    n_global = None
    err = None
    try:
        if rank == 0:
            from .core.symcsr import SymCSRMatrix
            from .io.mtx import read_mtx
            from .part import partition_rows, read_partition_file
            from .part.subdomain import SubdomainExtractor

            t0 = time.perf_counter()
            m = read_mtx(args.A, gzipped=args.gzip, binary=args.binary,
                         idxsize=args.idxsize)
            log(f"read {args.A}: {m.nrows}x{m.ncols}, {m.nnz} stored entries "
                f"({time.perf_counter() - t0:.2f}s)")
            A = SymCSRMatrix.from_mtxfile(m)
            if args.partition:
                part = read_partition_file(args.partition, A.n,
                                           binary=args.binary_partition,
                                           gzipped=args.gzip,
                                           idxsize=args.idxsize)
                if int(part.max()) >= nparts:
                    raise AcgError(ErrCode.INVALID_VALUE,
                                   f"partition file has {int(part.max()) + 1} parts, "
                                   f"running with {nparts} ranks")
            else:
                part = partition_rows(A, nparts, seed=args.seed,
                                      method=args.partition_method)
            t0 = time.perf_counter()
            ex = SubdomainExtractor(A, part, nparts, eps=args.epsilon)
            log(f"structure pass for {nparts} subdomains "
                f"({time.perf_counter() - t0:.2f}s)")
            n_global = A.n

            # RHS (reference acg-hip.c:1786-2087)
            if args.manufactured_solution:
                rng = np.random.default_rng(args.seed)
                xsol = rng.standard_normal(A.n)
                xsol /= np.linalg.norm(xsol)
                b_global = A.dsymv(xsol)
            elif args.b:
                # --binary applies to b/x0 too (reference acg-hip.c:1796)
                mb = read_mtx(args.b, gzipped=args.gzip, binary=args.binary,
                              idxsize=args.idxsize)
                b_global = _vector_from_mtx(mb, A.n, "b")
            else:
                b_global = np.ones(A.n, dtype=np.float64)
            if args.x0:
                mx = read_mtx(args.x0, gzipped=args.gzip, binary=args.binary,
                              idxsize=args.idxsize)
                x0_global = _vector_from_mtx(mx, A.n, "x0")
    except Exception as e:  # collective error agreement (acgerrmpi)
        err = e
    collective_raise(comm, err)

    # ---- scatter: parts streamed one at a time, fields as chunked
    # tensors (reference acgsymcsrmatrix_scatter field-by-field with
    # MPI_Send64 chunks, graph.c:1529-1893; never whole-object pickles)
    t0 = time.perf_counter()
    if comm and comm.size > 1:
        S = comm.scatter_systems((lambda p: ex.build(p)) if rank == 0 else None)
        b_local = comm.scatter_rows(
            (lambda p: b_global[ex.owned_globals[p]]) if rank == 0 else None)
        has_x0 = comm.bcast_object(x0_global is not None if rank == 0 else None)
        x0_local = comm.scatter_rows(
            (lambda p: x0_global[ex.owned_globals[p]]) if rank == 0 else None) \
            if has_x0 else None
        n_global = comm.bcast_object(n_global)
    else:
        S = ex.build(0)
        b_local = b_global[S.owned_global]
        x0_local = None if x0_global is None else x0_global[S.owned_global]
    log(f"scattered subdomains ({time.perf_counter() - t0:.2f}s)")
    if args.verbose and args.verbose > 1:
        S.dump(file=sys.stderr)
    if comm and comm.size > 1:
        # collective dry-run audit of the halo pattern: fail loudly at
        # setup (pairing symmetry, ghost-tail global-id agreement) instead
        # of deadlocking in the first exchange
        from .dist.verify import verify_halo

        verify_halo(S, comm)
        log("halo audit passed")

    s_scale = None
    if args.jacobi_scale:
        from .solvers.precond import jacobi_scale_system

        S, s_scale = jacobi_scale_system(S, comm, device)
        b_local = b_local * s_scale
        if x0_local is not None:
            x0_local = x0_local / s_scale
        log("jacobi scaling applied (D^-1/2 A D^-1/2)")

    if args.output_comm_matrix and comm:
        counts = comm.gather_object(
            {int(q): int(c) for q, c in zip(S.halo.recipients, S.halo.sendcounts)})
        if rank == 0:
            # to STDOUT like the reference (before the solution vector)
            print("%%MatrixMarket matrix coordinate integer general")
            entries = [(p, q, c) for p, row in enumerate(counts)
                       for q, c in row.items()]
            print(f"{nparts} {nparts} {len(entries)}")
            for p, q, c in entries:
                print(f"{p + 1} {q + 1} {c}")

    import torch as _t

    b = _t.from_numpy(np.ascontiguousarray(b_local))
    x = _t.zeros(S.nowned + S.nghost, dtype=_t.float64)
    if x0_local is not None:
        x[:S.nowned] = _t.from_numpy(np.ascontiguousarray(x0_local))

    # ---- solve (reference acg-hip.c:2192-2247)
    err = None
    res = None
    solver = None
    try:
        if solver_name in ("scipy", "scipy-pipelined"):
            from .solvers.oracle import solve_scipy

            res, xnp = solve_scipy(S, comm, b.numpy(), x[:S.nowned].numpy(),
                                   maxits=args.max_iterations,
                                   res_rtol=args.residual_rtol,
                                   res_atol=args.residual_atol,
                                   pipelined=solver_name.endswith("pipelined"))
            x[:S.nowned] = _t.from_numpy(xnp)
        elif gpu_solver:
            from .solvers.hip import CGSolverHIP

            t0 = time.perf_counter()
            solver = CGSolverHIP(S, comm=comm, device=device,
                                 profile=args.profile)
            fmt = ("Block-SELL" if solver.bsell is not None else
                   "sigma-SELL" if solver.sell_perm is not None else
                   "SELL" if solver.sell is not None else "CSR-vector")
            log(f"GPU solver init: operator format {fmt} "
                f"({time.perf_counter() - t0:.2f}s)")
            b = b.to(device)
            x = x.to(device)
            diff_requested = args.diff_atol > 0 or args.diff_rtol > 0
            if diff_requested and solver_name != "acg":
                # reference parity: the GPU pipelined/device solvers REJECT
                # diff tolerances (cghip.c:427, 1212) rather than silently
                # iterating to maxits; classic implements them (see
                # CGSolverHIP.solve diff mode)
                raise AcgError(ErrCode.NOT_SUPPORTED,
                               f"--diff-atol/--diff-rtol are not supported "
                               f"by {solver_name} (use acg or cpu)")
            if args.warmup:
                if solver_name == "acg-pipelined":
                    solver.solve_pipelined(b, x.clone(), maxits=args.warmup,
                                           res_rtol=0.0)
                elif solver_name == "acg-jacobi":
                    solver.solve_jacobi(b, x.clone(), maxits=args.warmup,
                                        res_rtol=0.0)
                elif solver_name == "acg-device":
                    # warm the actual cooperative kernel (reference warms the
                    # kernel it will time, cg-kernels-hip.hip:1925-1989)
                    solver.solve_device(b, x.clone(), maxits=args.warmup,
                                        res_rtol=0.0)
                else:
                    solver.solve(b, x.clone(), maxits=args.warmup, res_rtol=0.0)
            if solver_name == "acg-pipelined":
                res = solver.solve_pipelined(b, x, maxits=args.max_iterations,
                                             res_atol=args.residual_atol,
                                             res_rtol=args.residual_rtol)
            elif solver_name == "acg-jacobi":
                res = solver.solve_jacobi(b, x, maxits=args.max_iterations,
                                          res_atol=args.residual_atol,
                                          res_rtol=args.residual_rtol)
            elif solver_name == "acg-device":
                res = solver.solve_device(b, x, maxits=args.max_iterations,
                                          res_atol=args.residual_atol,
                                          res_rtol=args.residual_rtol)
            else:
                res = solver.solve(b, x, maxits=args.max_iterations,
                                   res_atol=args.residual_atol,
                                   res_rtol=args.residual_rtol,
                                   diff_atol=args.diff_atol,
                                   diff_rtol=args.diff_rtol)
        else:
            from .solvers.cpu import CGSolverCPU

            solver = CGSolverCPU(S, comm=comm)
            if solver_name == "cpu-pipelined":
                res = solver.solve_pipelined(b, x, maxits=args.max_iterations,
                                             res_atol=args.residual_atol,
                                             res_rtol=args.residual_rtol)
            elif solver_name == "cpu-jacobi":
                res = solver.solve_jacobi(b, x, maxits=args.max_iterations,
                                          res_atol=args.residual_atol,
                                          res_rtol=args.residual_rtol)
            else:
                res = solver.solve(b, x, maxits=args.max_iterations,
                                   res_atol=args.residual_atol,
                                   res_rtol=args.residual_rtol,
                                   diff_atol=args.diff_atol,
                                   diff_rtol=args.diff_rtol)
    except Exception as e:
        err = e
    collective_raise(comm, err)

    # ---- report (reference acgsolverhip_fwritempi, acg-hip.c:2247)
    from .solvers.profiling import write_stats

    write_stats(res, S, comm, file=sys.stderr)

    # true-residual integrity check: ||b - A x|| of the returned iterate,
    # independent of the solver's residual recursion (bench.py computes the
    # same; silent corruption anywhere would make these diverge)
    if res is not None:
        try:
            if gpu_solver:
                tchk = _t.zeros(S.nowned, dtype=_t.float64, device=device)
                solver._spmv_overlapped(x, tchk)
            else:
                from .solvers.cpu import CGSolverCPU

                chk = solver if isinstance(solver, CGSolverCPU) \
                    else CGSolverCPU(S, comm=comm)
                tchk = _t.zeros(S.nowned, dtype=_t.float64)
                chk._spmv(x, tchk)
            rl2 = float(_t.sum((b[:S.nowned] - tchk) ** 2))
            bl2 = float(_t.sum(b[:S.nowned] ** 2))
            if comm:
                import torch.distributed as dist

                rb = _t.tensor([rl2, bl2], dtype=_t.float64,
                               device=tchk.device if commkind == "rccl" else "cpu")
                dist.all_reduce(rb, op=dist.ReduceOp.SUM)
                rl2, bl2 = float(rb[0]), float(rb[1])
            if rank == 0 and bl2 > 0:
                lbl = " (scaled system)" if s_scale is not None else ""
                print(f"true residual{lbl}: ||b-Ax||/||b|| = "
                      f"{(rl2 / bl2) ** 0.5:.6e}", file=sys.stderr)
        except Exception as e:  # side-check only: never fail the solve over it
            if rank == 0:
                print(f"true residual: unavailable ({type(e).__name__}: {e})",
                      file=sys.stderr)

    x_host = x[:S.nowned].cpu().numpy()
    if s_scale is not None:
        x_host = x_host * s_scale  # map D^1/2 x back to x

    # manufactured-solution error norms (reference acg-hip.c:2354-2362)
    if args.manufactured_solution:
        if comm:
            xg = comm.gather_vector(x_host, S.owned_global, n_global)
        else:
            xg = np.empty(n_global)
            xg[S.owned_global] = x_host
        if rank == 0:
            enorm = np.linalg.norm(xg - xsol)
            print(f"manufactured solution: ||x-x*|| = {enorm:.6e} "
                  f"(||x*|| = 1)", file=sys.stderr)

    # solution to stdout as mtx array (reference acg-hip.c:2364-2403)
    if not args.quiet:
        if comm:
            xg = comm.gather_vector(x_host, S.owned_global, n_global)
        else:
            xg = np.empty(n_global)
            xg[S.owned_global] = x_host
        if rank == 0:
            from .io.mtx import vector_to_mtx, write_mtx

            write_mtx(sys.stdout, vector_to_mtx(xg), numfmt=numfmt)
    if comm:
        comm.finalize()
    # exit 2 on non-convergence whenever ANY stopping criterion was active
    # (residual atol/rtol or diff atol/rtol); 0 when none was requested
    # (fixed-iteration run)
    criterion = (args.residual_rtol > 0 or args.residual_atol > 0
                 or args.diff_atol > 0 or args.diff_rtol > 0)
    return 0 if (res is None or res.converged or not criterion) else 2


if __name__ == "__main__":
    sys.exit(main())
