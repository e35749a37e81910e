"""MI355X HIP CG solvers: classic and pipelined, with stream overlap.

Reference: acg/cghip.c (acgsolverhip_init :140-330, _solvempi :402-1159,
_solve_pipelined :1187-1933).  The structure here is the MI355X-native
re-design:

- All matrix/vector state is torch tensors on the GPU; the hot ops are the
  hand-written gfx950 kernels in ops/kernels.hip (no hipBLAS/hipSPARSE).
- Scalars (alpha/beta numerators/denominators) live in an 8-slot fp64
  device slab; every coefficient is computed *on device* inside the fused
  kernels.  The only D2H per iteration is the 8-byte residual norm for the
  host convergence test (reference cghip.c:996-1001).
- (p,t) is fused into the SpMV kernels (dot accumulated while t is
  produced), and the r/x updates + the new (r,r) are one fused kernel --
  the classic iteration runs 2 SpMV + 2 tiny prep kernels + 1 fused
  update + 1 daypx + 2 one-double RCCL allreduces.
- Streams: compute work on the caller's current stream; the halo
  (pack + grouped RCCL send/recv into the ghost tail) on a side
  ``comm_stream`` ordered by events exactly like the reference's
  preadytosend/preceived discipline (cghip.c:887-931): SpMV(matA) overlaps
  the exchange, SpMV(matO) waits for it.
"""

from __future__ import annotations

import math
import os
import time

import numpy as np
import torch

from ..dist.halo import HaloExchange
from ..ops import gpu_ops as ops
from ..part.subdomain import LocalSystem
from .base import SolveResult, cg_flops_per_iter


class CGSolverHIP:
    """Distributed CG on one MI355X per rank (classic + pipelined)."""

    def __init__(self, local: LocalSystem, comm=None, device=None,
                 lanes: int | None = None, use_sell: bool = True,
                 profile: bool = False, matfree: bool = False,
                 force_format: str | None = None):
        self.local = local
        self.comm = comm
        if device is None:
            device = comm.device if comm is not None and comm.device is not None \
                else torch.device("cuda", 0)
        self.device = torch.device(device)
        L = local

        def up(a, dtype=None):
            t = torch.from_numpy(np.ascontiguousarray(a))
            return t.to(self.device, non_blocking=True)

        self.n = L.nowned
        self.nlocal = L.nowned + L.nghost
        self.sell = None
        self.sellO = None
        self.sell_perm = None
        self.A_rowptr = self.A_colidx = self.A_vals = None
        self.O_rowptr = self.O_colidx = self.O_vals = None
        self.lanesA = self.lanesO = lanes or 16
        self.bsell = None
        self.hybrid = None  # (rowlist int32, bins) row-binned CSR
        # matrix-free analytic operator (dof=1 stencils, opt-in): the SpMV
        # reads NO matrix data -- see ops/kernels.hip k_stencil_spmv.
        self.matfree = None
        if matfree:
            mf = getattr(L, "mf_tables", None)
            if mf is None:
                raise ValueError(
                    "matfree=True needs a device-generated dof=1 stencil "
                    "system (gen.device_slab with spec dof==1)")
            self.matfree = mf
        if hasattr(L, "A_sell"):
            # device-generated system (gen.device_slab): SELL already in HBM
            self.sell = L.A_sell
            self.sellO = L.O_sell
            self.bsell = getattr(L, "A_bsell", None)
            if self.sell is None and self.matfree is None:
                raise ValueError("system generated with operator=False "
                                 "requires matfree=True")
        else:
            self.A_rowptr = up(L.A_rowptr)
            self.A_colidx = up(L.A_colidx)
            self.A_vals = up(L.A_vals)
            self.O_rowptr = up(L.O_rowptr)
            self.O_colidx = up(L.O_colidx)
            self.O_vals = up(L.O_vals)
            mean_nnz = L.nnzA / max(L.nowned, 1)
            self.lanesA = lanes or ops.pick_lanes(mean_nnz)
            mean_nnzO = L.nnzO / max(L.nborder, 1)
            self.lanesO = lanes or ops.pick_lanes(mean_nnzO)
            self._pick_format(L, up, use_sell, force_format)
        # megafused pipelined iteration needs SELL everywhere + int32 cols.
        # Measured on MI355X: for wide rows (~80 nnz, Queen-shaped) the fused
        # epilogue's 6 vector streams cost the SpMV more x-gather locality
        # than the saved q round-trip is worth (815us fused vs 805us split);
        # for narrow rows (7-pt Poisson) the q elimination is ~8% of the
        # iteration's traffic.  Auto-enable below ~16 nnz/row.
        self.can_megafuse = (
            self.matfree is not None
            or (self.sell is not None and self.sell_perm is None
                and self.sell[1].dtype == torch.int32
                and (L.nnzO == 0 or (self.sellO is not None
                                     and self.sellO[1].dtype == torch.int32))))
        # matrix-free: megafusion always pays (the SpMV reads no matrix, so
        # the vector-traffic saving has no x-gather-locality cost to trade)
        self.megafuse_auto = (self.can_megafuse
                              and (self.matfree is not None
                                   or L.nnzA / max(L.nowned, 1) <= 16.0))
        self.halo = HaloExchange(L.halo, L.nowned, self.device, comm)
        self.scal = ops.alloc_scalars(self.device)
        self.partials = ops.alloc_partials(self.device)
        from .profiling import EventProfiler

        self.prof = EventProfiler(profile)
        self.comm_stream = torch.cuda.Stream(self.device)
        self.allred_stream = torch.cuda.Stream(self.device)
        self.copy_stream = torch.cuda.Stream(self.device)
        self._rr_host = torch.zeros(1, dtype=torch.float64, pin_memory=True)
        self._ev_p = torch.cuda.Event()
        self._ev_recv = torch.cuda.Event()
        self._ev_rr = torch.cuda.Event()
        # always-on counters (reference cghip.h:109-118)
        self.niterations_total = 0
        # persistent per-method workspace vectors + captured iteration
        # graphs: repeated solves on one solver instance replay the graphs
        # captured on the FIRST solve (capture costs ~5-10 ms; re-capturing
        # inside every solve costs 2-10% at typical step counts).  Solution
        # state lives in the internal "xi" buffer so graph-referenced
        # addresses never change; the caller's x is copied in/out.
        self._ws: dict = {}
        self._graphs: dict = {}
        self._nstag = 0

    def _pick_format(self, L, up, use_sell: bool, force: str | None) -> None:
        """Choose the matA operator format (reference analog: the choice
        between hipsparse ALG_DEFAULT and the hand merge-path kernel,
        cghip.c:533-585 / cg-kernels-hip.hip:348).

        Auto ladder by measured fit:
          1. plain SELL-C-64 when padding waste <= 0.3 (stencil/FEM rows),
          2. sigma-SELL (window 16) when waste <= 0.5 (mildly irregular),
          3. row-binned hybrid CSR otherwise (power-law rows; MEASURED on
             MI355X 1M-row power-law: hybrid 665 us/it vs single-lane CSR
             855 vs wide-sigma SELL 2627 -- slice-length imbalance makes
             wide sigma lose badly, so it is force-only),
        plus Block-SELL whenever dense dof x dof block structure is found
        (density >= 0.75) -- it wins on index bytes.
        ``force`` in {csr, sell, sigma, bsell, hybrid} overrides for A/B
        measurement (bench --format)."""
        from ..ops.torch_ref import bsell_from_csr, sell_from_csr

        if L.nowned == 0:
            return

        def mk_hybrid(cut: int | None = None):
            """SELL+CSR split (short rows SELL, long tail binned CSR).
            MEASURED (MI355X 1M-row power-law): the pure binned hybrid's
            4/8-lane short-row bins cost 400 us/it of its 587 us SpMV --
            SELL's lockstep 512 B line loads serve those rows instead.
            Costs ~1.75x operator memory (CSR kept for the long rows).
            ACG_HYBRID_CUT overrides the split length (tuning sweeps)."""
            if cut is None:
                cut = int(os.environ.get("ACG_HYBRID_CUT", "192"))
            window = int(os.environ.get("ACG_HYBRID_WINDOW", "0"))
            bucket = int(os.environ.get("ACG_HYBRID_BUCKET", "8"))
            sp_, cols, svals, perm, rowlist, bins = ops.build_sellcsr_hybrid(
                L.A_rowptr, L.A_colidx, L.A_vals, cut=cut, window=window,
                bucket=bucket)
            self.hybrid = {
                "sellptr": up(sp_) if sp_ is not None else None,
                "cols": up(cols) if sp_ is not None else None,
                "svals": up(svals) if sp_ is not None else None,
                "perm": up(perm) if sp_ is not None else None,
                "rowlist": up(rowlist) if len(rowlist) else None,
                "bins": bins,
            }
            if L.nnzO > 0:
                # matO rows inherit the same power-law tail: bin them too
                rlO, binsO = ops.build_row_bins(L.O_rowptr)
                self.hybrid["rowlistO"] = up(rlO)
                self.hybrid["binsO"] = binsO

        def mk_binned():
            rowlist, bins = ops.build_row_bins(L.A_rowptr)
            self.hybrid = {"sellptr": None, "cols": None, "svals": None,
                           "perm": None, "rowlist": up(rowlist),
                           "bins": bins}

        def mk_sell(sigma):
            out = sell_from_csr(L.A_rowptr, L.A_colidx, L.A_vals, sigma=sigma)
            if sigma > 1:
                sellptr, scols, svals, perm = out
                self.sell_perm = up(perm)
            else:
                sellptr, scols, svals = out
                self.sell_perm = None
            self.sell = (up(sellptr), up(scols), up(svals))
            if L.nnzO > 0:
                optr, ocols, ovals = sell_from_csr(L.O_rowptr, L.O_colidx,
                                                   L.O_vals)
                self.sellO = (up(optr), up(ocols), up(ovals))
            return (int(sellptr[-1]) - L.nnzA) / max(L.nnzA, 1)

        def mk_bsell():
            for dof_try in (4, 3, 2):  # kernel dispatch covers 2/3/4
                out = bsell_from_csr(L.A_rowptr, L.A_colidx, L.A_vals, dof_try)
                if out is None:
                    continue
                bptr_h, bcol_h, bvals_h, density = out
                if density >= 0.75:
                    self.bsell = (up(bptr_h), up(bcol_h), up(bvals_h), dof_try)
                    return True
            return False

        if force == "csr":
            return
        if force == "hybrid":
            mk_hybrid()
            return
        if force == "binned":
            mk_binned()
            return
        if force == "sell":
            mk_sell(1)
        elif force == "sigma":
            mk_sell(4096)
        elif force == "bsell":
            if not mk_bsell():
                raise ValueError("force_format=bsell: no dense dof x dof "
                                 "block structure (density < 0.75)")
        elif use_sell:  # auto ladder
            waste = mk_sell(1)
            if waste > 0.3:
                self.sell = self.sellO = self.sell_perm = None
                waste = mk_sell(16)
                if waste > 0.5:
                    self.sell = self.sellO = self.sell_perm = None
                    mk_hybrid()
            mk_bsell()
        if self.sell is not None or self.bsell is not None:
            self.A_rowptr = self.A_colidx = self.A_vals = None  # free CSR

    def _workspace(self, key: str, names) -> dict:
        """Per-method persistent vectors, base-address STAGGERED: equal-size
        allocations come back congruent mod large powers of two, so the
        fused update's 6+ streams hit the same L2 set / HBM channel at every
        index -- measured as a ~10% per-instance placement lottery
        (535-589 us/it on identical Queen solvers).  A 4160 B (64 B-odd)
        offset per vector de-correlates the streams deterministically."""
        ws = self._ws.setdefault(key, {})
        for nm, nghost in names:
            if nm not in ws:
                n = self.nlocal if nghost else self.n
                off = 520 * self._nstag  # 520 doubles = 4096 + 64 bytes
                self._nstag += 1
                buf = torch.zeros(n + off, dtype=torch.float64,
                                  device=self.device)
                ws[nm] = buf[off:] if off else buf
        return ws

    # -- pieces -----------------------------------------------------------

    def _allreduce_slot(self, slot: int, count: int = 1):
        if self.comm is not None and self.comm.size > 1:
            self.comm.allreduce_(self.scal[slot:slot + count])

    def _spmv_overlapped(self, xfull: torch.Tensor, y: torch.Tensor,
                         fuse_dotslot: int = -1):
        """halo(x) on comm stream overlapped with SpMV(matA); SpMV(matO)
        after the ghost tail arrives.  Mirrors reference cghip.c:887-931."""
        L = self.local
        have_halo = self.comm is not None and self.comm.size > 1
        cur = torch.cuda.current_stream(self.device)
        halo_span = None
        if have_halo:
            self._ev_p.record(cur)
            self.comm_stream.wait_event(self._ev_p)
            halo_span = self.prof.span("halo", self.comm_stream)
            halo_span.__enter__()  # closed after halo.end() below
            with torch.cuda.stream(self.comm_stream):
                self.halo.begin(xfull)
        fuse = dict(partials=self.partials,
                    scal=self.scal if fuse_dotslot >= 0 else None,
                    dotslot=fuse_dotslot)
        with self.prof.span("spmvA"):
            # the matA pass OVERWRITES the dot slot (dot_accum=False), so
            # no zeroing prep kernel is needed; matO accumulates on top
            if self.matfree is not None:
                ops.stencil_spmv(self.matfree, self.n, 0, xfull, y,
                                 mato=False, dot_accum=False, **fuse)
            elif self.bsell is not None:
                bptr, bcol, bvals, dof = self.bsell
                ops.spmv_bsell(bptr, bcol, bvals, self.n // dof, dof,
                               xfull, y, dot_accum=False, **fuse)
            elif self.sell is not None:
                sellptr, scols, svals = self.sell
                ops.spmv_sell(sellptr, scols, svals, self.n, xfull, y,
                              accum=False, perm=self.sell_perm,
                              dot_accum=False, **fuse)
            elif self.hybrid is not None:
                h = self.hybrid
                have_sell = h["sellptr"] is not None
                if have_sell:
                    ops.spmv_sell(h["sellptr"], h["cols"], h["svals"],
                                  self.n, xfull, y, accum=False,
                                  perm=h["perm"], dot_accum=False, **fuse)
                if h["rowlist"] is not None:
                    # rows are disjoint from the SELL set: y writes stay
                    # overwrite-mode; the fused dot ACCUMULATES on top of
                    # the SELL part's contribution
                    ops.spmv_binned(self.A_rowptr, self.A_colidx,
                                    self.A_vals, h["rowlist"], h["bins"],
                                    xfull, y, accum=False,
                                    dot_accum=have_sell, **fuse)
            else:
                ops.spmv(self.A_rowptr, self.A_colidx, self.A_vals, xfull, y,
                         lanes=self.lanesA, accum=False, dot_accum=False,
                         **fuse)
        if have_halo:
            with torch.cuda.stream(self.comm_stream):
                self.halo.end()
                self._ev_recv.record(self.comm_stream)
            if halo_span is not None:
                halo_span.__exit__(None, None, None)
            cur.wait_event(self._ev_recv)
        if L.nborder > 0 and self.local.nnzO > 0:
            with self.prof.span("spmvO"):
                if self.matfree is not None:
                    ops.stencil_spmv(self.matfree, L.nborder, L.ninterior,
                                     xfull, y, mato=True, **fuse)
                elif self.sellO is not None:
                    optr, ocols, ovals = self.sellO
                    ops.spmv_sell(optr, ocols, ovals, L.nborder, xfull, y,
                                  rowbase=L.ninterior, accum=True, **fuse)
                elif self.hybrid is not None and "rowlistO" in self.hybrid:
                    ops.spmv_binned(self.O_rowptr, self.O_colidx, self.O_vals,
                                    self.hybrid["rowlistO"],
                                    self.hybrid["binsO"], xfull, y,
                                    rowbase=L.ninterior, accum=True, **fuse)
                else:
                    ops.spmv(self.O_rowptr, self.O_colidx, self.O_vals, xfull, y,
                             rowbase=L.ninterior, lanes=self.lanesO, accum=True,
                             **fuse)

    def _host_scalar(self, slot: int) -> float:
        cur = torch.cuda.current_stream(self.device)
        self._ev_rr.record(cur)
        self.copy_stream.wait_event(self._ev_rr)
        with torch.cuda.stream(self.copy_stream):
            self._rr_host.copy_(self.scal[slot:slot + 1], non_blocking=True)
        self.copy_stream.synchronize()
        return float(self._rr_host[0])

    def _vec(self, nghost: bool = False) -> torch.Tensor:
        return torch.zeros(self.nlocal if nghost else self.n,
                           dtype=torch.float64, device=self.device)

    # -- classic CG -------------------------------------------------------

    def solve(self, b: torch.Tensor, x: torch.Tensor, maxits: int = 100,
              res_atol: float = 0.0, res_rtol: float = 1e-9,
              use_graph: bool | None = None,
              fold_daypx: bool | None = None,
              diff_atol: float = 0.0, diff_rtol: float = 0.0) -> SolveResult:
        """Classic CG (reference acgsolverhip_solvempi, cghip.c:402-1159).

        ``x`` must be an nlocal vector (ghost tail included); ``b`` nowned.
        The host convergence test runs every iteration on the lag-1
        pipeline (see solve_pipelined).

        ``use_graph`` default None = measured policy (same as pipelined):
        OFF serial (eager wins by ~8 us/it), ON multi-GPU -- the WHOLE
        distributed iteration (halo fork, split SpMV, both allreduces,
        fused update, daypx) is captured once and replayed so the
        ~100 us/it of host-side c10d/launch work leaves the critical
        path.  Unlike pipelined no in-graph scalar snapshot is needed:
        the body ENDS with scal[S_RR] holding the allreduced new (r,r)
        (daypx reads but never writes it), so the post-replay lagged D2H
        reads a rank-consistent value by stream order.  Any capture
        failure (e.g. gloo) permanently falls back to eager;
        ACG_DIST_GRAPH=0 is the kill switch.

        ``diff_atol``/``diff_rtol`` stop on the solution change
        ``|alpha|*||p|| <= max(diff_atol, diff_rtol*||b||)`` -- a
        beyond-reference capability (the reference GPU solvers REJECT diff
        tolerances, cghip.c:427); engaging it switches to a blocking
        per-iteration host test (no lag pipeline, no graph replay).
        """
        res = SolveResult(solver="cg-hip", maxits=maxits, res_atol=res_atol,
                          res_rtol=res_rtol,
                          nranks=self.comm.size if self.comm else 1)
        n = self.n
        S = ops
        scal = self.scal
        ws = self._workspace("classic", [("r", False), ("t", False),
                                         ("p", True), ("xi", True)])
        r, t, p, xi = ws["r"], ws["t"], ws["p"], ws["xi"]
        xi.copy_(x)
        torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        # bnrm2
        S.dot(b, b, self.partials, scal, S.S_BNRM2, n=n)
        self._allreduce_slot(S.S_BNRM2)
        # r0 = b - A x0;  p = r0
        self._spmv_overlapped(xi, t)
        torch.sub(b[:n], t, out=r)
        p[:n] = r
        S.dot(r, r, self.partials, scal, S.S_RR, n=n)
        self._allreduce_slot(S.S_RR)
        bnrm2sqr = self._host_scalar(S.S_BNRM2)
        rr = self._host_scalar(S.S_RR)
        res.bnrm2 = math.sqrt(max(bnrm2sqr, 0.0))
        res.r0nrm2 = math.sqrt(max(rr, 0.0))
        rtol2 = max(res_atol, res_rtol * res.bnrm2) ** 2
        dtol = max(diff_atol, diff_rtol * res.bnrm2)
        if rtol2 > 0 and rr <= rtol2 and dtol == 0.0:
            res.converged = True
            res.rnrm2 = math.sqrt(rr)
            res.tsolve = time.perf_counter() - t0
            return res
        converged = False
        serial = self.comm is None or self.comm.size == 1
        if dtol > 0:
            converged = self._solve_diffmode(res, r, t, p, xi, rtol2, dtol,
                                             maxits)
            return self._finish_classic(res, x, xi, converged, rtol2, t0)
        # daypx folded into the BSELL SpMV (serial matA-only): the gather
        # computes beta*p_old + r on the fly and the row side materialises
        # p_new into a ping-pong buffer, eliminating the 3n-stream daypx
        # kernel.  MEASURED NEGATIVE and therefore opt-in: doubling the
        # gather footprint (p_old AND r) costs +60 us/it on Queen vs the
        # ~15 us the daypx saves (553.6 vs 493.8 us/it interleaved) --
        # the SpMV gather is the roofline resource, not kernel count.
        fold = (serial and self.bsell is not None and self.local.nnzO == 0
                and fold_daypx is True)
        if fold:
            p2 = self._workspace("classic", [("p2", True)])["p2"]
            scal[S.S_RR_PREV] = math.inf
        if use_graph is None:
            use_graph = not serial  # measured policy (see docstring)
        graph_ok = use_graph and serial and not self.prof.enabled and not fold
        graph = self._graphs.get("classic") if graph_ok else None
        # multi-GPU: capture the whole distributed iteration (see docstring)
        dist_key = "classic:dist"
        dist_failed_key = "classic:capture_failed"
        dist_graph_ok = (use_graph and not serial and not self.prof.enabled
                         and not fold
                         and getattr(self.comm, "can_capture", False)
                         and os.environ.get("ACG_DIST_GRAPH", "1") != "0"
                         and not self._graphs.get(dist_failed_key, False))
        dgraph = self._graphs.get(dist_key) if dist_graph_ok else None
        # lag-2 convergence pipeline (+ optional hipGraph replay), mirroring
        # solve_pipelined (the host test runs for every iteration; the host
        # reads the value two iterations late).  Classic's rr copy lands at
        # the END of its iteration, so LAG=1 would make the host wake
        # exactly when the GPU drains -- LAG=2 keeps one full iteration
        # queued and the GPU never idles (measured 582 -> ~545 us/it).
        LAG = 2
        hostbuf = [torch.zeros(1, dtype=torch.float64, pin_memory=True)
                   for _ in range(LAG + 1)]
        evdone = [torch.cuda.Event() for _ in range(LAG + 1)]

        def body(k: int = 0):
            if fold:
                bptr, bcol, bvals, dof = self.bsell
                pold, pnew = (p, p2) if k % 2 == 0 else (p2, p)
                with self.prof.span("spmvA"):
                    S.spmv_bsell_daypx(bptr, bcol, bvals, n // dof, dof,
                                       pold, r, pnew, t, scal,
                                       self.partials, S.S_PT)
                with self.prof.span("update_classic"):
                    S.cg_fused_update(r, xi, pnew, t, scal, self.partials, n)
                return
            # halo+split SpMV with the (p,t) reduction fused into both
            # passes (matA overwrites the slot, matO accumulates)
            self._spmv_overlapped(p, t, fuse_dotslot=S.S_PT)
            with self.prof.span("allreduce"):
                self._allreduce_slot(S.S_PT)
            # fused r/x update (alpha = rr/pt on device) + combined
            # finalize (rr -> rr_prev rotation + new (r,r))
            with self.prof.span("update_classic"):
                S.cg_fused_update(r, xi, p, t, scal, self.partials, n)
            with self.prof.span("allreduce"):
                self._allreduce_slot(S.S_RR)
            # p = (rr/rr_prev) p + r
            with self.prof.span("daypx"):
                S.daypx_ratio(p, r, scal, S.S_RR, S.S_RR_PREV, n=n)

        def read_rr(j):
            evdone[j % (LAG + 1)].synchronize()
            return float(hostbuf[j % (LAG + 1)][0])

        def check(j):
            nonlocal rr, converged
            rr = read_rr(j)
            if not math.isfinite(rr):
                raise FloatingPointError(f"CG diverged: rr={rr} at it {j + 1}")
            if rtol2 > 0 and rr <= rtol2:
                converged = True
                res.niterations = j + 1
                return True
            return False

        k = 0
        cur = torch.cuda.current_stream(self.device)
        while k < maxits:
            if k >= LAG and check(k - LAG):
                break
            if graph is not None:
                graph.replay()
            elif dgraph is not None and k > 0:
                dgraph.replay()
            else:
                body(k)
                if graph_ok and k == 1:
                    graph = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(graph):
                        body()
                    self._graphs["classic"] = graph
                elif dist_graph_ok and k == 2 and dgraph is None:
                    # RCCL channels are warm after 2 eager iterations;
                    # capture records without executing (rank-local)
                    try:
                        gg = torch.cuda.CUDAGraph()
                        with torch.cuda.graph(gg):
                            body()
                        dgraph = gg
                        self._graphs[dist_key] = gg
                    except Exception as e:
                        self._graphs[dist_failed_key] = True
                        torch.cuda.synchronize(self.device)
                        if self.comm is None or self.comm.rank == 0:
                            print(f"[acg_amd] classic distributed graph "
                                  f"capture unavailable ({type(e).__name__});"
                                  f" eager iterations",
                                  file=__import__("sys").stderr)
            # lagged rr D2H on the SAME stream: cross-stream event chains
            # measured ~20 us each in fire->launch latency (45 us/iter of
            # the iteration period); the in-order 8-byte copy costs ~3 us
            # on the critical path and makes the overwrite race impossible
            # by stream ordering.
            j = k % (LAG + 1)
            hostbuf[j].copy_(scal[S.S_RR:S.S_RR + 1], non_blocking=True)
            evdone[j].record(cur)
            k += 1
            res.niterations = k
        if not converged:
            for j in range(max(maxits - LAG, 0), maxits):
                if check(j):
                    break
        return self._finish_classic(res, x, xi, converged, rtol2, t0)

    def _solve_diffmode(self, res, r, t, p, xi, rtol2: float, dtol: float,
                        maxits: int) -> bool:
        """Classic iterations with the solution-change stopping criterion
        ``|alpha|*||p|| <= dtol`` (CPU-solver semantics, cg.c:1059-1068).
        Blocking host test each iteration: alpha = rr_prev/pt comes from
        the device scalar slab AFTER the fused update rotated rr ->
        rr_prev; ||p||^2 is one extra fused-dot + allreduce per iteration
        (S_GAMMA used as scratch -- classic never touches it)."""
        S = ops
        n = self.n
        scal = self.scal
        converged = False
        for k in range(maxits):
            self._spmv_overlapped(p, t, fuse_dotslot=S.S_PT)
            self._allreduce_slot(S.S_PT)
            S.dot(p, p, self.partials, scal, S.S_GAMMA, n=n)
            self._allreduce_slot(S.S_GAMMA)
            S.cg_fused_update(r, xi, p, t, scal, self.partials, n)
            self._allreduce_slot(S.S_RR)
            res.niterations = k + 1
            rr_old = self._host_scalar(S.S_RR_PREV)
            pt = self._host_scalar(S.S_PT)
            pp = self._host_scalar(S.S_GAMMA)
            rr = self._host_scalar(S.S_RR)
            if not math.isfinite(rr):
                raise FloatingPointError(f"CG diverged: rr={rr} at it {k + 1}")
            alpha = rr_old / pt if pt != 0.0 else 0.0
            dx = abs(alpha) * math.sqrt(max(pp, 0.0))
            if (rtol2 > 0 and rr <= rtol2) or dx <= dtol:
                converged = True
                break
            S.daypx_ratio(p, r, scal, S.S_RR, S.S_RR_PREV, n=n)
        return converged

    def _finish_classic(self, res, x, xi, converged: bool, rtol2: float,
                        t0: float):
        S = ops
        torch.cuda.synchronize(self.device)
        res.tsolve = time.perf_counter() - t0
        x.copy_(xi)
        rr = self._host_scalar(S.S_RR)
        res.rnrm2 = math.sqrt(max(rr, 0.0))
        res.converged = converged or (rtol2 > 0 and rr <= rtol2)
        nnz_full = self.local.nnzA + self.local.nnzO
        res.nflops = res.niterations * cg_flops_per_iter(nnz_full, self.n)
        res.halo_bytes_sent = self.halo.bytes_sent
        res.halo_msgs_sent = self.halo.nmsgs_sent
        self.niterations_total += res.niterations
        if self.prof.enabled:
            from .profiling import annotate_op_stats

            idxb = None
            if self.matfree is not None:
                idxb = -8.0  # matrix-free: no vals (8 B) and no cols read
            elif self.bsell is not None:
                idxb = 4.0 / (self.bsell[3] ** 2)
            annotate_op_stats(res, self.local, self.prof.collect(),
                              idx_bytes_per_nnz=idxb)
            self.prof.reset()
        return res

    # -- monolithic device-side CG ---------------------------------------

    def solve_device(self, b: torch.Tensor, x: torch.Tensor, maxits: int = 100,
                     res_atol: float = 0.0, res_rtol: float = 1e-9) -> SolveResult:
        """Whole CG solve in ONE cooperative kernel launch (reference
        acgsolverhip_solve_device, §3.4; single-GPU only, as in the
        reference's HIP build -- cg-kernels-hip.hip:1832).  Zero
        per-iteration launch overhead: wins in the launch-bound regime
        (small/medium systems)."""
        if self.comm is not None and self.comm.size > 1:
            from ..utils.errors import AcgError, ErrCode

            raise AcgError(ErrCode.NOT_SUPPORTED,
                           "device-side CG is single-GPU (reference parity)")
        if self.sell is None or self.sell_perm is not None:
            from ..utils.errors import AcgError, ErrCode

            raise AcgError(ErrCode.NOT_SUPPORTED,
                           "device-side CG requires the unpermuted SELL format")
        res = SolveResult(solver="cg-hip-device", maxits=maxits,
                          res_atol=res_atol, res_rtol=res_rtol, nranks=1)
        n = self.n
        r = self._vec()
        t = self._vec()
        p = self._vec(nghost=True)
        out2 = torch.zeros(2, dtype=torch.int32, device=self.device)
        barrier_state = torch.zeros(ops.BAR_STATE_WORDS, dtype=torch.int32,
                                    device=self.device)
        sellptr, scols, svals = self.sell
        torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        ops.cg_device(sellptr, scols, svals, n, b, x, r, p, t, self.scal,
                      self.partials, out2, barrier_state, maxits, res_atol,
                      res_rtol)
        torch.cuda.synchronize(self.device)
        res.tsolve = time.perf_counter() - t0
        S = ops
        res.bnrm2 = math.sqrt(max(float(self.scal[S.S_BNRM2]), 0.0))
        res.r0nrm2 = math.sqrt(max(float(self.scal[S.S_RR_PREV]), 0.0))
        res.rnrm2 = math.sqrt(max(float(self.scal[S.S_RR]), 0.0))
        res.niterations = int(out2[0])
        conv = int(out2[1])
        if conv < 0:
            raise RuntimeError("device CG grid barrier timed out (residency?)")
        res.converged = bool(conv)
        nnz_full = self.local.nnzA + self.local.nnzO
        res.nflops = res.niterations * cg_flops_per_iter(nnz_full, n)
        self.niterations_total += res.niterations
        return res

    # -- Jacobi-preconditioned CG (beyond reference: aCG is strictly
    # unpreconditioned, PCNONE even in its PETSc oracle cgpetsc.c:181-193;
    # opt-in so every parity bench stays unpreconditioned) ----------------

    def solve_jacobi(self, b: torch.Tensor, x: torch.Tensor,
                     maxits: int = 100, res_atol: float = 0.0,
                     res_rtol: float = 1e-9) -> SolveResult:
        """Jacobi-PCG composed from the existing device-scalar kernels:
        alpha = (r,z)/(p,t) via axpy_ratio, beta = rz'/rz via the
        RR/RR_PREV rotation, z = D^-1 r as one elementwise multiply.
        Convergence is tested on the TRUE residual 2-norm every iteration
        (blocking host read -- PCG trades per-iteration polish for a
        several-fold iteration-count cut on ill-conditioned systems)."""
        L = self.local
        if getattr(L, "A_rowptr", None) is None or not hasattr(L, "owned_global"):
            from ..utils.errors import AcgError, ErrCode

            raise AcgError(ErrCode.NOT_SUPPORTED,
                           "jacobi PCG needs the host CSR arrays "
                           "(file/extracted systems; not device-generated "
                           "slabs)")
        res = SolveResult(solver="cg-hip-jacobi", maxits=maxits,
                          res_atol=res_atol, res_rtol=res_rtol,
                          nranks=self.comm.size if self.comm else 1)
        n = self.n
        S = ops
        scal = self.scal
        if "jacobi_dinv" not in self._ws:
            rows = np.repeat(np.arange(n, dtype=np.int64),
                             np.diff(L.A_rowptr))
            dmask = rows == L.A_colidx
            d = np.zeros(n, dtype=np.float64)
            d[rows[dmask]] = L.A_vals[dmask]
            if (d == 0).any():
                from ..utils.errors import AcgError, ErrCode

                raise AcgError(ErrCode.INVALID_VALUE,
                               "jacobi preconditioner needs a full diagonal")
            self._ws["jacobi_dinv"] = torch.from_numpy(1.0 / d).to(self.device)
        dinv = self._ws["jacobi_dinv"]
        ws = self._workspace("jacobi", [("r", False), ("t", False),
                                        ("z", False), ("p", True),
                                        ("xi", True)])
        r, t, z, p, xi = ws["r"], ws["t"], ws["z"], ws["p"], ws["xi"]
        xi.copy_(x)
        torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        S.dot(b, b, self.partials, scal, S.S_BNRM2, n=n)
        self._allreduce_slot(S.S_BNRM2)
        self._spmv_overlapped(xi, t)
        torch.sub(b[:n], t, out=r)
        torch.mul(r, dinv, out=z)
        p[:n] = z
        S.dot(r, z, self.partials, scal, S.S_RR, n=n)
        self._allreduce_slot(S.S_RR)
        S.dot(r, r, self.partials, scal, S.S_GAMMA, n=n)
        self._allreduce_slot(S.S_GAMMA)
        res.bnrm2 = math.sqrt(max(self._host_scalar(S.S_BNRM2), 0.0))
        rr = self._host_scalar(S.S_GAMMA)
        res.r0nrm2 = math.sqrt(max(rr, 0.0))
        rtol2 = max(res_atol, res_rtol * res.bnrm2) ** 2
        converged = rtol2 > 0 and rr <= rtol2
        k = 0
        while not converged and k < maxits:
            self._spmv_overlapped(p, t, fuse_dotslot=S.S_PT)
            self._allreduce_slot(S.S_PT)
            # ONE fused kernel: alpha = rz/pt (device), r/x updates,
            # z = dinv*r, next rz (-> S_RR, rotated) and true rr
            # (-> S_GAMMA) -- replaces 6 launches and ~2x the traffic
            S.pcg_fused_update(r, xi, p, t, z, dinv, scal, self.partials, n)
            self._allreduce_slot(S.S_RR)
            self._allreduce_slot(S.S_GAMMA)
            rr = self._host_scalar(S.S_GAMMA)
            k += 1
            res.niterations = k
            if not math.isfinite(rr):
                raise FloatingPointError(f"jacobi PCG diverged at it {k}")
            if rtol2 > 0 and rr <= rtol2:
                converged = True
                break
            # p = (rz/rz_prev) p + z
            S.daypx_ratio(p, z, scal, S.S_RR, S.S_RR_PREV, n=n)
        torch.cuda.synchronize(self.device)
        res.tsolve = time.perf_counter() - t0
        x.copy_(xi)
        res.rnrm2 = math.sqrt(max(rr, 0.0))
        res.converged = converged
        nnz_full = L.nnzA + L.nnzO
        res.nflops = res.niterations * (cg_flops_per_iter(nnz_full, n) + 3.0 * n)
        res.halo_bytes_sent = self.halo.bytes_sent
        res.halo_msgs_sent = self.halo.nmsgs_sent
        self.niterations_total += res.niterations
        return res

    # -- pipelined CG -----------------------------------------------------

    def solve_pipelined(self, b: torch.Tensor, x: torch.Tensor, maxits: int = 100,
                        res_atol: float = 0.0, res_rtol: float = 1e-9,
                        use_graph: bool | None = None,
                        megafuse: bool | None = None) -> SolveResult:
        """Pipelined (Ghysels-Vanroose) CG: ONE 2-double allreduce per
        iteration, overlapped with the halo + SpMV of q = A w
        (reference acgsolverhip_solve_pipelined, cghip.c:1187-1933).

        ``use_graph``: capture the steady-state iteration into a hipGraph
        and replay it (captured once per solver, cached).  Default None =
        measured policy: OFF on one GPU (the eager 1-3 launch body is ~8
        us/it cheaper than hipGraphLaunch, profiles/RESULTS.md), ON for
        multi-GPU where the eager iteration's ~100 us of host-side c10d /
        launch work rivals the per-rank GPU time -- there the WHOLE
        iteration (side-stream allreduce fork, RCCL halo, split SpMV,
        fused update) is captured, with automatic permanent fallback to
        eager if the backend cannot capture (e.g. gloo).  Replay
        semantics: the update is enqueued before the host reads the
        previous gamma, so on the converging iteration 1-2 extra (valid)
        updates have already been applied to x; reported
        niterations/rnrm2 match the eager path.
        """
        res = SolveResult(solver="cg-hip-pipelined", maxits=maxits,
                          res_atol=res_atol, res_rtol=res_rtol,
                          nranks=self.comm.size if self.comm else 1)
        n = self.n
        S = ops
        L = self.local
        scal = self.scal
        mega = self.megafuse_auto if megafuse is None else (megafuse and self.can_megafuse)
        wskey = f"pipelined{'_mega' if mega else ''}"
        names = [("r", True), ("w", True), ("z", False), ("t", False),
                 ("p", False), ("tmp", False), ("xi", True)]
        names += [("w2", True)] if mega else [("q", False)]
        ws = self._workspace(wskey, names)
        r, w, z, t, p, tmp, xi = (ws["r"], ws["w"], ws["z"], ws["t"],
                                  ws["p"], ws["tmp"], ws["xi"])
        # megafused path double-buffers w (SpMV gathers w_old while the
        # fused epilogue writes w_new) and stages border q in qpart;
        # the separate q vector exists only on the fallback path.
        w2 = ws.get("w2")
        q = ws.get("q")
        if mega and "qpart" not in ws:
            ws["qpart"] = torch.zeros(max(L.nborder, 1), dtype=torch.float64,
                                      device=self.device)
        qpart = ws.get("qpart")
        xi.copy_(x)
        # first=True multiplies z/t/p by beta=0: stale non-finite values
        # from an aborted earlier solve must not poison 0*x
        for v in (z, t, p):
            v.zero_()
        torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        S.dot(b, b, self.partials, scal, S.S_BNRM2, n=n)
        self._allreduce_slot(S.S_BNRM2)
        self._spmv_overlapped(xi, tmp)
        torch.sub(b[:n], tmp, out=r[:n])
        self._spmv_overlapped(r, w)  # w = A r
        res.bnrm2 = math.sqrt(max(self._host_scalar(S.S_BNRM2), 0.0))
        rtol2 = max(res_atol, res_rtol * res.bnrm2) ** 2
        # initial local gamma/delta; per-iteration dots are fused into the
        # 6-vector update kernel, so no standalone dot pass ever runs again
        S.dot2(r, w, self.partials, scal, n)
        converged = False
        gamma_host = None
        serial = self.comm is None or self.comm.size == 1
        if use_graph is None:
            # measured policy: eager wins on one GPU (~8 us/it cheaper than
            # hipGraphLaunch); captured iterations win the host-bound
            # distributed loop (capture failure falls back to eager)
            use_graph = not serial
        graph_ok = use_graph and serial and not self.prof.enabled
        graph = self._graphs.get(wskey) if graph_ok and not mega else None
        # megafused: even/odd w ping-pong graphs
        graphs = (self._graphs.get(wskey, [None, None])
                  if graph_ok and mega else [None, None])

        def mega_body(wa, wb, first):
            """One megafused iteration: halo(wa) || matA pass (SpMV + update
            of interior rows), then matO pass finishing border rows."""
            have_halo = not serial
            cur = torch.cuda.current_stream(self.device)
            if have_halo:
                self._ev_p.record(cur)
                self.comm_stream.wait_event(self._ev_p)
                with torch.cuda.stream(self.comm_stream):
                    self.halo.begin(wa)
            border_base = L.ninterior if L.nnzO > 0 else n
            if self.matfree is not None:
                nbA = S.stencil_pipe(self.matfree, n, 0, border_base, wa,
                                     qpart, z, t, p, xi, r, wb, scal, first,
                                     self.partials, 0, mato=False)
            else:
                sp, sc, sv = self.sell
                nbA = S.sell_pipe(sp, sc, sv, n, 0, border_base, wa, qpart,
                                  z, t, p, xi, r, wb, scal, first,
                                  self.partials, 0, mato=False)
            nb = nbA
            if have_halo:
                with torch.cuda.stream(self.comm_stream):
                    self.halo.end()
                    self._ev_recv.record(self.comm_stream)
                cur.wait_event(self._ev_recv)
            if L.nnzO > 0:
                if self.matfree is not None:
                    nbO = S.stencil_pipe(self.matfree, L.nborder, L.ninterior,
                                         L.ninterior, wa, qpart, z, t, p, xi,
                                         r, wb, scal, first, self.partials,
                                         nbA, mato=True)
                else:
                    op_, oc, ov = self.sellO
                    nbO = S.sell_pipe(op_, oc, ov, L.nborder, L.ninterior,
                                      L.ninterior, wa, qpart, z, t, p, xi, r,
                                      wb, scal, first, self.partials, nbA,
                                      mato=True)
                nb += nbO
            S.pipelined_finalize(self.partials, nb, scal, first)
        # lag-1 convergence pipeline: gamma_k is copied to the host as soon
        # as its allreduce lands, but the host *reads* it one iteration
        # later -- the test still runs for every iteration, the host just
        # stays an iteration ahead of the GPU instead of blocking on each
        # iteration's tail (measured: the blocking check serializes
        # host<->GPU and costs ~8% at Queen scale).  On the detecting
        # iteration 1-2 extra (valid) updates have been applied to x;
        # reported niterations/rnrm2 correspond to the detected gamma.
        LAG = 1
        hostbuf = [torch.zeros(1, dtype=torch.float64, pin_memory=True)
                   for _ in range(LAG + 1)]
        evdone = [torch.cuda.Event() for _ in range(LAG + 1)]

        def issue_gamma_copy(k):
            # same-stream D2H (cross-stream event chains cost ~20 us each
            # in fire->launch latency; the in-order 8-byte copy ~3 us) --
            # also makes the gamma-overwrite race impossible by ordering
            cur = torch.cuda.current_stream(self.device)
            j = k % (LAG + 1)
            hostbuf[j].copy_(scal[S.S_GAMMA:S.S_GAMMA + 1], non_blocking=True)
            evdone[j].record(cur)

        def read_gamma(j):
            evdone[j % (LAG + 1)].synchronize()
            return float(hostbuf[j % (LAG + 1)][0])

        def check(j):
            """Host convergence test on gamma_j; True => converged at j."""
            nonlocal gamma_host, converged
            gamma_host = read_gamma(j)
            if j == 0:
                res.r0nrm2 = math.sqrt(max(gamma_host, 0.0))
            if not math.isfinite(gamma_host):
                raise FloatingPointError(f"pipelined CG diverged at it {j}")
            if rtol2 > 0 and gamma_host <= rtol2:
                converged = True
                res.rnrm2 = math.sqrt(max(gamma_host, 0.0))
                res.niterations = j
                return True
            return False

        ev_gd = torch.cuda.Event()
        ev_ar = torch.cuda.Event()
        # Multi-GPU graph capture: the eager distributed iteration costs
        # ~100 us of host work (c10d op lists for the halo, allreduce
        # setup, kernel launches) which rivals the GPU time per iteration
        # at 8-GPU Queen scale.  The ENTIRE steady-state iteration --
        # side-stream allreduce fork, halo send/recv, split SpMV, fused
        # update + finalize -- is captured once and replayed (RCCL
        # collectives and event forks are capturable; the 8-byte gamma
        # D2H stays outside, same-stream after the replay).  Capture is
        # rank-local and records without executing, so no rank-sync is
        # involved; ANY capture failure (e.g. the gloo test backend
        # cannot capture) permanently falls back to the eager path for
        # this solver, and mixed replay/eager ranks remain correct
        # because replay issues the identical collective sequence.
        dist_key = wskey + ":dist"  # dist bodies INCLUDE the allreduce --
        dist_failed_key = wskey + ":capture_failed"  # never mix with serial
        # ACG_DIST_GRAPH=0 is the operational kill switch for the captured
        # distributed iteration (falls back to the eager path everywhere)
        dist_graph_ok = (use_graph and not serial and not self.prof.enabled
                         and getattr(self.comm, "can_capture", False)
                         and os.environ.get("ACG_DIST_GRAPH", "1") != "0"
                         and not self._graphs.get(dist_failed_key, False))
        dgraph = self._graphs.get(dist_key) if dist_graph_ok and not mega else None
        dgraphs = (self._graphs.get(dist_key, [None, None])
                   if dist_graph_ok and mega else [None, None])

        # in-graph snapshot of the ALLREDUCED gamma: the replayed body both
        # allreduces gamma_{k-1} and overwrites it (finalize), so the host
        # copy after replay must read this intermediate, not scal itself --
        # otherwise each rank would test its LOCAL gamma and ranks could
        # break at different iterations (collective deadlock).
        if "gsnap" not in ws:
            ws["gsnap"] = torch.zeros(1, dtype=torch.float64,
                                      device=self.device)
        gsnap = ws["gsnap"]

        def body_dist(wa=None, wb=None):
            """One steady-state (first=False) distributed iteration in the
            shape the capture needs (allreduce included)."""
            if mega:
                self._allreduce_slot(S.S_GAMMA, 2)
                gsnap.copy_(scal[S.S_GAMMA:S.S_GAMMA + 1])
                mega_body(wa, wb, False)
            else:
                cur = torch.cuda.current_stream(self.device)
                ev_gd.record(cur)
                self.allred_stream.wait_event(ev_gd)
                with torch.cuda.stream(self.allred_stream):
                    self._allreduce_slot(S.S_GAMMA, 2)
                    ev_ar.record(self.allred_stream)
                self._spmv_overlapped(w, q)
                cur.wait_event(ev_ar)
                gsnap.copy_(scal[S.S_GAMMA:S.S_GAMMA + 1])
                S.pipelined_fused(z, t, p, xi, r, w, q, scal, self.partials,
                                  n, False)

        def try_capture_dist(kk):
            nonlocal dgraph
            if self._graphs.get(dist_failed_key):
                return
            try:
                gg = torch.cuda.CUDAGraph()
                if mega:
                    wa, wb = (w, w2) if kk % 2 == 0 else (w2, w)
                    with torch.cuda.graph(gg):
                        body_dist(wa, wb)
                    dgraphs[kk % 2] = gg
                    self._graphs[dist_key] = dgraphs
                else:
                    with torch.cuda.graph(gg):
                        body_dist()
                    dgraph = gg
                    self._graphs[dist_key] = gg
            except Exception as e:
                self._graphs[dist_failed_key] = True
                torch.cuda.synchronize(self.device)
                if (self.comm is None or self.comm.rank == 0):
                    print(f"[acg_amd] distributed graph capture unavailable "
                          f"({type(e).__name__}); eager iterations",
                          file=__import__("sys").stderr)

        k = 0
        while k < maxits:
            first = (k == 0)
            if not serial and dist_graph_ok:
                g = dgraphs[k % 2] if mega else dgraph
                if g is not None and k > 0:
                    # whole iteration (incl. allreduce) is in the graph
                    if k >= LAG and check(k - LAG):
                        break
                    g.replay()
                    # same slot/value as the eager path: the ALLREDUCED
                    # previous gamma, snapshotted inside the replay
                    cur = torch.cuda.current_stream(self.device)
                    j = k % (LAG + 1)
                    hostbuf[j].copy_(gsnap, non_blocking=True)
                    evdone[j].record(cur)
                    k += 1
                    res.niterations = k
                    continue
            # ONE 2-double allreduce per iteration (gamma,delta adjacent).
            # Multi-GPU non-megafused: the allreduce rides a dedicated side
            # stream so SpMV(q = A w) -- which does not read the scalars --
            # overlaps it; only the fused update waits (this is the point
            # of Ghysels-Vanroose pipelining: reference cghip.c:1750-1811).
            overlap_ar = (not serial) and not mega
            if overlap_ar:
                cur = torch.cuda.current_stream(self.device)
                ev_gd.record(cur)
                self.allred_stream.wait_event(ev_gd)
                with torch.cuda.stream(self.allred_stream):
                    with self.prof.span("allreduce", self.allred_stream):
                        self._allreduce_slot(S.S_GAMMA, 2)
                    ev_ar.record(self.allred_stream)
            else:
                with self.prof.span("allreduce"):
                    self._allreduce_slot(S.S_GAMMA, 2)
            # lagged host test of the previous iteration's gamma (read
            # BEFORE issuing this iteration's copy: LAG+1 buffers rotate)
            if k >= LAG and check(k - LAG):
                break
            if overlap_ar:
                # D2H of gamma chains off the allreduce, not the main stream
                # (the SpMV below must NOT wait for it -- that is the
                # overlap; only the fused update is ordered after it)
                j = k % (LAG + 1)
                self.copy_stream.wait_event(ev_ar)
                with torch.cuda.stream(self.copy_stream):
                    hostbuf[j].copy_(scal[S.S_GAMMA:S.S_GAMMA + 1],
                                     non_blocking=True)
                    evdone[j].record(self.copy_stream)
            else:
                issue_gamma_copy(k)  # same-stream: ordered by construction
            if mega:
                wa, wb = (w, w2) if k % 2 == 0 else (w2, w)
                g = graphs[k % 2]
                if g is not None and k > 0:  # graphs are first=False bodies
                    g.replay()
                else:
                    mega_body(wa, wb, first)
                    if graph_ok and graphs[k % 2] is None and k in (2, 3):
                        gg = torch.cuda.CUDAGraph()
                        with torch.cuda.graph(gg):
                            mega_body(wa, wb, False)
                        graphs[k % 2] = gg
                        self._graphs[wskey] = graphs
            elif graph is not None and k > 0:
                graph.replay()
            else:
                self._spmv_overlapped(w, q)
                if overlap_ar:
                    # the fused update reads gamma/delta (allreduce) and its
                    # finalize overwrites them (in-flight D2H copy): order
                    # after both
                    cur2 = torch.cuda.current_stream(self.device)
                    cur2.wait_event(ev_ar)
                    cur2.wait_event(evdone[k % (LAG + 1)])
                with self.prof.span("update"):
                    S.pipelined_fused(z, t, p, xi, r, w, q, scal, self.partials,
                                      n, first)
                if graph_ok and graph is None and k == 2:
                    # steady state (first=False): capture SpMV + fused update
                    graph = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(graph):
                        self._spmv_overlapped(w, q)
                        S.pipelined_fused(z, t, p, xi, r, w, q, scal,
                                          self.partials, n, False)
                    self._graphs[wskey] = graph
            if dist_graph_ok and not serial and (
                    (mega and k in (2, 3) and dgraphs[k % 2] is None)
                    or (not mega and k == 2 and dgraph is None)):
                try_capture_dist(k)
            k += 1
            res.niterations = k
        if not converged:
            # drain the lagged checks for the tail iterations
            for j in range(max(maxits - LAG, 0), maxits):
                if check(j):
                    break
        torch.cuda.synchronize(self.device)
        res.tsolve = time.perf_counter() - t0
        x.copy_(xi)
        if not converged and gamma_host is not None:
            res.rnrm2 = math.sqrt(max(gamma_host, 0.0))
        res.converged = converged
        nnz_full = self.local.nnzA + self.local.nnzO
        res.nflops = res.niterations * (cg_flops_per_iter(nnz_full, n) + 8.0 * n)
        res.halo_bytes_sent = self.halo.bytes_sent
        res.halo_msgs_sent = self.halo.nmsgs_sent
        self.niterations_total += res.niterations
        if self.prof.enabled:
            from .profiling import annotate_op_stats

            idxb = None
            if self.matfree is not None:
                idxb = -8.0  # matrix-free: no vals (8 B) and no cols read
            elif self.bsell is not None:
                idxb = 4.0 / (self.bsell[3] ** 2)
            annotate_op_stats(res, self.local, self.prof.collect(),
                              idx_bytes_per_nnz=idxb)
            self.prof.reset()
        return res
