"""Solver result / statistics structures.

Reference: struct acgsolverhip counters (cghip.h:109-118) and the
statistics block printed by acgsolverhip_fwritempi (cghip.c:2003-2270):
iterations, b/r0/r 2-norms, per-op seconds/flops/bytes, halo traffic.
"""

from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class OpStats:
    seconds: float = 0.0
    count: int = 0
    flops: float = 0.0
    bytes: float = 0.0

    @property
    def gflops_rate(self) -> float:
        return self.flops / self.seconds / 1e9 if self.seconds > 0 else 0.0

    @property
    def gbytes_rate(self) -> float:
        return self.bytes / self.seconds / 1e9 if self.seconds > 0 else 0.0


@dataclass
class SolveResult:
    converged: bool = False
    niterations: int = 0
    bnrm2: float = 0.0
    r0nrm2: float = 0.0
    rnrm2: float = 0.0
    tsolve: float = 0.0
    maxits: int = 0
    res_atol: float = 0.0
    res_rtol: float = 0.0
    solver: str = ""
    nflops: float = 0.0
    nbytes: float = 0.0
    ops: dict = field(default_factory=dict)  # name -> OpStats
    halo_bytes_sent: int = 0
    halo_msgs_sent: int = 0
    nranks: int = 1

    @property
    def iters_per_s(self) -> float:
        return self.niterations / self.tsolve if self.tsolve > 0 else 0.0

    @property
    def gflops(self) -> float:
        return self.nflops / self.tsolve / 1e9 if self.tsolve > 0 else 0.0

    def summary(self) -> str:
        lines = [
            f"solver: {self.solver}",
            f"ranks: {self.nranks}",
            f"iterations: {self.niterations} (max {self.maxits})",
            f"converged: {self.converged}",
            f"b 2-norm: {self.bnrm2:.6e}",
            f"initial residual 2-norm: {self.r0nrm2:.6e}",
            f"final residual 2-norm: {self.rnrm2:.6e}",
            f"solve time: {self.tsolve:.6f} s"
            f" ({self.iters_per_s:.2f} it/s, {self.gflops:.1f} Gflop/s)",
        ]
        if self.halo_bytes_sent:
            per_it = self.halo_bytes_sent / max(self.niterations, 1)
            lines.append(
                f"halo: {self.halo_bytes_sent} B sent total, {per_it:.0f} B/it, "
                f"{self.halo_msgs_sent / max(self.niterations, 1):.1f} msg/it")
        for name, op in self.ops.items():
            if op.count:
                lines.append(
                    f"  {name}: {op.seconds:.4f} s / {op.count} calls"
                    f" ({op.gbytes_rate:.1f} GB/s, {op.gflops_rate:.1f} Gflop/s)")
        return "\n".join(lines)


def cg_flops_per_iter(nnz_full: int, n: int) -> float:
    """Flops per classic-CG iteration: SpMV 2*nnz + 2 dots + 3 axpy-likes."""
    return 2.0 * nnz_full + 10.0 * n


def cg_bytes_per_iter(nnz_full: int, n: int, colbytes: int = 4) -> float:
    """Approximate HBM traffic per iteration (fp64 vals + colidx + vectors)."""
    spmv = nnz_full * (8 + colbytes) + 8.0 * n * 3
    vecs = 8.0 * n * 9
    return spmv + vecs
