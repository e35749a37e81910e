"""Per-op GPU profiling + distributed statistics report.

Reference: the ACG_ENABLE_PROFILING machinery (acgEventRecord pairs around
every BLAS/SpMV/comm call, cghip.c:71-75, 604-626, 1073-1111) and the
statistics block printed by acgsolverhip_fwritempi (cghip.c:2003-2270):
per-op seconds/flops/bytes/GB/s, per-rank halo traffic, Gflop/s.

Here: :class:`EventProfiler` wraps op classes in hipEvent pairs
(torch.cuda.Event, only when enabled -- zero overhead otherwise);
:func:`write_stats` renders the reduced multi-rank report.
"""

from __future__ import annotations

import sys

import torch

from .base import OpStats, SolveResult


class EventProfiler:
    """hipEvent-pair timing per op class (enabled => ~4 us/op overhead)."""

    def __init__(self, enabled: bool = False):
        self.enabled = enabled
        self.pairs: dict[str, list] = {}

    def span(self, name: str, stream=None):
        if not self.enabled:
            return _NullSpan()
        return _Span(self, name, stream)

    def collect(self) -> dict:
        """Sum elapsed ms per op class (synchronizes)."""
        out = {}
        if not self.enabled:
            return out
        torch.cuda.synchronize()
        for name, pairs in self.pairs.items():
            secs = sum(a.elapsed_time(b) for a, b in pairs) / 1e3
            st = OpStats(seconds=secs, count=len(pairs))
            out[name] = st
        return out

    def reset(self):
        self.pairs.clear()


class _NullSpan:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


class _Span:
    def __init__(self, prof: EventProfiler, name: str, stream):
        self.prof = prof
        self.name = name
        self.stream = stream or torch.cuda.current_stream()

    def __enter__(self):
        self.e0 = torch.cuda.Event(enable_timing=True)
        self.e0.record(self.stream)
        return self

    def __exit__(self, *a):
        e1 = torch.cuda.Event(enable_timing=True)
        e1.record(self.stream)
        self.prof.pairs.setdefault(self.name, []).append((self.e0, e1))
        return False


def annotate_op_stats(res: SolveResult, local, ops_times: dict,
                      idx_bytes_per_nnz: float | None = None) -> None:
    """Attach analytic flop/byte counts to measured per-op seconds
    (reference: always-on counters, cghip.h:109-118).

    ``idx_bytes_per_nnz``: index bytes per nonzero of the operator format
    actually used (4 for int32 CSR/SELL, 4/dof^2 for Block-SELL)."""
    n = local.nowned
    it = max(res.niterations, 1)
    nnzA, nnzO = local.nnzA, local.nnzO
    if idx_bytes_per_nnz is not None:
        colb = idx_bytes_per_nnz
    else:
        _ac = getattr(local, "A_colidx", None)
        colb = _ac.dtype.itemsize if _ac is not None and hasattr(_ac, "dtype") else 4
    model = {
        "spmvA": (2.0 * nnzA, nnzA * (8 + colb) + 16.0 * n),
        "spmvO": (2.0 * nnzO, nnzO * (8 + colb) + 16.0 * local.nborder),
        "update": (6.0 * n, 13.0 * 8 * n),
        "update_classic": (5.0 * n, 5.0 * 8 * n),
        "daypx": (2.0 * n, 3.0 * 8 * n),
        "dot": (2.0 * n, 2.0 * 8 * n),
        "halo": (0.0, 8.0 * (local.halo.sendsize + local.halo.recvsize)),
        "allreduce": (0.0, 16.0),
    }
    for name, st in ops_times.items():
        fl, by = model.get(name, (0.0, 0.0))
        st.flops = fl * it
        st.bytes = by * it
    res.ops = ops_times


def write_stats(res: SolveResult, local, comm=None, file=None) -> None:
    """Distributed statistics report (reference acgsolverhip_fwritempi).

    Every rank calls this; rank 0 prints.  Includes per-rank halo traffic
    (B/iteration, messages/iteration) like cghip.c:2184-2270."""
    file = file or sys.stderr
    halo_row = {
        "rank": local.rank,
        "nowned": local.nowned,
        "nghost": local.nghost,
        "neighbours": int(local.halo.nrecipients),
        "sent_B_per_it": 8 * local.halo.sendsize,
        "recv_B_per_it": 8 * local.halo.recvsize,
        "msgs_per_it": int(local.halo.nrecipients),
    }
    rows = comm.gather_object(halo_row) if comm else [halo_row]
    if rows is None:
        return
    print(res.summary(), file=file)
    if res.ops:
        print("per-op timing (rank 0):", file=file)
        for name, st in sorted(res.ops.items(), key=lambda kv: -kv[1].seconds):
            print(f"  {name:14s} {st.seconds * 1e3:9.3f} ms  {st.count:6d} calls"
                  f"  {st.gbytes_rate:8.1f} GB/s  {st.gflops_rate:8.1f} Gflop/s",
                  file=file)
    if comm and comm.size > 1:
        print("per-rank halo traffic:", file=file)
        print("  rank   owned    ghost  nbrs   sent B/it   recv B/it  msg/it",
              file=file)
        for h in rows:
            print(f"  {h['rank']:4d} {h['nowned']:8d} {h['nghost']:8d} "
                  f"{h['neighbours']:5d} {h['sent_B_per_it']:11d} "
                  f"{h['recv_B_per_it']:11d} {h['msgs_per_it']:7d}", file=file)
