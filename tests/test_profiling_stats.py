"""Profiling subsystem, stats report, dumps, binary partition files (CPU)."""

import io

import numpy as np

from acg_amd.gen import STENCIL_5PT_2D, stencil_global
from acg_amd.part import (extract_subdomains, partition_rows,
                          read_partition_file, write_partition_file)
from acg_amd.solvers.base import OpStats, SolveResult
from acg_amd.solvers.profiling import annotate_op_stats, write_stats


def _system(nparts=1):
    A = stencil_global(12, 12, 1, STENCIL_5PT_2D)
    part = partition_rows(A, nparts)
    return A, extract_subdomains(A, part, nparts)


def test_write_stats_serial():
    A, systems = _system()
    S = systems[0]
    res = SolveResult(solver="test", niterations=10, tsolve=0.5,
                      bnrm2=1.0, r0nrm2=1.0, rnrm2=1e-9, converged=True)
    ops = {"spmvA": OpStats(seconds=0.3, count=10),
           "update": OpStats(seconds=0.1, count=10)}
    annotate_op_stats(res, S, ops)
    assert res.ops["spmvA"].flops > 0
    assert res.ops["spmvA"].bytes > 0
    buf = io.StringIO()
    write_stats(res, S, comm=None, file=buf)
    out = buf.getvalue()
    assert "per-op timing" in out and "spmvA" in out


def test_dump():
    A, systems = _system(nparts=2)
    buf = io.StringIO()
    systems[0].dump(file=buf)
    s = buf.getvalue()
    assert "LocalSystem(rank=0/2" in s and "halo:" in s


def test_partition_file_binary_roundtrip(tmp_path):
    part = np.random.default_rng(0).integers(0, 4, size=100).astype(np.int32)
    for binary in (False, True):
        p = tmp_path / f"part_{binary}.mtx"
        write_partition_file(p, part, binary=binary)
        back = read_partition_file(p, 100, binary=binary)
        np.testing.assert_array_equal(back, part)


def test_event_profiler_cpu_noop():
    from acg_amd.solvers.profiling import EventProfiler

    prof = EventProfiler(enabled=False)
    with prof.span("x"):
        pass
    assert prof.collect() == {}
