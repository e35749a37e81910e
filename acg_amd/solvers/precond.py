"""Jacobi (diagonal) preconditioning as a system transform.

Jacobi-PCG is algebraically CG on the symmetrically scaled system
``(D^-1/2 A D^-1/2) (D^1/2 x) = D^-1/2 b``.  Applying the scaling ONCE at
setup therefore composes diagonal preconditioning with EVERY solver in
the stack unchanged -- classic, pipelined (Ghysels-Vanroose's own paper
is about the preconditioned variant; ours becomes it), megafused,
graph-captured multi-GPU iterations, every operator format -- instead of
needing a dedicated PCG loop per variant (solve_jacobi remains as the
explicit-M reference implementation).

Distributed: the off-diagonal entry a_ij needs d_j^-1/2 for GHOST
columns j, obtained with ONE setup-time halo exchange of the scale
vector over the production engine.

Beyond reference: aCG has no preconditioning at all (PCNONE even in its
PETSc oracle, cgpetsc.c:181-193).  Opt-in (`--jacobi-scale`): parity
benches stay unpreconditioned.  Tolerances apply to the SCALED residual
(the preconditioned norm); the CLI's true-residual line reports the
unscaled truth.
"""

from __future__ import annotations

from dataclasses import replace

import numpy as np

from ..part.subdomain import LocalSystem
from ..utils.errors import AcgError, ErrCode


def jacobi_scale_system(S: LocalSystem, comm=None, device=None):
    """Return (S_scaled, s_owned): the symmetrically scaled system and the
    per-owned-row scale ``s = diag(A)^-1/2`` (``x = s * x_scaled`` maps
    solutions back; ``b_scaled = s * b``)."""
    n = S.nowned
    rows = np.repeat(np.arange(n, dtype=np.int64), np.diff(S.A_rowptr))
    dmask = rows == S.A_colidx
    d = np.zeros(n, dtype=np.float64)
    d[rows[dmask]] = S.A_vals[dmask]
    if (d <= 0).any():
        raise AcgError(ErrCode.INVALID_VALUE,
                       "jacobi scaling needs a strictly positive diagonal")
    s = 1.0 / np.sqrt(d)
    # full local scale vector incl. ghost tail (one halo exchange)
    s_full = np.zeros(S.nowned + S.nghost, dtype=np.float64)
    s_full[:n] = s
    if S.nghost > 0:
        if comm is None or comm.size == 1:
            raise AcgError(ErrCode.INVALID_VALUE,
                           "system has ghosts but no communicator")
        import torch

        from ..dist.halo import HaloExchange

        dev = device if (device is not None
                         and getattr(comm, "kind", "gloo") == "rccl") else "cpu"
        t = torch.from_numpy(s_full).to(dev)
        HaloExchange(S.halo, n, dev, comm).exchange(t)
        s_full = t.cpu().numpy()
    A_vals = S.A_vals * s_full[rows] * s_full[S.A_colidx]
    if S.nnzO > 0:
        orows = S.ninterior + np.repeat(
            np.arange(S.nborder, dtype=np.int64), np.diff(S.O_rowptr))
        O_vals = S.O_vals * s_full[orows] * s_full[S.O_colidx]
    else:
        O_vals = S.O_vals
    return replace(S, A_vals=A_vals, O_vals=O_vals), s
