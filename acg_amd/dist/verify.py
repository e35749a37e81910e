"""Distributed halo-pattern consistency verification (dry-run asserts).

Run once before the first real exchange (bench warmup / CLI verbose): a
collective cross-rank audit of the halo pattern so the FIRST multi-GPU run
fails loudly at setup instead of deadlocking or silently corrupting ghosts.

The reference has no equivalent (its debug story is acghalo_fwrite dumps,
halo.c:356); this is the "correct by construction, checked by collective
audit" piece of the MI355X build, where N>=2 paths may meet real hardware
for the first time at scale.

Checks (all ranks gather their HaloDescriptor to rank 0):
  1. pairing symmetry: p sends to q  <=>  q receives from p,
     with equal counts;
  2. global-id agreement: the global row ids p gathers for q
     (owned_global[sendidx]) equal the global ids of q's ghost-tail
     segment for p (ghost_global[rdispls:+count]) -- order included,
     because ghosts are received IN PLACE into the sorted tail;
  3. every ghost is owned by its claimed sender (sanity of the partition).
Failures raise AcgError on ALL ranks (collective_raise semantics).
"""

from __future__ import annotations

import numpy as np

from ..utils.errors import AcgError, ErrCode, collective_raise


def halo_descriptor(S) -> dict:
    """Rank-local summary shipped to root for the audit (small: O(halo)).

    Systems without global-id arrays (device-generated slabs hold no
    owned_global/ghost_global -- they would be O(n) host memory at the
    1B-row scale) get a counts-only descriptor; the audit then checks
    pairing symmetry and counts but skips the gid comparison."""
    h = S.halo
    og = getattr(S, "owned_global", None)
    gg = getattr(S, "ghost_global", None)
    have_gids = og is not None and gg is not None
    send_gids = {}
    for i in range(h.nrecipients):
        lo = int(h.sdispls[i])
        hi = lo + int(h.sendcounts[i])
        idx = np.asarray(h.sendidx[lo:hi], dtype=np.int64)
        if len(idx) and (idx.min() < 0 or idx.max() >= S.nowned):
            raise AcgError(ErrCode.INVALID_VALUE,
                           f"rank {S.rank}: sendidx outside owned range")
        send_gids[int(h.recipients[i])] = \
            np.asarray(og)[idx] if have_gids else hi - lo
    recv_gids = {}
    for i in range(h.nsenders):
        lo = int(h.rdispls[i])
        hi = lo + int(h.recvcounts[i])
        recv_gids[int(h.senders[i])] = \
            np.asarray(gg[lo:hi], dtype=np.int64) if have_gids else hi - lo
    return {"rank": S.rank, "send": send_gids, "recv": recv_gids,
            "nowned": S.nowned, "nghost": S.nghost}


def _audit(descs: list) -> None:
    nparts = len(descs)
    by_rank = {d["rank"]: d for d in descs}
    if sorted(by_rank) != list(range(nparts)):
        raise AcgError(ErrCode.INVALID_VALUE,
                       f"halo audit: ranks {sorted(by_rank)} != 0..{nparts - 1}")
    def _count(v):
        return v if isinstance(v, (int, np.integer)) else len(v)

    for p in range(nparts):
        dp = by_rank[p]
        for q, gids in dp["send"].items():
            dq = by_rank.get(q)
            if dq is None or p not in dq["recv"]:
                raise AcgError(ErrCode.INVALID_VALUE,
                               f"halo audit: {p} sends to {q} but {q} does "
                               f"not expect {p}")
            want = dq["recv"][p]
            if _count(gids) != _count(want):
                raise AcgError(ErrCode.INVALID_VALUE,
                               f"halo audit: {p}->{q} count {_count(gids)} "
                               f"!= expected {_count(want)}")
            if (not isinstance(gids, (int, np.integer))
                    and not isinstance(want, (int, np.integer))
                    and not np.array_equal(gids, want)):
                k = int(np.argmax(gids != want))
                raise AcgError(ErrCode.INVALID_VALUE,
                               f"halo audit: {p}->{q} global-id mismatch at "
                               f"slot {k}: sends {gids[k]}, {q} expects "
                               f"{want[k]} (in-place ghost tail order)")
        for q in dp["recv"]:
            dq = by_rank.get(q)
            if dq is None or p not in dq["send"]:
                raise AcgError(ErrCode.INVALID_VALUE,
                               f"halo audit: {p} expects from {q} but {q} "
                               f"does not send to {p}")


def verify_halo(S, comm) -> None:
    """Collective halo audit; call on every rank before the first solve.

    Raises AcgError on every rank if any check fails; no-op serial."""
    if comm is None or comm.size == 1:
        return
    err = None
    desc = None
    try:
        desc = halo_descriptor(S)
    except Exception as e:
        err = e
    collective_raise(comm, err)
    descs = comm.gather_object(desc)
    err = None
    if comm.rank == 0:
        try:
            _audit(descs)
        except Exception as e:
            err = e
    collective_raise(comm, err)
