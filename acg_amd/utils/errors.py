"""Error handling (reference: acg/error.{c,h}).

The reference defines an error-code enum for every backend plus
``acgerrmpi()`` (error.h:124-149), a *collective* error check that
all-reduces error codes across ranks so that a failure on one rank
aborts every rank instead of deadlocking in a later collective.

Here errors are Python exceptions; the collective agreement is
:func:`collective_raise`, which max-reduces a per-rank error flag over the
process group before anyone proceeds past a failure point.
"""

from __future__ import annotations

import enum


class ErrCode(enum.IntEnum):
    """Error codes, mirroring ACG_ERR_* (reference acg/error.h:54-103)."""

    SUCCESS = 0
    ERRNO = 1
    EOF = 2
    INVALID_VALUE = 3
    NOT_SUPPORTED = 4
    OVERFLOW = 5
    INVALID_FORMAT = 6
    HIP = 7
    RCCL = 8
    NOT_CONVERGED = 9
    FEXCEPT = 10


class AcgError(Exception):
    """Framework error carrying an :class:`ErrCode`."""

    def __init__(self, code: ErrCode, msg: str = ""):
        self.code = ErrCode(code)
        super().__init__(f"{self.code.name}: {msg}" if msg else self.code.name)


class NotConvergedError(AcgError):
    def __init__(self, msg: str = ""):
        super().__init__(ErrCode.NOT_CONVERGED, msg)


def errcodestr(code: int) -> str:
    """Human-readable string for an error code (acgerrcodestr, error.h)."""
    try:
        return ErrCode(code).name
    except ValueError:
        return f"unknown error {code}"


def collective_raise(comm, exc: BaseException | None) -> None:
    """Collective error agreement (reference acgerrmpi, acg/error.c:149).

    Every rank calls this with either ``None`` (no local error) or the
    exception it hit.  The error flag is max-reduced over ``comm``; if any
    rank failed, *all* ranks raise, so no rank hangs waiting in a later
    collective for a dead peer.

    ``comm`` is an :class:`acg_amd.dist.comm.Comm` (or None for serial).
    """
    code = 0
    if exc is not None:
        code = int(getattr(exc, "code", ErrCode.ERRNO))
        code = code if code > 0 else int(ErrCode.ERRNO)
    if comm is not None and comm.size > 1:
        code = comm.allreduce_max_int(code)
    if exc is not None:
        raise exc
    if code != 0:
        raise AcgError(ErrCode(code), "error on a remote rank")
