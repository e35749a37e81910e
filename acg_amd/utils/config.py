"""Index-width policy (reference: acg/config.h ACG_IDX_SIZE / acgidx_t).

The reference fixes ``acgidx_t`` at build time (32 or 64 bits,
config.h:62-94).  Here widths are chosen *per array* at runtime:

- global row ids: always int64 (the 2048^3 Poisson has 8.6e9 rows),
- CSR/SELL row pointers: int64,
- local column indices: int32 whenever nowned+nghost < 2^31 (halves
  index bandwidth on the SpMV hot path vs a 64-bit build), int64 otherwise,
- binary Matrix Market files: the stored width is the file's ``idxsize``
  (32|64), chosen by the writer exactly like the reference's build flag.
"""

from __future__ import annotations

import numpy as np

IDX_GLOBAL = np.int64


def col_dtype(ncols_local: int):
    """Column-index dtype for a local operator with this many columns."""
    return np.int32 if ncols_local < 2**31 else np.int64


def idx_dtype(idxsize: int):
    """File/binary index dtype for a given --idxsize (reference acgidx_t)."""
    if idxsize == 32:
        return np.int32
    if idxsize == 64:
        return np.int64
    raise ValueError(f"idxsize must be 32 or 64, got {idxsize}")
