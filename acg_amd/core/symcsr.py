"""Symmetric sparse matrix in packed upper-triangular CSR form.

Reference: acg/symcsrmatrix.{c,h} (struct acgsymcsrmatrix, symcsrmatrix.h:62;
COO assembly _init_real_double, symcsrmatrix.c:66; packed->full conversion
_dsymv_init, symcsrmatrix.c:760-845; CPU dsymv, symcsrmatrix.c:863-997).

The matrix is held as numpy arrays on the host (assembly/partitioning are
host-side preprocessing); the GPU solvers consume the *full* CSR produced
by :meth:`SymCSRMatrix.to_full_csr` (or the distributed split from
acg_amd.part.subdomain) as torch tensors.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

from ..utils.errors import AcgError, ErrCode


@dataclass
class FullCSR:
    """Full (both triangles) CSR operator, ready for SpMV."""

    nrows: int
    ncols: int
    rowptr: np.ndarray  # int64[nrows+1]
    colidx: np.ndarray  # int32 or int64 [nnz]
    vals: np.ndarray  # float64[nnz]

    @property
    def nnz(self) -> int:
        return int(self.rowptr[-1])

    def spmv(self, x: np.ndarray) -> np.ndarray:
        """CPU reference SpMV y = A x (scipy-backed)."""
        import scipy.sparse as sp

        A = sp.csr_matrix((self.vals, self.colidx, self.rowptr), shape=(self.nrows, self.ncols))
        return A @ x


class SymCSRMatrix:
    """Sparse symmetric matrix, packed upper triangle in CSR.

    Only entries with ``row <= col`` are stored (the strict lower triangle
    is implicit by symmetry), matching the reference's packed storage.
    """

    def __init__(self, n: int, rowptr: np.ndarray, colidx: np.ndarray, vals: np.ndarray):
        self.n = int(n)
        self.rowptr = np.asarray(rowptr, dtype=np.int64)
        self.colidx = np.asarray(colidx, dtype=np.int64)
        self.vals = np.asarray(vals, dtype=np.float64)

    @property
    def nnz_stored(self) -> int:
        return int(self.rowptr[-1])

    @property
    def nnz_full(self) -> int:
        ndiag = int(np.count_nonzero(self.colidx == self._rows()))
        return 2 * self.nnz_stored - ndiag

    def _rows(self) -> np.ndarray:
        return np.repeat(np.arange(self.n, dtype=np.int64), np.diff(self.rowptr))

    # -- assembly ---------------------------------------------------------

    @classmethod
    def from_coo(cls, n: int, rowidx: np.ndarray, colidx: np.ndarray, vals: np.ndarray,
                 symmetric_input: bool = True) -> "SymCSRMatrix":
        """Assemble from COO entries (reference _init_real_double).

        ``symmetric_input=True`` means the COO lists each off-diagonal pair
        once (standard MTX ``symmetric``); entries are canonicalised to the
        upper triangle (row <= col), duplicates summed.

        Uses the native C++/OpenMP assembly (acg_amd.host) when built;
        falls back to vectorised numpy.
        """
        i = np.asarray(rowidx, dtype=np.int64)
        j = np.asarray(colidx, dtype=np.int64)
        v = np.asarray(vals, dtype=np.float64)
        if i.shape != j.shape or i.shape != v.shape:
            raise AcgError(ErrCode.INVALID_VALUE, "COO array length mismatch")
        # validate BOTH indices up front: negative numpy indices wrap
        # silently and the native scatter would write past its buffers
        if len(i) and (int(i.min()) < 0 or int(j.min()) < 0
                       or int(i.max()) >= n or int(j.max()) >= n):
            raise AcgError(ErrCode.INVALID_VALUE,
                           f"COO index outside [0, {n})")
        try:
            from ..host import _acg_host as H

            rowptr, cols, vv = H.coo_to_sym_csr(n, i, j, v)
            return cls(n, rowptr, cols, vv)
        except ImportError:
            pass
        # canonicalise to upper triangle
        lo = i > j
        iu = np.where(lo, j, i)
        ju = np.where(lo, i, j)
        # sort by (row, col) and sum duplicates
        order = np.lexsort((ju, iu))
        iu, ju, v = iu[order], ju[order], v[order]
        if len(iu):
            new = np.empty(len(iu), dtype=bool)
            new[0] = True
            new[1:] = (iu[1:] != iu[:-1]) | (ju[1:] != ju[:-1])
            grp = np.cumsum(new) - 1
            iu2 = iu[new]
            ju2 = ju[new]
            v2 = np.zeros(int(grp[-1]) + 1, dtype=np.float64)
            np.add.at(v2, grp, v)
        else:
            iu2, ju2, v2 = iu, ju, v
        rowptr = np.zeros(n + 1, dtype=np.int64)
        np.add.at(rowptr, iu2 + 1, 1)
        np.cumsum(rowptr, out=rowptr)
        return cls(n, rowptr, ju2, v2)

    @classmethod
    def from_mtxfile(cls, m) -> "SymCSRMatrix":
        """Build from an io.MtxFile (must be matrix/coordinate/symmetric)."""
        if m.object != "matrix" or m.format != "coordinate":
            raise AcgError(ErrCode.INVALID_FORMAT, "need matrix coordinate file")
        if m.symmetry != "symmetric":
            raise AcgError(ErrCode.INVALID_FORMAT, "need a symmetric matrix")
        if m.nrows != m.ncols:
            raise AcgError(ErrCode.INVALID_VALUE, "matrix must be square")
        vals = np.asarray(m.a, dtype=np.float64)
        return cls.from_coo(m.nrows, m.rowidx, m.colidx, vals)

    # -- packed -> full conversion (reference _dsymv_init) ---------------

    def to_full_csr(self, eps: float = 0.0) -> FullCSR:
        """Expand the packed upper triangle to a full CSR operator.

        ``eps`` is added to every diagonal entry (reference --epsilon
        diagonal shift, symcsrmatrix.c:760-845).  Native C++/OpenMP
        expansion (O(nnz), parallel) when built; numpy lexsort fallback.
        """
        try:
            from ..host import _acg_host as H

            col32 = self.n < 2**31
            rowptr, cols, vv = H.sym_expand_full(self.n, self.rowptr,
                                                 self.colidx, self.vals,
                                                 eps, col32)
            return FullCSR(self.n, self.n, rowptr, cols, vv)
        except ImportError:
            pass
        rows_u = self._rows()
        cols_u = self.colidx
        vals_u = self.vals
        diag = rows_u == cols_u
        offd = ~diag
        # full COO: upper entries + mirrored strict-lower entries
        fi = np.concatenate([rows_u, cols_u[offd]])
        fj = np.concatenate([cols_u, rows_u[offd]])
        fv = np.concatenate([vals_u, vals_u[offd]])
        if eps:
            dmask = np.concatenate([diag, np.zeros(int(offd.sum()), dtype=bool)])
            fv = fv.copy()
            fv[dmask] += eps
        order = np.lexsort((fj, fi))
        fi, fj, fv = fi[order], fj[order], fv[order]
        rowptr = np.zeros(self.n + 1, dtype=np.int64)
        np.add.at(rowptr, fi + 1, 1)
        np.cumsum(rowptr, out=rowptr)
        colidx = fj.astype(np.int32) if self.n < 2**31 else fj
        return FullCSR(self.n, self.n, rowptr, colidx, fv)

    # -- CPU reference ops ------------------------------------------------

    def dsymv(self, x: np.ndarray, y: np.ndarray | None = None,
              alpha: float = 1.0, beta: float = 0.0) -> np.ndarray:
        """y = alpha*A*x + beta*y using the packed storage (CPU oracle)."""
        import scipy.sparse as sp

        rows_u = self._rows()
        U = sp.csr_matrix((self.vals, (rows_u, self.colidx)), shape=(self.n, self.n))
        d = U.diagonal()
        full = U + U.T
        full.setdiag(d)
        out = alpha * (full @ x)
        if y is not None and beta != 0.0:
            out += beta * y
        return out

    def to_scipy_full(self):
        import scipy.sparse as sp

        f = self.to_full_csr()
        return sp.csr_matrix((f.vals, f.colidx, f.rowptr), shape=(f.nrows, f.ncols))
