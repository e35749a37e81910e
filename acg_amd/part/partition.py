"""Row/graph partitioning (reference: acg/metis.{c,h}, acg/graph.c:510-528).

The reference delegates to METIS (METIS_PartGraphRecursive/Kway with seed
control, metis.c:80-436).  METIS is not available in this image, so this
module implements its role natively:

- ``block``: contiguous balanced row blocks (optimal for banded/structured
  matrices such as the Poisson and Queen-like stencil operators the
  benchmarks use — minimises halo for banded orderings).
- ``rgb``: recursive graph bisection via BFS level structures from a
  pseudo-peripheral vertex, with weighted splits for arbitrary part
  counts — a METIS-recursive stand-in for general irregular graphs.

A precomputed partition vector can also be read from / written to a
Matrix Market integer array file, compatible with the reference's
``--partition`` option and ``mtxpartition`` tool output (1-based parts).
"""

from __future__ import annotations

import numpy as np

from ..utils.errors import AcgError, ErrCode


def _full_adjacency(A):
    """Symmetric adjacency (both triangles, no diagonal) as scipy CSR.

    Reference analog: metis.c:225-248 builds both-triangle adjacency from
    the packed upper-triangular input.
    """
    import scipy.sparse as sp

    rows = np.repeat(np.arange(A.n, dtype=np.int64), np.diff(A.rowptr))
    cols = A.colidx
    offd = rows != cols
    r, c = rows[offd], cols[offd]
    i = np.concatenate([r, c])
    j = np.concatenate([c, r])
    G = sp.csr_matrix((np.ones(len(i), dtype=np.int8), (i, j)), shape=(A.n, A.n))
    return G


def _pseudo_peripheral(G, start: int) -> int:
    """Find a pseudo-peripheral vertex by repeated BFS."""
    from scipy.sparse.csgraph import breadth_first_order

    v = start
    last_ecc = -1
    for _ in range(4):
        order, _pred = breadth_first_order(G, v, directed=False, return_predecessors=True)
        u = order[-1]
        ecc = len(order)
        if ecc == last_ecc:
            break
        last_ecc = ecc
        v = int(u)
    return v


def _bisect(G, nodes: np.ndarray, rng,
            nleft: int | None = None) -> tuple[np.ndarray, np.ndarray]:
    """Split ``nodes`` along a BFS level structure: first ``nleft`` (default
    half) in BFS order go left -- weighted splits support arbitrary part
    counts like METIS-recursive."""
    from scipy.sparse.csgraph import breadth_first_order

    sub = G[nodes][:, nodes]
    n = len(nodes)
    half = n // 2 if nleft is None else int(nleft)
    # BFS over the subgraph (may be disconnected: loop over components)
    visited = np.zeros(n, dtype=bool)
    orderall = np.empty(n, dtype=np.int64)
    filled = 0
    while filled < n:
        unv = np.where(~visited)[0]
        start = _pseudo_peripheral(sub, int(unv[rng.integers(len(unv))] if len(unv) > 1 else unv[0]))
        if visited[start]:
            start = int(unv[0])
        order = breadth_first_order(sub, start, directed=False, return_predecessors=False)
        order = order[~visited[order]]
        orderall[filled:filled + len(order)] = order
        visited[order] = True
        filled += len(order)
    left = nodes[orderall[:half]]
    right = nodes[orderall[half:]]
    return left, right


def partition_rows(A, nparts: int, seed: int = 0, method: str = "auto") -> np.ndarray:
    """Partition matrix rows into ``nparts`` parts.

    Reference analog: acgsymcsrmatrix_partition_rows (symcsrmatrix.c:656)
    -> metis_partgraphsym (metis.c:80).  Returns int32 part[n] in
    [0, nparts).
    """
    n = A.n
    if nparts <= 0:
        raise AcgError(ErrCode.INVALID_VALUE, f"nparts={nparts}")
    if nparts == 1:
        return np.zeros(n, dtype=np.int32)
    if method == "auto":
        method = "block"
    if method == "block":
        # contiguous balanced blocks
        part = ((np.arange(n, dtype=np.int64) * nparts) // n).astype(np.int32)
        return part
    if method == "rgb":
        if nparts > n:
            raise AcgError(ErrCode.INVALID_VALUE, "more parts than rows")
        G = _full_adjacency(A)
        rng = np.random.default_rng(seed)
        part = np.zeros(n, dtype=np.int32)
        groups = [(np.arange(n, dtype=np.int64), 0, nparts)]
        while groups:
            nodes, base, k = groups.pop()
            if k == 1:
                part[nodes] = base
                continue
            # weighted bisection (kl:kr) supports arbitrary nparts, like
            # METIS_PartGraphRecursive
            kl = (k + 1) // 2
            kr = k - kl
            left, right = _bisect(G, nodes, rng, (len(nodes) * kl) // k)
            groups.append((left, base, kl))
            groups.append((right, base + kl, kr))
        return part
    raise AcgError(ErrCode.NOT_SUPPORTED, f"partition method {method!r}")


def read_partition_file(path, n: int | None = None, binary: bool = False,
                        gzipped: bool = False, idxsize: int = 64) -> np.ndarray:
    """Read a partition vector (mtx integer array, 1-based parts).

    Reference: --partition / --binary-partition handling at
    hip/acg-hip.c:1513-1641 and the mtxpartition tool output format.
    """
    from ..io.mtx import read_mtx

    m = read_mtx(path, binary=binary, gzipped=gzipped, idxsize=idxsize)
    part = np.asarray(m.a, dtype=np.int64)
    if n is not None and len(part) != n:
        raise AcgError(ErrCode.INVALID_VALUE,
                       f"partition length {len(part)} != matrix rows {n}")
    return (part - 1).astype(np.int32)


def write_partition_file(path, part: np.ndarray, binary: bool = False,
                         idxsize: int = 64) -> None:
    """Write a 1-based partition vector as mtx integer array (mtxpartition)."""
    from ..io.mtx import MtxFile, write_mtx

    part = np.asarray(part)
    m = MtxFile(object="matrix", format="array", field_="integer",
                symmetry="general", nrows=len(part), ncols=1, nnz=len(part),
                a=part.astype(np.int64) + 1)
    write_mtx(path, m, binary=binary, idxsize=idxsize)
