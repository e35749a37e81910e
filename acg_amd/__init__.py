"""acg_amd — an MI355X-native distributed conjugate-gradient framework.

A from-scratch re-design of the capabilities of ParCoreLab/aCG
(reference: /root/reference) for AMD Instinct MI355X (gfx950):

- PyTorch-ROCm driver layer: vectors and CSR arrays live in torch tensors.
- Hand-written CDNA4 HIP kernels for the hot path (CSR SpMV, fused
  dot/axpy/daypx, pipelined 6-vector update, halo pack/unpack).
- RCCL over xGMI (torch.distributed, backend "nccl") for the neighbour
  halo exchange (grouped send/recv) and the 1-2 double all-reduce,
  on side HIP streams so communication overlaps the split SpMV.
- Own row/graph partitioner (no METIS dependency in this image).

Layer map (mirrors reference SURVEY.md §1):
  utils/      error codes, timing, numfmt    (acg/error.*, time.h, fmtspec.*)
  io/         Matrix Market text/gz/binary   (acg/mtxfile.*)
  core/       SymCSRMatrix, ghost-tail vecs  (acg/symcsrmatrix.*, vector.*)
  part/       graph partition + subdomains   (acg/graph.*, metis.*)
  dist/       comm wrapper + halo exchange   (acg/comm.*, halo.*)
  ops/        gfx950 HIP kernels             (acg/cg-kernels-hip.hip, halo-kernels-hip.hip)
  solvers/    CG / pipelined CG, CPU + GPU   (acg/cg.*, cghip.*)
  cli.py      acg-compatible driver          (hip/acg-hip.c)
"""

__version__ = "0.2.0"

from .utils.errors import AcgError  # noqa: F401
