#!/usr/bin/env python3
"""Offline partitioner (reference: mtxpartition/mtxpartition.c): read a
symmetric matrix, partition its rows into --parts parts, write the 1-based
part vector as a Matrix Market integer array to stdout (or --output)."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from acg_amd.core.symcsr import SymCSRMatrix  # noqa: E402
from acg_amd.io.mtx import read_mtx  # noqa: E402
from acg_amd.part import partition_rows, write_partition_file  # noqa: E402


def main(argv=None):
    p = argparse.ArgumentParser(prog="mtxpartition")
    p.add_argument("input")
    p.add_argument("--parts", type=int, required=True)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--method", choices=("block", "rgb", "ml", "auto"),
                   default="auto")
    p.add_argument("-z", "--gzip", action="store_true")
    p.add_argument("--binary", action="store_true")
    p.add_argument("--output", default=None)
    args = p.parse_args(argv)
    m = read_mtx(args.input, gzipped=args.gzip, binary=args.binary)
    A = SymCSRMatrix.from_mtxfile(m)
    part = partition_rows(A, args.parts, seed=args.seed, method=args.method)
    if args.output:
        write_partition_file(args.output, part)
    else:
        import numpy as np

        from acg_amd.io.mtx import MtxFile, write_mtx

        mf = MtxFile(object="matrix", format="array", field_="integer",
                     symmetry="general", nrows=len(part), ncols=1,
                     nnz=len(part), a=np.asarray(part, dtype=np.int64) + 1)
        write_mtx(sys.stdout, mf)
    return 0


if __name__ == "__main__":
    sys.exit(main())
