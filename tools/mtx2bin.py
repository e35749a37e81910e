#!/usr/bin/env python3
"""Convert Matrix Market text files to the binary format (reference:
mtx2bin/mtx2bin.c).  Binary layout: text header + size line followed by raw
rowidx[nnz], colidx[nnz] (--idxsize 32|64) and a[nnz] float64 arrays,
1-based indices, struct-of-arrays."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from acg_amd.io.mtx import read_mtx, write_mtx  # noqa: E402


def main(argv=None):
    p = argparse.ArgumentParser(prog="mtx2bin")
    p.add_argument("input", help="Matrix Market text file (or .gz)")
    p.add_argument("output", nargs="?", default=None,
                   help="output path (default: stdout)")
    p.add_argument("-z", "--gzip", action="store_true")
    p.add_argument("--idxsize", type=int, choices=(32, 64), default=64)
    args = p.parse_args(argv)
    m = read_mtx(args.input, gzipped=args.gzip)
    if args.output:
        write_mtx(args.output, m, binary=True, idxsize=args.idxsize)
    else:
        write_mtx(sys.stdout.buffer, m, binary=True, idxsize=args.idxsize)
    return 0


if __name__ == "__main__":
    sys.exit(main())
