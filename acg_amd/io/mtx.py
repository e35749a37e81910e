"""Matrix Market I/O (reference: acg/mtxfile.{c,h}).

Supports the same surface as the reference reader/writer:

- text Matrix Market (``matrix coordinate real|integer|pattern
  general|symmetric`` and ``vector``/``matrix array`` files),
- gzip-compressed text (reference acgmtxfile_gzread, mtxfile.h:371),
- the reference's *binary* format (mtxfile.c:730-752): the usual text
  header and size line followed by raw ``rowidx[nnz]``, ``colidx[nnz]``
  (C ``acgidx_t``: int32 or int64, build-time choice — here a parameter)
  and ``a[nnz]`` float64 arrays, struct-of-arrays, 1-based indices.

Indices are stored 0-based in memory; files are 1-based as per the format.
Parsing uses numpy (pandas fast path when available) instead of the
reference's hand-rolled C parser.
"""

from __future__ import annotations

import gzip
import io as _io
from dataclasses import dataclass, field

import numpy as np

from ..utils.errors import AcgError, ErrCode

# header vocabularies (reference mtxfile.h:144-179)
OBJECTS = ("matrix", "vector")
FORMATS = ("coordinate", "array")
FIELDS = ("real", "integer", "complex", "pattern")
SYMMETRIES = ("general", "symmetric", "skew-symmetric", "hermitian")


@dataclass
class MtxFile:
    """Raw Matrix Market data (reference struct acgmtxfile, mtxfile.h:214)."""

    object: str = "matrix"
    format: str = "coordinate"
    field_: str = "real"
    symmetry: str = "general"
    nrows: int = 0
    ncols: int = 0
    nnz: int = 0  # number of *stored* entries
    rowidx: np.ndarray | None = None  # 0-based, int64
    colidx: np.ndarray | None = None  # 0-based, int64
    a: np.ndarray | None = None  # float64 (or int64 for integer field)
    comments: list = field(default_factory=list)

    @property
    def is_symmetric(self) -> bool:
        return self.symmetry == "symmetric"


def _open(path, mode="rt", gzipped=False):
    if gzipped or str(path).endswith(".gz"):
        return gzip.open(path, mode)
    return open(path, mode)


def _parse_header_line(line: str):
    parts = line.strip().split()
    if len(parts) < 3 or parts[0] != "%%MatrixMarket":
        raise AcgError(ErrCode.INVALID_FORMAT, f"bad MatrixMarket header: {line!r}")
    obj = parts[1]
    fmt = parts[2]
    if obj not in OBJECTS:
        raise AcgError(ErrCode.INVALID_FORMAT, f"bad object {obj!r}")
    if fmt not in FORMATS:
        raise AcgError(ErrCode.INVALID_FORMAT, f"bad format {fmt!r}")
    fld = parts[3] if len(parts) > 3 else "real"
    sym = parts[4] if len(parts) > 4 else "general"
    if fld not in FIELDS:
        raise AcgError(ErrCode.INVALID_FORMAT, f"bad field {fld!r}")
    if sym not in SYMMETRIES:
        raise AcgError(ErrCode.INVALID_FORMAT, f"bad symmetry {sym!r}")
    return obj, fmt, fld, sym


def _read_header(f):
    """Read banner, comments and the size line from a text stream."""
    line = f.readline()
    if isinstance(line, bytes):
        line = line.decode()
    obj, fmt, fld, sym = _parse_header_line(line)
    comments = []
    while True:
        line = f.readline()
        if isinstance(line, bytes):
            line = line.decode()
        if not line:
            raise AcgError(ErrCode.EOF, "EOF before size line")
        if line.startswith("%"):
            comments.append(line.rstrip("\n"))
            continue
        if line.strip():
            break
    sizes = line.split()
    if obj == "matrix" and fmt == "coordinate":
        if len(sizes) != 3:
            raise AcgError(ErrCode.INVALID_FORMAT, f"bad size line: {line!r}")
        nrows, ncols, nnz = int(sizes[0]), int(sizes[1]), int(sizes[2])
    elif fmt == "array":
        if obj == "vector" and len(sizes) == 1:
            nrows, ncols, nnz = int(sizes[0]), 1, int(sizes[0])
        else:
            if len(sizes) != 2:
                raise AcgError(ErrCode.INVALID_FORMAT, f"bad size line: {line!r}")
            nrows, ncols = int(sizes[0]), int(sizes[1])
            nnz = nrows * ncols
    elif obj == "vector" and fmt == "coordinate":
        if len(sizes) != 2:
            raise AcgError(ErrCode.INVALID_FORMAT, f"bad size line: {line!r}")
        nrows, ncols, nnz = int(sizes[0]), 1, int(sizes[1])
    else:
        raise AcgError(ErrCode.NOT_SUPPORTED, f"{obj}/{fmt}")
    return obj, fmt, fld, sym, nrows, ncols, nnz, comments


def _parse_body_text(text: bytes | str, ncols_expected: int) -> np.ndarray:
    """Parse whitespace-separated numeric rows into a 2-D float64 array."""
    if isinstance(text, bytes):
        text = text.decode()
    if not text.strip():
        return np.zeros((0, max(ncols_expected, 1)), dtype=np.float64)
    try:
        import pandas as pd

        df = pd.read_csv(
            _io.StringIO(text),
            sep=r"\s+",
            header=None,
            comment="%",
            dtype=np.float64,
            engine="c",
        )
        arr = df.to_numpy()
    except ImportError:  # pragma: no cover
        arr = np.loadtxt(_io.StringIO(text), dtype=np.float64, comments="%", ndmin=2)
    if arr.ndim == 1:
        arr = arr.reshape(-1, ncols_expected if ncols_expected else 1)
    return arr


def read_mtx(path, gzipped: bool = False, binary: bool = False, idxsize: int = 64) -> MtxFile:
    """Read a Matrix Market file (reference acgmtxfile_read, mtxfile.h:352-416).

    ``binary`` selects the reference's binary layout; ``idxsize`` (32|64)
    is the width of the stored index type in a binary file (the
    reference's ``acgidx_t`` compile-time choice, config.h:62-94).
    """
    mode = "rb" if binary else "rt"
    with _open(path, mode, gzipped) as f:
        obj, fmt, fld, sym, nrows, ncols, nnz, comments = _read_header(f)
        m = MtxFile(obj, fmt, fld, sym, nrows, ncols, nnz, comments=comments)
        if binary:
            if fmt == "array" and fld in ("real", "integer"):
                # binary array: raw values after the size line (integer
                # values stored as acgidx_t, real as float64)
                vdt = np.float64 if fld == "real" else (np.int32 if idxsize == 32 else np.int64)
                raw = f.read(nnz * np.dtype(vdt).itemsize)
                if len(raw) < nnz * np.dtype(vdt).itemsize:
                    raise AcgError(ErrCode.EOF, "binary array body truncated")
                m.a = np.frombuffer(raw, dtype=vdt, count=nnz).astype(
                    np.float64 if fld == "real" else np.int64)
                return m
            if not (obj == "matrix" and fmt == "coordinate" and fld in ("real", "integer")):
                raise AcgError(ErrCode.NOT_SUPPORTED, "binary supports matrix/coordinate real|integer")
            idt = np.int32 if idxsize == 32 else np.int64
            raw = f.read(2 * nnz * np.dtype(idt).itemsize + 8 * nnz)
            need = 2 * nnz * np.dtype(idt).itemsize + 8 * nnz
            if len(raw) < need:
                raise AcgError(ErrCode.EOF, f"binary body truncated: {len(raw)} < {need}")
            o = 0
            ri = np.frombuffer(raw, dtype=idt, count=nnz, offset=o).astype(np.int64)
            o += nnz * np.dtype(idt).itemsize
            ci = np.frombuffer(raw, dtype=idt, count=nnz, offset=o).astype(np.int64)
            o += nnz * np.dtype(idt).itemsize
            vdt = np.float64 if fld == "real" else np.int64
            a = np.frombuffer(raw, dtype=vdt, count=nnz, offset=o).copy()
            m.rowidx, m.colidx, m.a = ri - 1, ci - 1, a
            return m
        body = f.read()
        if obj == "matrix" and fmt == "coordinate":
            want = {"pattern": 2, "complex": 4}.get(fld, 3)
            arr = _parse_body_text(body, want)
            if arr.shape[0] < nnz:
                raise AcgError(ErrCode.EOF, f"expected {nnz} entries, got {arr.shape[0]}")
            arr = arr[:nnz]
            m.rowidx = arr[:, 0].astype(np.int64) - 1
            m.colidx = arr[:, 1].astype(np.int64) - 1
            if fld == "pattern":
                m.a = np.ones(nnz, dtype=np.float64)
            elif fld == "integer":
                m.a = arr[:, 2].astype(np.int64)
            elif fld == "complex":
                m.a = arr[:, 2] + 1j * arr[:, 3]
            else:
                m.a = np.ascontiguousarray(arr[:, 2])
        elif fmt == "array":
            arr = _parse_body_text(body, 1).reshape(-1)
            if arr.shape[0] < nnz:
                raise AcgError(ErrCode.EOF, f"expected {nnz} values, got {arr.shape[0]}")
            m.a = np.ascontiguousarray(arr[:nnz], dtype=np.float64)
        elif obj == "vector" and fmt == "coordinate":
            arr = _parse_body_text(body, 2)
            m.rowidx = arr[:, 0].astype(np.int64) - 1
            m.a = np.ascontiguousarray(arr[:, 1])
        else:
            raise AcgError(ErrCode.NOT_SUPPORTED, f"{obj}/{fmt}")
        return m


def _savetxt(f, arr, fmt: str, bytes_mode: bool, chunk: int = 1 << 20) -> None:
    """Fast text writer: numpy savetxt in chunks (avoids per-row python)."""
    import io as _io2

    for lo in range(0, len(arr), chunk):
        buf = _io2.StringIO()
        np.savetxt(buf, arr[lo:lo + chunk], fmt=fmt)
        s = buf.getvalue()
        f.write(s.encode() if bytes_mode else s)


def _savetxt_mixed(f, i, j, v, vfmt: str, bytes_mode: bool) -> None:
    """int int float rows (coordinate real) via chunked savetxt."""
    import io as _io2

    chunk = 1 << 20
    for lo in range(0, len(i), chunk):
        buf = _io2.StringIO()
        np.savetxt(buf, np.column_stack([i[lo:lo + chunk].astype(np.float64),
                                         j[lo:lo + chunk].astype(np.float64),
                                         v[lo:lo + chunk]]),
                   fmt=f"%d %d {vfmt}")
        s = buf.getvalue()
        f.write(s.encode() if bytes_mode else s)


def write_mtx(path_or_file, m: MtxFile, binary: bool = False, idxsize: int = 64,
              numfmt=None, gzipped: bool = False) -> None:
    """Write a Matrix Market file (reference mtxfile_fwrite_double, mtx2bin).

    ``numfmt`` is an optional :class:`acg_amd.utils.numfmt.FmtSpec` applied
    to real values in text output (reference --numfmt).
    """
    own = not hasattr(path_or_file, "write")
    if own:
        f = _open(path_or_file, "wb" if (binary or gzipped) else "wt", gzipped)
    else:
        f = path_or_file

    def _w(s: str):
        f.write(s.encode() if binary or gzipped else s)

    try:
        hdr = f"%%MatrixMarket {m.object} {m.format} {m.field_} {m.symmetry}\n"
        _w(hdr)
        for c in m.comments:
            _w(c + "\n")
        if m.object == "matrix" and m.format == "coordinate":
            _w(f"{m.nrows} {m.ncols} {m.nnz}\n")
            if binary:
                if m.field_ not in ("real", "integer"):
                    # matches the reader and mtx2bin (--double/--integer):
                    # the reference binary layout stores a value array
                    raise AcgError(ErrCode.NOT_SUPPORTED,
                                   "binary supports real|integer fields")
                idt = np.int32 if idxsize == 32 else np.int64
                f.write((m.rowidx + 1).astype(idt).tobytes())
                f.write((m.colidx + 1).astype(idt).tobytes())
                f.write(np.asarray(m.a, dtype=np.float64 if m.field_ == "real" else np.int64).tobytes())
            else:
                vfmt = numfmt._py if numfmt else "%.17g"
                if m.field_ == "pattern":
                    body = np.column_stack([m.rowidx + 1, m.colidx + 1])
                    _savetxt(f, body, "%d %d", binary or gzipped)
                elif m.field_ == "integer":
                    body = np.column_stack([m.rowidx + 1, m.colidx + 1,
                                            np.asarray(m.a, dtype=np.int64)])
                    _savetxt(f, body, "%d %d %d", binary or gzipped)
                elif numfmt is not None and numfmt._hex:
                    for i, j, v in zip(m.rowidx, m.colidx, m.a):
                        _w(f"{i + 1} {j + 1} {numfmt.format(v)}\n")
                else:
                    _savetxt_mixed(f, m.rowidx + 1, m.colidx + 1, m.a, vfmt,
                                   binary or gzipped)
        elif m.format == "array":
            if m.object == "vector":
                _w(f"{m.nrows}\n")
            else:
                _w(f"{m.nrows} {m.ncols}\n")
            if binary:
                vdt = np.float64 if m.field_ == "real" else \
                    (np.int32 if idxsize == 32 else np.int64)
                f.write(np.asarray(m.a, dtype=vdt).tobytes())
            else:
                vfmt = numfmt._py if numfmt else "%.17g"
                if m.field_ == "integer":
                    _savetxt(f, np.asarray(m.a, dtype=np.int64), "%d",
                             binary or gzipped)
                elif numfmt is not None and numfmt._hex:
                    for v in m.a:
                        _w(f"{numfmt.format(v)}\n")
                else:
                    _savetxt(f, np.asarray(m.a, dtype=np.float64), vfmt,
                             binary or gzipped)
        else:
            raise AcgError(ErrCode.NOT_SUPPORTED, f"{m.object}/{m.format}")
    finally:
        if own:
            f.close()


def vector_to_mtx(x: np.ndarray, field_: str = "real") -> MtxFile:
    """Wrap a dense vector as a ``matrix array`` MtxFile (as acg-hip writes x)."""
    x = np.asarray(x)
    return MtxFile(
        object="matrix", format="array", field_=field_, symmetry="general",
        nrows=x.shape[0], ncols=1, nnz=x.shape[0], a=x,
    )
