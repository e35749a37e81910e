from .mtx import MtxFile, read_mtx, write_mtx  # noqa: F401
