from .symcsr import SymCSRMatrix  # noqa: F401
