from .base import SolveResult  # noqa: F401
from .cpu import CGSolverCPU  # noqa: F401


def get_solver_hip():
    from .hip import CGSolverHIP

    return CGSolverHIP
