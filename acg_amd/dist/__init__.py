from .comm import Comm  # noqa: F401
from .halo import HaloExchange  # noqa: F401
