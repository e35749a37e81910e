from .stencil import (  # noqa: F401
    STENCIL_5PT_2D,
    STENCIL_7PT_3D,
    STENCIL_27PT_3D,
    queen_like_spec,
    stencil_global,
    stencil_local_slab,
)
