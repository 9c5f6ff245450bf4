from turboprune_amd.ops.mask_layers import (  # noqa: F401
    ConvMask,
    Conv1dMask,
    LinearMask,
    MASKED_LAYER_TYPES,
    masked_modules,
)
from turboprune_amd.ops import functional  # noqa: F401
from turboprune_amd.ops._backend import has_extension  # noqa: F401
