"""esr_amd.ops — compute ops (torch-vectorized CPU oracles + gfx950 HIP paths)."""

from . import events  # noqa: F401
from .events import (  # noqa: F401
    events_to_image,
    events_to_channels,
    events_to_stack_no_polarity,
    events_to_stack_polarity,
    events_to_voxel,
    stack_to_count,
    redistribute_stack,
    redistribute_count,
    event_formatting,
    normalize_events,
    scaled_count_encoding,
)
from .dcn import modulated_deform_conv2d, DeformAlign2d  # noqa: F401
from .convgru import ConvGRUCell  # noqa: F401
from .native import native_available  # noqa: F401
