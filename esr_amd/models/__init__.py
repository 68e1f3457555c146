from .registry import register_model, build_model, get_model_cls, list_models  # noqa: F401
from .esrnet import ESRNet  # noqa: F401
from .unet import UNetRecurrent, SRUNetRecurrent, MultiResUNet  # noqa: F401
from .interp import FrameInterpolator  # noqa: F401
from . import blocks, aux_blocks  # noqa: F401
