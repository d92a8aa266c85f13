from .mmdit import Flux, FluxConfig, ZImage, ZImageConfig  # noqa: F401
from .sd_unet import SDUNet, UNetConfig  # noqa: F401
from .wan import WanConfig, WanDiT  # noqa: F401
from .registry import MODELS  # noqa: F401
