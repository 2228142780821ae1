from .saturation import SaturationScalingConfig  # noqa: F401
from .scale_to_zero import (  # noqa: F401
    ModelScaleToZeroConfig,
    ScaleToZeroConfigData,
    parse_scale_to_zero_configmap,
)
from .config import Config  # noqa: F401
from .loader import load_config  # noqa: F401
