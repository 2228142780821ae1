from .enforcer import Enforcer  # noqa: F401
from .limiter import (  # noqa: F401
    DefaultLimiter,
    ModelScalingRequest,
    ResourceConstraints,
    ResourcePool,
)
from .inventory import TypeInventory  # noqa: F401
from .greedy_saturation import GreedyBySaturation  # noqa: F401
from .optimizer import CostAwareOptimizer  # noqa: F401
