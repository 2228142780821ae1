from .actuator import Actuator  # noqa: F401
from .direct import DirectActuator  # noqa: F401
