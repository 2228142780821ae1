from .executor import PollingExecutor  # noqa: F401
from .manager import Manager, Runnable  # noqa: F401
