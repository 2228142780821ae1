"""Inferno — model-based (queueing-theory) optimizer library.

Parity: reference pkg/{core,analyzer,solver,manager,config}.
The reference keeps this library dormant (not wired into the live loop);
here it ships as a first-class, wireable analyzer with the same math:
M/M/1/K and state-dependent M/M/1/K queueing, binary-search SLO sizing,
greedy capacity-constrained solving — and its ServiceParms (α, β, γ) come
from MI355X measurements via wva_amd.calibration instead of offline
NVIDIA/MI300X fits.
"""
from .queueing import MM1KModel, MM1StateDependentModel, binary_search  # noqa: F401
from .queue_analyzer import (  # noqa: F401
    AnalysisMetrics,
    Configuration,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from .system import Allocation, System  # noqa: F401
from .solver import Solver  # noqa: F401
from .manager import Manager  # noqa: F401
