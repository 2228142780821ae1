from .vllm_sim import ReplicaSim, RequestSpec, ServiceProfile  # noqa: F401
from .cluster_sim import ClusterSim, ModelSim  # noqa: F401
from .sim_source import SimMetricsSource  # noqa: F401
from .workload import constant_qps, ramp_qps, burst_qps  # noqa: F401
