from .objects import (  # noqa: F401
    ConfigMap,
    Container,
    Deployment,
    EndpointPool,
    EnvVar,
    InferencePool,
    Lease,
    Node,
    Pod,
    Service,
    ServicePort,
)
from .fake import FakeCluster, WatchEvent  # noqa: F401
