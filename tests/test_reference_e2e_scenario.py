"""The reference's e2e acceptance scenario, reproduced byte-for-byte on
the emulated cluster (test/e2e/config.go:80-92): load 8 req/s
with 100-in/50-out tokens against a single variant whose replicas run
`--max-num-seqs 5`, and require scale-up within 600 simulated seconds.
The tiny max-num-seqs makes one replica saturate at ~<8 req/s, so the
saturation path MUST scale — the same pass/fail contract the reference's
Kind suite asserts.
"""
import pytest

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.config.config import Config
from wva_amd.config.saturation import SaturationScalingConfig
from wva_amd.emulator.cluster_sim import ClusterSim
from wva_amd.emulator.sim_source import SimMetricsSource
from wva_amd.emulator.vllm_sim import ServiceProfile
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import Container, Deployment, Node, PodTemplateSpec

MODEL = "default/default"
NS = "default"
VARIANT = "vllm-sim"


def build(analyzer_name: str, backend: str = "fake"):
    from prometheus_client import CollectorRegistry
    from wva_amd.app import build_app

    cluster = FakeCluster()
    cluster.create(Node(
        metadata=ObjectMeta(
            name="n0",
            labels={
                "amd.com/gpu.product": "AMD-Instinct-MI355X-288GB",
                "amd.com/gpu.memory": "294912",
            },
        ),
        allocatable={"amd.com/gpu": "8"},
    ))
    cluster.create(Deployment(
        metadata=ObjectMeta(name=VARIANT, namespace=NS),
        replicas=1,
        selector={"app": VARIANT},
        template=PodTemplateSpec(
            labels={"app": VARIANT},
            containers=[Container(
                args=["--max-num-seqs", "5"],  # e2e config.go:86
                requests={"amd.com/gpu": "1"},
            )],
        ),
    ))
    cluster.create(VariantAutoscaling(
        metadata=ObjectMeta(
            name=VARIANT, namespace=NS,
            labels={"inference.optimization/acceleratorName": "MI355X"},
        ),
        spec=VariantAutoscalingSpec(
            scale_target_ref=CrossVersionObjectReference(name=VARIANT),
            model_id=MODEL,
        ),
    ))

    # llm-d-inference-sim-like slow replica: the e2e emulator's service
    # rate (the HPA tutorial's fitted α=20.58, β=0.41 on "A100")
    profile = ServiceProfile(
        alpha_ms=20.58, beta_ms=0.41, max_num_seqs=5, num_gpu_blocks=2000
    )
    sim = ClusterSim(cluster, warm_start=True)
    sim.register_variant(MODEL, NS, VARIANT, profile)
    sim.reconcile_deployments()

    config = Config()
    config.update_saturation_config(
        SaturationScalingConfig.from_dict({"analyzerName": analyzer_name})
    )
    config.mark_bootstrap_complete()

    app_cluster = cluster
    teardown = []
    if backend == "rest":
        from wva_amd.kube.cache import CachedCluster
        from wva_amd.kube.rest import RestCluster
        from k8s_test_server import K8sTestServer

        server = K8sTestServer(cluster).start()
        cache = CachedCluster(RestCluster(server.url)).start()
        assert cache.wait_for_sync(10)
        teardown = [cache.stop, server.stop]
        app_cluster = cache

    app = build_app(
        app_cluster, config, source=SimMetricsSource(sim),
        metrics_registry=CollectorRegistry(), start_engines=False,
    )
    if backend == "rest":
        cache = app_cluster
        orig_opt = app.saturation_engine.optimize
        app.saturation_engine.optimize = (
            lambda: (cache.wait_caught_up(), orig_opt())[1]
        )
    app._teardown = teardown  # type: ignore[attr-defined]
    return cluster, sim, app


@pytest.mark.parametrize("backend", ["fake", "rest"])
@pytest.mark.parametrize("analyzer", ["saturation", "percentage"])
def test_scale_up_within_600s(analyzer, backend):
    """8 req/s at 100/50 tokens vs max-num-seqs=5 replicas: one replica
    holds ~5 concurrent (≈ 5/(50·ITL(5)) ≈ 4.4 req/s) — saturation must
    drive scale-up well within the reference's 600 s ceiling, on BOTH
    analyzer paths (V2 token and V1 percentage)."""
    cluster, sim, app = build(analyzer, backend)
    model = sim.model(MODEL, NS)

    scaled_at = None
    for tick in range(20):  # 20 × 30 s = 600 s budget (engine.go:147)
        for _ in range(120):
            sim.generate_arrivals(model, lambda t: 8.0, 0.25, 100, 50)
            sim.advance(0.25)
        app.saturation_engine.optimize()
        app.va_reconciler.reconcile(NS, VARIANT)
        d = app.decision_cache.get(NS, VARIANT)
        if d is not None and d.target_replicas > 1:
            scaled_at = (tick + 1) * 30
            cluster.scale("Deployment", NS, VARIANT, d.target_replicas)
            sim.reconcile_deployments()
            break

    assert scaled_at is not None, "no scale-up within 600 simulated seconds"
    assert scaled_at <= 600
    va = cluster.get("VariantAutoscaling", NS, VARIANT)
    assert va.status.desired_optimized_alloc.num_replicas > 1
    for fn in getattr(app, "_teardown", []):
        fn()
