"""Multi-variant / multi-model scenarios (BASELINE configs #3 and #4):
two accelerator variants through the cost-optimizer path, and two models
sharing the cluster with independent decisions.
"""
from prometheus_client import CollectorRegistry

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.app import build_app
from wva_amd.config.config import Config
from wva_amd.config.saturation import SaturationScalingConfig
from wva_amd.emulator.cluster_sim import ClusterSim
from wva_amd.emulator.sim_source import SimMetricsSource
from wva_amd.emulator.vllm_sim import ServiceProfile
from wva_amd.emulator.workload import constant_qps
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import Container, Deployment, Node, PodTemplateSpec

NS = "default"
LLAMA = "meta-llama/Llama-3.1-8B"
MIXTRAL = "mistralai/Mixtral-8x7B"


def gpu_node(name, product, mem, gpus=8):
    return Node(
        metadata=ObjectMeta(
            name=name,
            labels={"amd.com/gpu.product": product, "amd.com/gpu.memory": mem},
        ),
        allocatable={"amd.com/gpu": str(gpus)},
    )


def add_variant(cluster, sim, name, model, accel, profile, cost="10.0",
                replicas=1, gpus_per_replica=1):
    cluster.create(Deployment(
        metadata=ObjectMeta(name=name, namespace=NS),
        replicas=replicas,
        selector={"app": name},
        template=PodTemplateSpec(
            labels={"app": name},
            containers=[Container(
                args=["--max-num-seqs", str(profile.max_num_seqs),
                      "--block-size", str(profile.block_size)],
                requests={"amd.com/gpu": str(gpus_per_replica)},
            )],
        ),
    ))
    cluster.create(VariantAutoscaling(
        metadata=ObjectMeta(
            name=name, namespace=NS,
            labels={"inference.optimization/acceleratorName": accel},
        ),
        spec=VariantAutoscalingSpec(
            scale_target_ref=CrossVersionObjectReference(name=name),
            model_id=model,
            variant_cost=cost,
        ),
    ))
    sim.register_variant(model, NS, name, profile)


def build(analyzer="saturation"):
    cluster = FakeCluster()
    cluster.create(gpu_node("mi355x-0", "AMD-Instinct-MI355X-288GB", "294912", 32))
    cluster.create(gpu_node("mi300x-0", "AMD-MI300X-192G", "196608", 32))
    sim = ClusterSim(cluster, pod_ready_delay_s=0.0)
    source = SimMetricsSource(sim)
    config = Config()
    config.update_saturation_config(
        SaturationScalingConfig.from_dict({"analyzerName": analyzer}
                                          if analyzer else {})
    )
    config.mark_bootstrap_complete()
    app = build_app(cluster, config, source=source,
                    metrics_registry=CollectorRegistry(), start_engines=False)
    return cluster, sim, app


def drive(sim, model, qps, seconds, dt=0.25):
    prof = constant_qps(qps)
    for _ in range(int(seconds / dt)):
        sim.generate_arrivals(model, prof, dt)
        sim.advance(dt)


SMALL = dict(alpha_ms=50.0, beta_ms=2.0, max_num_seqs=8, num_gpu_blocks=500)


class TestTwoAcceleratorVariants:
    def test_cost_optimizer_prefers_cost_efficient_variant(self):
        """MI355X twice the capacity at 1.5x the cost → more cost-efficient
        per token; scale-up lands there (BASELINE config #3)."""
        cluster, sim, app = build()
        mi355x = ServiceProfile(alpha_ms=25.0, beta_ms=1.0, max_num_seqs=16,
                                num_gpu_blocks=1000)
        mi300x = ServiceProfile(**SMALL)
        add_variant(cluster, sim, "llama-mi355x", LLAMA, "MI355X", mi355x,
                    cost="45.0")
        add_variant(cluster, sim, "llama-mi300x", LLAMA, "MI300X", mi300x,
                    cost="30.0")
        sim.reconcile_deployments()
        model = sim.model(LLAMA, NS)
        drive(sim, model, qps=40, seconds=20)
        app.saturation_engine.optimize()
        d355 = app.decision_cache.get(NS, "llama-mi355x")
        d300 = app.decision_cache.get(NS, "llama-mi300x")
        assert d355 is not None and d300 is not None
        total_added = (d355.target_replicas - 1) + (d300.target_replicas - 1)
        assert total_added >= 1
        # cost efficiency: MI355X k2≈16·125=2000 tokens @45 (0.0225/token),
        # MI300X k2≈8·125=1000 @30 (0.03/token) → MI355X takes the scale-up
        assert d355.target_replicas - 1 >= d300.target_replicas - 1

    def test_both_variants_get_metrics(self):
        cluster, sim, app = build()
        add_variant(cluster, sim, "llama-mi355x", LLAMA, "MI355X",
                    ServiceProfile(**SMALL), cost="45.0")
        add_variant(cluster, sim, "llama-mi300x", LLAMA, "MI300X",
                    ServiceProfile(**SMALL), cost="30.0")
        sim.reconcile_deployments()
        model = sim.model(LLAMA, NS)
        drive(sim, model, qps=5, seconds=10)
        app.saturation_engine.optimize()
        accels = set()
        for fam in app.emitter.registry.collect():
            if fam.name == "wva_desired_replicas":
                for s in fam.samples:
                    accels.add(s.labels["accelerator_type"])
        assert accels == {"MI355X", "MI300X"}


class TestMultiModel:
    def test_independent_decisions_per_model(self):
        """Llama overloaded, Mixtral idle → only Llama scales up
        (BASELINE config #4 shape)."""
        cluster, sim, app = build()
        add_variant(cluster, sim, "llama", LLAMA, "MI355X",
                    ServiceProfile(**SMALL))
        add_variant(cluster, sim, "mixtral", MIXTRAL, "MI355X",
                    ServiceProfile(**SMALL), gpus_per_replica=8)
        sim.reconcile_deployments()
        llama = sim.model(LLAMA, NS)
        mixtral = sim.model(MIXTRAL, NS)
        prof_hot = constant_qps(20)
        prof_idle = constant_qps(0.2)
        for _ in range(80):
            sim.generate_arrivals(llama, prof_hot, 0.25)
            sim.generate_arrivals(mixtral, prof_idle, 0.25)
            sim.advance(0.25)
        app.saturation_engine.optimize()
        d_llama = app.decision_cache.get(NS, "llama")
        d_mixtral = app.decision_cache.get(NS, "mixtral")
        assert d_llama.target_replicas >= 2
        assert d_mixtral.target_replicas <= 1

    def test_v1_path_multi_model(self):
        cluster, sim, app = build(analyzer="")
        add_variant(cluster, sim, "llama", LLAMA, "MI355X",
                    ServiceProfile(**SMALL))
        add_variant(cluster, sim, "mixtral", MIXTRAL, "MI355X",
                    ServiceProfile(**SMALL))
        sim.reconcile_deployments()
        llama = sim.model(LLAMA, NS)
        drive(sim, llama, qps=20, seconds=10)
        app.saturation_engine.optimize()
        d = app.decision_cache.get(NS, "llama")
        assert d is not None and d.target_replicas >= 2


class TestLimitedMode:
    def test_limiter_caps_scale_up_to_cluster_gpus(self):
        cluster = FakeCluster()
        cluster.create(gpu_node("n1", "AMD-Instinct-MI355X-288GB", "294912",
                                gpus=3))
        sim = ClusterSim(cluster, pod_ready_delay_s=0.0)
        source = SimMetricsSource(sim)
        config = Config()
        config.update_saturation_config(SaturationScalingConfig.from_dict(
            {"enableLimiter": True}  # V1 path + limiter
        ))
        config.mark_bootstrap_complete()
        app = build_app(cluster, config, source=source,
                        metrics_registry=CollectorRegistry(),
                        start_engines=False)
        add_variant(cluster, sim, "llama", LLAMA, "MI355X",
                    ServiceProfile(**SMALL))
        sim.reconcile_deployments()
        model = sim.model(LLAMA, NS)
        drive(sim, model, qps=20, seconds=10)
        app.saturation_engine.optimize()
        d = app.decision_cache.get(NS, "llama")
        # V1 scales +1 per tick; limiter allows it (3 GPUs, 1 used)
        assert d.target_replicas <= 3
