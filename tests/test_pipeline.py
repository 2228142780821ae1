"""Pipeline tests: enforcer, cost-aware optimizer, inventory, greedy
allocation, default limiter. Mirrors reference
internal/engines/pipeline/*_test.go coverage (1883 LoC).
"""
import pytest

from wva_amd.analyzers.interfaces import (
    AnalyzerResult,
    VariantCapacity,
    VariantDecision,
    VariantReplicaState,
    VariantSaturationAnalysis,
)
from wva_amd.config.scale_to_zero import (
    ModelScaleToZeroConfig,
)
from wva_amd.discovery.gpu_operator import (
    K8sGpuOperatorDiscovery,
    normalize_accelerator_name,
)
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import Container, Node, Pod
from wva_amd.api.types import ObjectMeta
from wva_amd.pipeline import (
    CostAwareOptimizer,
    DefaultLimiter,
    GreedyBySaturation,
    ModelScalingRequest,
    TypeInventory,
)
from wva_amd.pipeline.enforcer import Enforcer


def stz(enabled=None, retention=""):
    data = {}
    if enabled is not None:
        data["default"] = ModelScaleToZeroConfig(
            enable_scale_to_zero=enabled, retention_period=retention
        )
    return data


class TestEnforcer:
    def test_scale_to_zero_on_idle(self):
        e = Enforcer(lambda m, n, r: 0.0)
        targets, applied = e.enforce_policy(
            "m", "ns", {"a": 2, "b": 1}, [], stz(enabled=True)
        )
        assert applied and targets == {"a": 0, "b": 0}

    def test_keeps_targets_with_traffic(self):
        e = Enforcer(lambda m, n, r: 42.0)
        targets, applied = e.enforce_policy(
            "m", "ns", {"a": 2}, [], stz(enabled=True)
        )
        assert not applied and targets == {"a": 2}

    def test_query_error_fails_safe(self):
        def boom(m, n, r):
            raise RuntimeError("prometheus down")

        e = Enforcer(boom)
        targets, applied = e.enforce_policy(
            "m", "ns", {"a": 2}, [], stz(enabled=True)
        )
        assert not applied and targets == {"a": 2}

    def test_min_replica_on_cheapest(self):
        e = Enforcer(lambda m, n, r: 0.0)
        analyses = [
            VariantSaturationAnalysis(variant_name="pricey", cost=50),
            VariantSaturationAnalysis(variant_name="cheap", cost=5),
        ]
        targets, applied = e.enforce_policy(
            "m", "ns", {"cheap": 0, "pricey": 0}, analyses, stz(enabled=False)
        )
        assert applied and targets == {"cheap": 1, "pricey": 0}

    def test_min_replica_not_needed(self):
        e = Enforcer(lambda m, n, r: 0.0)
        targets, applied = e.enforce_policy(
            "m", "ns", {"a": 1}, [], stz(enabled=False)
        )
        assert not applied and targets == {"a": 1}

    def test_min_replica_tie_break(self):
        e = Enforcer(lambda m, n, r: 0.0)
        analyses = [
            VariantSaturationAnalysis(variant_name="bbb", cost=10),
            VariantSaturationAnalysis(variant_name="aaa", cost=10),
        ]
        targets, _ = e.enforce_policy(
            "m", "ns", {"bbb": 0, "aaa": 0}, analyses, {}
        )
        assert targets == {"aaa": 1, "bbb": 0}


def vc(name, cost, cap, accel="MI355X"):
    return VariantCapacity(
        variant_name=name,
        cost=cost,
        per_replica_capacity=cap,
        accelerator_name=accel,
    )


def req(required=0.0, spare=0.0, capacities=(), states=()):
    return ModelScalingRequest(
        model_id="m",
        namespace="ns",
        result=AnalyzerResult(
            required_capacity=required,
            spare_capacity=spare,
            variant_capacities=list(capacities),
        ),
        variant_states=list(states),
    )


def st(name, current, gpus=1):
    return VariantReplicaState(
        variant_name=name, current_replicas=current, gpus_per_replica=gpus
    )


class TestCostAwareOptimizer:
    def test_scale_up_most_cost_efficient(self):
        # cheap: 10/1000 = 0.01; pricey: 50/10000 = 0.005 → pricey is more
        # cost-efficient per token and takes the scale-up
        opt = CostAwareOptimizer()
        decisions = opt.optimize(
            [
                req(
                    required=5000,
                    capacities=[vc("cheap", 10, 1000), vc("pricey", 50, 10000)],
                    states=[st("cheap", 1), st("pricey", 1)],
                )
            ]
        )
        by = {d.variant_name: d for d in decisions}
        assert by["pricey"].target_replicas == 2  # ceil(5000/10000) = 1 added
        assert by["cheap"].target_replicas == 1
        assert by["pricey"].action == "scale-up"

    def test_scale_up_spillover(self):
        opt = CostAwareOptimizer()
        decisions = opt.optimize(
            [
                req(
                    required=5000,
                    capacities=[vc("a", 10, 1000)],
                    states=[st("a", 1)],
                )
            ]
        )
        assert decisions[0].target_replicas == 1 + 5  # ceil(5000/1000)

    def test_scale_down_most_expensive_first(self):
        opt = CostAwareOptimizer()
        decisions = opt.optimize(
            [
                req(
                    spare=2500,
                    capacities=[vc("cheap", 10, 1000), vc("pricey", 50, 1000)],
                    states=[st("cheap", 2), st("pricey", 2)],
                )
            ]
        )
        by = {d.variant_name: d for d in decisions}
        # floor(2500/1000)=2 removed from pricey first
        assert by["pricey"].target_replicas == 0
        assert by["cheap"].target_replicas == 2

    def test_cheapest_protected_when_last(self):
        opt = CostAwareOptimizer()
        decisions = opt.optimize(
            [
                req(
                    spare=100000,
                    capacities=[vc("only", 10, 1000)],
                    states=[st("only", 3)],
                )
            ]
        )
        assert decisions[0].target_replicas == 1

    def test_cheapest_not_protected_when_others_have_replicas(self):
        opt = CostAwareOptimizer()
        decisions = opt.optimize(
            [
                req(
                    spare=1000,
                    capacities=[vc("cheap", 10, 1000), vc("pricey", 50, 100000)],
                    states=[st("cheap", 2), st("pricey", 1)],
                )
            ]
        )
        by = {d.variant_name: d for d in decisions}
        # pricey: floor(1000/100000)=0 — skipped; cheap: floor(1000/1000)=1
        assert by["cheap"].target_replicas == 1
        assert by["pricey"].target_replicas == 1

    def test_no_result_skipped(self):
        opt = CostAwareOptimizer()
        assert opt.optimize([ModelScalingRequest(model_id="m")]) == []

    def test_steady_state(self):
        opt = CostAwareOptimizer()
        decisions = opt.optimize(
            [req(capacities=[vc("a", 10, 1000)], states=[st("a", 2)])]
        )
        assert decisions[0].action == "no-change"
        assert decisions[0].target_replicas == 2


def mi355x_node(name, gpus=8):
    return Node(
        metadata=ObjectMeta(
            name=name,
            labels={
                "amd.com/gpu.product": "AMD-Instinct-MI355X-288GB",
                "amd.com/gpu.memory": "294912",
            },
        ),
        allocatable={"amd.com/gpu": str(gpus)},
    )


def mi300x_node(name, gpus=8):
    return Node(
        metadata=ObjectMeta(
            name=name,
            labels={
                "amd.com/gpu.product": "AMD-MI300X-192G",
                "amd.com/gpu.memory": "196608",
            },
        ),
        allocatable={"amd.com/gpu": str(gpus)},
    )


class TestDiscovery:
    def test_normalize(self):
        assert normalize_accelerator_name("AMD-Instinct-MI355X-288GB") == "MI355X"
        assert normalize_accelerator_name("AMD-MI300X-192G") == "MI300X"
        assert normalize_accelerator_name("NVIDIA-A100-PCIE-80GB") == "A100"
        assert normalize_accelerator_name("Intel-Gaudi-3-128GB") == "Gaudi-3"
        assert normalize_accelerator_name("MI355X") == "MI355X"

    def test_capacity_discovery(self):
        c = FakeCluster()
        c.create(mi355x_node("n1"))
        c.create(mi355x_node("n2"))
        c.create(mi300x_node("n3"))
        d = K8sGpuOperatorDiscovery(c, node_selector={})
        inv = d.discover()
        assert inv["n1"]["AMD-Instinct-MI355X-288GB"].count == 8
        assert inv["n1"]["AMD-Instinct-MI355X-288GB"].memory == "294912"
        assert len(inv) == 3

    def test_usage_discovery(self):
        c = FakeCluster()
        c.create(mi355x_node("n1"))
        pod = Pod(
            metadata=ObjectMeta(name="p1", namespace="ns"),
            containers=[Container(requests={"amd.com/gpu": "2"})],
            node_name="n1",
        )
        c.create(pod)
        d = K8sGpuOperatorDiscovery(c, node_selector={})
        assert d.discover_usage() == {"MI355X": 2}

    def test_init_container_max(self):
        c = FakeCluster()
        c.create(mi355x_node("n1"))
        pod = Pod(
            metadata=ObjectMeta(name="p1", namespace="ns"),
            containers=[Container(requests={"amd.com/gpu": "1"})],
            init_containers=[Container(requests={"amd.com/gpu": "4"})],
            node_name="n1",
        )
        c.create(pod)
        d = K8sGpuOperatorDiscovery(c, node_selector={})
        assert d.discover_usage() == {"MI355X": 4}

    def test_node_selector_sharding(self):
        c = FakeCluster()
        n = mi355x_node("n1")
        n.metadata.labels["pool"] = "a"
        c.create(n)
        c.create(mi355x_node("n2"))
        d = K8sGpuOperatorDiscovery(c, node_selector={"pool": "a"})
        assert list(d.discover().keys()) == ["n1"]

    def test_max_hive(self):
        c = FakeCluster()
        c.create(mi355x_node("n1", gpus=8))
        c.create(mi355x_node("n2", gpus=4))
        d = K8sGpuOperatorDiscovery(c, node_selector={})
        assert d.max_hive_by_type() == {"MI355X": 8}


def decision(name, current, target, accel="MI355X", gpus=1, spare=0.0, cost=10.0):
    return VariantDecision(
        variant_name=name,
        namespace="ns",
        model_id="m",
        accelerator_name=accel,
        current_replicas=current,
        target_replicas=target,
        gpus_per_replica=gpus,
        spare_capacity=spare,
        cost=cost,
    )


class TestLimiter:
    def _cluster(self, mi355x_gpus=8):
        c = FakeCluster()
        c.create(mi355x_node("n1", gpus=mi355x_gpus))
        return c

    def _limiter(self, cluster):
        disc = K8sGpuOperatorDiscovery(cluster, node_selector={})
        inv = TypeInventory("gpu", disc)
        return DefaultLimiter("gpu-limiter", inv, GreedyBySaturation())

    def test_no_limit_when_capacity_available(self):
        lim = self._limiter(self._cluster())
        d = decision("v", current=1, target=3)
        lim.limit([d])
        assert d.target_replicas == 3
        assert not d.was_limited
        assert d.gpus_allocated == 2

    def test_partial_allocation(self):
        lim = self._limiter(self._cluster(mi355x_gpus=4))
        # current 2 uses 2 GPUs; target 6 wants +4 but only 2 remain
        d = decision("v", current=2, target=6)
        lim.limit([d])
        assert d.target_replicas == 4
        assert d.was_limited
        assert d.limited_by == "gpu-limiter"

    def test_most_saturated_first(self):
        lim = self._limiter(self._cluster(mi355x_gpus=3))
        hot = decision("hot", current=1, target=3, spare=0.0)
        cold = decision("cold", current=1, target=3, spare=0.9)
        lim.limit([cold, hot])
        # 3 GPUs - 2 used = 1 available → hot gets it
        assert hot.target_replicas == 2
        assert cold.target_replicas == 1

    def test_tp8_whole_replica_truncation(self):
        lim = self._limiter(self._cluster(mi355x_gpus=8))
        # TP=8 replica: 0 current, want 2 replicas = 16 GPUs, only 8 free
        d = decision("tp8", current=0, target=2, gpus=8)
        lim.limit([d])
        assert d.target_replicas == 1  # one whole 8-GPU hive
        assert d.gpus_allocated == 8
        assert d.was_limited

    def test_hive_feasibility_tp8_on_4gpu_nodes(self):
        """A TP=8 variant can never fit on nodes with 4 allocatable GPUs,
        even if the cluster-wide total is 8."""
        c = FakeCluster()
        c.create(mi355x_node("n1", gpus=4))
        c.create(mi355x_node("n2", gpus=4))
        lim = self._limiter(c)
        d = decision("tp8", current=0, target=1, gpus=8)
        lim.limit([d])
        assert d.target_replicas == 0
        assert d.was_limited

    def test_type_isolation(self):
        c = FakeCluster()
        c.create(mi355x_node("n1", gpus=2))
        c.create(mi300x_node("n2", gpus=8))
        lim = self._limiter(c)
        d = decision("v", current=0, target=4, accel="MI355X")
        lim.limit([d])
        # only 2 MI355X GPUs despite 8 MI300X available
        assert d.target_replicas == 2

    def test_compute_constraints(self):
        c = self._cluster()
        lim = self._limiter(c)
        rc = lim.compute_constraints({"MI355X": 3})
        assert rc.pools["MI355X"].limit == 8
        assert rc.pools["MI355X"].used == 3
        assert rc.pools["MI355X"].available == 5
        assert rc.total_avail == 5

    def test_missing_accelerator_raises(self):
        lim = self._limiter(self._cluster())
        d = decision("v", current=0, target=1, accel="")
        with pytest.raises(ValueError):
            lim.limit([d])
