"""E2E matrix growth toward the reference's suites:

* limiter under GPU exhaustion (test/e2e/limiter_test.go analog)
* target-condition lifecycle: Deployment deleted → TargetResolved=False
  → recreated → True (target_condition_test.go analog)
* multi-controller-instance isolation at the engine level
* KEDA-shaped assertion: wva_desired_replicas queried over the metrics
  HTTP endpoint the way prometheus-adapter / KEDA's prometheus trigger
  would (smoke_test.go's external-metrics assertion analog)
"""
import os
import urllib.request

import pytest

from wva_amd.api import conditions as cond
from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.emulator.vllm_sim import ServiceProfile
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec

from test_e2e_emulated import MODEL, NS, VARIANT, make_stack, run_sim

OVERLOAD_PROFILE = dict(
    alpha_ms=50.0, beta_ms=2.0, max_num_seqs=8, num_gpu_blocks=500
)


class TestLimiterUnderExhaustion:
    def _exhausted_stack(self, cluster_gpus=3):
        """1-replica variant on a tiny GPU pool, limiter ON, V2 token
        analyzer (jumps straight to required capacity — the flow that
        can actually exceed the pool in one tick), under heavy
        overload."""
        from wva_amd.config.saturation import SaturationScalingConfig

        prof = ServiceProfile(**OVERLOAD_PROFILE)
        cluster, sim, app = make_stack(
            replicas=1, profile=prof, analyzer="saturation"
        )
        app.config.update_saturation_config(
            SaturationScalingConfig.from_dict(
                {"analyzerName": "saturation", "enableLimiter": True}
            )
        )
        # shrink the node to an exhausted pool (replaces the 8-GPU node)
        node = cluster.get("Node", "", "mi355x-0")
        node.allocatable = {"amd.com/gpu": str(cluster_gpus)}
        node.metadata.resource_version = 0
        cluster.update(node)
        app.config.set_limited_mode_enabled(True)
        return cluster, sim, app

    def test_scale_up_truncated_to_inventory(self):
        cluster, sim, app = self._exhausted_stack(cluster_gpus=3)
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=40, seconds=10)  # wants >> 3 replicas
        app.saturation_engine.optimize()
        d = app.decision_cache.get(NS, VARIANT)
        assert d is not None
        # actual target capped at what 3 GPUs fund (1 GPU per replica);
        # the decision message records the truncation (the cache entry
        # is slim by design — reference engines/common cache semantics)
        assert d.target_replicas <= 3
        assert d.target_replicas >= 2  # still scaled up within budget
        assert "limited" in d.optimization_ready_message

    def test_exhaustion_marks_limited(self):
        cluster, sim, app = self._exhausted_stack(cluster_gpus=2)
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=40, seconds=10)
        app.saturation_engine.optimize()
        d = app.decision_cache.get(NS, VARIANT)
        assert d is not None and d.target_replicas <= 2
        assert "limited" in d.optimization_ready_message

    def test_zero_free_gpus_freezes_scale_up(self):
        """Pool fully used by the current replica: no scale-up possible,
        but the current replica is never revoked (floor at current)."""
        cluster, sim, app = self._exhausted_stack(cluster_gpus=1)
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=40, seconds=10)
        app.saturation_engine.optimize()
        d = app.decision_cache.get(NS, VARIANT)
        assert d is not None
        assert d.target_replicas == 1
        # the full truncation is reported as a limit, not "no scale-up"
        assert "limited" in d.optimization_ready_message


class TestTargetConditionLifecycle:
    def test_deployment_delete_then_recreate(self):
        """TargetResolved flips False on Deployment deletion and back to
        True on recreation — through the manager's Deployment watch
        (reference target_condition_test.go + predicates.go:149-167)."""
        import time as _time

        prof = ServiceProfile(**OVERLOAD_PROFILE)
        cluster, sim, app = make_stack(replicas=1, profile=prof)
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=5, seconds=5)
        app.saturation_engine.optimize()
        app.va_reconciler.reconcile(NS, VARIANT)
        va = cluster.get("VariantAutoscaling", NS, VARIANT)
        assert cond.is_condition_true(va, "TargetResolved")

        app.manager.start()
        try:
            saved = cluster.get("Deployment", NS, VARIANT)
            cluster.delete("Deployment", NS, VARIANT)
            deadline = _time.time() + 5
            while _time.time() < deadline:
                va = cluster.get("VariantAutoscaling", NS, VARIANT)
                if cond.is_condition_false(va, "TargetResolved"):
                    break
                _time.sleep(0.05)
            assert cond.is_condition_false(va, "TargetResolved")

            saved.metadata.resource_version = 0
            cluster.create(saved)
            deadline = _time.time() + 5
            while _time.time() < deadline:
                va = cluster.get("VariantAutoscaling", NS, VARIANT)
                if cond.is_condition_true(va, "TargetResolved"):
                    break
                _time.sleep(0.05)
            assert cond.is_condition_true(va, "TargetResolved")
        finally:
            app.manager.stop()


class TestControllerInstanceIsolation:
    def _second_va(self, cluster, name, instance=None):
        labels = {"inference.optimization/acceleratorName": "MI355X"}
        if instance:
            labels["wva.llmd.ai/controller-instance"] = instance
        cluster.create(Deployment(
            metadata=ObjectMeta(name=name, namespace=NS),
            replicas=1,
            selector={"app": name},
            template=PodTemplateSpec(
                labels={"app": name},
                containers=[Container(requests={"amd.com/gpu": "1"})],
            ),
        ))
        cluster.create(VariantAutoscaling(
            metadata=ObjectMeta(name=name, namespace=NS, labels=labels),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name=name),
                model_id=MODEL,
            ),
        ))

    def test_engine_skips_foreign_instance_vas(self, monkeypatch):
        """Two VAs: one unlabeled (ours), one labeled for controller
        instance "other" — with CONTROLLER_INSTANCE unset, the engine
        must only decide for the unlabeled VA (predicates.go:184-243)."""
        monkeypatch.delenv("CONTROLLER_INSTANCE", raising=False)
        prof = ServiceProfile(**OVERLOAD_PROFILE)
        cluster, sim, app = make_stack(replicas=1, profile=prof)
        self._second_va(cluster, "vllm-other", instance="other")
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=20, seconds=10)
        app.saturation_engine.optimize()
        assert app.decision_cache.get(NS, VARIANT) is not None
        assert app.decision_cache.get(NS, "vllm-other") is None

    def test_instance_scoped_engine_owns_only_labeled(self, monkeypatch):
        monkeypatch.setenv("CONTROLLER_INSTANCE", "other")
        prof = ServiceProfile(**OVERLOAD_PROFILE)
        cluster, sim, app = make_stack(replicas=1, profile=prof)
        self._second_va(cluster, "vllm-other", instance="other")
        sim.register_variant(MODEL, NS, "vllm-other", ServiceProfile(
            **OVERLOAD_PROFILE))
        sim.reconcile_deployments()
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=20, seconds=10)
        app.saturation_engine.optimize()
        # this engine instance only owns the labeled VA
        assert app.decision_cache.get(NS, VARIANT) is None
        assert app.decision_cache.get(NS, "vllm-other") is not None


class TestKedaShapedMetricsQuery:
    def test_wva_desired_replicas_over_http(self):
        """Scrape /metrics over real HTTP and extract
        wva_desired_replicas{variant_name,namespace,accelerator_type}
        exactly as prometheus-adapter (HPA external metrics) or KEDA's
        prometheus trigger would consume it after a Prometheus scrape."""
        from wva_amd.collector.pod_scraping_source import parse_prometheus_text
        from wva_amd.runtime.http import ProbeServer

        prof = ServiceProfile(**OVERLOAD_PROFILE)
        cluster, sim, app = make_stack(replicas=1, profile=prof)
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=20, seconds=10)
        app.saturation_engine.optimize()

        server = ProbeServer(
            "127.0.0.1:0",
            healthz=lambda: True,
            readyz=lambda: True,
            registry=app.emitter.registry,
        )
        server.start()
        try:
            with urllib.request.urlopen(
                f"http://127.0.0.1:{server.port}/metrics", timeout=5
            ) as resp:
                text = resp.read().decode()
        finally:
            server.stop()

        samples = [
            v for v in parse_prometheus_text(text)
            if v.labels.get("__name__") == "wva_desired_replicas"
            and v.labels.get("variant_name") == VARIANT
            and v.labels.get("namespace") == NS
        ]
        assert samples, f"no wva_desired_replicas in:\n{text[:800]}"
        sample = samples[0]
        assert sample.labels["accelerator_type"] == "MI355X"
        assert sample.value >= 2  # the scale-up signal HPA/KEDA consumes
        # the companion gauges KEDA dashboards read
        names = {v.labels.get("__name__") for v in parse_prometheus_text(text)}
        assert {"wva_current_replicas", "wva_desired_ratio"} <= names


class TestWatchNamespaceScoping:
    def test_engine_ignores_other_namespaces(self):
        """WATCH_NAMESPACE set: the engine only decides for VAs in that
        namespace (cmd/main.go:289-297 single-namespace cache analog)."""
        from wva_amd.kube.objects import Node

        prof = ServiceProfile(**OVERLOAD_PROFILE)
        cluster, sim, app = make_stack(replicas=1, profile=prof)
        # a second VA in another namespace
        other_ns = "other-team"
        cluster.create(Deployment(
            metadata=ObjectMeta(name="vllm-b", namespace=other_ns),
            replicas=1,
            selector={"app": "vllm-b"},
            template=PodTemplateSpec(
                labels={"app": "vllm-b"},
                containers=[Container(requests={"amd.com/gpu": "1"})],
            ),
        ))
        cluster.create(VariantAutoscaling(
            metadata=ObjectMeta(
                name="vllm-b", namespace=other_ns,
                labels={"inference.optimization/acceleratorName": "MI355X"},
            ),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name="vllm-b"),
                model_id=MODEL,
            ),
        ))
        app.config.infra.watch_namespace = NS
        model = sim.model(MODEL, NS)
        run_sim(sim, model, qps=20, seconds=10)
        app.saturation_engine.optimize()
        assert app.decision_cache.get(NS, VARIANT) is not None
        assert app.decision_cache.get(other_ns, "vllm-b") is None


class TestMultiNamespaceConfigIsolation:
    def test_ns_local_thresholds_change_decisions(self):
        """Two namespaces, same model + identical load: the namespace
        with an ultra-sensitive ns-local saturation config scales up
        while the global-config namespace doesn't — the e2e proof of
        the global vs namespace-local override chain (config.go:360,
        configmap_reconciler.go:154-168)."""
        from wva_amd.kube.objects import ConfigMap

        prof_kwargs = dict(
            alpha_ms=20.0, beta_ms=1.0, max_num_seqs=64,
            num_gpu_blocks=20_000,
        )
        prof = ServiceProfile(**prof_kwargs)
        cluster, sim, app = make_stack(replicas=2, profile=prof)
        other_ns = "team-b"
        cluster.create(Deployment(
            metadata=ObjectMeta(name="vllm-b", namespace=other_ns),
            replicas=2,
            selector={"app": "vllm-b"},
            template=PodTemplateSpec(
                labels={"app": "vllm-b"},
                containers=[Container(requests={"amd.com/gpu": "1"})],
            ),
        ))
        cluster.create(VariantAutoscaling(
            metadata=ObjectMeta(
                name="vllm-b", namespace=other_ns,
                labels={"inference.optimization/acceleratorName": "MI355X"},
            ),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name="vllm-b"),
                model_id=MODEL,
            ),
        ))
        sim.register_variant(MODEL, other_ns, "vllm-b",
                             ServiceProfile(**prof_kwargs))
        sim.reconcile_deployments()

        # ns-local override for team-b: ANY kv usage is "saturated"
        # (kvCacheThreshold 0.01) → scale-up at the slightest load
        app.datastore.namespace_track(other_ns)
        cluster.create(ConfigMap(
            metadata=ObjectMeta(name="wva-saturation-scaling-config",
                                namespace=other_ns),
            data={"default": (
                "kvCacheThreshold: 0.01\n"
                "queueLengthThreshold: 1\n"
                "kvSpareTrigger: 0.01\n"
                "queueSpareTrigger: 1\n"
            )},
        ))
        app.configmap_reconciler.reconcile(
            other_ns, "wva-saturation-scaling-config"
        )

        # identical light load on both namespaces
        for ns_, variant in ((NS, VARIANT), (other_ns, "vllm-b")):
            m = sim.model(MODEL, ns_)
            run_sim(sim, m, qps=3, seconds=10)
        app.saturation_engine.optimize()

        d_global = app.decision_cache.get(NS, VARIANT)
        d_local = app.decision_cache.get(other_ns, "vllm-b")
        assert d_global is not None and d_local is not None
        # global thresholds (0.80/5): light load → no scale-up
        assert d_global.target_replicas <= 2
        # ns-local hair-trigger config → scale-up
        assert d_local.target_replicas > 2


class TestFullZeroCycle:
    def test_scale_to_zero_then_back_from_zero(self):
        """The complete zero cycle in one scenario: idle model scales
        to 0 (enforcer retention query), then queued EPP traffic brings
        it back 0→1 (scale-from-zero engine) — the two reference
        engines cooperating end to end."""
        from wva_amd.config.scale_to_zero import ModelScaleToZeroConfig
        from wva_amd.kube.objects import (
            InferencePool, Pod, PodStatus, Service, ServicePort,
        )

        cluster, sim, app = make_stack(replicas=1)
        model = sim.model(MODEL, NS)
        app.config.update_scale_to_zero_config({
            "default": ModelScaleToZeroConfig(
                enable_scale_to_zero=True, retention_period="1m"
            )
        })
        # EPP infra for the comeback path
        cluster.create(Service(
            metadata=ObjectMeta(name="pool-epp", namespace=NS),
            selector={"app": "epp"},
            ports=[ServicePort(name="metrics", port=9090)],
        ))
        cluster.create(Pod(
            metadata=ObjectMeta(name="epp-0", namespace=NS,
                                labels={"app": "epp"}),
            status=PodStatus(phase="Running", ready=True,
                             pod_ip="10.1.0.9"),
        ))
        app.datastore.scrape_fetch = (
            lambda url, headers, timeout: sim.epp_metrics_text(NS)
        )
        cluster.create(InferencePool(
            metadata=ObjectMeta(name="pool", namespace=NS),
            selector={"app": VARIANT},
            epp_service_name="pool-epp",
        ))
        app.inferencepool_reconciler.reconcile(NS, "pool")

        # phase 1: idle past retention → scale to zero
        run_sim(sim, model, qps=0, seconds=90)
        app.saturation_engine.optimize()
        d = app.decision_cache.get(NS, VARIANT)
        assert d is not None and d.target_replicas == 0
        cluster.scale("Deployment", NS, VARIANT, 0)  # HPA applies 0
        sim.reconcile_deployments()

        # phase 2: traffic arrives with zero replicas → EPP queue grows
        run_sim(sim, model, qps=2, seconds=3)
        assert len(model.scheduler_queue) > 0

        app.scale_from_zero_engine.optimize()
        deploy = cluster.get("Deployment", NS, VARIANT)
        assert deploy.replicas == 1  # DirectActuator 0→1
        va = cluster.get("VariantAutoscaling", NS, VARIANT)
        assert cond.is_condition_true(va, "ScaleFromZeroMode")
