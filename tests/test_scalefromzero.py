"""ScaleFromZero engine unit suite (reference scalefromzero/engine.go
:73-358 — processed so far only through e2e; these pin the per-VA
decision ladder, tolerance paths, concurrency bound and the
target_model_name label fallback (#2309))."""
import threading

import pytest

from wva_amd.actuator.direct import DirectActuator
from wva_amd.api import conditions as cond
from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.config.config import Config
from wva_amd.datastore.datastore import Datastore
from wva_amd.engines.common import DecisionCache, DecisionTrigger
from wva_amd.engines.scalefromzero import ScaleFromZeroEngine
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import (
    Container,
    Deployment,
    EndpointPicker,
    EndpointPool,
    Pod,
    PodStatus,
    PodTemplateSpec,
    Service,
    ServicePort,
)

NS = "default"
MODEL = "meta-llama/Llama-3.1-8B"


def epp_text(queue_size, label="target_model_name", model=MODEL):
    return (
        f'inference_extension_flow_control_queue_size{{{label}="{model}"}} '
        f"{queue_size}\n"
    )


class Stack:
    def __init__(self, replicas=0, epp_response=None):
        self.cluster = FakeCluster()
        self.cluster.create(Deployment(
            metadata=ObjectMeta(name="v", namespace=NS),
            replicas=replicas,
            selector={"app": "v"},
            template=PodTemplateSpec(
                labels={"app": "v"},
                containers=[Container(requests={"amd.com/gpu": "1"})],
            ),
        ))
        self.cluster.create(VariantAutoscaling(
            metadata=ObjectMeta(
                name="v", namespace=NS,
                labels={"inference.optimization/acceleratorName": "MI355X"},
            ),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name="v"),
                model_id=MODEL,
            ),
        ))
        self.cluster.create(Service(
            metadata=ObjectMeta(name="epp", namespace=NS),
            selector={"app": "epp"},
            ports=[ServicePort(name="metrics", port=9090)],
        ))
        self.cluster.create(Pod(
            metadata=ObjectMeta(name="epp-0", namespace=NS,
                                labels={"app": "epp"}),
            status=PodStatus(phase="Running", ready=True, pod_ip="10.0.0.1"),
        ))
        self.fetches = []

        def fetch(url, headers, timeout):
            self.fetches.append(url)
            if isinstance(epp_response, Exception):
                raise epp_response
            return epp_response if epp_response is not None else ""

        self.config = Config()
        self.config.mark_bootstrap_complete()
        self.datastore = Datastore(self.cluster, scrape_fetch=fetch)
        self.cache = DecisionCache()
        self.trigger = DecisionTrigger()
        self.engine = ScaleFromZeroEngine(
            cluster=self.cluster,
            config=self.config,
            datastore=self.datastore,
            direct_actuator=DirectActuator(self.cluster),
            decision_cache=self.cache,
            decision_trigger=self.trigger,
            interval_seconds=999,
        )

    def add_pool(self):
        self.datastore.pool_set(EndpointPool(
            name="pool", namespace=NS, selector={"app": "v"},
            endpoint_picker=EndpointPicker(
                service_name="epp", namespace=NS, metrics_port_number=9090,
            ),
        ))

    def replicas(self):
        return self.cluster.get("Deployment", NS, "v").replicas


class TestScaleFromZeroLadder:
    def test_pending_scales_0_to_1(self):
        s = Stack(epp_response=epp_text(3))
        s.add_pool()
        s.engine.optimize()
        assert s.replicas() == 1
        d = s.cache.get(NS, "v")
        assert d is not None and d.target_replicas == 1
        va = s.cluster.get("VariantAutoscaling", NS, "v")
        assert cond.is_condition_true(va, "ScaleFromZeroMode")
        assert va.status.desired_optimized_alloc.accelerator == "MI355X"

    def test_zero_queue_no_action(self):
        s = Stack(epp_response=epp_text(0))
        s.add_pool()
        s.engine.optimize()
        assert s.replicas() == 0
        assert s.cache.get(NS, "v") is None

    def test_active_va_ignored(self):
        s = Stack(replicas=2, epp_response=epp_text(9))
        s.add_pool()
        s.engine.optimize()
        assert s.replicas() == 2  # not an inactive VA

    def test_no_pool_no_action(self):
        s = Stack(epp_response=epp_text(9))  # pool never registered
        s.engine.optimize()
        assert s.replicas() == 0

    def test_scrape_failure_tolerated(self):
        s = Stack(epp_response=RuntimeError("epp down"))
        s.add_pool()
        s.engine.optimize()  # must not raise
        assert s.replicas() == 0

    def test_2309_model_name_fallback(self):
        """EPP metrics without target_model_name fall back to matching
        model_name with empty target (registration #2309 parity)."""
        s = Stack(epp_response=epp_text(5, label="model_name"))
        s.add_pool()
        s.engine.optimize()
        assert s.replicas() == 1

    def test_other_models_queue_ignored(self):
        s = Stack(epp_response=epp_text(7, model="someone/else"))
        s.add_pool()
        s.engine.optimize()
        assert s.replicas() == 0


class TestConcurrencyBound:
    def test_parallel_processing_bounded(self):
        """Many inactive VAs process concurrently but never more than
        SCALE_FROM_ZERO_ENGINE_MAX_CONCURRENCY at once
        (scalefromzero/engine.go:150-177)."""
        s = Stack(epp_response=epp_text(1))
        s.add_pool()
        N = 12
        for i in range(1, N):
            s.cluster.create(Deployment(
                metadata=ObjectMeta(name=f"v{i}", namespace=NS),
                replicas=0,
                selector={"app": "v"},
                template=PodTemplateSpec(
                    labels={"app": "v"},
                    containers=[Container(requests={"amd.com/gpu": "1"})],
                ),
            ))
            s.cluster.create(VariantAutoscaling(
                metadata=ObjectMeta(name=f"v{i}", namespace=NS),
                spec=VariantAutoscalingSpec(
                    scale_target_ref=CrossVersionObjectReference(
                        name=f"v{i}"),
                    model_id=MODEL,
                ),
            ))
        s.config.set_scale_from_zero_max_concurrency(3)

        live = 0
        peak = 0
        lock = threading.Lock()
        orig = s.engine._process_inactive_variant

        def tracked(va):
            nonlocal live, peak
            with lock:
                live += 1
                peak = max(peak, live)
            try:
                import time
                time.sleep(0.02)
                return orig(va)
            finally:
                with lock:
                    live -= 1

        s.engine._process_inactive_variant = tracked
        s.engine.optimize()
        assert peak <= 3
        # every VA still processed: all scaled to 1
        for i in range(1, N):
            assert s.cluster.get("Deployment", NS, f"v{i}").replicas == 1
