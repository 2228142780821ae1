"""Cluster simulator: deployment controller + model load balancing.

Plays the role of the reference's Kind-emulator stack (SURVEY §4:
deploy/kind-emulator + llm-d-inference-sim): drives the FakeCluster's
Deployments into Pods with a configurable readiness delay (model-load time;
2-7 min on real 70B pods, seconds here), hosts one ReplicaSim per ready
pod, routes arriving requests least-loaded across ready replicas, and
maintains the model-level EPP flow-control queue for scale-from-zero.

Time is simulated: advance(dt) moves the whole world forward, decoupled
from wall clock so bench can run hours of autoscaling in seconds.
"""
from __future__ import annotations

import itertools
import random
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..api.types import ObjectMeta
from ..kube.fake import FakeCluster
from ..kube.objects import Container, Deployment, Pod, PodStatus
from .vllm_sim import CompletedRequest, ReplicaSim, RequestSpec, ServiceProfile
from .workload import QPSProfile


@dataclass
class ModelSim:
    """One model's serving state across its variants."""

    model_id: str
    namespace: str
    # deployment name → service profile for its replicas
    profiles: Dict[str, ServiceProfile] = field(default_factory=dict)
    # model-level scheduler (EPP flow-control) queue
    scheduler_queue: List[RequestSpec] = field(default_factory=list)
    completed: List[CompletedRequest] = field(default_factory=list)
    dropped: int = 0
    submitted: int = 0


class ClusterSim:
    def __init__(
        self,
        cluster: FakeCluster,
        pod_ready_delay_s: float = 30.0,
        seed: int = 0,
        warm_start: bool = False,
    ):
        self.cluster = cluster
        self.pod_ready_delay_s = pod_ready_delay_s
        # warm_start: pods created at sim time 0 are ready immediately
        # (the serving stack pre-exists the benchmark window)
        self.warm_start = warm_start
        self.now = 0.0
        self.models: Dict[str, ModelSim] = {}
        # pod name → (ReplicaSim, ready_at, deployment_name, namespace)
        self.replicas: Dict[str, Tuple[ReplicaSim, float, str, str]] = {}
        self._pod_counter = itertools.count()
        self._arrival_residual: Dict[str, float] = {}
        random.seed(seed)

    # --- model/variant registration ---

    def register_model(
        self, model_id: str, namespace: str
    ) -> ModelSim:
        key = f"{model_id}|{namespace}"
        if key not in self.models:
            self.models[key] = ModelSim(model_id=model_id, namespace=namespace)
        return self.models[key]

    def register_variant(
        self,
        model_id: str,
        namespace: str,
        deployment_name: str,
        profile: ServiceProfile,
    ) -> None:
        self.register_model(model_id, namespace).profiles[deployment_name] = profile

    def model(self, model_id: str, namespace: str) -> ModelSim:
        return self.models[f"{model_id}|{namespace}"]

    # --- deployment controller ---

    def _variant_of_deployment(
        self, deploy: Deployment
    ) -> Optional[ModelSim]:
        for m in self.models.values():
            if m.namespace == deploy.namespace and deploy.name in m.profiles:
                return m
        return None

    def reconcile_deployments(self) -> None:
        """Drive pods toward spec.replicas; update deployment status."""
        for deploy in self.cluster.list("Deployment"):
            model = self._variant_of_deployment(deploy)
            if model is None:
                continue
            profile = model.profiles[deploy.name]
            pods = [
                name
                for name, (_, _, dname, ns) in self.replicas.items()
                if dname == deploy.name and ns == deploy.namespace
            ]
            want = deploy.replicas
            # scale up: create pods
            while len(pods) < want:
                pod_name = f"{deploy.name}-{next(self._pod_counter):05x}"
                sim = ReplicaSim(pod_name, profile)
                ready_at = self.now + self.pod_ready_delay_s
                if self.warm_start and self.now == 0.0:
                    ready_at = 0.0
                self.replicas[pod_name] = (
                    sim, ready_at, deploy.name, deploy.namespace
                )
                labels = dict(deploy.template.labels) or {"app": deploy.name}
                self.cluster.create(Pod(
                    metadata=ObjectMeta(
                        name=pod_name,
                        namespace=deploy.namespace,
                        labels=labels,
                        owner_references=[
                            {"kind": "Deployment", "name": deploy.name}
                        ],
                    ),
                    containers=[
                        Container(requests=dict(
                            deploy.template.containers[0].requests
                        ) if deploy.template.containers else {})
                    ],
                    node_name=self._pick_node(),
                    status=PodStatus(phase="Running", ready=False),
                ))
                pods.append(pod_name)
            # scale down: delete newest pods first
            while len(pods) > want:
                victim = pods.pop()
                sim, _, _, ns = self.replicas.pop(victim)
                # re-queue unfinished work onto the model scheduler queue
                for r in list(sim.waiting) + list(sim.running):
                    model.scheduler_queue.append(r.spec)
                try:
                    self.cluster.delete("Pod", ns, victim)
                except KeyError:
                    pass
            # readiness + status
            ready = 0
            for pod_name in pods:
                sim, ready_at, _, ns = self.replicas[pod_name]
                is_ready = self.now >= ready_at
                if is_ready:
                    ready += 1
                pod = self.cluster.try_get("Pod", ns, pod_name)
                if pod is not None and pod.status.ready != is_ready:
                    pod.status.ready = is_ready
                    # blind write (no resourceVersion precondition): the sim
                    # plays kubelet/controller-manager and must not conflict
                    # with concurrent WVA writes through the API surface
                    pod.metadata.resource_version = 0
                    self.cluster.update(pod)
            deploy.status.replicas = len(pods)
            deploy.status.ready_replicas = ready
            deploy.status.available_replicas = ready
            deploy.metadata.resource_version = 0
            self.cluster.update(deploy)

    def _pick_node(self) -> str:
        nodes = self.cluster.list("Node")
        return nodes[0].name if nodes else ""

    # --- traffic ---

    def ready_replicas_of_model(self, model: ModelSim) -> List[ReplicaSim]:
        out = []
        for name, (sim, ready_at, dname, ns) in self.replicas.items():
            if ns != model.namespace or dname not in model.profiles:
                continue
            if self.now >= ready_at:
                out.append(sim)
        return out

    # EPP flow-control admission bound: a replica accepts new work only up
    # to max_num_seqs + this many waiting requests; the rest stays in the
    # model-level scheduler queue (inference_extension_flow_control_*).
    # Mirrors llm-d's EPP, and keeps vllm:num_requests_waiting a truthful
    # saturation signal instead of an unbounded dump.
    QUEUE_ALLOWANCE = 10

    def _admission_cap(self, replica: ReplicaSim) -> float:
        return replica.profile.max_num_seqs + self.QUEUE_ALLOWANCE

    def submit_request(self, model: ModelSim, spec: RequestSpec) -> None:
        model.submitted += 1
        ready = self.ready_replicas_of_model(model)
        candidates = [r for r in ready if r.load() < self._admission_cap(r)]
        if not candidates:
            model.scheduler_queue.append(spec)
            return
        target = min(candidates, key=lambda r: r.load())
        target.submit(spec)

    def _drain_scheduler_queue(self, model: ModelSim) -> None:
        ready = self.ready_replicas_of_model(model)
        if not ready:
            return
        while model.scheduler_queue:
            candidates = [r for r in ready if r.load() < self._admission_cap(r)]
            if not candidates:
                return
            spec = model.scheduler_queue.pop(0)
            target = min(candidates, key=lambda r: r.load())
            target.submit(spec)

    def generate_arrivals(
        self,
        model: ModelSim,
        profile: QPSProfile,
        dt: float,
        input_tokens: int = 100,
        output_tokens: int = 50,
    ) -> int:
        """Poisson-ish arrivals for dt at rate profile(now)."""
        key = f"{model.model_id}|{model.namespace}"
        rate = profile(self.now)
        expected = rate * dt + self._arrival_residual.get(key, 0.0)
        n = int(expected)
        self._arrival_residual[key] = expected - n
        for _ in range(n):
            self.submit_request(
                model,
                RequestSpec(
                    input_tokens=input_tokens,
                    output_tokens=output_tokens,
                    arrival_time=self.now,
                ),
            )
        return n

    # --- time ---

    def advance(self, dt: float) -> None:
        """Advance simulated time: deployment controller + replica decode."""
        self.reconcile_deployments()
        for model in self.models.values():
            self._drain_scheduler_queue(model)
        for name, (sim, ready_at, _, _) in self.replicas.items():
            if self.now >= ready_at:
                completed = sim.step(self.now, dt)
                if completed:
                    model = self._model_of_replica(name)
                    if model is not None:
                        model.completed.extend(completed)
        self.now += dt

    def _model_of_replica(self, pod_name: str) -> Optional[ModelSim]:
        entry = self.replicas.get(pod_name)
        if entry is None:
            return None
        _, _, dname, ns = entry
        for m in self.models.values():
            if m.namespace == ns and dname in m.profiles:
                return m
        return None

    # --- EPP metrics text (for PodScrapingSource fetch hook) ---

    def epp_metrics_text(self, namespace: str) -> str:
        lines = []
        for model in self.models.values():
            if model.namespace != namespace:
                continue
            size = len(model.scheduler_queue)
            byts = sum(
                4 * s.input_tokens for s in model.scheduler_queue
            )
            lines.append(
                f'inference_extension_flow_control_queue_size'
                f'{{model_name="{model.model_id}",'
                f'target_model_name="{model.model_id}"}} {size}'
            )
            lines.append(
                f'inference_extension_flow_control_queue_bytes'
                f'{{model_name="{model.model_id}",'
                f'target_model_name="{model.model_id}"}} {byts}'
            )
        return "\n".join(lines) + "\n"
