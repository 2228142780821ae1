"""Model analyzer — VA → Inferno System adapter.

Parity: reference internal/modelanalyzer/analyzer.go:13-34 + the SystemData
adapters in internal/utils/utils.go:125-315 (the dormant SLO path). Builds
an inferno Server from a VariantAutoscaling + observed load and returns all
feasible allocations across registered accelerators.

Beyond parity, InfernoAnalyzer exposes the library through the common
Analyzer interface (interfaces/analyzer.go) so it is selectable via
analyzerName: "inferno" — the reference never wires its Inferno library
into the live loop; here it is a first-class analyzer whose capacity is
SLO-derived (replicas sized by queueing model) instead of
saturation-derived.
"""
from __future__ import annotations

from typing import Dict, List

from ..api.types import VariantAutoscaling, utcnow
from ..config.saturation import SaturationScalingConfig
from ..inferno.system import Allocation, System
from ..inferno.types import ServerLoadSpec, ServerSpec
from .interfaces import (
    AnalyzerInput,
    AnalyzerResult,
    VariantCapacity,
)


class ModelAnalyzer:
    """All feasible allocations for one VA's model across accelerators."""

    def __init__(self, system: System):
        self.system = system

    def analyze_model(
        self,
        va: VariantAutoscaling,
        arrival_rate_per_min: float,
        avg_in_tokens: int,
        avg_out_tokens: int,
        service_class: str = "default",
        current_accelerator: str = "",
        current_num_replicas: int = 0,
    ) -> Dict[str, Allocation]:
        """Returns accelerator name → feasible Allocation (empty if none)."""
        server_name = f"{va.namespace}/{va.name}"
        self.system.add_server(ServerSpec(
            name=server_name,
            service_class=service_class,
            model=va.spec.model_id,
            load=ServerLoadSpec(
                arrival_rate=arrival_rate_per_min,
                avg_in_tokens=avg_in_tokens,
                avg_out_tokens=avg_out_tokens,
            ),
            current_accelerator=current_accelerator,
            current_num_replicas=current_num_replicas,
        ))
        result: Dict[str, Allocation] = {}
        for acc_name in self.system.accelerators:
            alloc = self.system.create_allocation(server_name, acc_name)
            if alloc is not None:
                result[acc_name] = alloc
        return result


class InfernoAnalyzer:
    """Inferno exposed through the common Analyzer interface.

    Capacity semantics: per-variant capacity = SLO-constrained sustainable
    request rate per replica × replicas (requests/s units); demand = the
    observed arrival rate. Signals use the same required/spare convention
    as the V2 analyzer so the CostAwareOptimizer consumes them unchanged.
    """

    def __init__(self, system: System, service_class: str = "default",
                 enable_tuner: bool = True):
        self.system = system
        self.service_class = service_class
        # model_id → (arrival_rate req/s, avg_in, avg_out) fed by the engine
        self._observed_load: Dict[str, tuple] = {}
        # online EKF tuners per (model, accelerator) refining the
        # ConfigMap-seeded α/β/γ from live (TTFT, ITL) observations
        # (reference tuner.go:29-143 — dormant there, live here)
        self.enable_tuner = enable_tuner
        self._tuners: Dict[tuple, object] = {}

    def name(self) -> str:
        return "inferno-slo"

    def observe_load(
        self, model_id: str, arrival_rate_per_s: float, avg_in: float, avg_out: float
    ) -> None:
        self._observed_load[model_id] = (arrival_rate_per_s, avg_in, avg_out)

    def observe_latency(
        self,
        model_id: str,
        accelerator: str,
        per_replica_rate: float,
        avg_in: float,
        avg_out: float,
        ttft_ms: float,
        itl_ms: float,
    ) -> bool:
        """One EKF step on the (model, accelerator) service parameters
        from a live (TTFT, ITL) observation at the per-REPLICA request
        rate (the queueing model is single-server). Accepted updates are
        written back into the System's perf record, so the very next
        sizing pass uses the refined α/β/γ. Returns True if accepted
        (NIS gate), False if rejected as an outlier."""
        if not self.enable_tuner:
            return False
        perf = self.system.perf.get((model_id, accelerator))
        if perf is None or per_replica_rate <= 0:
            return False
        from ..inferno.queue_analyzer import ServiceParms
        from ..inferno.tuner import Observation, ServiceParmsTuner, TunerConfig

        key = (model_id, accelerator)
        tuner = self._tuners.get(key)
        if tuner is None:
            tuner = ServiceParmsTuner(
                ServiceParms(
                    alpha=perf.service_parms.alpha,
                    beta=perf.service_parms.beta,
                    gamma=perf.service_parms.gamma,
                ),
                TunerConfig(
                    max_batch_size=max(perf.max_batch_size, 1),
                    max_queue_size=max(perf.max_batch_size, 1) * 10,
                ),
            )
            self._tuners[key] = tuner
        # saturation gate: when the observed per-replica rate reaches
        # the CURRENT model's maximum, the real queue is unbounded and
        # the observed TTFT cannot be explained by the bounded-queue
        # model — update from ITL alone (still identifies α, β at the
        # running batch) instead of letting backlogged TTFT corrupt θ
        itl_only = False
        try:
            from ..inferno.queue_analyzer import (
                Configuration as _QCfg,
                QueueAnalyzer as _QA,
                RequestSize as _QReq,
            )

            probe = _QA(
                _QCfg(
                    max_batch_size=tuner.config.max_batch_size,
                    max_queue_size=tuner.config.max_queue_size,
                    service_parms=tuner.parms(),
                ),
                _QReq(avg_input_tokens=avg_in, avg_output_tokens=avg_out),
            )
            itl_only = per_replica_rate >= 0.9 * probe.rate_max
        except (ValueError, ZeroDivisionError):
            itl_only = True
        accepted = tuner.update(Observation(
            request_rate=per_replica_rate,
            avg_input_tokens=avg_in,
            avg_output_tokens=avg_out,
            ttft_ms=ttft_ms,
            itl_ms=itl_ms,
        ), itl_only=itl_only)
        if accepted:
            p = tuner.parms()
            perf.service_parms.alpha = p.alpha
            perf.service_parms.beta = p.beta
            perf.service_parms.gamma = p.gamma
        return accepted

    def analyze(self, input: AnalyzerInput) -> AnalyzerResult:
        cfg = input.config
        scale_up = 0.85
        scale_down = 0.70
        if isinstance(cfg, SaturationScalingConfig):
            cfg.apply_defaults()
            scale_up = cfg.scale_up_threshold or 0.85
            scale_down = cfg.scale_down_boundary or 0.70

        arrival, avg_in, avg_out = self._observed_load.get(
            input.model_id, (0.0, 100.0, 50.0)
        )
        # fall back to replica-metrics token averages when present
        if input.replica_metrics:
            ins = [m.avg_input_tokens for m in input.replica_metrics if m.avg_input_tokens > 0]
            outs = [m.avg_output_tokens for m in input.replica_metrics if m.avg_output_tokens > 0]
            if ins:
                avg_in = sum(ins) / len(ins)
            if outs:
                avg_out = sum(outs) / len(outs)

        variant_capacities: List[VariantCapacity] = []
        total_supply = total_demand = 0.0
        for vs in input.variant_states:
            accel = ""
            cost = 10.0
            for m in input.replica_metrics:
                if m.variant_name == vs.variant_name:
                    accel = m.accelerator_name
                    cost = m.cost
                    break
            per_replica_rate = self._per_replica_rate(
                input.model_id, accel, avg_in, avg_out
            )
            ready = max(vs.current_replicas - vs.pending_replicas, 0)
            supply = ready * per_replica_rate
            variant_capacities.append(VariantCapacity(
                variant_name=vs.variant_name,
                accelerator_name=accel,
                cost=cost,
                replica_count=ready,
                pending_replicas=vs.pending_replicas,
                per_replica_capacity=per_replica_rate,
                total_capacity=supply,
                total_demand=arrival,  # model-level demand on the variant set
                utilization=arrival / supply if supply > 0 else 0.0,
            ))
            total_supply += supply
        total_demand = arrival

        anticipated = sum(
            (vc.replica_count + vc.pending_replicas) * vc.per_replica_capacity
            for vc in variant_capacities
        )
        required = max(total_demand / scale_up - anticipated, 0.0)
        spare = max(total_supply - total_demand / scale_down, 0.0)
        return AnalyzerResult(
            analyzer_name=self.name(),
            model_id=input.model_id,
            namespace=input.namespace,
            analyzed_at=utcnow(),
            variant_capacities=variant_capacities,
            total_supply=total_supply,
            total_demand=total_demand,
            utilization=total_demand / total_supply if total_supply else 0.0,
            required_capacity=required,
            spare_capacity=spare,
        )

    def _per_replica_rate(
        self, model_id: str, accelerator: str, avg_in: float, avg_out: float
    ) -> float:
        """SLO-constrained sustainable requests/s per replica from the
        queueing model (QueueAnalyzer.size)."""
        from ..inferno.queue_analyzer import (
            Configuration,
            QueueAnalyzer,
            RequestSize,
            ServiceParms,
            TargetPerf,
        )
        from ..inferno.types import MAX_QUEUE_TO_BATCH_RATIO

        perf = self.system.perf.get((model_id, accelerator))
        if perf is None:
            return 0.0
        # SLO target resolution: the configured class first, then any
        # other class carrying this model, highest priority (lowest
        # number) first — so a multi-class ConfigMap (premium/freemium,
        # reference chart wva-configmap-service-class.yaml) works
        # without per-VA class labels
        target = None
        classes = []
        sc = self.system.service_classes.get(self.service_class)
        if sc is not None:
            classes.append(sc)
        classes.extend(sorted(
            (c for n, c in self.system.service_classes.items()
             if sc is None or n != sc.name),
            key=lambda c: c.priority,
        ))
        for c in classes:
            for t in c.model_targets:
                if t.model == model_id:
                    target = t
                    break
            if target is not None:
                break
        if target is None:
            return 0.0
        K = max(int(avg_out), 1)
        N = max(perf.max_batch_size * perf.at_tokens // K, 1) \
            if perf.at_tokens else perf.max_batch_size
        try:
            qa = QueueAnalyzer(
                Configuration(
                    max_batch_size=N,
                    max_queue_size=N * MAX_QUEUE_TO_BATCH_RATIO,
                    service_parms=ServiceParms(
                        alpha=perf.service_parms.alpha,
                        beta=perf.service_parms.beta,
                        gamma=perf.service_parms.gamma,
                    ),
                ),
                RequestSize(avg_input_tokens=avg_in, avg_output_tokens=float(K)),
            )
            _, metrics, _ = qa.size(TargetPerf(
                target_ttft=target.slo_ttft,
                target_itl=target.slo_itl,
                target_tps=target.slo_tps,
            ))
        except (ValueError, ZeroDivisionError):
            return 0.0
        return metrics.throughput
