"""Inferno SystemData JSON schema.

Parity: reference pkg/config/types.go:6-157 and defaults.go:12-35 — same
field names on the wire (JSON camelCase) so reference system-data files
load unchanged. Defaults: SLOPercentile 0.95, MaxQueueToBatchRatio 10,
AcceleratorSwitchFactor (transition penalty) 0.1.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List

# defaults (pkg/config/defaults.go)
SLO_PERCENTILE = 0.95
MAX_QUEUE_TO_BATCH_RATIO = 10
ACCELERATOR_SWITCH_FACTOR = 0.1

# best-effort saturation policies (solver/greedy.go bestEffort)
POLICY_PRIORITY_EXHAUSTIVE = "PriorityExhaustive"
POLICY_PRIORITY_ROUND_ROBIN = "PriorityRoundRobin"
POLICY_ROUND_ROBIN = "RoundRobin"
POLICY_NONE = "None"


@dataclass
class AcceleratorSpec:
    name: str = ""
    type: str = ""  # capacity pool type (e.g. "MI355X")
    multiplicity: int = 1  # devices per unit (xGMI hive size for TP variants)
    mem_size: int = 0  # GB
    mem_bw: int = 0  # GB/s
    power: float = 0.0
    cost: float = 0.0

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "AcceleratorSpec":
        return cls(
            name=d.get("name", ""),
            type=d.get("type", d.get("name", "")),
            multiplicity=int(d.get("multiplicity", 1)),
            mem_size=int(d.get("memSize", 0)),
            mem_bw=int(d.get("memBW", 0)),
            power=float(d.get("power", 0.0)),
            cost=float(d.get("cost", 0.0)),
        )


@dataclass
class ServiceParmsSpec:
    alpha: float = 0.0
    beta: float = 0.0
    gamma: float = 0.0

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ServiceParmsSpec":
        return cls(
            alpha=float(d.get("alpha", 0.0)),
            beta=float(d.get("beta", 0.0)),
            gamma=float(d.get("gamma", 0.0)),
        )

    @classmethod
    def from_itl_fit(
        cls, alpha_ms: float, beta_itl_ms: float,
        avg_input_tokens: float, avg_output_tokens: float,
    ) -> "ServiceParmsSpec":
        """Convert a RAW ITL fit (ITL(n) = α + β_itl·n, the calibration
        harness' model — profiles/calibration_*.json) into the Inferno
        parameter convention, whose iteration time multiplies β by
        tokensCompute = (in+out)/(out+1) (queueanalyzer.go:261-279).
        Feeding β_itl directly overestimates service time ~3× at the
        default 100/50 token mix and makes the SLO sizer over-provision
        by the same factor."""
        tokens_compute = (avg_input_tokens + avg_output_tokens) / (
            avg_output_tokens + 1.0
        )
        return cls(
            alpha=alpha_ms,
            beta=beta_itl_ms / max(tokens_compute, 1e-9),
            gamma=0.0,
        )

    @classmethod
    def from_itl_surface(
        cls, alpha_ms: float, beta_eff_ms: float,
        gamma_ms_per_ctx_token: float,
        avg_input_tokens: float, avg_output_tokens: float,
    ) -> "ServiceParmsSpec":
        """Convert a measured (batch, context) ITL SURFACE fit
        (calibration.itl_benchmark.fit_itl_surface:
        ITL = α + n·(β_eff + γ·ctx)) into the reference parameter
        convention. γ maps directly: the model's tokensMemory term
        (I + O/2) IS the average per-request context footprint, so the
        per-context-token slope measured on hardware is the reference's
        γ. β_eff divides out tokensCompute like from_itl_fit."""
        tokens_compute = (avg_input_tokens + avg_output_tokens) / (
            avg_output_tokens + 1.0
        )
        return cls(
            alpha=alpha_ms,
            beta=beta_eff_ms / max(tokens_compute, 1e-9),
            gamma=gamma_ms_per_ctx_token,
        )


@dataclass
class ModelAcceleratorPerfData:
    name: str = ""  # model name
    acc: str = ""  # accelerator name
    acc_count: int = 1  # instances of accelerator per replica
    max_batch_size: int = 0
    at_tokens: int = 0  # token count at which maxBatchSize was measured
    service_parms: ServiceParmsSpec = field(default_factory=ServiceParmsSpec)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ModelAcceleratorPerfData":
        # two parameter conventions:
        #  * decodeParms/serviceParms: the reference's SystemData fields
        #    (β multiplied by tokensCompute inside the queueing model)
        #  * itlFit: a RAW calibration fit (ITL(n) = α + β·n, the
        #    profiles/calibration_*.json numbers) with the request token
        #    mix it should be converted at — feeding a raw ITL β as
        #    decodeParms overestimates service time ~3× (see
        #    ServiceParmsSpec.from_itl_fit)
        itl_fit = d.get("itlFit")
        itl_surface = d.get("itlSurface")
        if itl_surface:
            # full 3-parameter measured surface (fit_itl_surface):
            # includes the per-context-token γ the fixed-context fit
            # absorbs into β — profiles/calibration_8b_gamma.json
            parms = ServiceParmsSpec.from_itl_surface(
                float(itl_surface.get("alpha", 0.0)),
                float(itl_surface.get("betaEff", 0.0)),
                float(itl_surface.get("gamma", 0.0)),
                float(itl_surface.get("avgInputTokens", 100.0)),
                float(itl_surface.get("avgOutputTokens", 50.0)),
            )
        elif itl_fit:
            parms = ServiceParmsSpec.from_itl_fit(
                float(itl_fit.get("alpha", 0.0)),
                float(itl_fit.get("beta", 0.0)),
                float(itl_fit.get("avgInputTokens", 100.0)),
                float(itl_fit.get("avgOutputTokens", 50.0)),
            )
        else:
            parms = ServiceParmsSpec.from_dict(
                d.get("decodeParms") or d.get("serviceParms") or {}
            )
        return cls(
            name=d.get("name", ""),
            acc=d.get("acc", ""),
            acc_count=int(d.get("accCount", 1)),
            max_batch_size=int(d.get("maxBatchSize", 0)),
            at_tokens=int(d.get("atTokens", 0)),
            service_parms=parms,
        )


@dataclass
class ModelTarget:
    model: str = ""
    slo_itl: float = 0.0  # msec
    slo_ttft: float = 0.0  # msec (queueing + prefill)
    slo_tps: float = 0.0  # tokens/sec

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ModelTarget":
        # accepts the reference chart's service-class keys: slo-tpot
        # (time-per-output-token = ITL), slo-ttft, slo-tps
        itl = d.get("slo-tpot", d.get("slo-itl", d.get("SLO_ITL", 0.0)))
        ttft = d.get("slo-ttft", d.get("slo-ttw", d.get("SLO_TTFT", 0.0)))
        return cls(
            model=d.get("model", ""),
            slo_itl=float(itl or 0.0),
            slo_ttft=float(ttft or 0.0),
            slo_tps=float(d.get("slo-tps", d.get("SLO_TPS", 0.0)) or 0.0),
        )


@dataclass
class ServiceClassSpec:
    name: str = ""
    priority: int = 0  # lower value = higher priority
    model_targets: List[ModelTarget] = field(default_factory=list)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ServiceClassSpec":
        return cls(
            name=d.get("name", ""),
            priority=int(d.get("priority", 0)),
            model_targets=[ModelTarget.from_dict(t) for t in d.get("data", [])],
        )


@dataclass
class ServerLoadSpec:
    arrival_rate: float = 0.0  # requests/min
    avg_in_tokens: int = 0
    avg_out_tokens: int = 0

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ServerLoadSpec":
        return cls(
            arrival_rate=float(d.get("arrivalRate", 0.0)),
            avg_in_tokens=int(d.get("avgInTokens", d.get("avgInputTokens", 0))),
            avg_out_tokens=int(d.get("avgOutTokens", d.get("avgOutputTokens", 0))),
        )


@dataclass
class ServerSpec:
    name: str = ""
    service_class: str = ""
    model: str = ""
    load: ServerLoadSpec = field(default_factory=ServerLoadSpec)
    max_batch_size: int = 0  # 0 = derive from perf data
    min_num_replicas: int = 0
    current_accelerator: str = ""
    current_num_replicas: int = 0

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ServerSpec":
        return cls(
            name=d.get("name", ""),
            service_class=d.get("class", d.get("serviceClass", "")),
            model=d.get("model", ""),
            load=ServerLoadSpec.from_dict(d.get("load") or {}),
            max_batch_size=int(d.get("maxBatchSize", 0)),
            min_num_replicas=int(d.get("minNumReplicas", 0)),
            current_accelerator=d.get("currentAccelerator", ""),
            current_num_replicas=int(d.get("currentNumReplicas", 0)),
        )


@dataclass
class CapacitySpec:
    """Available accelerator units per type (wire shape
    ``{"count": {"MI355X": 16}}``; a bare type→count map is accepted)."""

    counts: Dict[str, int] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, d: Any) -> "CapacitySpec":
        if not isinstance(d, dict):
            return cls()
        src = d.get("count", d)
        if not isinstance(src, dict):
            return cls()
        counts: Dict[str, int] = {}
        for k, v in src.items():
            try:
                counts[str(k)] = int(v)
            except (TypeError, ValueError):
                continue
        return cls(counts=counts)


@dataclass
class OptimizerSpec:
    unlimited: bool = False
    delayed_best_effort: bool = False
    saturation_policy: str = POLICY_NONE
    use_cplex: bool = False  # parity placeholder; no MIP path

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "OptimizerSpec":
        return cls(
            unlimited=bool(d.get("unlimited", False)),
            delayed_best_effort=bool(d.get("delayedBestEffort", False)),
            saturation_policy=d.get("saturationPolicy", POLICY_NONE),
        )


@dataclass
class SystemData:
    accelerators: List[AcceleratorSpec] = field(default_factory=list)
    models: List[ModelAcceleratorPerfData] = field(default_factory=list)
    service_classes: List[ServiceClassSpec] = field(default_factory=list)
    servers: List[ServerSpec] = field(default_factory=list)
    capacity: Dict[str, int] = field(default_factory=dict)
    optimizer: OptimizerSpec = field(default_factory=OptimizerSpec)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "SystemData":
        spec = d.get("spec", d)
        return cls(
            accelerators=[
                AcceleratorSpec.from_dict(a)
                for a in (spec.get("accelerators") or {}).get("data", [])
            ] if isinstance(spec.get("accelerators"), dict) else [
                AcceleratorSpec.from_dict(a) for a in spec.get("accelerators", [])
            ],
            models=[
                ModelAcceleratorPerfData.from_dict(m)
                for m in (spec.get("models") or {}).get("data", [])
            ] if isinstance(spec.get("models"), dict) else [
                ModelAcceleratorPerfData.from_dict(m) for m in spec.get("models", [])
            ],
            service_classes=[
                ServiceClassSpec.from_dict(s)
                for s in (spec.get("serviceClasses") or {}).get("data", [])
            ] if isinstance(spec.get("serviceClasses"), dict) else [
                ServiceClassSpec.from_dict(s) for s in spec.get("serviceClasses", [])
            ],
            servers=[
                ServerSpec.from_dict(s)
                for s in (spec.get("servers") or {}).get("data", [])
            ] if isinstance(spec.get("servers"), dict) else [
                ServerSpec.from_dict(s) for s in spec.get("servers", [])
            ],
            capacity=CapacitySpec.from_dict(spec.get("capacity") or {}).counts,
            optimizer=OptimizerSpec.from_dict(spec.get("optimizer") or {}),
        )


def parse_service_class_configmap(data: Dict[str, str]):
    """Parse the reference's service-classes ConfigMap data section: each
    key is a YAML doc {name, priority, data: [{model, slo-tpot, slo-ttft}]}
    (charts/.../wva-configmap-service-class.yaml)."""
    import yaml

    out = []
    for key in sorted(data or {}):
        try:
            parsed = yaml.safe_load(data[key]) or {}
        except Exception:  # noqa: BLE001
            continue
        if not isinstance(parsed, dict) or "name" not in parsed:
            continue
        out.append(ServiceClassSpec.from_dict(parsed))
    return out


def parse_accelerator_configmap(data: Dict[str, str]) -> List[AcceleratorSpec]:
    """Parse the accelerator ConfigMap (the reference's acceleratorCm in
    utils.go:142-156): each data key is an accelerator NAME, each value a
    YAML doc {device, cost, [multiplicity], [memSize], [memBW]}; cost may
    be a quoted decimal string exactly as the reference parses it.

      MI355X: |
        device: AMD-Instinct-MI355X-288GB
        cost: "50.0"
        multiplicity: 1
        memSize: 288
    """
    import yaml

    out: List[AcceleratorSpec] = []
    for key in sorted(data or {}):
        try:
            parsed = yaml.safe_load(data[key]) or {}
        except Exception:  # noqa: BLE001
            continue
        if not isinstance(parsed, dict):
            continue
        try:
            cost = float(parsed.get("cost", 0.0))
        except (TypeError, ValueError):
            # reference: unparseable cost → skip accelerator (utils.go:144-148)
            continue
        out.append(AcceleratorSpec(
            name=key,
            type=str(parsed.get("device", key)),
            multiplicity=int(parsed.get("multiplicity", 1)),
            mem_size=int(parsed.get("memSize", 0)),
            mem_bw=int(parsed.get("memBW", 0)),
            cost=cost,
        ))
    return out


def parse_model_perf_configmap(data: Dict[str, str]) -> List[ModelAcceleratorPerfData]:
    """Parse the model-perf ConfigMap: each data key holds one YAML doc
    (or a list of docs) in the SystemData perf schema — the on-cluster
    home for this repo's measured calibration records
    (profiles/calibration_tp{N}.json → alpha/beta per (model, acc,
    accCount)):

      llama-8b-tp1.yaml: |
        name: meta-llama/Llama-3.1-8B
        acc: MI355X
        accCount: 1
        maxBatchSize: 256
        atTokens: 50
        decodeParms: {alpha: 4.9, beta: 0.025}
    """
    import yaml

    out: List[ModelAcceleratorPerfData] = []
    for key in sorted(data or {}):
        try:
            parsed = yaml.safe_load(data[key])
        except Exception:  # noqa: BLE001
            continue
        docs = parsed if isinstance(parsed, list) else [parsed]
        for doc in docs:
            if not isinstance(doc, dict) or not doc.get("name"):
                continue
            out.append(ModelAcceleratorPerfData.from_dict(doc))
    return out
