"""Inferno core: System registry + Allocation sizing.

Parity: reference pkg/core/{system,accelerator,model,server,serviceclass,
allocation}.go. Differences by design: the reference uses a package-global
singleton registry (system.go:10-60); here System is an explicit object.

Allocation sizing (allocation.go:27-155): per (server, accelerator) build a
state-dependent queue analyzer from ServiceParms, Size() to the SLO targets
→ rate* per replica, replicas = ceil(totalRate / rate*), cost =
accCost × instancesPerReplica × replicas. Zero-load path allocates
minNumReplicas at base cost. Transition penalty multiplies value by
(1 + AcceleratorSwitchFactor) when the accelerator changes.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict

from .queue_analyzer import (
    Configuration,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from .types import (
    ACCELERATOR_SWITCH_FACTOR,
    MAX_QUEUE_TO_BATCH_RATIO,
    AcceleratorSpec,
    ModelAcceleratorPerfData,
    ServerSpec,
    ServiceClassSpec,
    SystemData,
)


@dataclass
class Allocation:
    accelerator: str = ""
    num_replicas: int = 0
    batch_size: int = 0
    cost: float = 0.0
    value: float = 0.0
    itl: float = 0.0  # expected average token decode time (msec)
    ttft: float = 0.0  # expected queueing + prefill (msec)
    rho: float = 0.0
    max_arrv_rate_per_replica: float = 0.0  # req/msec

    def deepcopy(self) -> "Allocation":
        return Allocation(**self.__dict__)


@dataclass
class AllocationDiff:
    accelerator_from: str = ""
    accelerator_to: str = ""
    replicas_from: int = 0
    replicas_to: int = 0

    @property
    def is_change(self) -> bool:
        return (
            self.accelerator_from != self.accelerator_to
            or self.replicas_from != self.replicas_to
        )


class Server:
    def __init__(self, spec: ServerSpec, priority: int):
        self.spec = spec
        self.priority = priority
        self.all_allocations: Dict[str, Allocation] = {}
        self.allocation: Optional[Allocation] = None

    @property
    def name(self) -> str:
        return self.spec.name

    def set_allocation(self, alloc: Optional[Allocation]) -> None:
        self.allocation = alloc

    def remove_allocation(self) -> None:
        self.allocation = None


class System:
    """Registry of accelerators, model perf data, service classes, servers
    and capacity; generates candidate allocations and applies solutions."""

    def __init__(self, data: Optional[SystemData] = None):
        self.accelerators: Dict[str, AcceleratorSpec] = {}
        # (model name, accelerator) → perf data
        self.perf: Dict[tuple, ModelAcceleratorPerfData] = {}
        self.service_classes: Dict[str, ServiceClassSpec] = {}
        self.servers: Dict[str, Server] = {}
        self.capacity: Dict[str, int] = {}
        if data is not None:
            self.load(data)

    # --- registration ---

    def load(self, data: SystemData) -> None:
        for acc in data.accelerators:
            self.accelerators[acc.name] = acc
        for perf in data.models:
            self.perf[(perf.name, perf.acc)] = perf
        for sc in data.service_classes:
            self.service_classes[sc.name] = sc
        for srv in data.servers:
            self.add_server(srv)
        self.capacity = dict(data.capacity)

    def add_server(self, spec: ServerSpec) -> Server:
        sc = self.service_classes.get(spec.service_class)
        priority = sc.priority if sc is not None else 0
        server = Server(spec, priority)
        self.servers[spec.name] = server
        return server

    def set_capacity(self, counts: Dict[str, int]) -> None:
        self.capacity = dict(counts)

    # --- allocation generation (allocation.go:27-155) ---

    def _target_for(self, server: Server):
        sc = self.service_classes.get(server.spec.service_class)
        if sc is None:
            return None
        for t in sc.model_targets:
            if t.model == server.spec.model:
                return t
        return None

    def create_allocation(
        self, server_name: str, acc_name: str
    ) -> Optional[Allocation]:
        server = self.servers.get(server_name)
        acc = self.accelerators.get(acc_name)
        if server is None or acc is None:
            return None
        load = server.spec.load
        if load.arrival_rate < 0 or load.avg_in_tokens < 0 or load.avg_out_tokens < 0:
            return None
        perf = self.perf.get((server.spec.model, acc_name))
        if perf is None:
            return None
        target = self._target_for(server)
        if target is None:
            return None

        # zero-traffic path (allocation.go zeroLoadAllocation)
        if load.arrival_rate == 0 or load.avg_out_tokens == 0:
            replicas = max(server.spec.min_num_replicas, 1)
            cost = acc.cost * perf.acc_count * replicas
            alloc = Allocation(
                accelerator=acc_name,
                num_replicas=replicas,
                batch_size=perf.max_batch_size,
                cost=cost,
            )
            alloc.value = cost
            return alloc

        K = load.avg_out_tokens
        if server.spec.max_batch_size > 0:
            N = server.spec.max_batch_size
        else:
            N = max(perf.max_batch_size * perf.at_tokens // max(K, 1), 1)
        max_queue = N * MAX_QUEUE_TO_BATCH_RATIO

        try:
            analyzer = QueueAnalyzer(
                Configuration(
                    max_batch_size=N,
                    max_queue_size=max_queue,
                    service_parms=ServiceParms(
                        alpha=perf.service_parms.alpha,
                        beta=perf.service_parms.beta,
                        gamma=perf.service_parms.gamma,
                    ),
                ),
                RequestSize(
                    avg_input_tokens=float(load.avg_in_tokens),
                    avg_output_tokens=float(K),
                ),
            )
            _, metrics, _ = analyzer.size(TargetPerf(
                target_ttft=target.slo_ttft,
                target_itl=target.slo_itl,
                target_tps=target.slo_tps,
            ))
        except (ValueError, ZeroDivisionError):
            return None

        rate_star = metrics.throughput  # requests/sec per replica
        if target.slo_tps == 0:
            total_rate = load.arrival_rate / 60.0  # req/min → req/sec
        else:
            total_rate = target.slo_tps / K
        num_replicas = max(
            int(math.ceil(total_rate / rate_star)), server.spec.min_num_replicas, 1
        )

        cost = acc.cost * perf.acc_count * num_replicas

        try:
            metrics = analyzer.analyze(total_rate / num_replicas)
        except ValueError:
            return None

        alloc = Allocation(
            accelerator=acc_name,
            num_replicas=num_replicas,
            batch_size=N,
            cost=cost,
            itl=metrics.avg_token_time,
            ttft=metrics.avg_wait_time + metrics.avg_prefill_time,
            rho=metrics.rho,
            max_arrv_rate_per_replica=rate_star / 1000.0,
        )
        alloc.value = cost
        return alloc

    def generate_all_allocations(self) -> None:
        """For every server, candidate allocations on all accelerators with
        perf data, value = cost × transition penalty when the accelerator
        differs from the current one (allocation.go TransitionPenalty)."""
        for server in self.servers.values():
            server.all_allocations = {}
            for acc_name in self.accelerators:
                alloc = self.create_allocation(server.name, acc_name)
                if alloc is None:
                    continue
                if (
                    server.spec.current_accelerator
                    and server.spec.current_accelerator != acc_name
                ):
                    alloc.value = alloc.cost * (1.0 + ACCELERATOR_SWITCH_FACTOR)
                server.all_allocations[acc_name] = alloc

    def units_per_replica(self, model_name: str, acc_name: str) -> int:
        perf = self.perf.get((model_name, acc_name))
        acc = self.accelerators.get(acc_name)
        if perf is None or acc is None:
            return 0
        return perf.acc_count * acc.multiplicity

    def total_cost(self) -> float:
        return sum(
            s.allocation.cost for s in self.servers.values() if s.allocation
        )
