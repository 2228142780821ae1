#!/usr/bin/env python3
"""Control-plane hot-path micro-benchmark (CPU).

Measures the saturation-engine tick latency — the reference's hot loop
(collector → analyzer → optimizer → actuator) — at cluster
scale: N models × V variants × R replicas each.

    python scripts/bench_controlplane.py --models 20 --variants 2 --replicas 10
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from prometheus_client import CollectorRegistry  # noqa: E402

from wva_amd.api.types import (  # noqa: E402
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.app import build_app  # noqa: E402
from wva_amd.config.config import Config  # noqa: E402
from wva_amd.config.saturation import SaturationScalingConfig  # noqa: E402
from wva_amd.emulator.cluster_sim import ClusterSim  # noqa: E402
from wva_amd.emulator.sim_source import SimMetricsSource  # noqa: E402
from wva_amd.emulator.vllm_sim import ServiceProfile  # noqa: E402
from wva_amd.emulator.workload import constant_qps  # noqa: E402
from wva_amd.kube.fake import FakeCluster  # noqa: E402
from wva_amd.kube.objects import (  # noqa: E402
    Container,
    Deployment,
    Node,
    PodTemplateSpec,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--models", type=int, default=20)
    p.add_argument("--variants", type=int, default=2)
    p.add_argument("--replicas", type=int, default=10)
    p.add_argument("--ticks", type=int, default=10)
    p.add_argument("--analyzer", default="saturation")
    p.add_argument("--backend", default="fake",
                   choices=["fake", "rest", "cached"],
                   help="cluster client the CONTROLLER uses: in-memory, "
                        "uncached REST, or REST behind the informer cache")
    args = p.parse_args()

    cluster = FakeCluster()
    cluster.create(Node(
        metadata=ObjectMeta(
            name="mi355x-0",
            labels={"amd.com/gpu.product": "AMD-Instinct-MI355X-288GB",
                    "amd.com/gpu.memory": "294912"},
        ),
        allocatable={"amd.com/gpu": "4096"},
    ))
    sim = ClusterSim(cluster, pod_ready_delay_s=0.0, warm_start=True)
    profile = ServiceProfile(max_num_seqs=97)

    for m in range(args.models):
        model_id = f"model-{m}"
        for v in range(args.variants):
            name = f"m{m}-v{v}"
            cluster.create(Deployment(
                metadata=ObjectMeta(name=name, namespace="default"),
                replicas=args.replicas,
                selector={"app": name},
                template=PodTemplateSpec(
                    labels={"app": name},
                    containers=[Container(requests={"amd.com/gpu": "1"})],
                ),
            ))
            cluster.create(VariantAutoscaling(
                metadata=ObjectMeta(
                    name=name, namespace="default",
                    labels={"inference.optimization/acceleratorName": "MI355X"},
                ),
                spec=VariantAutoscalingSpec(
                    scale_target_ref=CrossVersionObjectReference(name=name),
                    model_id=model_id,
                ),
            ))
            sim.register_variant(model_id, "default", name, profile)
    sim.reconcile_deployments()

    config = Config()
    config.update_saturation_config(SaturationScalingConfig.from_dict(
        {"analyzerName": args.analyzer} if args.analyzer else {}
    ))
    config.mark_bootstrap_complete()

    app_cluster = cluster
    server = cache = None
    if args.backend in ("rest", "cached"):
        import sys as _sys, os as _os
        _sys.path.insert(0, _os.path.join(
            _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))),
            "tests"))
        from k8s_test_server import K8sTestServer
        from wva_amd.kube.rest import RestCluster

        server = K8sTestServer(cluster).start()
        app_cluster = RestCluster(server.url)
        if args.backend == "cached":
            from wva_amd.kube.cache import CachedCluster

            cache = CachedCluster(app_cluster).start()
            assert cache.wait_for_sync(30)
            assert cache.wait_caught_up(30)
            app_cluster = cache

    app = build_app(app_cluster, config, source=SimMetricsSource(sim),
                    metrics_registry=CollectorRegistry(), start_engines=False)

    # brief traffic so metrics exist
    qps = constant_qps(50)
    for model in sim.models.values():
        for _ in range(8):
            sim.generate_arrivals(model, qps, 0.25)
    sim.advance(2.0)

    app.saturation_engine.optimize()  # warm
    times = []
    for _ in range(args.ticks):
        t0 = time.perf_counter()
        app.saturation_engine.optimize()
        times.append((time.perf_counter() - t0) * 1000.0)
    times.sort()
    pods = args.models * args.variants * args.replicas
    vas = args.models * args.variants
    line = (
        f"engine tick over {args.models} models / {vas} VAs / {pods} pods "
        f"({args.analyzer or 'v1'}, backend={args.backend}): "
        f"median {times[len(times)//2]:.1f} ms, "
        f"p90 {times[int(len(times)*0.9)]:.1f} ms"
    )
    if server is not None:
        gets = server.request_counts.get("GET", 0)
        line += f", api-server GETs total {gets}"
    print(line)
    if cache is not None:
        cache.stop()
    if server is not None:
        server.stop()


if __name__ == "__main__":
    main()
