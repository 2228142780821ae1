"""Process entry point — `python -m wva_amd`.

Parity: reference cmd/main.go flag surface (metrics/probe addresses,
leader election timings, watch namespace, verbosity) + env/config-file
loading via the same keys (PROMETHEUS_BASE_URL, GLOBAL_OPT_INTERVAL,
WVA_SCALE_TO_ZERO, ...). In this environment there is no Kubernetes API
server; `--emulated` starts the full controller stack against an in-memory
cluster seeded from a scenario file, which is the supported mode (the
FakeCluster implements the client surface a REST-backed client would).
"""
from __future__ import annotations

import argparse
import os
import signal
import sys
import time

from .app import build_app
from .config.loader import load_config
from .emulator.cluster_sim import ClusterSim
from .emulator.sim_source import SimMetricsSource
from .kube.fake import FakeCluster
from .utils.logging import get_logger, setup_logging

log = get_logger("main")


def parse_flags(argv=None):
    p = argparse.ArgumentParser(prog="wva-amd")
    p.add_argument("--metrics-bind-address", default=None)
    p.add_argument("--health-probe-bind-address", default=None)
    p.add_argument("--leader-elect", action="store_true", default=None)
    p.add_argument("--watch-namespace", default=None)
    p.add_argument("--config", default=None, help="YAML config file path")
    p.add_argument("--v", type=int, default=None, help="log verbosity (0-5)")
    p.add_argument(
        "--emulated",
        action="store_true",
        help="run against the in-memory emulated cluster",
    )
    p.add_argument(
        "--kube-api-url",
        default=None,
        help="Kubernetes API server URL (REST mode); in-cluster config is "
             "auto-detected via KUBERNETES_SERVICE_HOST when omitted",
    )
    p.add_argument("--kube-token", default=None)
    p.add_argument("--kube-ca-cert", default=None)
    p.add_argument("--kube-insecure-skip-verify", action="store_true")
    return p.parse_args(argv)


def main(argv=None) -> int:
    # install signal handlers FIRST: a SIGTERM during startup (informer
    # sync, ConfigMap bootstrap) must still shut down cleanly
    stop = []
    signal.signal(signal.SIGINT, lambda *_: stop.append(1))
    signal.signal(signal.SIGTERM, lambda *_: stop.append(1))

    args = parse_flags(argv)
    flags = {
        "METRICS_BIND_ADDRESS": args.metrics_bind_address,
        "HEALTH_PROBE_BIND_ADDRESS": args.health_probe_bind_address,
        "LEADER_ELECT": args.leader_elect,
        "WATCH_NAMESPACE": args.watch_namespace,
        "V": args.v,
    }
    config = load_config(
        flags=flags,
        config_path=args.config,
        require_prometheus=not args.emulated,
    )
    setup_logging(config.infra.logger_verbosity or 2)

    source = None
    if args.emulated:
        cluster = FakeCluster()
        sim = ClusterSim(cluster)
        source = SimMetricsSource(sim)
        log.info("running in emulated mode (in-memory cluster)")
    elif args.kube_api_url or os.environ.get("KUBERNETES_SERVICE_HOST"):
        from .kube.cache import CachedCluster
        from .kube.rest import RestCluster

        if args.kube_api_url:
            rest = RestCluster(
                args.kube_api_url,
                token=args.kube_token,
                ca_cert_path=args.kube_ca_cert,
                insecure_skip_verify=args.kube_insecure_skip_verify,
            )
            log.info("REST mode against %s", args.kube_api_url)
        else:
            rest = RestCluster.in_cluster()
            log.info("in-cluster REST mode")
        # informer-style read cache in front of the API server (the
        # controller-runtime cache analog, cmd/main.go:289-297): engine
        # ticks read locally; writes pass through
        cluster = CachedCluster(rest).start()
        if not cluster.wait_for_sync(60.0):
            log.error("informer cache failed to sync within 60s")
            return 1
    else:
        cluster = FakeCluster()
        log.warning(
            "no --emulated, --kube-api-url or in-cluster environment: "
            "running against an empty in-memory cluster"
        )

    app = build_app(cluster, config, source=source, serve_http=True)
    app.configmap_reconciler.bootstrap_initial_configmaps()
    app.start()
    log.info("manager started (leader_elect=%s)", config.infra.enable_leader_election)

    try:
        while not stop:
            time.sleep(0.5)
        log.info("signal received, shutting down")
    finally:
        app.stop()
        cache_stop = getattr(cluster, "stop", None)
        if cache_stop is not None and hasattr(cluster, "wait_for_sync"):
            cache_stop()  # CachedCluster: stop pump + close REST watches
    return 0


if __name__ == "__main__":
    sys.exit(main())
