"""fp8 KV capacity end-to-end through the control plane:
`kv-cache-dtype=fp8` parsed from deployment args → measured-profile
registry (profiles/calibration_8b{,_fp8}.json — real MI355X numbers:
2.0M bf16 vs 4.0M fp8 tokens) → capacity store → V2 analyzer, so the
calibrated fp8 capacity actually changes autoscaling decisions.
"""
import os

import pytest

from wva_amd.analyzers.capacity_store import (
    CapacityKnowledgeStore,
    MeasuredProfile,
    load_calibration_dir,
    normalize_kv_dtype,
)
from wva_amd.api.types import ObjectMeta
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PROFILES = os.path.join(REPO, "profiles")
MODEL = "meta-llama/Llama-3.1-8B"
NS = "default"


def deploy_with_args(name, args):
    return Deployment(
        metadata=ObjectMeta(name=name, namespace=NS),
        replicas=0,
        selector={"app": name},
        template=PodTemplateSpec(
            labels={"app": name},
            containers=[Container(args=args, requests={"amd.com/gpu": "1"})],
        ),
    )


class TestKvDtypeNormalization:
    def test_families(self):
        assert normalize_kv_dtype("fp8") == "fp8"
        assert normalize_kv_dtype("fp8_e4m3") == "fp8"
        assert normalize_kv_dtype("fp8_e5m2") == "fp8"
        assert normalize_kv_dtype("auto") == "bf16"
        assert normalize_kv_dtype("bfloat16") == "bf16"
        assert normalize_kv_dtype("") == "bf16"


class TestMeasuredProfileRegistry:
    def test_load_committed_calibration_profiles(self):
        """The committed MI355X measurements load as-is; fp8 carries the
        measured 2x capacity (4,001,424 vs 2,000,704 tokens)."""
        store = CapacityKnowledgeStore()
        n = load_calibration_dir(store, PROFILES)
        assert n >= 3  # 8b bf16, 8b fp8, 70b
        bf16 = store.measured_profile(MODEL, "MI355X", 1, "auto")
        fp8 = store.measured_profile(MODEL, "MI355X", 1, "fp8_e4m3")
        assert bf16 is not None and fp8 is not None
        assert bf16.total_kv_capacity_tokens == 2_000_704
        assert fp8.total_kv_capacity_tokens == 4_001_424
        assert fp8.total_kv_capacity_tokens > 1.9 * bf16.total_kv_capacity_tokens

    def test_deployment_fallback_uses_measured_k1(self):
        store = CapacityKnowledgeStore()
        load_calibration_dir(store, PROFILES)
        store.load_from_deployment(
            NS, MODEL, "v-fp8", "MI355X", 1,
            deploy_with_args("v-fp8", [
                "--kv-cache-dtype", "fp8", "--max-num-seqs", "256",
            ]),
        )
        store.load_from_deployment(
            NS, MODEL, "v-bf16", "MI355X", 1,
            deploy_with_args("v-bf16", ["--max-num-seqs", "256"]),
        )
        rec_fp8 = store.get(NS, MODEL, "v-fp8")
        rec_bf16 = store.get(NS, MODEL, "v-bf16")
        assert rec_fp8.learned_from == "measured-profile"
        assert rec_fp8.total_kv_capacity_tokens == 4_001_424
        assert rec_bf16.total_kv_capacity_tokens == 2_000_704

    def test_explicit_blocks_override_wins(self):
        store = CapacityKnowledgeStore()
        load_calibration_dir(store, PROFILES)
        store.load_from_deployment(
            NS, MODEL, "v", "MI355X", 1,
            deploy_with_args("v", [
                "--kv-cache-dtype", "fp8",
                "--num-gpu-blocks-override", "1000", "--block-size", "16",
            ]),
        )
        rec = store.get(NS, MODEL, "v")
        assert rec.total_kv_capacity_tokens == 16_000
        assert rec.learned_from == "deployment"

    def test_live_never_overwritten(self):
        from wva_amd.analyzers.capacity_store import CapacityRecord

        store = CapacityKnowledgeStore()
        load_calibration_dir(store, PROFILES)
        store.update(NS, MODEL, "v", CapacityRecord(
            accelerator_name="MI355X", gpu_count=1,
            total_kv_capacity_tokens=123, learned_from="live",
        ))
        store.load_from_deployment(
            NS, MODEL, "v", "MI355X", 1,
            deploy_with_args("v", ["--kv-cache-dtype", "fp8"]),
        )
        assert store.get(NS, MODEL, "v").total_kv_capacity_tokens == 123


class TestFp8ChangesDecisions:
    def _zero_replica_estimate(self, kv_args):
        """Per-replica capacity the V2 analyzer would grant a
        ZERO-replica variant of this deployment shape (the path where
        only the store estimate exists — analyzer.go:418-437)."""
        from wva_amd.analyzers.saturation_v2 import SaturationAnalyzerV2

        store = CapacityKnowledgeStore()
        load_calibration_dir(store, PROFILES)
        analyzer = SaturationAnalyzerV2(store)
        store.load_from_deployment(
            NS, MODEL, "v", "MI355X", 1, deploy_with_args("v", kv_args)
        )
        rec = store.get(NS, MODEL, "v")
        return analyzer._estimate_stored_capacity(
            rec, MODEL, kv_cache_threshold=0.8,
            model_avg_input=6000.0, model_avg_output=500.0,
        )

    def test_fp8_doubles_long_context_estimate(self):
        """Long-context workload (k1-bound): the fp8 variant's estimated
        per-replica capacity is ~2x the bf16 one — HALF the replicas for
        the same token demand. This is the measured fp8 story changing a
        real autoscaling decision."""
        bf16 = self._zero_replica_estimate(["--max-num-seqs", "4096",
                                            "--max-model-len", "131072"])
        fp8 = self._zero_replica_estimate([
            "--kv-cache-dtype", "fp8", "--max-num-seqs", "4096",
            "--max-model-len", "131072",
        ])
        assert bf16 > 0 and fp8 > 0
        assert fp8 == pytest.approx(2 * bf16, rel=0.05), (bf16, fp8)

    def test_optimizer_prefers_fewer_fp8_replicas(self):
        """CostAwareOptimizer sizing: same demand, fp8 per-replica
        capacity 2x → ceil() asks for about half the replicas."""
        import math

        bf16 = self._zero_replica_estimate(["--max-num-seqs", "4096",
                                            "--max-model-len", "131072"])
        fp8 = self._zero_replica_estimate([
            "--kv-cache-dtype", "fp8", "--max-num-seqs", "4096",
            "--max-model-len", "131072",
        ])
        demand = 6 * bf16  # needs 6 bf16 replicas at threshold
        assert math.ceil(demand / bf16) == 6
        assert math.ceil(demand / fp8) == 3


class TestBuildAppSeedsProfiles:
    def test_build_app_loads_profile_dir(self):
        from prometheus_client import CollectorRegistry

        from wva_amd.app import build_app
        from wva_amd.config.config import Config
        from wva_amd.kube.fake import FakeCluster

        config = Config()
        config.mark_bootstrap_complete()
        app = build_app(
            FakeCluster(), config,
            source=_NullSource(), metrics_registry=CollectorRegistry(),
            start_engines=False, measured_profiles_dir=PROFILES,
        )
        assert app.capacity_store.measured_profile(
            MODEL, "MI355X", 1, "fp8"
        ) is not None


class _NullSource:
    def name(self):
        return "prometheus"

    def query_list(self):
        from wva_amd.collector.query_template import QueryList

        return QueryList()

    def refresh(self, spec):
        return {}

    def get(self, query, params):
        return None
