"""Capacity-store ConfigMap persistence (checkpoint/resume improvement;
SURVEY §5: the reference loses all live-learned capacity records on
restart and degrades until re-learned — here they round-trip through
the `wva-capacity-store` ConfigMap with ages preserved).
"""
import json
import time

import pytest

from wva_amd.analyzers.capacity_store import (
    CAPACITY_STORE_CONFIG_MAP_NAME,
    CapacityKnowledgeStore,
    CapacityRecord,
    CapacityStorePersistence,
    restore_store,
    snapshot_store,
)
from wva_amd.analyzers.deployment_parser import VLLMEngineParams
from wva_amd.kube.fake import FakeCluster

NS = "wva-system"
MODEL = "meta-llama/Llama-3.1-8B"


def live_record(tokens=2_000_704, effective=900_000):
    return CapacityRecord(
        accelerator_name="MI355X",
        gpu_count=1,
        num_gpu_blocks=tokens // 16,
        block_size=16,
        total_kv_capacity_tokens=tokens,
        effective_capacity=effective,
        vllm_params=VLLMEngineParams(max_num_seqs=256, kv_cache_dtype="fp8"),
        learned_from="live",
    )


class TestSnapshotRestore:
    def test_roundtrip_preserves_fields_and_age(self):
        store = CapacityKnowledgeStore()
        store.update(NS, MODEL, "v0", live_record())
        rec = store.get(NS, MODEL, "v0")
        rec_age_before = time.monotonic() - rec.learned_at

        snap = snapshot_store(store)
        fresh = CapacityKnowledgeStore()
        assert restore_store(fresh, snap) == 1
        got = fresh.get(NS, MODEL, "v0")
        assert got.total_kv_capacity_tokens == 2_000_704
        assert got.effective_capacity == 900_000
        assert got.learned_from == "live"
        assert got.vllm_params.kv_cache_dtype == "fp8"
        age_after = time.monotonic() - got.learned_at
        assert age_after == pytest.approx(rec_age_before, abs=1.0)

    def test_restore_never_overwrites_local_learning(self):
        store = CapacityKnowledgeStore()
        store.update(NS, MODEL, "v0", live_record(effective=111))
        snap_other = {"records": {
            f"{NS}|{MODEL}|v0": {
                "accelerator_name": "MI355X", "gpu_count": 1,
                "total_kv_capacity_tokens": 5, "effective_capacity": 5,
                "learned_from": "live", "age_seconds": 0,
            },
        }}
        assert restore_store(store, snap_other) == 0
        assert store.get(NS, MODEL, "v0").effective_capacity == 111

    def test_snapshot_json_serializable(self):
        store = CapacityKnowledgeStore()
        store.update(NS, MODEL, "v0", live_record())
        json.dumps(snapshot_store(store))  # must not raise

    def test_restore_tolerates_garbage(self):
        store = CapacityKnowledgeStore()
        assert restore_store(store, {"records": {
            "bad": {"gpu_count": "not-an-int"},
        }}) == 0
        assert restore_store(store, {}) == 0
        assert restore_store(store, None) == 0


class TestPersistenceOverCluster:
    def test_persist_and_restore_cycle(self):
        cluster = FakeCluster()
        store = CapacityKnowledgeStore()
        store.update(NS, MODEL, "v0", live_record())
        p = CapacityStorePersistence(cluster, store, NS,
                                     write_interval_seconds=0.0)
        assert p.maybe_persist() is True
        cm = cluster.get("ConfigMap", NS, CAPACITY_STORE_CONFIG_MAP_NAME)
        assert f"{NS}|{MODEL}|v0" in cm.data["records"]

        # "restart": a new process's empty store restores from the CM
        store2 = CapacityKnowledgeStore()
        p2 = CapacityStorePersistence(cluster, store2, NS)
        assert p2.restore() == 1
        assert store2.get(NS, MODEL, "v0").total_kv_capacity_tokens \
            == 2_000_704

    def test_unchanged_snapshot_not_rewritten(self):
        cluster = FakeCluster()
        store = CapacityKnowledgeStore()
        store.update(NS, MODEL, "v0", live_record())
        p = CapacityStorePersistence(cluster, store, NS,
                                     write_interval_seconds=0.0)
        assert p.maybe_persist() is True
        rv1 = cluster.get(
            "ConfigMap", NS, CAPACITY_STORE_CONFIG_MAP_NAME
        ).metadata.resource_version
        assert p.maybe_persist() is False  # no change → no write
        rv2 = cluster.get(
            "ConfigMap", NS, CAPACITY_STORE_CONFIG_MAP_NAME
        ).metadata.resource_version
        assert rv1 == rv2

    def test_write_interval_respected(self):
        cluster = FakeCluster()
        store = CapacityKnowledgeStore()
        store.update(NS, MODEL, "v0", live_record())
        p = CapacityStorePersistence(cluster, store, NS,
                                     write_interval_seconds=3600.0)
        assert p.maybe_persist() is True  # first write immediate
        store.update(NS, MODEL, "v1", live_record(tokens=4_001_424))
        assert p.maybe_persist() is False  # within interval

    def test_restart_resume_over_rest(self):
        """Full restart cycle through the REST path: engine learns →
        persists → a NEW app (fresh store) against the same API server
        starts with the records already present."""
        from wva_amd.kube.rest import RestCluster
        from k8s_test_server import K8sTestServer

        backing = FakeCluster()
        server = K8sTestServer(backing).start()
        try:
            rest = RestCluster(server.url)
            store = CapacityKnowledgeStore()
            store.update(NS, MODEL, "v0", live_record())
            p = CapacityStorePersistence(rest, store, NS,
                                         write_interval_seconds=0.0)
            assert p.maybe_persist() is True

            rest2 = RestCluster(server.url)
            store2 = CapacityKnowledgeStore()
            p2 = CapacityStorePersistence(rest2, store2, NS)
            assert p2.restore() == 1
            rec = store2.get(NS, MODEL, "v0")
            assert rec.learned_from == "live"
            assert rec.vllm_params.max_num_seqs == 256
            rest.close()
            rest2.close()
        finally:
            server.stop()


class TestEngineIntegration:
    def test_engine_tick_persists(self, monkeypatch):
        """build_app wires persistence; an engine tick after learning
        writes the ConfigMap (controller namespace)."""
        import sys
        sys.path.insert(0, "tests")
        monkeypatch.setenv("POD_NAMESPACE", NS)
        from test_e2e_emulated import (
            MODEL as EMODEL, NS as ENS, VARIANT, make_stack, run_sim,
        )
        from wva_amd.emulator.vllm_sim import ServiceProfile

        prof = ServiceProfile(
            alpha_ms=50.0, beta_ms=2.0, max_num_seqs=8, num_gpu_blocks=500
        )
        cluster, sim, app = make_stack(
            replicas=1, profile=prof, analyzer="saturation"
        )
        app.saturation_engine.capacity_persistence.write_interval_seconds = 0.0
        model = sim.model(EMODEL, ENS)
        run_sim(sim, model, qps=10, seconds=10)
        app.saturation_engine.optimize()
        cm = cluster.try_get(
            "ConfigMap", NS, CAPACITY_STORE_CONFIG_MAP_NAME
        )
        assert cm is not None
        assert f"{ENS}|{EMODEL}|{VARIANT}" in cm.data["records"]


class TestK2HistoryPersistence:
    def test_history_rides_with_capacity_records(self):
        from wva_amd.analyzers.saturation_v2 import SaturationAnalyzerV2
        from wva_amd.kube.fake import FakeCluster

        cluster = FakeCluster()
        store = CapacityKnowledgeStore()
        an = SaturationAnalyzerV2(store)
        # seed a learned k2 history entry
        an._compute_capacity_history = {}
        from wva_amd.analyzers.history import RollingAverage

        ra = RollingAverage(10)
        for v in (30_000.0, 31_000.0, 32_000.0):
            ra.add(v)
        an._compute_capacity_history["m|MI355X|le500"] = ra
        store.update(NS, MODEL, "v0", live_record())

        p = CapacityStorePersistence(cluster, store, NS,
                                     write_interval_seconds=0.0,
                                     analyzer=an)
        assert p.maybe_persist() is True

        # restart: a fresh analyzer restores the history with its ages
        store2 = CapacityKnowledgeStore()
        an2 = SaturationAnalyzerV2(store2)
        p2 = CapacityStorePersistence(cluster, store2, NS, analyzer=an2)
        restored = p2.restore()
        assert restored >= 2  # 1 record + 1 history key
        ra2 = an2._compute_capacity_history["m|MI355X|le500"]
        assert ra2.average() == pytest.approx(31_000.0)

    def test_local_history_not_overwritten(self):
        from wva_amd.analyzers.saturation_v2 import SaturationAnalyzerV2

        an = SaturationAnalyzerV2(CapacityKnowledgeStore())
        from wva_amd.analyzers.history import RollingAverage

        ra = RollingAverage(10)
        ra.add(99.0)
        an._compute_capacity_history["k"] = ra
        assert an.history_restore({"k": {"values": [1.0], "age_seconds": 0}}) == 0
        assert an._compute_capacity_history["k"].average() == 99.0
