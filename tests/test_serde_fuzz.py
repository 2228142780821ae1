"""Property-based serde round-trips for every kind the REST path speaks.

The wire contract (kube/serde.py) is what separates the controller from
a real API server: hypothesis-generated objects must survive
encode→decode→encode byte-stably for every registered kind, including
awkward label values, unicode, and zero/empty fields — the fuzz layer
the reference gets for free from client-go's generated deepcopy/codecs.
"""
import string

import pytest
from hypothesis import given, settings, strategies as st

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.kube import serde
from wva_amd.kube.objects import (
    ConfigMap,
    Container,
    Deployment,
    EndpointPool,
    Node,
    Pod,
    PodStatus,
    PodTemplateSpec,
    Secret,
    Service,
    ServicePort,
)

NAME = st.text(
    alphabet=string.ascii_lowercase + string.digits + "-",
    min_size=1, max_size=40,
).filter(lambda s: s[0].isalnum() and s[-1].isalnum())
LABEL_KEY = st.text(
    alphabet=string.ascii_lowercase + string.digits + "-._/",
    min_size=1, max_size=40,
).filter(lambda s: s[0].isalnum() and s[-1].isalnum() and s.count("/") <= 1)
LABEL_VAL = st.text(
    alphabet=string.ascii_letters + string.digits + "-._",
    min_size=0, max_size=40,
)
LABELS = st.dictionaries(LABEL_KEY, LABEL_VAL, max_size=4)
FREE_TEXT = st.text(max_size=120)  # unicode payloads (data values)


def roundtrip(obj):
    kind = obj.kind
    enc1 = serde.encode(obj)
    dec = serde.decode(kind, enc1)
    enc2 = serde.encode(dec)
    assert enc1 == enc2, f"{kind} not byte-stable"
    return dec


class TestSerdeRoundtrips:
    @given(name=NAME, ns=NAME, labels=LABELS,
           data=st.dictionaries(LABEL_KEY, FREE_TEXT, max_size=4))
    @settings(max_examples=100, deadline=None)
    def test_configmap(self, name, ns, labels, data):
        dec = roundtrip(ConfigMap(
            metadata=ObjectMeta(name=name, namespace=ns, labels=labels),
            data=data,
        ))
        assert dec.data == data and dec.metadata.labels == labels

    @given(name=NAME, ns=NAME,
           data=st.dictionaries(LABEL_KEY, FREE_TEXT, max_size=4))
    @settings(max_examples=100, deadline=None)
    def test_secret_base64(self, name, ns, data):
        dec = roundtrip(Secret(
            metadata=ObjectMeta(name=name, namespace=ns), data=data,
        ))
        assert dec.data == data  # decoded values, wire is base64

    @given(name=NAME, ns=NAME, replicas=st.integers(0, 10_000),
           labels=LABELS,
           args=st.lists(st.text(
               alphabet=string.printable.strip(), min_size=0, max_size=30
           ), max_size=6),
           requests=st.dictionaries(
               st.sampled_from(["amd.com/gpu", "nvidia.com/gpu", "cpu"]),
               st.sampled_from(["1", "2", "8"]), max_size=2))
    @settings(max_examples=100, deadline=None)
    def test_deployment(self, name, ns, replicas, labels, args, requests):
        dec = roundtrip(Deployment(
            metadata=ObjectMeta(name=name, namespace=ns, labels=labels),
            replicas=replicas,
            selector=dict(labels),
            template=PodTemplateSpec(
                labels=dict(labels),
                containers=[Container(args=args, requests=requests)],
            ),
        ))
        assert dec.replicas == replicas
        assert dec.template.containers[0].args == args
        assert dec.template.containers[0].requests == requests

    @given(name=NAME, ns=NAME, phase=st.sampled_from(
        ["Pending", "Running", "Succeeded", "Failed"]),
        ready=st.booleans(), ip=st.sampled_from(["", "10.0.0.1", "::1"]))
    @settings(max_examples=100, deadline=None)
    def test_pod(self, name, ns, phase, ready, ip):
        dec = roundtrip(Pod(
            metadata=ObjectMeta(name=name, namespace=ns),
            status=PodStatus(phase=phase, ready=ready, pod_ip=ip),
        ))
        assert dec.status.ready == ready and dec.status.pod_ip == ip

    @given(name=NAME, labels=LABELS,
           alloc=st.dictionaries(
               st.sampled_from(["amd.com/gpu", "cpu", "memory"]),
               st.sampled_from(["0", "8", "64", "294912"]), max_size=3))
    @settings(max_examples=100, deadline=None)
    def test_node(self, name, labels, alloc):
        dec = roundtrip(Node(
            metadata=ObjectMeta(name=name, labels=labels),
            allocatable=alloc,
        ))
        assert dec.allocatable == alloc

    @given(name=NAME, ns=NAME, port=st.integers(1, 65535),
           port_name=st.sampled_from(["metrics", "http", "vllm", ""]))
    @settings(max_examples=100, deadline=None)
    def test_service(self, name, ns, port, port_name):
        dec = roundtrip(Service(
            metadata=ObjectMeta(name=name, namespace=ns),
            selector={"app": name},
            ports=[ServicePort(name=port_name, port=port)],
        ))
        assert dec.ports[0].port == port

    @given(name=NAME, ns=NAME, model=st.text(
        alphabet=string.ascii_letters + string.digits + "-./_",
        min_size=1, max_size=60),
        cost=st.sampled_from(["10.0", "50", "0.5", "999.99"]),
        replicas=st.integers(0, 4096), accel=st.sampled_from(
            ["MI355X", "MI300X", "H100", "xx"]))
    @settings(max_examples=150, deadline=None)
    def test_variantautoscaling_full(self, name, ns, model, cost,
                                     replicas, accel):
        va = VariantAutoscaling(
            metadata=ObjectMeta(name=name, namespace=ns),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name=name),
                model_id=model,
                variant_cost=cost,
            ),
        )
        va.status.desired_optimized_alloc.accelerator = accel
        va.status.desired_optimized_alloc.num_replicas = replicas
        dec = roundtrip(va)
        assert dec.spec.model_id == model
        assert dec.spec.variant_cost == cost
        assert dec.status.desired_optimized_alloc.num_replicas == replicas

    @given(name=NAME, ns=NAME)
    @settings(max_examples=50, deadline=None)
    def test_all_kinds_have_resource_paths(self, name, ns):
        for kind, (_, _, (prefix, plural, namespaced)) in serde.SERDE.items():
            path = serde.resource_path(kind, ns if namespaced else None, name)
            assert plural in path and path.startswith("/" + prefix.split("/")[0])


class TestRemainingKindRoundtrips:
    """InferencePool / Namespace / ServiceMonitor / Lease — the kinds the
    controller watches but the original fuzz suite didn't cover."""

    @settings(max_examples=100, deadline=None)
    @given(name=NAME, ns=NAME, selector=LABELS,
           port=st.integers(min_value=1, max_value=65535),
           epp=NAME)
    def test_inferencepool(self, name, ns, selector, port, epp):
        from wva_amd.kube.objects import InferencePool

        roundtrip(InferencePool(
            metadata=ObjectMeta(name=name, namespace=ns),
            selector=dict(selector), target_port=port,
            epp_service_name=epp,
        ))

    @settings(max_examples=100, deadline=None)
    @given(name=NAME, labels=LABELS,
           annotations=st.dictionaries(LABEL_KEY, FREE_TEXT, max_size=3))
    def test_namespace_with_exclusion_annotation(self, name, labels,
                                                 annotations):
        from wva_amd.kube.objects import Namespace

        meta = ObjectMeta(name=name, labels=dict(labels))
        meta.annotations = dict(annotations)
        dec = roundtrip(Namespace(metadata=meta))
        assert dec.metadata.annotations == dict(annotations)

    @settings(max_examples=100, deadline=None)
    @given(name=NAME, ns=NAME)
    def test_servicemonitor_if_registered(self, name, ns):
        try:
            from wva_amd.kube.objects import ServiceMonitor
        except ImportError:
            pytest.skip("no ServiceMonitor kind")
        roundtrip(ServiceMonitor(
            metadata=ObjectMeta(name=name, namespace=ns),
        ))

    @settings(max_examples=100, deadline=None)
    @given(name=NAME, ns=NAME, holder=NAME,
           duration=st.integers(min_value=1, max_value=3600))
    def test_lease(self, name, ns, holder, duration):
        try:
            from wva_amd.kube.objects import Lease
        except ImportError:
            pytest.skip("no Lease kind")
        lease = Lease(
            metadata=ObjectMeta(name=name, namespace=ns),
            holder_identity=holder,
            lease_duration_seconds=duration,
        )
        dec = roundtrip(lease)
        assert dec.holder_identity == holder


class TestMalformedWirePayloads:
    """A degraded API server / proxy can hand the REST client anything;
    decode must degrade to defaults, never crash inside a watch pump."""

    @pytest.mark.parametrize("payload", [
        {}, {"metadata": None}, {"spec": "nope"}, {"status": []},
        {"metadata": 7}, {"spec": ["x"]}, None,
    ])
    def test_every_kind_survives_garbage(self, payload):
        for kind in serde.SERDE:
            obj = serde.decode(kind, payload)
            assert obj.kind == kind
            # still round-trippable after the degradation
            serde.encode(obj)

    @settings(max_examples=150, deadline=None)
    @given(garbage=st.dictionaries(
        st.sampled_from(["metadata", "spec", "status", "data", "x"]),
        st.none() | st.integers() | st.text(max_size=5)
        | st.lists(st.integers(), max_size=2),
        max_size=4,
    ))
    def test_va_decode_fuzz(self, garbage):
        obj = serde.decode("VariantAutoscaling", garbage)
        assert obj.kind == "VariantAutoscaling"
