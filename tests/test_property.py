"""Property-based tests (hypothesis) for the parsing and queueing-math
surfaces where adversarial inputs are realistic: vLLM arg strings,
durations, PromQL escaping, queueing-model invariants.
"""
from hypothesis import given, settings, strategies as st

from wva_amd.analyzers.deployment_parser import (
    parse_vllm_args,
    split_shell_string,
)
from wva_amd.collector.query_template import escape_promql_value
from wva_amd.config.scale_to_zero import parse_go_duration
from wva_amd.inferno.queueing import MM1KModel, MM1StateDependentModel
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=200))
def test_split_shell_string_never_raises(s):
    tokens = split_shell_string(s)
    assert all(isinstance(t, str) for t in tokens)


@settings(max_examples=200, deadline=None)
@given(st.lists(st.text(max_size=40), max_size=12))
def test_parse_vllm_args_never_raises(args):
    d = Deployment(template=PodTemplateSpec(containers=[Container(args=args)]))
    p = parse_vllm_args(d)
    assert p.effective_max_batched_tokens > 0
    assert p.block_size > 0


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=100))
def test_promql_escape_roundtrip_safe(s):
    escaped = escape_promql_value(s)
    # escaped value must not contain an unescaped double quote
    i = 0
    while i < len(escaped):
        if escaped[i] == "\\":
            i += 2
            continue
        assert escaped[i] != '"'
        i += 1


@settings(max_examples=100, deadline=None)
@given(
    st.integers(min_value=1, max_value=50),
    st.floats(min_value=0.001, max_value=10.0, allow_nan=False),
    st.floats(min_value=0.01, max_value=10.0, allow_nan=False),
)
def test_mm1k_invariants(K, lam, mu):
    m = MM1KModel(K)
    m.solve(lam, mu)
    if not m.is_valid:
        return
    assert abs(sum(m.p) - 1.0) < 1e-6
    assert 0 <= m.avg_num_in_system <= K
    assert m.throughput <= lam + 1e-9
    assert m.throughput <= mu + 1e-9  # single server can't beat service rate
    assert m.avg_wait_time >= 0


@settings(max_examples=100, deadline=None)
@given(
    st.integers(min_value=1, max_value=30),
    st.lists(
        st.floats(min_value=0.01, max_value=5.0, allow_nan=False),
        min_size=1, max_size=8,
    ),
    st.floats(min_value=0.0, max_value=20.0, allow_nan=False),
)
def test_state_dependent_invariants(K, rates, lam):
    m = MM1StateDependentModel(K, rates)
    m.solve(lam)
    if not m.is_valid:
        return
    assert abs(sum(m.p) - 1.0) < 1e-6
    assert 0 <= m.avg_num_in_servers <= min(len(rates), K) + 1e-9
    assert m.throughput <= lam + 1e-9


@settings(max_examples=100, deadline=None)
@given(st.text(alphabet="0123456789hms.", max_size=12))
def test_parse_duration_never_hangs(s):
    try:
        v = parse_go_duration(s)
        assert isinstance(v, float)
    except ValueError:
        pass


@settings(max_examples=60, deadline=None)
@given(
    st.text(alphabet=st.characters(whitelist_categories=("Ll", "Nd"),
                                   max_codepoint=122), min_size=1, max_size=20),
    st.dictionaries(
        st.text(alphabet="abcdefxyz.-/", min_size=1, max_size=15),
        st.text(alphabet="abcdefxyz0123456789.-", max_size=15),
        max_size=4,
    ),
    st.integers(min_value=0, max_value=50),
    st.integers(min_value=0, max_value=10 ** 9),
)
def test_serde_deployment_roundtrip_property(name, labels, replicas, rv):
    from wva_amd.api.types import ObjectMeta
    from wva_amd.kube import serde
    from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec

    d = Deployment(
        metadata=ObjectMeta(name=name, namespace="ns", labels=labels,
                            resource_version=rv),
        replicas=replicas,
        selector=dict(labels),
        template=PodTemplateSpec(labels=dict(labels),
                                 containers=[Container(args=["--x"])]),
    )
    wire = serde.encode(d)
    back = serde.decode("Deployment", wire)
    assert back.metadata.name == name
    assert back.metadata.labels == labels
    assert back.replicas == replicas
    assert back.metadata.resource_version == rv
    assert serde.encode(back) == wire


@settings(max_examples=60, deadline=None)
@given(
    st.floats(min_value=0.1, max_value=100.0, allow_nan=False),
    st.floats(min_value=0.0, max_value=5.0, allow_nan=False),
    st.integers(min_value=1, max_value=512),
)
def test_service_profile_itl_monotone(alpha, beta, batch):
    from wva_amd.emulator.vllm_sim import ServiceProfile

    p = ServiceProfile(alpha_ms=alpha, beta_ms=beta)
    assert p.itl_ms(batch) >= p.itl_ms(max(batch - 1, 1)) - 1e-9
    assert p.itl_ms(batch) >= alpha - 1e-9


@settings(max_examples=60, deadline=None)
@given(
    st.integers(min_value=50, max_value=500),
    st.floats(min_value=1e-6, max_value=1e-2, allow_nan=False),
    st.floats(min_value=1e4, max_value=1e9, allow_nan=False),
)
def test_state_dependent_overflow_rescaling(K, rate, lam):
    """Extreme λ/μ ratios at large K drive the birth-death recursion
    through the MaxFloat64 rescaling loop
    (mm1modelstatedependent.go:70-113 analog) — probabilities must
    still normalize and stay finite."""
    import math as _math

    m = MM1StateDependentModel(K, [rate])
    m.solve(lam)
    assert m.is_valid
    assert all(_math.isfinite(p) and p >= 0 for p in m.p)
    assert abs(sum(m.p) - 1.0) < 1e-6
    # at λ ≫ μ the system is pinned at capacity
    assert m.avg_num_in_system > 0.9 * K
    assert _math.isfinite(m.throughput)
