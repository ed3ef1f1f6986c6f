"""Property-based fuzzing (hypothesis): the CEL parser and the strict
config decoder must never raise anything but their typed errors, for any
input."""

import string

from hypothesis import given, settings, strategies as st

from k8s_dra_driver_amd.allocator.cel import CelError, evaluate, matches
from k8s_dra_driver_amd.api.types import (
    API_GROUP_VERSION,
    ConfigError,
    decode_config,
)

DEVICE = {
    "name": "gpu-0",
    "basic": {
        "attributes": {
            "gpu.amd.com/type": {"string": "gpu"},
            "gpu.amd.com/index": {"int": 0},
            "gpu.amd.com/ok": {"bool": True},
        },
        "capacity": {"gpu.amd.com/memory": {"value": "288Gi"}},
    },
}


@settings(max_examples=300, deadline=None)
@given(st.text(alphabet=string.printable, max_size=120))
def test_cel_arbitrary_text_never_crashes(expr):
    """Any garbage: either a bool comes back or CelError is raised;
    matches() maps both failure modes to False."""
    try:
        result = evaluate(expr, DEVICE, "gpu.amd.com")
        assert isinstance(result, bool)
    except CelError:
        pass
    assert matches(expr, DEVICE, "gpu.amd.com") in (True, False)


@settings(max_examples=200, deadline=None)
@given(
    st.recursive(
        st.one_of(
            st.none(),
            st.booleans(),
            st.integers(-(2**40), 2**40),
            st.text(max_size=20),
        ),
        lambda children: st.one_of(
            st.lists(children, max_size=4),
            st.dictionaries(st.text(max_size=10), children, max_size=4),
        ),
        max_leaves=12,
    )
)
def test_decode_config_arbitrary_json_never_crashes(obj):
    try:
        decode_config(obj)
    except ConfigError:
        pass
    except (TypeError, AttributeError) as e:  # would be a decoder bug
        raise AssertionError(f"untyped error from decode_config: {e!r}")


@settings(max_examples=100, deadline=None)
@given(
    strategy=st.sampled_from(["TimeSlicing", "SharedCompute"]),
    interval=st.sampled_from(["Default", "Short", "Medium", "Long"]),
    pct=st.integers(1, 100),
)
def test_valid_gpu_configs_always_decode(strategy, interval, pct):
    cfg = decode_config(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "GpuConfig",
            "sharing": {
                "strategy": strategy,
                "timeSlicingConfig": {"interval": interval},
                "sharedComputeConfig": {"defaultCuSharePercent": pct},
            },
        }
    )
    cfg.normalize()
    cfg.validate()


# ---------------------------------------------------------------------------
# kubeconfig loader: any YAML input -> KubeAuthError or success, never
# an unhandled exception type (round-2 auth surface hardening)
# ---------------------------------------------------------------------------
_yamlish = st.recursive(
    st.one_of(
        st.none(),
        st.booleans(),
        st.integers(-10, 10),
        st.text(alphabet=string.printable, max_size=20),
    ),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(
            st.sampled_from(
                [
                    "current-context", "contexts", "clusters", "users",
                    "name", "context", "cluster", "user", "server",
                    "token", "tokenFile", "username", "password", "exec",
                    "certificate-authority", "certificate-authority-data",
                    "client-certificate-data", "client-key-data",
                    "client-certificate", "client-key",
                    "insecure-skip-tls-verify", "command", "args", "env",
                    "apiVersion", "auth-provider",
                ]
            ),
            children,
            max_size=6,
        ),
    ),
    max_leaves=25,
)


@settings(max_examples=200, deadline=None)
@given(doc=_yamlish)
def test_kubeconfig_loader_never_crashes(tmp_path_factory, doc):
    import yaml as _yaml

    from k8s_dra_driver_amd.kube.auth import KubeAuthError, load_kubeconfig

    d = tmp_path_factory.mktemp("kc")
    path = d / "kc"
    path.write_text(_yaml.safe_dump(doc))
    try:
        conn = load_kubeconfig(str(path))
        conn.ssl_verify()
    except KubeAuthError:
        pass  # the typed error is the contract


@settings(max_examples=100, deadline=None)
@given(doc1=_yamlish, doc2=_yamlish)
def test_kubeconfig_merge_never_crashes(tmp_path_factory, doc1, doc2):
    import yaml as _yaml

    from k8s_dra_driver_amd.kube.auth import KubeAuthError, load_kubeconfig

    d = tmp_path_factory.mktemp("kcm")
    p1, p2 = d / "a", d / "b"
    p1.write_text(_yaml.safe_dump(doc1))
    p2.write_text(_yaml.safe_dump(doc2))
    try:
        conn = load_kubeconfig(f"{p1}:{p2}")
        conn.ssl_verify()
    except KubeAuthError:
        pass
