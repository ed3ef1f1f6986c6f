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
