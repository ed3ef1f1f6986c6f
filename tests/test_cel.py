"""CEL-subset evaluator tests, including the reference's real selector
expressions translated to gpu.amd.com."""

import pytest

from k8s_dra_driver_amd.allocator.cel import CelError, evaluate, matches


def dev(attrs=None, caps=None):
    attributes = {}
    for k, (t, v) in (attrs or {}).items():
        attributes[f"gpu.amd.com/{k}"] = {t: v}
    capacity = {}
    for k, v in (caps or {}).items():
        capacity[f"gpu.amd.com/{k}"] = {"value": v}
    return {
        "name": "gpu-0",
        "basic": {"attributes": attributes, "capacity": capacity},
    }


GPU = dev(
    attrs={
        "type": ("string", "gpu"),
        "productName": ("string", "AMD Instinct MI355X"),
        "index": ("int", 2),
        "partitionable": ("bool", True),
        "architecture": ("string", "gfx950"),
    },
    caps={"memory": "288Gi", "computeUnits": "256"},
)


class TestBasics:
    def test_driver_and_type(self):
        # the DeviceClass selector (deviceclass-gpu.yaml:10 analog)
        assert evaluate(
            "device.driver == 'gpu.amd.com' && "
            "device.attributes['gpu.amd.com'].type == 'gpu'",
            GPU,
            "gpu.amd.com",
        )
        assert not evaluate(
            "device.driver == 'gpu.nvidia.com'", GPU, "gpu.amd.com"
        )

    def test_product_regex_and_index(self):
        # gpu-test6.yaml:25-30 pattern
        expr = (
            "device.attributes['gpu.amd.com'].productName.lowerAscii()"
            ".matches('^.*mi355x.*$') && "
            "(device.attributes['gpu.amd.com'].index == 0 || "
            " device.attributes['gpu.amd.com'].index == 2)"
        )
        assert evaluate(expr, GPU, "gpu.amd.com")

    def test_int_and_bool_attrs(self):
        assert evaluate(
            "device.attributes['gpu.amd.com'].index >= 1", GPU, "gpu.amd.com"
        )
        assert evaluate(
            "device.attributes['gpu.amd.com'].partitionable", GPU, "gpu.amd.com"
        )
        assert evaluate(
            "!device.attributes['gpu.amd.com'].partitionable == false",
            GPU,
            "gpu.amd.com",
        )

    def test_in_operator(self):
        assert evaluate(
            "device.attributes['gpu.amd.com'].index in [0, 2, 4]",
            GPU,
            "gpu.amd.com",
        )

    def test_string_methods(self):
        assert evaluate(
            "device.attributes['gpu.amd.com'].architecture.startsWith('gfx')",
            GPU,
            "gpu.amd.com",
        )
        assert evaluate(
            "device.attributes['gpu.amd.com'].productName.contains('Instinct')",
            GPU,
            "gpu.amd.com",
        )


class TestCapacityQuantities:
    def test_quantity_compare(self):
        assert evaluate(
            "device.capacity['gpu.amd.com'].memory >= quantity('200Gi')",
            GPU,
            "gpu.amd.com",
        )
        assert not evaluate(
            "device.capacity['gpu.amd.com'].memory > quantity('1Ti')",
            GPU,
            "gpu.amd.com",
        )

    def test_quantity_equality(self):
        assert evaluate(
            "device.capacity['gpu.amd.com'].computeUnits == quantity('256')",
            GPU,
            "gpu.amd.com",
        )


class TestErrorSemantics:
    def test_missing_attribute_is_error(self):
        with pytest.raises(CelError):
            evaluate(
                "device.attributes['gpu.amd.com'].bogus == 1", GPU, "gpu.amd.com"
            )

    def test_matches_maps_errors_to_false(self):
        assert not matches(
            "device.attributes['gpu.amd.com'].bogus == 1", GPU, "gpu.amd.com"
        )
        assert not matches("][", GPU, "gpu.amd.com")

    def test_missing_domain_is_error(self):
        with pytest.raises(CelError):
            evaluate(
                "device.attributes['other.com'].x == 1", GPU, "gpu.amd.com"
            )

    def test_non_bool_result_rejected(self):
        with pytest.raises(CelError):
            evaluate("device.attributes['gpu.amd.com'].index", GPU, "gpu.amd.com")


class TestCelErrorAbsorption:
    """Kubernetes CEL ||/&& are commutative error absorbers:
    true || error -> true, false && error -> false. The evaluator must
    match, or selectors that guard optional attributes reject devices the
    real scheduler would accept."""

    def test_true_or_error_is_true(self):
        assert evaluate(
            "device.attributes['gpu.amd.com'].type == 'gpu' || "
            "device.attributes['gpu.amd.com'].missingAttr == 1",
            GPU,
            "gpu.amd.com",
        )
        # commutative: error first
        assert evaluate(
            "device.attributes['gpu.amd.com'].missingAttr == 1 || "
            "device.attributes['gpu.amd.com'].type == 'gpu'",
            GPU,
            "gpu.amd.com",
        )

    def test_false_and_error_is_false(self):
        assert not evaluate(
            "device.attributes['gpu.amd.com'].type == 'cpu' && "
            "device.attributes['gpu.amd.com'].missingAttr == 1",
            GPU,
            "gpu.amd.com",
        )
        assert not evaluate(
            "device.attributes['gpu.amd.com'].missingAttr == 1 && "
            "device.attributes['gpu.amd.com'].type == 'cpu'",
            GPU,
            "gpu.amd.com",
        )

    def test_unresolved_error_still_raises(self):
        with pytest.raises(CelError):
            evaluate(
                "device.attributes['gpu.amd.com'].type == 'cpu' || "
                "device.attributes['gpu.amd.com'].missingAttr == 1",
                GPU,
                "gpu.amd.com",
            )
        with pytest.raises(CelError):
            evaluate(
                "device.attributes['gpu.amd.com'].type == 'gpu' && "
                "device.attributes['gpu.amd.com'].missingAttr == 1",
                GPU,
                "gpu.amd.com",
            )

    def test_non_bool_logical_operand_is_error(self):
        with pytest.raises(CelError):
            evaluate(
                "device.attributes['gpu.amd.com'].productName && true",
                GPU,
                "gpu.amd.com",
            )
