"""Structured allocator + xGMI topology scoring tests (BASELINE config #5)."""

import pytest

from k8s_dra_driver_amd.allocator.structured import AllocationError, Allocator
from k8s_dra_driver_amd.hal.model import AllocatableDevice
from k8s_dra_driver_amd.topology.xgmi import (
    is_fully_connected,
    pair_score,
    pick_best_subset,
    subset_score,
)


def published_devices(lib):
    out = []
    for g in lib.enumerate():
        if g.partitions:
            out.extend(
                AllocatableDevice.from_partition(g, p).to_device()
                for p in g.partitions
            )
        else:
            out.append(AllocatableDevice.from_gpu(g).to_device())
    return out


@pytest.fixture
def devices(fake_lib):
    return published_devices(fake_lib)


def request(name="gpu", cls="gpu.amd.com", count=1, selectors=None, mode=None):
    r = {"name": name, "deviceClassName": cls, "count": count}
    if selectors:
        r["selectors"] = [{"cel": {"expression": e}} for e in selectors]
    if mode:
        r["allocationMode"] = mode
    return r


class TestTopologyScoring:
    def test_full_mesh_pairs(self, devices):
        assert pair_score(devices[0], devices[1]) == 10
        assert is_fully_connected(devices[:4])
        assert subset_score(devices[:4]) == 6 * 10

    def test_partitions_of_same_parent_score_highest(self, fake_lib):
        fake_lib.set_compute_partition(0, "CPX")
        fake_lib.set_memory_partition(0, "NPS4")
        devs = published_devices(fake_lib)
        p0 = next(d for d in devs if d["name"] == "gpu-0-cpx-0")
        p1 = next(d for d in devs if d["name"] == "gpu-0-cpx-1")  # same domain
        p2 = next(d for d in devs if d["name"] == "gpu-0-cpx-2")
        g1 = next(d for d in devs if d["name"] == "gpu-1")
        assert pair_score(p0, p1) == 120
        assert pair_score(p0, p2) == 100
        assert pair_score(p0, g1) <= 10

    def test_pick_best_subset_prefers_same_parent(self, fake_lib):
        fake_lib.set_compute_partition(0, "CPX")
        fake_lib.set_compute_partition(1, "CPX")
        devs = [d for d in published_devices(fake_lib) if "cpx" in d["name"]]
        chosen = pick_best_subset(devs, 4)
        parents = {
            d["basic"]["attributes"]["gpu.amd.com/parentUUID"]["string"]
            for d in chosen
        }
        assert len(parents) == 1  # all from one die


class TestAllocator:
    def test_single_gpu(self, devices):
        alloc = Allocator()
        res = alloc.allocate(
            {"devices": {"requests": [request()]}}, devices, pool="node-a"
        )
        assert len(res) == 1
        assert res[0].device.startswith("gpu-")
        assert res[0].driver == "gpu.amd.com"

    def test_count_and_exclusion(self, devices):
        alloc = Allocator()
        res = alloc.allocate(
            {"devices": {"requests": [request(count=4)]}},
            devices,
            pool="node-a",
            in_use={"gpu-0", "gpu-1"},
        )
        names = {r.device for r in res}
        assert len(names) == 4
        assert not names & {"gpu-0", "gpu-1"}

    def test_insufficient_devices(self, devices):
        alloc = Allocator()
        with pytest.raises(AllocationError, match="need 9"):
            alloc.allocate(
                {"devices": {"requests": [request(count=9)]}},
                devices,
                pool="node-a",
            )

    def test_cel_selector_filters(self, devices):
        alloc = Allocator()
        res = alloc.allocate(
            {
                "devices": {
                    "requests": [
                        request(
                            selectors=[
                                "device.attributes['gpu.amd.com'].index in [3, 5]"
                            ],
                            count=2,
                        )
                    ]
                }
            },
            devices,
            pool="node-a",
        )
        assert {r.device for r in res} == {"gpu-3", "gpu-5"}

    def test_match_attribute_constraint(self, fake_lib):
        """gpu-test4 pattern: partitions constrained to one parent die."""
        fake_lib.set_compute_partition(0, "CPX")
        fake_lib.set_compute_partition(1, "CPX")
        devs = published_devices(fake_lib)
        alloc = Allocator()
        spec = {
            "devices": {
                "requests": [
                    request("a", cls="partition.gpu.amd.com"),
                    request("b", cls="partition.gpu.amd.com"),
                    request("c", cls="partition.gpu.amd.com"),
                ],
                "constraints": [
                    {"requests": [], "matchAttribute": "gpu.amd.com/parentUUID"}
                ],
            }
        }
        res = alloc.allocate(spec, devs, pool="node-a")
        parents = set()
        by_name = {d["name"]: d for d in devs}
        for r in res:
            parents.add(
                by_name[r.device]["basic"]["attributes"][
                    "gpu.amd.com/parentUUID"
                ]["string"]
            )
        assert len(parents) == 1

    def test_allocation_mode_all(self, fake_lib):
        fake_lib.set_compute_partition(2, "CPX")
        devs = published_devices(fake_lib)
        alloc = Allocator()
        res = alloc.allocate(
            {
                "devices": {
                    "requests": [
                        request(
                            "all-parts",
                            cls="partition.gpu.amd.com",
                            mode="All",
                            selectors=[
                                "device.attributes['gpu.amd.com'].parentIndex == 2"
                            ],
                        )
                    ]
                }
            },
            devs,
            pool="node-a",
        )
        assert len(res) == 8

    def test_unknown_class(self, devices):
        with pytest.raises(AllocationError, match="unknown DeviceClass"):
            Allocator().allocate(
                {"devices": {"requests": [request(cls="nope")]}},
                devices,
                pool="node-a",
            )

    def test_allocate_into_claim_carries_config(self, devices):
        alloc = Allocator()
        claim = {
            "metadata": {"uid": "u1"},
            "spec": {
                "devices": {
                    "requests": [request()],
                    "config": [
                        {
                            "requests": [],
                            "opaque": {"driver": "gpu.amd.com", "parameters": {}},
                        }
                    ],
                }
            },
        }
        alloc.allocate_into_claim(
            claim, devices, pool="node-a", node_name="node-a"
        )
        a = claim["status"]["allocation"]
        assert a["devices"]["results"][0]["driver"] == "gpu.amd.com"
        assert a["devices"]["config"][0]["source"] == "FromClaim"
        assert a["nodeSelector"]["nodeSelectorTerms"]


class TestTopologyAwarePlacement:
    def test_four_gpu_claim_lands_on_connected_subset(self, devices):
        """BASELINE config #5: 4 xGMI-adjacent GPUs via CEL + scoring."""
        alloc = Allocator()
        spec = {
            "devices": {
                "requests": [
                    request(
                        "quad",
                        count=4,
                        selectors=[
                            "device.attributes['gpu.amd.com'].xgmiLinkCount >= 4"
                        ],
                    )
                ]
            }
        }
        res = alloc.allocate(spec, devices, pool="node-a")
        by_name = {d["name"]: d for d in devices}
        chosen = [by_name[r.device] for r in res]
        assert is_fully_connected(chosen)


def test_combination_cap_is_logged(caplog):
    """'No silent caps': when the per-request combination fan-out cap
    bites, the allocator says so (round-1 weak finding)."""
    import logging

    from k8s_dra_driver_amd.allocator.structured import Allocator

    devices = [
        {
            "name": f"g{i}",
            "basic": {
                "attributes": {
                    "gpu.amd.com/type": {"string": "gpu"},
                },
                "capacity": {},
            },
        }
        for i in range(80)
    ]
    spec = {
        "devices": {
            "requests": [
                {"name": "pair", "deviceClassName": "any.gpu.amd.com", "count": 2}
            ]
        }
    }
    with caplog.at_level(logging.WARNING, logger="k8s_dra_driver_amd.allocator.structured"):
        Allocator().allocate(spec, devices, pool="p")
    assert any("combination cap" in r.message for r in caplog.records)


class TestDeviceTaints:
    """DRA device taints: NoSchedule devices need a matching toleration;
    adminAccess (monitoring) bypasses."""

    def _devs(self):
        def d(name, taints=None):
            out = {
                "name": name,
                "basic": {
                    "attributes": {"gpu.amd.com/type": {"string": "gpu"}},
                    "capacity": {},
                },
            }
            if taints:
                out["taints"] = taints
            return out

        sick = [{"key": "gpu.amd.com/unhealthy", "effect": "NoSchedule"}]
        return [d("gpu-0", sick), d("gpu-1")]

    def test_tainted_device_skipped(self):
        from k8s_dra_driver_amd.allocator.structured import Allocator

        spec = {
            "devices": {
                "requests": [{"name": "g", "deviceClassName": "gpu.amd.com"}]
            }
        }
        res = Allocator().allocate(spec, self._devs(), pool="p")
        assert res[0].device == "gpu-1"

    def test_toleration_admits(self):
        from k8s_dra_driver_amd.allocator.structured import Allocator

        spec = {
            "devices": {
                "requests": [
                    {
                        "name": "g",
                        "deviceClassName": "gpu.amd.com",
                        "count": 2,
                        "tolerations": [
                            {
                                "key": "gpu.amd.com/unhealthy",
                                "operator": "Exists",
                            }
                        ],
                    }
                ]
            }
        }
        res = Allocator().allocate(spec, self._devs(), pool="p")
        assert sorted(r.device for r in res) == ["gpu-0", "gpu-1"]

    def test_admin_access_bypasses_taints(self):
        from k8s_dra_driver_amd.allocator.structured import Allocator

        spec = {
            "devices": {
                "requests": [
                    {
                        "name": "mon",
                        "deviceClassName": "gpu.amd.com",
                        "count": 2,
                        "adminAccess": True,
                    }
                ]
            }
        }
        res = Allocator().allocate(spec, self._devs(), pool="p")
        assert len(res) == 2

    def test_untolerated_two_needed_fails(self):
        import pytest as _pytest

        from k8s_dra_driver_amd.allocator.structured import (
            AllocationError,
            Allocator,
        )

        spec = {
            "devices": {
                "requests": [
                    {"name": "g", "deviceClassName": "gpu.amd.com", "count": 2}
                ]
            }
        }
        with _pytest.raises(AllocationError):
            Allocator().allocate(spec, self._devs(), pool="p")
