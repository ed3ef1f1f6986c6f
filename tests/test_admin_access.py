"""adminAccess (monitoring claims) + DeviceClass-attached config tests."""

from k8s_dra_driver_amd.allocator.structured import Allocator, DeviceClass
from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.model import AllocatableDevice


def devices(lib):
    return [AllocatableDevice.from_gpu(g).to_device() for g in lib.enumerate()]


def test_admin_access_ignores_in_use(fake_lib):
    devs = devices(fake_lib)
    alloc = Allocator()
    all_names = {d["name"] for d in devs}
    spec = {
        "devices": {
            "requests": [
                {
                    "name": "monitor",
                    "deviceClassName": "gpu.amd.com",
                    "allocationMode": "All",
                    "adminAccess": True,
                }
            ]
        }
    }
    # every device is already held by normal claims
    res = alloc.allocate(spec, devs, pool="n", in_use=all_names)
    assert len(res) == 8
    assert all(r.admin_access for r in res)
    assert res[0].to_obj()["adminAccess"] is True


def test_admin_results_not_counted_in_use():
    from k8s_dra_driver_amd.controller.scheduler import ClaimScheduler
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.kube.resourceslice import ResourceSlicePublisher

    kube = InMemoryKube()
    lib = FakeDeviceLib()
    lib.open()
    ResourceSlicePublisher(
        kube, driver_name="gpu.amd.com", node_name="n"
    ).publish(devices(lib))
    # an allocated admin claim covering every GPU
    kube.put_resource_claim(
        {
            "metadata": {"namespace": "d", "name": "mon", "uid": "mon"},
            "spec": {
                "devices": {
                    "requests": [
                        {
                            "name": "m",
                            "deviceClassName": "gpu.amd.com",
                            "allocationMode": "All",
                            "adminAccess": True,
                        }
                    ]
                }
            },
            "status": {
                "allocation": {
                    "devices": {
                        "results": [
                            {
                                "request": "m",
                                "driver": "gpu.amd.com",
                                "pool": "n",
                                "device": f"gpu-{i}",
                                "adminAccess": True,
                            }
                            for i in range(8)
                        ]
                    }
                }
            },
        }
    )
    # a normal claim must still get a device
    kube.put_resource_claim(
        {
            "metadata": {"namespace": "d", "name": "w", "uid": "w"},
            "spec": {
                "devices": {
                    "requests": [
                        {"name": "gpu", "deviceClassName": "gpu.amd.com", "count": 1}
                    ]
                }
            },
        }
    )
    assert ClaimScheduler(kube).reconcile_once() == ["w"]


def test_admin_claim_does_not_block_repartition(tmp_path):
    from tests.test_devicestate import make_claim, make_state

    state, lib = make_state(tmp_path)
    admin_claim = make_claim("mon", ["gpu-0"])
    admin_claim["status"]["allocation"]["devices"]["results"][0][
        "adminAccess"
    ] = True
    state.prepare(admin_claim)
    assert state.claims_holding_gpu(0) == []  # monitoring holds nothing
    # a partition claim on the same GPU succeeds
    cfg = {
        "source": "FromClaim",
        "requests": [],
        "opaque": {
            "driver": "gpu.amd.com",
            "parameters": {
                "apiVersion": API_GROUP_VERSION,
                "kind": "PartitionConfig",
                "computePartition": "CPX",
                "memoryPartition": "NPS1",
                "allowDynamicRepartition": True,
            },
        },
    }
    devs = state.prepare(make_claim("carve", ["gpu-0"], configs=[cfg]))
    assert len(devs) == 8


def test_device_class_config_merged_from_class():
    alloc = Allocator(
        {
            "shared.gpu.amd.com": DeviceClass(
                "shared.gpu.amd.com",
                selectors=["device.driver == 'gpu.amd.com'"],
                config=[
                    {
                        "opaque": {
                            "driver": "gpu.amd.com",
                            "parameters": {
                                "apiVersion": API_GROUP_VERSION,
                                "kind": "GpuConfig",
                                "sharing": {"strategy": "SharedCompute"},
                            },
                        }
                    }
                ],
            )
        }
    )
    lib = FakeDeviceLib()
    lib.open()
    claim = {
        "metadata": {"uid": "u"},
        "spec": {
            "devices": {
                "requests": [
                    {"name": "g", "deviceClassName": "shared.gpu.amd.com"}
                ]
            }
        },
    }
    alloc.allocate_into_claim(claim, devices(lib), pool="n")
    cfg = claim["status"]["allocation"]["devices"]["config"]
    assert cfg[0]["source"] == "FromClass"
    assert cfg[0]["requests"] == ["g"]
    assert (
        cfg[0]["opaque"]["parameters"]["sharing"]["strategy"] == "SharedCompute"
    )
