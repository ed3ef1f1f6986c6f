"""Orphan cleanup tests (the leak the reference documents as TODO,
driver.go:156-168)."""

from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.plugin.driver import ClaimRef, Driver


def make_stack(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
    )
    driver.startup()
    return kube, driver


def put_claim(kube, uid, name="c1"):
    kube.put_resource_claim(
        {
            "metadata": {"namespace": "d", "name": name, "uid": uid},
            "status": {
                "allocation": {
                    "devices": {
                        "results": [
                            {
                                "request": "gpu",
                                "driver": "gpu.amd.com",
                                "pool": "n",
                                "device": "gpu-0",
                            }
                        ]
                    }
                }
            },
        }
    )


def test_deleted_claim_is_cleaned(tmp_path):
    kube, driver = make_stack(tmp_path)
    put_claim(kube, "uid-1")
    assert not driver.node_prepare_resources([ClaimRef("d", "c1", "uid-1")])[
        "uid-1"
    ].error
    # the user deletes the ResourceClaim; kubelet never calls unprepare
    del kube.resource_claims["d/c1"]
    cleaned = driver.cleanup_orphans()
    assert cleaned == ["uid-1"]
    assert driver.state.checkpoints.read("uid-1") is None
    assert driver.state.cdi.list_claim_spec_uids() == []


def test_recreated_claim_new_uid_cleans_old(tmp_path):
    kube, driver = make_stack(tmp_path)
    put_claim(kube, "uid-old")
    driver.node_prepare_resources([ClaimRef("d", "c1", "uid-old")])
    put_claim(kube, "uid-new")  # same name, new incarnation
    cleaned = driver.cleanup_orphans()
    assert cleaned == ["uid-old"]


def test_live_claim_untouched(tmp_path):
    kube, driver = make_stack(tmp_path)
    put_claim(kube, "uid-1")
    driver.node_prepare_resources([ClaimRef("d", "c1", "uid-1")])
    assert driver.cleanup_orphans() == []
    assert driver.state.checkpoints.read("uid-1") is not None


def test_specless_cdi_file_cleaned(tmp_path):
    kube, driver = make_stack(tmp_path)
    # simulate a crash that left a claim CDI spec with no checkpoint
    driver.state.cdi.create_claim_spec("ghost-uid", ["gpu-0"])
    cleaned = driver.cleanup_orphans()
    assert "ghost-uid" in cleaned
    assert driver.state.cdi.list_claim_spec_uids() == []


def test_prepare_failure_emits_event(tmp_path):
    kube, driver = make_stack(tmp_path)
    put_claim(kube, "uid-ev")
    kube.resource_claims["d/c1"]["status"]["allocation"]["devices"]["results"][0][
        "device"
    ] = "gpu-404"
    kube.put_resource_claim(kube.resource_claims["d/c1"])
    res = driver.node_prepare_resources([ClaimRef("d", "c1", "uid-ev")])["uid-ev"]
    assert "not found" in res.error
    assert kube.events, "no Event emitted"
    ev = kube.events[-1]
    assert ev["reason"] == "PrepareFailed"
    assert ev["involvedObject"]["name"] == "c1"
    assert ev["type"] == "Warning"
