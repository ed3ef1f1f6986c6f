"""End-to-end validation of the committed demo specs (demo/specs/quickstart)
against the full driver pipeline: YAML -> allocator (scheduler role) ->
gRPC prepare (kubelet role) -> CDI. The reference validates these only
manually on a kind cluster; here they are CI-checked.
"""

import glob
import os

import pytest
import yaml

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.allocator.structured import Allocator
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.plugin.driver import ClaimRef, Driver

SPEC_DIR = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "demo",
    "specs",
    "quickstart",
)


def load_docs(name):
    with open(os.path.join(SPEC_DIR, name)) as f:
        return [d for d in yaml.safe_load_all(f) if d]


def claim_specs_from(docs):
    """ResourceClaim + ResourceClaimTemplate -> claim spec dicts."""
    out = []
    for d in docs:
        if d.get("kind") == "ResourceClaim":
            out.append((d["metadata"]["name"], d["spec"]))
        elif d.get("kind") == "ResourceClaimTemplate":
            out.append((d["metadata"]["name"], d["spec"]["spec"]))
    return out


@pytest.fixture
def cluster(tmp_path):
    """A simulated single-node cluster: fake HAL + plugin + allocator."""
    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="demo-node",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
    )
    driver.startup()
    alloc = Allocator()

    class Cluster:
        def __init__(self):
            self.lib, self.kube, self.driver, self.alloc = lib, kube, driver, alloc
            self.in_use = set()
            self._seq = 0

        def devices(self):
            return [
                d
                for s in kube.list_resource_slices(DRIVER_NAME)
                for d in s["spec"]["devices"]
            ]

        def schedule_and_prepare(self, spec, name):
            self._seq += 1
            uid = f"{name}-{self._seq}"
            claim = {
                "metadata": {"namespace": "demo", "name": name, "uid": uid},
                "spec": spec,
            }
            self.alloc.allocate_into_claim(
                claim,
                self.devices(),
                pool="demo-node",
                in_use=self.in_use,
                node_name="demo-node",
            )
            for r in claim["status"]["allocation"]["devices"]["results"]:
                self.in_use.add(r["device"])
            self.kube.put_resource_claim(claim)
            res = self.driver.node_prepare_resources(
                [ClaimRef("demo", name, uid)]
            )[uid]
            if res.error:
                raise RuntimeError(res.error)
            return uid, res.devices

    return Cluster()


def test_gpu_test1_two_dedicated_gpus(cluster):
    specs = claim_specs_from(load_docs("gpu-test1.yaml"))
    assert len(specs) == 1
    _, devs1 = cluster.schedule_and_prepare(specs[0][1], "pod1-gpu")
    _, devs2 = cluster.schedule_and_prepare(specs[0][1], "pod2-gpu")
    assert devs1[0]["device_name"] != devs2[0]["device_name"]


def test_gpu_test4_partitions_same_parent(cluster):
    cluster.lib.set_compute_partition(0, "CPX")
    cluster.lib.set_compute_partition(1, "CPX")
    cluster.driver.state.refresh_allocatable()
    specs = claim_specs_from(load_docs("gpu-test4.yaml"))
    _, devs = cluster.schedule_and_prepare(specs[0][1], "parts")
    assert len(devs) == 4
    prefixes = {d["device_name"].rsplit("-", 1)[0] for d in devs}
    assert len(prefixes) == 1  # all gpu-N-cpx-*


def test_gpu_test5_sharing_configs(cluster):
    specs = dict(claim_specs_from(load_docs("gpu-test5.yaml")))
    uid_ts, _ = cluster.schedule_and_prepare(
        specs["timeslicing-gpu"], "ts-claim"
    )
    uid_sc, _ = cluster.schedule_and_prepare(
        specs["sharedcompute-gpu"], "sc-claim"
    )
    ckpt_ts = cluster.driver.state.checkpoints.read(uid_ts)
    assert ckpt_ts.sharing_strategy == "TimeSlicing"
    assert ckpt_ts.timeslice_gpus
    ckpt_sc = cluster.driver.state.checkpoints.read(uid_sc)
    assert ckpt_sc.sharing_strategy == "SharedCompute"
    assert cluster.driver.state.shared_manager.get_session(
        ckpt_sc.shared_session_id
    )


def test_gpu_test6_cel_even_index(cluster):
    specs = claim_specs_from(load_docs("gpu-test6.yaml"))
    _, devs = cluster.schedule_and_prepare(specs[0][1], "even")
    idx = int(devs[0]["device_name"].split("-")[1])
    assert idx in (0, 2, 4, 6)


def test_gpu_test7_topology_quad(cluster):
    specs = claim_specs_from(load_docs("gpu-test7-topology.yaml"))
    _, devs = cluster.schedule_and_prepare(specs[0][1], "quad")
    assert len(devs) == 4


def test_partition_carve_dynamic(cluster):
    specs = claim_specs_from(load_docs("partition-carve.yaml"))
    uid, devs = cluster.schedule_and_prepare(specs[0][1], "carve")
    assert len(devs) == 8  # whole carved die
    g0 = cluster.lib.enumerate()[0]
    assert g0.compute_partition == "CPX"
    assert g0.memory_partition == "NPS4"
    cluster.driver.node_unprepare_resources([ClaimRef("demo", "carve", uid)])
    g0 = cluster.lib.enumerate()[0]
    assert g0.compute_partition == "SPX"


def test_all_demo_specs_parse():
    files = glob.glob(os.path.join(SPEC_DIR, "*.yaml"))
    assert len(files) >= 7
    for f in files:
        docs = load_docs(os.path.basename(f))
        assert docs, f
        for d in docs:
            assert "kind" in d and "apiVersion" in d, f


def test_gpu_test8_prospective_partitions(tmp_path):
    """Scheduler-driven dynamic partitioning demo: two pods claim CPX
    partitions directly from PROSPECTIVE devices (no PartitionConfig,
    no pre-carve); the first prepare carves, the drain restores SPX."""
    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    kube.api_versions = ["v1beta2", "v1beta1"]
    driver = Driver(
        lib,
        kube,
        node_name="demo-node",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
        prospective_partitions="cpx",
    )
    driver.startup()
    alloc = Allocator()
    specs = dict(claim_specs_from(load_docs("gpu-test8-prospective.yaml")))
    spec = specs["one-cpx-partition"]

    devices = [
        d
        for s in kube.list_resource_slices(DRIVER_NAME)
        for d in s["spec"]["devices"]
    ]
    assert any("-cpx-" in d["name"] for d in devices)

    uids, in_use = [], set()
    for i in range(2):
        uid = f"t8-{i}"
        results = alloc.allocate(
            {"devices": {"requests": spec["devices"]["requests"]}},
            devices,
            pool="demo-node",
            in_use=in_use,
        )
        assert "-cpx-" in results[0].device
        in_use.add(results[0].device)
        claim = {
            "metadata": {"namespace": "gpu-test8", "name": f"p{i}", "uid": uid},
            "status": {
                "allocation": {
                    "devices": {"results": [r.to_obj() for r in results]}
                }
            },
        }
        kube.put_resource_claim(claim)
        res = driver.node_prepare_resources(
            [ClaimRef("gpu-test8", f"p{i}", uid)]
        )
        assert res[uid].error == "", res[uid].error
        uids.append(uid)

    assert lib.enumerate()[0].compute_partition == "CPX"  # auto-carved
    for i, uid in enumerate(uids):
        driver.node_unprepare_resources([ClaimRef("gpu-test8", f"p{i}", uid)])
    assert lib.enumerate()[0].compute_partition == "SPX"  # drained
    driver.shutdown(unpublish=False)
