"""amd-dra-demo — run the committed demo specs without a cluster.

Boots the full system as real processes — mini apiserver (HTTP), the
kubelet plugin binary (fake or real HAL), the controller binary with the
topology-aware allocator — then plays the kubelet role for a demo YAML:
creates the ResourceClaims, waits for the controller to allocate them,
calls NodePrepareResources over the plugin's gRPC socket, and prints what
each pod would see (devices + CDI edits). The kind-cluster walkthrough of
the reference (demo/clusters/kind + kubectl apply) without kind, kubectl
or a container runtime.

    amd-dra-demo demo/specs/quickstart/gpu-test1.yaml
    amd-dra-demo --hal amdsmi demo/specs/quickstart/gpu-test7-topology.yaml
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import tempfile
import time
import uuid as uuidlib

import grpc
import yaml

from . import DRIVER_NAME
from .cdi.spec import read_spec_file
from .kube.miniapiserver import MiniApiServer
from .plugin.proto import V1BETA1


def _wait(pred, timeout, what):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if pred():
            return
        time.sleep(0.2)
    raise TimeoutError(f"timed out waiting for {what}")


class DemoHarness:
    def __init__(
        self,
        hal: str = "fake",
        verbose: bool = False,
        prospective: str = "",
    ):
        self.hal = hal
        self.verbose = verbose
        self.prospective = prospective
        self.tmp = tempfile.mkdtemp(prefix="amd-dra-demo-")
        self.api = MiniApiServer().start()
        if prospective:
            # counters need resource.k8s.io v1beta2+ (K8s 1.33)
            self.api.store.api_versions = ["v1beta2", "v1beta1"]
        self.api.store.put_node({"metadata": {"name": "demo-node", "uid": "demo-node-uid"}})
        self.kubeconfig = self.api.write_kubeconfig(os.path.join(self.tmp, "kubeconfig"))
        self.procs: list = []
        self.plugin_sock = ""

    def _spawn(self, name, module, *args):
        env = dict(
            os.environ,
            PYTHONPATH=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            KUBECONFIG=self.kubeconfig,
            NODE_NAME="demo-node",
        )
        out = None if self.verbose else subprocess.DEVNULL
        p = subprocess.Popen(
            [sys.executable, "-m", module, *args],
            env=env,
            stdout=out,
            stderr=subprocess.STDOUT if out is None else subprocess.DEVNULL,
        )
        self.procs.append((name, p))
        return p

    def start(self):
        plugin_dir = os.path.join(self.tmp, "plugins", DRIVER_NAME)
        registry = os.path.join(self.tmp, "plugins_registry")
        self.cdi_root = os.path.join(self.tmp, "cdi")
        plugin_args = [
            "--hal", self.hal,
            "--cdi-root", self.cdi_root,
            "--plugin-path", plugin_dir,
            "--plugin-registration-path", registry,
        ]
        if self.prospective:
            plugin_args += ["--prospective-partitions", self.prospective]
        self._spawn(
            "plugin", "k8s_dra_driver_amd.plugin.main", *plugin_args
        )
        self._spawn(
            "controller",
            "k8s_dra_driver_amd.controller.main",
            "--poll-interval", "1",
            "--metrics-port", "0",
            "--allocate-claims",
        )
        self.plugin_sock = os.path.join(plugin_dir, "plugin.sock")
        _wait(lambda: os.path.exists(self.plugin_sock), 30, "plugin socket")
        _wait(
            lambda: self.api.store.list_resource_slices(DRIVER_NAME),
            30,
            "ResourceSlice publication",
        )
        n = sum(
            len(s["spec"]["devices"])
            for s in self.api.store.list_resource_slices(DRIVER_NAME)
        )
        print(f"[demo] system up: {n} device(s) published on demo-node")

    def carve(self, gpu_name: str, compute: str, memory: str = "NPS1") -> None:
        """Pre-carve a GPU via a held claim (enables the partition demos,
        e.g. gpu-test4's four-partitions-one-die constraint)."""
        uid = f"carve-{gpu_name}"
        self.api.store.put_resource_claim(
            {
                "metadata": {"namespace": "demo", "name": uid, "uid": uid},
                "status": {
                    "allocation": {
                        "devices": {
                            "results": [
                                {
                                    "request": "gpu",
                                    "driver": DRIVER_NAME,
                                    "pool": "demo-node",
                                    "device": gpu_name,
                                }
                            ],
                            "config": [
                                {
                                    "source": "FromClaim",
                                    "requests": [],
                                    "opaque": {
                                        "driver": DRIVER_NAME,
                                        "parameters": {
                                            "apiVersion": "resource.gpu.amd.com/v1alpha1",
                                            "kind": "PartitionConfig",
                                            "computePartition": compute,
                                            "memoryPartition": memory,
                                            "allowDynamicRepartition": True,
                                        },
                                    },
                                }
                            ],
                        }
                    }
                },
            }
        )
        channel, prepare, _ = self._grpc()
        m = V1BETA1
        req = m.NodePrepareResourcesRequest()
        c = req.claims.add()
        c.namespace, c.name, c.uid = "demo", uid, uid
        resp = prepare(req, timeout=60)
        channel.close()
        if resp.claims[uid].error:
            raise RuntimeError(f"carve failed: {resp.claims[uid].error}")
        parts = [d.device_name for d in resp.claims[uid].devices]
        print(f"[demo] carved {gpu_name} -> {len(parts)} partition(s) ({compute}/{memory})")
        _wait(
            lambda: any(
                "-" in d["name"] and d["name"].startswith(gpu_name + "-")
                for s in self.api.store.list_resource_slices(DRIVER_NAME)
                for d in s["spec"]["devices"]
            ),
            30,
            "partition republication",
        )

    # -- kubelet role -------------------------------------------------------
    def _grpc(self):
        channel = grpc.insecure_channel(f"unix://{self.plugin_sock}")
        m = V1BETA1
        return channel, channel.unary_unary(
            f"/{m.service_name}/NodePrepareResources",
            request_serializer=lambda x: x.SerializeToString(),
            response_deserializer=m.NodePrepareResourcesResponse.FromString,
        ), channel.unary_unary(
            f"/{m.service_name}/NodeUnprepareResources",
            request_serializer=lambda x: x.SerializeToString(),
            response_deserializer=m.NodeUnprepareResourcesResponse.FromString,
        )

    def run_spec(self, path: str) -> int:
        with open(path) as f:
            docs = [d for d in yaml.safe_load_all(f) if d]
        templates = {}
        standalone = {}
        pods = []
        for d in docs:
            kind = d.get("kind")
            if kind == "ResourceClaimTemplate":
                templates[d["metadata"]["name"]] = d["spec"]["spec"]
            elif kind == "ResourceClaim":
                standalone[d["metadata"]["name"]] = d["spec"]
            elif kind == "Pod":
                pods.append(d)
        print(f"[demo] {os.path.basename(path)}: {len(pods)} pod(s), "
              f"{len(templates)} template(s), {len(standalone)} claim(s)")

        # create claims (template -> one claim per pod reference;
        # standalone claims shared by name)
        claim_uid_by_key = {}
        for pod in pods:
            pod_name = pod["metadata"]["name"]
            for ref in pod["spec"].get("resourceClaims") or []:
                if "resourceClaimTemplateName" in ref:
                    spec = templates[ref["resourceClaimTemplateName"]]
                    name = f"{pod_name}-{ref['name']}"
                elif "resourceClaimName" in ref:
                    if ref["resourceClaimName"] in claim_uid_by_key:
                        continue
                    spec = standalone[ref["resourceClaimName"]]
                    name = ref["resourceClaimName"]
                else:
                    continue
                uid = str(uuidlib.uuid4())
                self.api.store.put_resource_claim(
                    {
                        "metadata": {
                            "namespace": "demo",
                            "name": name,
                            "uid": uid,
                        },
                        "spec": spec,
                    }
                )
                claim_uid_by_key[name] = uid

        # wait for the controller to allocate all claims
        def all_allocated():
            for name in claim_uid_by_key:
                c = self.api.store.get_resource_claim("demo", name)
                if not (c.get("status") or {}).get("allocation"):
                    return False
            return True

        try:
            _wait(all_allocated, 30, "controller allocation")
        except TimeoutError:
            pending = [
                name
                for name in claim_uid_by_key
                if not (
                    self.api.store.get_resource_claim("demo", name).get(
                        "status"
                    )
                    or {}
                ).get("allocation")
            ]
            n_dev = sum(
                len(s["spec"]["devices"])
                for s in self.api.store.list_resource_slices(DRIVER_NAME)
            )
            print(
                f"[demo] FAILED: claim(s) {pending} not allocatable on "
                f"this node ({n_dev} published device(s)) — the spec "
                "likely needs more/other GPUs than this node has "
                "(e.g. gpu-test1 needs 2 whole GPUs)"
            )
            return 1
        print(f"[demo] {len(claim_uid_by_key)} claim(s) allocated by the controller")

        # kubelet role: prepare each pod's claims, print the pod view
        channel, prepare, unprepare = self._grpc()
        m = V1BETA1
        rc = 0
        prepared = []
        for pod in pods:
            pod_name = pod["metadata"]["name"]
            devices = []
            env = set()
            for ref in pod["spec"].get("resourceClaims") or []:
                name = (
                    f"{pod_name}-{ref['name']}"
                    if "resourceClaimTemplateName" in ref
                    else ref.get("resourceClaimName", "")
                )
                uid = claim_uid_by_key.get(name)
                if uid is None:
                    continue
                req = m.NodePrepareResourcesRequest()
                c = req.claims.add()
                c.namespace, c.name, c.uid = "demo", name, uid
                resp = prepare(req, timeout=30)
                r = resp.claims[uid]
                if r.error:
                    print(f"[demo] POD {pod_name}: PREPARE FAILED: {r.error}")
                    rc = 1
                    continue
                prepared.append((name, uid))
                for dev in r.devices:
                    devices.append(dev.device_name)
                # pull env from the claim CDI spec (what containerd injects)
                spec_path = os.path.join(
                    self.cdi_root, f"k8s.gpu.amd.com-claim-{uid}.json"
                )
                if os.path.exists(spec_path):
                    spec = read_spec_file(spec_path)
                    for d2 in spec.get("devices") or []:
                        env.update((d2.get("containerEdits") or {}).get("env") or [])
            print(
                f"[demo] POD {pod_name}: devices={devices} "
                f"env={sorted(e for e in env if not e.startswith('AMD_DRA_CLAIM_UID'))}"
            )
        # unprepare everything (pods exit)
        seen = set()
        for name, uid in prepared:
            if uid in seen:
                continue
            seen.add(uid)
            ureq = m.NodeUnprepareResourcesRequest()
            uc = ureq.claims.add()
            uc.namespace, uc.name, uc.uid = "demo", name, uid
            unprepare(ureq, timeout=30)
        channel.close()
        # release the claims so the next spec's allocation sees free
        # devices (the kubelet+GC role at pod deletion)
        with self.api.store._lock:
            for name in claim_uid_by_key:
                self.api.store.resource_claims.pop(f"demo/{name}", None)
        print(f"[demo] done rc={rc}")
        return rc

    def stop(self):
        for name, p in self.procs:
            p.terminate()
        for name, p in self.procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait(timeout=5)
        self.api.stop()


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("amd-dra-demo")
    ap.add_argument("specs", nargs="+", help="demo spec YAML file(s)")
    ap.add_argument("--hal", default="fake", choices=["fake", "amdsmi", "kfd"])
    ap.add_argument(
        "--carve",
        action="append",
        default=[],
        metavar="GPU:MODE[:NPS]",
        help="pre-carve a GPU before running specs, e.g. gpu-0:CPX:NPS4",
    )
    ap.add_argument(
        "--prospective",
        default="",
        choices=["", "cpx", "dpx", "qpx"],
        help="publish prospective partitions of this mode (scheduler-"
        "driven dynamic partitioning; serves resource.k8s.io/v1beta2) — "
        "required for gpu-test8",
    )
    ap.add_argument("-v", "--verbose", action="store_true")
    args = ap.parse_args(argv)
    harness = DemoHarness(
        hal=args.hal, verbose=args.verbose, prospective=args.prospective
    )
    rc = 0
    try:
        harness.start()
        for spec in args.carve:
            parts = spec.split(":")
            harness.carve(parts[0], parts[1], parts[2] if len(parts) > 2 else "NPS1")
        for spec in args.specs:
            rc |= harness.run_spec(spec)
    finally:
        harness.stop()
    return rc


if __name__ == "__main__":
    raise SystemExit(main())
