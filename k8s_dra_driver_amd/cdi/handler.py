"""CDI handler: base device spec + transient per-claim specs.

Reference parity (``cmd/nvidia-dra-plugin/cdi.go:40-298``): two spec classes

- a persistent **base** spec enumerating every allocatable device with its
  device-node edits (``CreateStandardDeviceSpecFile``, cdi.go:158-227), and
- transient **per-claim** specs carrying claim-scoped edits — sharing env,
  shm mounts — named by claim UID (``CreateClaimSpecFile``, cdi.go:229-279).

A prepared device resolves to two CDI ids (device + claim), exactly like the
reference (``device_state.go:307-321``).
"""

from __future__ import annotations

import os
import re
from typing import Dict, Iterable, List, Optional

from ..hal.model import AllocatableDevice
from .spec import (
    CDIDevice,
    CDISpec,
    ContainerEdits,
    DeviceNode,
    qualified_device_id,
    write_spec_file,
)

DEFAULT_CDI_ROOT = "/var/run/cdi"
DEVICE_KIND = "k8s.gpu.amd.com/device"
CLAIM_KIND = "k8s.gpu.amd.com/claim"

#: Guard env (reference: NVIDIA_VISIBLE_DEVICES=void, cdi.go:175-180):
#: tells any AMD container runtime hook that devices are CDI-managed and it
#: must not inject its own.
GUARD_ENV = "AMD_VISIBLE_DEVICES=void"

_name_re = re.compile(r"[^a-zA-Z0-9_.-]")


def _safe(name: str) -> str:
    return _name_re.sub("_", name)


class CDIHandler:
    """Writes/deletes CDI spec files under ``cdi_root``."""

    def __init__(
        self,
        cdi_root: str = DEFAULT_CDI_ROOT,
        *,
        dev_root: str = "",
        driver_version: str = "",
        rocm_mount: str = "",
    ):
        self.cdi_root = cdi_root
        # dev_root: prefix of host /dev (containerized-driver-root analog,
        # reference root.go:76-98); "" = host /dev directly.
        self.dev_root = dev_root.rstrip("/")
        self.driver_version = driver_version
        # rocm_mount: host ROCm path to bind into containers read-only —
        # the driver-library-injection analog (reference cdi.go:158-227)
        # for workload images that ship no ROCm userspace. "" = off
        # (images normally carry their own ROCm; the amdgpu/KFD ABI is
        # stable across versions).
        self.rocm_mount = rocm_mount.rstrip("/")

    # -- paths -------------------------------------------------------------
    def _base_spec_path(self) -> str:
        return os.path.join(self.cdi_root, "k8s.gpu.amd.com-device.json")

    def _claim_spec_path(self, claim_uid: str) -> str:
        return os.path.join(self.cdi_root, f"k8s.gpu.amd.com-claim-{_safe(claim_uid)}.json")

    # -- device edits --------------------------------------------------------
    def _device_nodes_for(self, dev: AllocatableDevice) -> List[DeviceNode]:
        gpu = dev.parent_gpu
        if dev.kind == "gpu":
            render_minor, card_minor = gpu.render_minor, gpu.card_minor
        else:
            part = dev.partition
            assert part is not None
            render_minor, card_minor = part.render_minor, part.card_minor
        nodes = [
            DeviceNode(
                path=f"/dev/dri/renderD{render_minor}",
                host_path=f"{self.dev_root}/dev/dri/renderD{render_minor}"
                if self.dev_root
                else "",
            )
        ]
        # card nodes are optional: some deployments expose only renderD*
        # (compute needs kfd + renderD only); referencing a missing node
        # would fail container creation.
        if card_minor >= 0:
            card_host = f"{self.dev_root}/dev/dri/card{card_minor}"
            if os.path.exists(card_host):
                nodes.append(
                    DeviceNode(
                        path=f"/dev/dri/card{card_minor}",
                        host_path=card_host if self.dev_root else "",
                    )
                )
        return nodes

    # -- base spec -----------------------------------------------------------
    def create_standard_spec(self, devices: Iterable[AllocatableDevice]) -> str:
        """Write the persistent base spec for all allocatable devices.

        Common edits inject ``/dev/kfd`` (needed by every ROCm process) and
        the guard env; per-device entries add the render/card nodes.
        Re-written on every (re)enumeration, including after repartition.
        """
        common = ContainerEdits(
            env=[GUARD_ENV],
            device_nodes=[
                DeviceNode(
                    path="/dev/kfd",
                    host_path=f"{self.dev_root}/dev/kfd" if self.dev_root else "",
                )
            ],
        )
        if self.driver_version:
            common.env.append(f"AMD_DRIVER_VERSION={self.driver_version}")
        if self.rocm_mount:
            from .spec import Mount

            common.mounts.append(
                Mount(host_path=self.rocm_mount, container_path="/opt/rocm")
            )
            common.env.append("ROCM_PATH=/opt/rocm")
        spec = CDISpec(kind=DEVICE_KIND, common_edits=common)
        for dev in devices:
            spec.devices.append(
                CDIDevice(
                    name=dev.canonical_name,
                    edits=ContainerEdits(device_nodes=self._device_nodes_for(dev)),
                )
            )
        path = self._base_spec_path()
        write_spec_file(spec, path)
        return path

    # -- claim specs -----------------------------------------------------------
    def create_claim_spec(
        self,
        claim_uid: str,
        device_names: List[str],
        claim_edits: Optional[ContainerEdits] = None,
        per_device_edits: Optional[Dict[str, ContainerEdits]] = None,
    ) -> str:
        """Write the transient per-claim spec (cdi.go:229-279 analog).

        Each prepared device gets a claim-scoped CDI device named
        ``<claimUID>-<device>`` carrying claim-level edits (sharing env, shm
        mounts) merged with any device-specific edits.
        """
        base = claim_edits or ContainerEdits()
        spec = CDISpec(kind=CLAIM_KIND)
        for name in device_names:
            edits = base
            if per_device_edits and name in per_device_edits:
                edits = base.merge(per_device_edits[name])
            spec.devices.append(
                CDIDevice(name=f"{_safe(claim_uid)}-{name}", edits=edits)
            )
        path = self._claim_spec_path(claim_uid)
        # durable=False: a lost claim spec is regenerated from the fsynced
        # checkpoint (prepare's cache-hit path calls _ensure_claim_spec)
        write_spec_file(spec, path, durable=False)
        return path

    def delete_claim_spec(self, claim_uid: str) -> None:
        """Idempotent delete (cdi.go:281-284 analog)."""
        try:
            os.unlink(self._claim_spec_path(claim_uid))
        except FileNotFoundError:
            pass

    def list_claim_spec_uids(self) -> List[str]:
        """UIDs with a claim spec on disk — used by the orphan-cleanup loop
        (a gap the reference left as TODO, driver.go:156-168)."""
        out = []
        prefix, suffix = "k8s.gpu.amd.com-claim-", ".json"
        try:
            names = os.listdir(self.cdi_root)
        except FileNotFoundError:
            return []
        for n in names:
            if n.startswith(prefix) and n.endswith(suffix):
                out.append(n[len(prefix) : -len(suffix)])
        return out

    # -- CDI ids returned to kubelet -------------------------------------------
    @staticmethod
    def device_id(device_name: str) -> str:
        return qualified_device_id(DEVICE_KIND, device_name)

    @staticmethod
    def claim_device_id(claim_uid: str, device_name: str) -> str:
        return qualified_device_id(CLAIM_KIND, f"{_safe(claim_uid)}-{device_name}")
