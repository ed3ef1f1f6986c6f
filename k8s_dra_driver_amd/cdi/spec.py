"""CDI (Container Device Interface) spec model + atomic writer.

Written from scratch against the CDI spec format (tags.cncf.io/
container-device-interface); there is no AMD analog of nvcdi to lean on
(reference leans on it at ``cmd/nvidia-dra-plugin/cdi.go:96-129`` —
SURVEY.md §2.3 N11 marks this "write from scratch").

What gets injected for an MI355X claim (vs /dev/nvidia* + driver libs in the
reference):

- ``/dev/kfd`` — the compute interface, one per node, needed by every claim;
- ``/dev/dri/renderD<minor>`` (+ ``/dev/dri/card<n>``) — per allocated GPU
  *or partition* (each active compute partition owns its own render node);
- env annotations for introspection.

ROCm userspace ships inside workload images and the amdgpu/KFD kernel ABI is
stable across it, so no driver-library mounts are required — a deliberate
divergence from the reference, which must inject the NVIDIA userspace to
match the kernel driver (``cdi.go:158-227``).
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import List

# CDI versions: 0.5.0 baseline; 0.6.0 adds annotations/class-level deviceNodes
# semantics we use. Stamp the minimum the spec actually needs (reference
# parity: cdi.go:218-222 MinimumRequiredVersion).
CDI_VERSION_BASE = "0.5.0"
CDI_VERSION_HOSTPATH = "0.6.0"  # containerEdits mounts with host paths/options


@dataclass
class DeviceNode:
    path: str
    host_path: str = ""
    dev_type: str = "c"
    permissions: str = "rw"

    def to_json(self) -> dict:
        out: dict = {"path": self.path, "type": self.dev_type}
        if self.host_path and self.host_path != self.path:
            out["hostPath"] = self.host_path
        if self.permissions:
            out["permissions"] = self.permissions
        return out


@dataclass
class Mount:
    host_path: str
    container_path: str
    options: List[str] = field(default_factory=lambda: ["ro", "nosuid", "nodev", "bind"])

    def to_json(self) -> dict:
        return {
            "hostPath": self.host_path,
            "containerPath": self.container_path,
            "options": list(self.options),
        }


@dataclass
class Hook:
    hook_name: str
    path: str
    args: List[str] = field(default_factory=list)

    def to_json(self) -> dict:
        return {"hookName": self.hook_name, "path": self.path, "args": list(self.args)}


@dataclass
class ContainerEdits:
    env: List[str] = field(default_factory=list)
    device_nodes: List[DeviceNode] = field(default_factory=list)
    mounts: List[Mount] = field(default_factory=list)
    hooks: List[Hook] = field(default_factory=list)

    def merge(self, other: "ContainerEdits") -> "ContainerEdits":
        return ContainerEdits(
            env=self.env + other.env,
            device_nodes=self.device_nodes + other.device_nodes,
            mounts=self.mounts + other.mounts,
            hooks=self.hooks + other.hooks,
        )

    def is_empty(self) -> bool:
        return not (self.env or self.device_nodes or self.mounts or self.hooks)

    def to_json(self) -> dict:
        out: dict = {}
        if self.env:
            out["env"] = list(self.env)
        if self.device_nodes:
            out["deviceNodes"] = [d.to_json() for d in self.device_nodes]
        if self.mounts:
            out["mounts"] = [m.to_json() for m in self.mounts]
        if self.hooks:
            out["hooks"] = [h.to_json() for h in self.hooks]
        return out


@dataclass
class CDIDevice:
    name: str
    edits: ContainerEdits

    def to_json(self) -> dict:
        return {"name": self.name, "containerEdits": self.edits.to_json()}


@dataclass
class CDISpec:
    kind: str  # e.g. "k8s.gpu.amd.com/device"
    devices: List[CDIDevice] = field(default_factory=list)
    common_edits: ContainerEdits = field(default_factory=ContainerEdits)

    def minimum_version(self) -> str:
        """Smallest cdiVersion that supports everything this spec uses."""
        all_edits = [self.common_edits] + [d.edits for d in self.devices]
        for e in all_edits:
            if e.mounts or e.hooks:
                return CDI_VERSION_HOSTPATH
            if any(d.host_path and d.host_path != d.path for d in e.device_nodes):
                return CDI_VERSION_HOSTPATH
        return CDI_VERSION_BASE

    def to_json(self) -> dict:
        out: dict = {
            "cdiVersion": self.minimum_version(),
            "kind": self.kind,
            "devices": [d.to_json() for d in self.devices],
        }
        if not self.common_edits.is_empty():
            out["containerEdits"] = self.common_edits.to_json()
        return out


def qualified_device_id(kind: str, name: str) -> str:
    """Fully-qualified CDI device id, e.g. k8s.gpu.amd.com/device=gpu-0
    (reference: cdi.go:286-298)."""
    return f"{kind}={name}"


def write_spec_file(spec: CDISpec, path: str, *, durable: bool = True) -> None:
    """Atomic JSON write (tmp + rename, fsync when durable) so containerd
    never reads a torn spec. The reference delegates to the CDI cache
    (cdi.go:225-227); writing directly keeps the hot path to one syscall
    sequence. Per-claim specs pass ``durable=False``: they are regenerated
    from the (fsynced) checkpoint after a crash, so rename atomicity is
    the only requirement."""
    from ..utils.atomicfile import atomic_write_text

    atomic_write_text(
        path,
        json.dumps(spec.to_json(), indent=2, sort_keys=True),
        durable=durable,
    )


def read_spec_file(path: str) -> dict:
    with open(path) as f:
        return json.load(f)
