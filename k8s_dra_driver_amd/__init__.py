"""k8s_dra_driver_amd — an MI355X-native Kubernetes Dynamic Resource Allocation driver.

A from-scratch AMD Instinct MI355X (gfx950) DRA driver with the capability
surface of NVIDIA's k8s-dra-driver (reference @ 2025-02-05), re-designed for
the AMD stack:

- device enumeration through ``libamd_smi`` + KFD sysfs topology (the NVML
  analog, cf. reference ``cmd/nvidia-dra-plugin/nvlib.go``),
- CDI specs that inject ``/dev/kfd`` + ``/dev/dri/renderD*`` (not
  ``/dev/nvidia*``, cf. reference ``cmd/nvidia-dra-plugin/cdi.go``),
- MI355X compute/memory partitioning (SPX/DPX/QPX/CPX x NPS1/NPS4) in place
  of MIG — including *dynamic* repartition, which the reference shipped
  disabled (``nvlib.go:560-669``),
- xGMI-fabric topology attributes for placement of multi-GPU claims
  (the IMEX/NVLink analog, cf. reference ``cmd/nvidia-dra-controller/imex.go``).

Package layout:

========================  =====================================================
``hal``                   hardware abstraction: amdsmi-backed + fake backends
``partition``             gfx950 partition catalog + dynamic repartition engine
``api``                   opaque per-claim config types (GpuConfig analog)
``cdi``                   CDI spec generation (from scratch; no nvcdi analog)
``state``                 claim prepare/unprepare state machine + checkpoints
``sharing``               time-slice config + shared-GPU supervisor (MPS analog)
``plugin``                kubelet DRA gRPC plugin + registration server
``controller``            cluster controller (ResourceSlice reconciler, labels)
``allocator``             structured-parameters allocator + CEL-subset engine
``topology``              xGMI adjacency model and placement scoring
``kube``                  minimal Kubernetes REST client + in-memory fake
``metrics``               Prometheus metrics for the prepare hot path
========================  =====================================================
"""

from .version import __version__  # noqa: F401

DRIVER_NAME = "gpu.amd.com"
DRIVER_DOMAIN = "gpu.amd.com"
