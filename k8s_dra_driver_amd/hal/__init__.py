from .model import (  # noqa: F401
    AllocatableDevice,
    GpuInfo,
    PartitionedDeviceInfo,
    XgmiLink,
)
from .base import DeviceLib, HalError, HalNotSupported, HalUnavailable  # noqa: F401
from .fake import FakeDeviceLib, FakeNodeConfig, FaultInjector  # noqa: F401


def new_device_lib(backend: str = "auto", **kw):
    """Construct a device library backend.

    ``auto`` picks amdsmi when the native extension can bind libamd_smi and a
    KFD topology exists, else raises (callers choose ``fake`` explicitly —
    silently falling back to fake hardware on a GPU node would mask breakage).
    """
    if backend == "fake":
        return FakeDeviceLib(**kw)
    if backend == "kfd":
        from .kfd import KfdDeviceLib

        return KfdDeviceLib(**kw)
    if backend in ("amdsmi", "auto"):
        from .amdsmi import AmdSmiDeviceLib

        return AmdSmiDeviceLib(**kw)
    raise ValueError(f"unknown HAL backend {backend!r}")
