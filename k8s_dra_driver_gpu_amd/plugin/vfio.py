"""VFIO passthrough: amdgpu <-> vfio-pci driver rebinding + CDI edits.

Parity with ``cmd/gpu-kubelet-plugin/vfio-device.go:50-356`` and
``vfio-cdi.go:51-118``: on prepare, unbind the GPU from amdgpu and bind it to
vfio-pci (after waiting for the GPU to be free, 1 s poll / 60 s timeout —
ref :46-47), detect the IOMMU group (legacy ``/dev/vfio/<group>``) or IOMMUFD
(``/dev/vfio/devices/vfioX`` + ``/dev/iommu``), and emit the CDI device
nodes; on unprepare, rebind to amdgpu.  The same sysfs mechanism works for
amdgpu as for the reference's nvidia driver.
"""

from __future__ import annotations

import logging
import os
import time
from typing import Callable, Optional

from ..api.configs import IOMMU_PREFER_IOMMUFD, VfioDeviceConfig
from ..cdi.spec import ContainerEdits, DeviceNode
from ..device.info import GpuInfo, VfioDeviceInfo

logger = logging.getLogger("amddra.vfio")

GPU_FREE_POLL = 1.0
GPU_FREE_TIMEOUT = 60.0  # ref vfio-device.go:46-47


class VfioError(RuntimeError):
    pass


class VfioPciManager:
    def __init__(
        self,
        sysfs_root: str = "/sys",
        dev_root: str = "/dev",
        busy_check: Optional[Callable[[str], bool]] = None,
        rebind_hook: Optional[Callable[[str, str], None]] = None,
    ):
        self.sysfs_root = sysfs_root
        self.dev_root = dev_root
        self._busy_check = busy_check or self._default_busy_check
        self._rebind_hook = rebind_hook

    # -- sysfs paths --------------------------------------------------------

    def pci_dev_dir(self, pci: str) -> str:
        return os.path.join(self.sysfs_root, "bus", "pci", "devices", pci)

    def driver_dir(self, driver: str) -> str:
        return os.path.join(self.sysfs_root, "bus", "pci", "drivers", driver)

    def current_driver(self, pci: str) -> str:
        link = os.path.join(self.pci_dev_dir(pci), "driver")
        try:
            return os.path.basename(os.readlink(link))
        except OSError:
            return ""

    def iommu_group(self, pci: str) -> str:
        link = os.path.join(self.pci_dev_dir(pci), "iommu_group")
        try:
            return os.path.basename(os.readlink(link))
        except OSError:
            return ""

    def is_sriov_vf(self, pci: str) -> bool:
        # SR-IOV VF guard (ref vfio-device.go:296-316): VFs have physfn
        return os.path.exists(os.path.join(self.pci_dev_dir(pci), "physfn"))

    def iommufd_available(self) -> bool:
        return os.path.exists(os.path.join(self.dev_root, "iommu"))

    def iommufd_cdev_name(self, pci: str) -> str:
        """Discover the per-device IOMMUFD cdev name (``vfioX``).

        The cdev index is per-device and UNRELATED to the IOMMU group
        number; the kernel publishes it as
        ``/sys/bus/pci/devices/<addr>/vfio-dev/vfio<X>`` once the device is
        bound to vfio-pci (ref vfio-device.go reads this dir and errors if
        missing). Raises VfioError if absent.
        """
        d = os.path.join(self.pci_dev_dir(pci), "vfio-dev")
        try:
            names = sorted(n for n in os.listdir(d) if n.startswith("vfio"))
        except OSError:
            names = []
        if not names:
            raise VfioError(
                f"IOMMUFD selected but {d} has no vfio cdev entry — "
                f"is {pci} bound to vfio-pci with VFIO_DEVICE_CDEV enabled?"
            )
        return names[0]

    # -- busy wait ----------------------------------------------------------

    def _default_busy_check(self, pci: str) -> bool:
        """A GPU is busy while KFD processes reference it."""
        proc_dir = os.path.join(self.sysfs_root, "class", "kfd", "kfd", "proc")
        try:
            return len(os.listdir(proc_dir)) > 0
        except OSError:
            return False

    def wait_for_gpu_free(self, pci: str, timeout: float = GPU_FREE_TIMEOUT) -> None:
        deadline = time.monotonic() + timeout
        while self._busy_check(pci):
            if time.monotonic() >= deadline:
                raise VfioError(f"GPU {pci} still busy after {timeout:.0f}s")
            time.sleep(GPU_FREE_POLL)

    # -- rebinding ----------------------------------------------------------

    def _write(self, path: str, value: str) -> None:
        with open(path, "w") as f:
            f.write(value)

    def rebind(self, pci: str, target_driver: str) -> None:
        cur = self.current_driver(pci)
        if cur == target_driver:
            return
        if self._rebind_hook is not None:
            self._rebind_hook(pci, target_driver)
            return
        if cur:
            self._write(os.path.join(self.driver_dir(cur), "unbind"), pci)
        # driver_override is the reliable mechanism for vfio-pci
        override = os.path.join(self.pci_dev_dir(pci), "driver_override")
        if os.path.exists(override):
            self._write(override, target_driver if target_driver != "amdgpu" else "\n")
        try:
            self._write(os.path.join(self.driver_dir(target_driver), "bind"), pci)
        except OSError:
            # fall back to drivers_probe
            probe = os.path.join(self.sysfs_root, "bus", "pci", "drivers_probe")
            if os.path.exists(probe):
                self._write(probe, pci)
        if self.current_driver(pci) != target_driver:
            raise VfioError(
                f"rebind of {pci} to {target_driver} failed "
                f"(now bound to {self.current_driver(pci) or 'nothing'})"
            )

    # -- prepare / unprepare ------------------------------------------------

    def prepare(self, gpu: GpuInfo, cfg: Optional[VfioDeviceConfig] = None) -> VfioDeviceInfo:
        cfg = cfg or VfioDeviceConfig()
        cfg.normalize()
        pci = gpu.pci_bus_id
        if self.is_sriov_vf(pci):
            raise VfioError(f"{pci} is an SR-IOV VF; passthrough of VFs is not supported")
        self.wait_for_gpu_free(pci)
        self.rebind(pci, "vfio-pci")
        group = self.iommu_group(pci)
        use_iommufd = (
            cfg.iommu is not None
            and cfg.iommu.backend_policy == IOMMU_PREFER_IOMMUFD
            and self.iommufd_available()
        )
        if use_iommufd:
            # cdev index is per-device, not the IOMMU group number
            cdev = self.iommufd_cdev_name(pci)
            vfio_path = os.path.join(self.dev_root, "vfio", "devices", cdev)
        else:
            vfio_path = os.path.join(self.dev_root, "vfio", group)
        return VfioDeviceInfo(
            uuid=gpu.uuid,
            pci_bus_id=pci,
            iommu_group=group,
            vfio_dev_path=vfio_path,
            product_name=gpu.product_name,
        )

    def unprepare(self, pci: str) -> None:
        self.rebind(pci, "amdgpu")

    # -- CDI ------------------------------------------------------------------

    def cdi_edits(self, info: VfioDeviceInfo, cfg: Optional[VfioDeviceConfig] = None) -> ContainerEdits:
        """ref vfio-cdi.go:51-118: vfio group node (+ /dev/vfio/vfio control
        node), IOMMUFD nodes when selected, and the visibility env."""
        cfg = cfg or VfioDeviceConfig()
        cfg.normalize()
        edits = ContainerEdits()
        edits.device_nodes.append(
            DeviceNode(path="/dev/vfio/vfio", host_path=os.path.join(self.dev_root, "vfio", "vfio"))
        )
        edits.device_nodes.append(
            DeviceNode(
                path=info.vfio_dev_path.replace(self.dev_root, "/dev", 1)
                if self.dev_root != "/dev"
                else info.vfio_dev_path,
                host_path=info.vfio_dev_path,
            )
        )
        if cfg.iommu and cfg.iommu.backend_policy == IOMMU_PREFER_IOMMUFD and cfg.iommu.enable_api_device:
            edits.device_nodes.append(
                DeviceNode(path="/dev/iommu", host_path=os.path.join(self.dev_root, "iommu"))
            )
        # the ROCm runtime must NOT grab this device in the container
        edits.env.append("AMD_VISIBLE_DEVICES=void")
        edits.env.append(f"VFIO_GROUP={info.iommu_group}")
        return edits
