"""Mock MI355X sysfs/dev tree for CPU-only CI.

The analog of the reference's mock-NVML harness
(``hack/ci/mock-nvml/setup-mock-gpu.sh:29-282``: a fake ``libnvidia-ml.so``
with YAML GPU profiles, mknod'd ``/dev/nvidia*`` nodes and a fake
``/proc/driver/nvidia`` tree).  Here the whole device layer reads sysfs, so
the mock is a generated sysfs+dev tree with MI355X-profile values that the
real ``SysfsBackend`` consumes unchanged — one code path, no library swap.

Partition dynamics are emulated: a watcher applies ``current_compute_partition``
writes by re-laying-out the per-GPU card/render/KFD entries exactly as the
amdgpu driver does on a mode switch (extra cards on the same PCI address, one
KFD node per partition).
"""

from __future__ import annotations

import os
import shutil
from dataclasses import dataclass, field
from typing import Dict, List

from ..api.configs import COMPUTE_MODE_PARTITIONS
from .info import MI355X_VRAM_BYTES
from .sysfs import IOLINK_TYPE_XGMI, SysfsBackend


@dataclass
class MockGpuProfile:
    name: str = "MI355X"
    vram_bytes: int = MI355X_VRAM_BYTES
    simd_count: int = 1024  # 256 CUs x 4 SIMDs
    gfx_target_version: int = 90500  # gfx950
    device_id: str = "0x75a0"
    available_compute: str = "SPX, DPX, QPX, CPX"
    available_memory: str = "NPS1, NPS2"
    xgmi_links_per_pair: int = 1


@dataclass
class MockTree:
    """Generates and mutates the fake sysfs/dev layout under `root`."""

    root: str
    num_gpus: int = 8
    profile: MockGpuProfile = field(default_factory=MockGpuProfile)
    hive_id: int = 0x1234ABCD5678EF01
    driver_version: str = "6.14.5"
    num_accessible: int = -1  # -1 = all; else only first N GPUs get dev nodes

    def __post_init__(self):
        self.sysfs_root = os.path.join(self.root, "sys")
        self.dev_root = os.path.join(self.root, "dev")
        # per-GPU state: current compute/memory modes
        self._compute: Dict[int, str] = {i: "SPX" for i in range(self.num_gpus)}
        self._memory: Dict[int, str] = {i: "NPS1" for i in range(self.num_gpus)}

    # ------------------------------------------------------------------

    def backend(self) -> SysfsBackend:
        return MockSysfsBackend(self)

    def setup(self) -> None:
        if os.path.exists(self.sysfs_root):
            shutil.rmtree(self.sysfs_root)
        if os.path.exists(self.dev_root):
            shutil.rmtree(self.dev_root)
        os.makedirs(os.path.join(self.sysfs_root, "module", "amdgpu"), exist_ok=True)
        with open(os.path.join(self.sysfs_root, "module", "amdgpu", "version"), "w") as f:
            f.write(self.driver_version + "\n")
        os.makedirs(os.path.join(self.dev_root, "dri"), exist_ok=True)
        self._touch(os.path.join(self.dev_root, "kfd"))
        self._layout()

    def pci_addr(self, gpu: int) -> str:
        return f"0000:{0x0c + gpu:02x}:00.0"

    def unique_id(self, gpu: int) -> str:
        return f"{0x5f0 + gpu:04x}deadbeef{gpu:04x}"

    # -- partition mutation (what the amdgpu driver does on a mode write) --

    def set_compute_partition(self, gpu: int, mode: str) -> None:
        mode = mode.upper()
        avail = [m.strip() for m in self.profile.available_compute.split(",")]
        if mode not in avail:
            raise OSError(f"invalid argument: {mode} not in {avail}")
        self._compute[gpu] = mode
        self._layout()

    def add_kfd_process(self, pid: int, gpu: int, vram_bytes: int = 1 << 20) -> None:
        """Emulate a process holding KFD VRAM on `gpu` (makes the quiesce
        check refuse partition switches, like a busy real GPU)."""
        # gpu_id convention mirrors _layout: 10000 + primary kfd node id
        node_id = None
        kfd_nodes = os.path.join(self.sysfs_root, "class", "kfd", "kfd", "topology", "nodes")
        for nd in sorted(os.listdir(kfd_nodes), key=lambda x: int(x) if x.isdigit() else 1 << 30):
            props_path = os.path.join(kfd_nodes, nd, "properties")
            if not os.path.exists(props_path):
                continue
            props = dict(
                line.split()[:2] for line in open(props_path) if len(line.split()) >= 2
            )
            pci = self.pci_addr(gpu)
            bus = int(pci.split(":")[1], 16)
            if int(props.get("location_id", "0") or 0) == (bus << 8) and int(
                props.get("simd_count", "0") or 0
            ) > 0:
                node_id = int(nd)
                break
        if node_id is None:
            raise RuntimeError(f"no kfd node for gpu {gpu}")
        pdir = os.path.join(self.sysfs_root, "class", "kfd", "kfd", "proc", str(pid))
        os.makedirs(pdir, exist_ok=True)
        with open(os.path.join(pdir, f"vram_{10000 + node_id}"), "w") as f:
            f.write(str(vram_bytes))

    def remove_kfd_process(self, pid: int) -> None:
        pdir = os.path.join(self.sysfs_root, "class", "kfd", "kfd", "proc", str(pid))
        shutil.rmtree(pdir, ignore_errors=True)

    def setup_vfio(self) -> None:
        """Create the PCI-driver rebind surface for VFIO demos/tests:
        per-GPU PCI device dirs with driver symlinks (amdgpu), IOMMU
        groups, and /dev/vfio nodes — consumed by vfio_manager()."""
        import glob as _glob

        drivers = os.path.join(self.sysfs_root, "bus", "pci", "drivers")
        for drv in ("amdgpu", "vfio-pci"):
            os.makedirs(os.path.join(drivers, drv), exist_ok=True)
        vdev = os.path.join(self.dev_root, "vfio")
        os.makedirs(vdev, exist_ok=True)
        if not os.path.exists(os.path.join(vdev, "vfio")):
            open(os.path.join(vdev, "vfio"), "w").close()
        for gpu in range(self.num_gpus):
            pci = self.pci_addr(gpu)
            devdir = os.path.join(self.sysfs_root, "bus", "pci", "devices", pci)
            os.makedirs(devdir, exist_ok=True)
            drv_link = os.path.join(devdir, "driver")
            if not os.path.islink(drv_link):
                os.symlink(os.path.join("..", "..", "drivers", "amdgpu"), drv_link)
            group = str(40 + gpu)
            gdir = os.path.join(self.sysfs_root, "kernel", "iommu_groups", group)
            os.makedirs(gdir, exist_ok=True)
            glink = os.path.join(devdir, "iommu_group")
            if not os.path.islink(glink):
                os.symlink(gdir, glink)
            if not os.path.exists(os.path.join(vdev, group)):
                open(os.path.join(vdev, group), "w").close()

    def _vfio_rebind(self, pci: str, driver: str) -> None:
        devdir = os.path.join(self.sysfs_root, "bus", "pci", "devices", pci)
        link = os.path.join(devdir, "driver")
        if os.path.islink(link):
            os.unlink(link)
        os.symlink(os.path.join("..", "..", "drivers", driver), link)

    def vfio_manager(self):
        """A VfioPciManager over this mock tree (functional rebind)."""
        from ..plugin.vfio import VfioPciManager

        self.setup_vfio()
        return VfioPciManager(
            sysfs_root=self.sysfs_root,
            dev_root=self.dev_root,
            busy_check=lambda pci: False,
            rebind_hook=self._vfio_rebind,
        )

    def set_memory_partition(self, gpu: int, mode: str) -> None:
        mode = mode.upper()
        avail = [m.strip() for m in self.profile.available_memory.split(",")]
        if mode not in avail:
            raise OSError(f"invalid argument: {mode} not in {avail}")
        self._memory[gpu] = mode
        self._layout()

    def compute_partition(self, gpu: int) -> str:
        return self._compute[gpu]

    # -- layout engine ----------------------------------------------------

    def _card_minor(self, gpu: int, part: int) -> int:
        # verified MI355X layout: amdgpu pre-allocates 8 card slots per GPU
        # (cards 0,8,16,... are the physical GPUs; +1..+7 partition slots)
        return 8 * gpu + part

    def _render_minor(self, gpu: int, part: int) -> int:
        return 128 + self._card_minor(gpu, part)

    def _layout(self) -> None:
        drm = os.path.join(self.sysfs_root, "class", "drm")
        kfd_nodes = os.path.join(self.sysfs_root, "class", "kfd", "kfd", "topology", "nodes")
        for d in (drm, kfd_nodes):
            if os.path.exists(d):
                shutil.rmtree(d)
            os.makedirs(d)
        # remove stale render dev nodes
        dri = os.path.join(self.dev_root, "dri")
        if os.path.exists(dri):
            shutil.rmtree(dri)
        os.makedirs(dri)

        kfd_node_id = 1  # node 0 reserved for the CPU node
        self._write_cpu_kfd_node(os.path.join(kfd_nodes, "0"))
        node_of: Dict[int, int] = {}  # gpu -> primary kfd node id
        gpu_nodes: List[int] = []
        node_meta: List[dict] = []

        for gpu in range(self.num_gpus):
            nparts = COMPUTE_MODE_PARTITIONS.get(self._compute[gpu], 1)
            for part in range(nparts):
                cm = self._card_minor(gpu, part)
                rm = self._render_minor(gpu, part)
                devdir = os.path.join(drm, f"card{cm}", "device")
                os.makedirs(devdir, exist_ok=True)
                self._write_pci_device(devdir, gpu, part, nparts)
                # renderD entry points at same device dir
                rd = os.path.join(drm, f"renderD{rm}")
                os.makedirs(rd, exist_ok=True)
                rdev = os.path.join(rd, "device")
                if not os.path.exists(rdev):
                    os.symlink(os.path.join("..", f"card{cm}", "device"), rdev)
                # dev nodes (only for accessible GPUs — multi-tenant nodes
                # expose every GPU's sysfs but only assigned /dev/dri nodes)
                if self.num_accessible < 0 or gpu < self.num_accessible:
                    self._touch(os.path.join(dri, f"card{cm}"))
                    self._touch(os.path.join(dri, f"renderD{rm}"))
                node_meta.append(
                    {"gpu": gpu, "part": part, "nparts": nparts, "render_minor": rm,
                     "node_id": kfd_node_id}
                )
                if part == 0:
                    node_of[gpu] = kfd_node_id
                gpu_nodes.append(kfd_node_id)
                kfd_node_id += 1

        # KFD nodes + full-mesh xGMI io_links between primary nodes
        for meta in node_meta:
            nd = os.path.join(kfd_nodes, str(meta["node_id"]))
            os.makedirs(os.path.join(nd, "io_links"), exist_ok=True)
            gpu, part, nparts = meta["gpu"], meta["part"], meta["nparts"]
            pci = self.pci_addr(gpu)
            domain = int(pci.split(":")[0], 16)
            bus = int(pci.split(":")[1], 16)
            location_id = bus << 8
            props = {
                "simd_count": self.profile.simd_count // nparts,
                "mem_banks_count": 1,
                "io_links_count": 0,
                "cpu_cores_count": 0,
                "gfx_target_version": self.profile.gfx_target_version,
                "drm_render_minor": meta["render_minor"],
                "location_id": location_id,
                "domain": domain,
                "hive_id": self.hive_id if self.num_gpus > 1 else 0,
            }
            li = 0
            if part == 0:
                for peer_gpu, peer_node in node_of.items():
                    if peer_gpu == meta["gpu"] or peer_node is None:
                        continue
                    for _ in range(self.profile.xgmi_links_per_pair):
                        ld = os.path.join(nd, "io_links", str(li))
                        os.makedirs(ld, exist_ok=True)
                        with open(os.path.join(ld, "properties"), "w") as f:
                            f.write(f"type {IOLINK_TYPE_XGMI}\n")
                            f.write(f"node_from {meta['node_id']}\n")
                            f.write(f"node_to {peer_node}\n")
                            f.write("weight 15\n")
                        li += 1
            props["io_links_count"] = li
            with open(os.path.join(nd, "properties"), "w") as f:
                for k, v in props.items():
                    f.write(f"{k} {v}\n")
            # KFD gpu_id (consumed by the busy/quiesce check)
            with open(os.path.join(nd, "gpu_id"), "w") as f:
                f.write(str(10000 + meta["node_id"]))

        # NOTE: the loop above only creates links gpu->peers with node ids
        # already assigned (node_of filled incrementally); re-walk to make the
        # mesh symmetric.
        self._symmetrize_links(kfd_nodes, node_of)

    def _symmetrize_links(self, kfd_nodes: str, node_of: Dict[int, int]) -> None:
        for gpu, node in node_of.items():
            nd = os.path.join(kfd_nodes, str(node), "io_links")
            existing = set()
            for d in os.listdir(nd) if os.path.exists(nd) else []:
                p = os.path.join(nd, d, "properties")
                with open(p) as f:
                    for line in f:
                        if line.startswith("node_to"):
                            existing.add(int(line.split()[1]))
            li = len(os.listdir(nd)) if os.path.exists(nd) else 0
            for peer_gpu, peer_node in node_of.items():
                if peer_gpu == gpu or peer_node in existing:
                    continue
                for _ in range(self.profile.xgmi_links_per_pair):
                    ld = os.path.join(nd, str(li))
                    os.makedirs(ld, exist_ok=True)
                    with open(os.path.join(ld, "properties"), "w") as f:
                        f.write(f"type {IOLINK_TYPE_XGMI}\n")
                        f.write(f"node_from {node}\n")
                        f.write(f"node_to {peer_node}\n")
                        f.write("weight 15\n")
                    li += 1

    def _write_pci_device(self, devdir: str, gpu: int, part: int, nparts: int) -> None:
        pci = self.pci_addr(gpu)
        files = {
            "vendor": "0x1002",
            "device": self.profile.device_id,
            "class": "0x038000",
            "uevent": f"DRIVER=amdgpu\nPCI_CLASS=38000\nPCI_SLOT_NAME={pci}\n",
            "unique_id": self.unique_id(gpu),
            "serial_number": f"MOCKSER{gpu:04d}",
            "vbios_version": "113-MOCK355-001",
            "numa_node": str(gpu // 4),
            "mem_info_vram_total": str(self.profile.vram_bytes // (nparts if part else 1)
                                       if part else self.profile.vram_bytes),
            "mem_info_vram_used": "0",
            "current_compute_partition": self._compute[gpu],
            "current_memory_partition": self._memory[gpu],
            "available_compute_partition": self.profile.available_compute,
            "available_memory_partition": self.profile.available_memory,
        }
        for name, content in files.items():
            with open(os.path.join(devdir, name), "w") as f:
                f.write(str(content) + ("\n" if not str(content).endswith("\n") else ""))
        if self.num_gpus > 1:
            hd = os.path.join(devdir, "xgmi_hive_info")
            os.makedirs(hd, exist_ok=True)
            with open(os.path.join(hd, "xgmi_hive_id"), "w") as f:
                f.write(f"{self.hive_id}\n")

    def _write_cpu_kfd_node(self, nd: str) -> None:
        os.makedirs(nd, exist_ok=True)
        with open(os.path.join(nd, "properties"), "w") as f:
            f.write("cpu_cores_count 128\nsimd_count 0\ndrm_render_minor -1\n")

    @staticmethod
    def _touch(path: str) -> None:
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "w") as f:
            f.write("")


class MockSysfsBackend(SysfsBackend):
    """SysfsBackend whose partition writes go through the MockTree layout
    engine (a plain file write cannot re-layout the tree the way the real
    amdgpu driver does)."""

    def __init__(self, tree: MockTree):
        super().__init__(sysfs_root=tree.sysfs_root, dev_root=tree.dev_root)
        self._tree = tree

    def set_compute_partition(self, minor: int, mode: str) -> None:
        # primary card minors are 8*gpu in the (verified-real) layout
        self._tree.set_compute_partition(minor // 8, mode)

    def set_memory_partition(self, minor: int, mode: str) -> None:
        self._tree.set_memory_partition(minor // 8, mode)
