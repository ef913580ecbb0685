"""DeviceLib — the device layer facade (the ``nvlib.go`` analog).

Parity targets from the reference (``cmd/gpu-kubelet-plugin/nvlib.go``):
enumeration of GPUs + partitions into allocatable devices (:196-339), dynamic
partition create/delete with rollback (:926-1154 createMigDevice/
deleteMigDevice), partition-spec lookup (:1284-1363 FindMigDevBySpec),
whole-GPU mode-toggle gating (:1156-1197, :1461-1506), and cached handles
(the NVML-handle-cache analog — enumeration results are cached and
invalidated on partition mutations, :118-124,882-923).

MI355X semantics differ from MIG and the model embraces that: a compute-mode
switch (SPX<->DPX/QPX/CPX) is a **whole-GPU transition** that requires the GPU
to be idle and re-enumerates KFD nodes, so "create partition" means "ensure
parent GPU is in the requested mode, then address partition #i", and "delete"
means "when the last partition claim is gone, return the GPU to SPX".
"""

from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import List, Optional, Tuple

from ..api.configs import (
    COMPUTE_MODE_PARTITIONS,
    MEMORY_MODE_MIN_PARTITIONS,
    NPS1,
    SPX,
)
from .info import GpuInfo, GpuTopology, PartitionInfo, format_partition_name
from .sysfs import SysfsBackend, SysfsError


class DeviceError(RuntimeError):
    pass


class DeviceBusyError(DeviceError):
    """A partition-mode switch was requested while processes hold the GPU
    (the reference's in-use rejection, nvlib.go:1472-1506). Retryable once
    workloads are quiesced."""


@dataclass(frozen=True)
class PartitionSpec:
    """Identity of a requested partition (the MigSpecTuple analog,
    ref mig.go:37-70)."""

    parent_uuid: str
    compute_mode: str
    index: int

    def canonical_name(self, parent_minor: int) -> str:
        return format_partition_name(parent_minor, self.compute_mode, self.index)


class DeviceLib:
    def __init__(self, backend: Optional[SysfsBackend] = None):
        self.backend = backend or SysfsBackend()
        self._lock = threading.RLock()
        self._cache: Optional[Tuple[List[GpuInfo], List[PartitionInfo], GpuTopology]] = None
        # accelerator profile table (amd-smi), read once on first use
        self._acc_profiles: Optional[dict] = None

    # -- enumeration (cached) ---------------------------------------------

    def invalidate(self) -> None:
        with self._lock:
            self._cache = None

    def _enumerate(self) -> Tuple[List[GpuInfo], List[PartitionInfo], GpuTopology]:
        with self._lock:
            if self._cache is None:
                self._cache = self.backend.enumerate()
            return self._cache

    def gpus(self) -> List[GpuInfo]:
        return list(self._enumerate()[0])

    def live_partitions(self) -> List[PartitionInfo]:
        return list(self._enumerate()[1])

    def topology(self) -> GpuTopology:
        return self._enumerate()[2]

    def gpu_by_uuid(self, uuid: str) -> Optional[GpuInfo]:
        for g in self.gpus():
            if g.uuid == uuid:
                return g
        return None

    def gpu_by_minor(self, minor: int) -> Optional[GpuInfo]:
        for g in self.gpus():
            if g.minor == minor:
                return g
        return None

    # -- partition support queries ----------------------------------------

    def supported_compute_modes(self, gpu: GpuInfo) -> List[str]:
        avail = self.backend.available_compute_partitions(gpu.minor)
        return [m for m in avail if m in COMPUTE_MODE_PARTITIONS]

    def supported_memory_modes(self, gpu: GpuInfo) -> List[str]:
        return self.backend.available_memory_partitions(gpu.minor)

    def accelerator_profile_error(
        self, gpu: GpuInfo, mode: str, memory_mode: str = ""
    ) -> Optional[str]:
        """Cross-check a requested partition switch against the platform's
        accelerator profile table (`amd-smi partition --accelerator`, the
        inspectMigProfilesAndPlacements analog). Returns an error string
        when the table explicitly rejects the combination; None when OK or
        unverifiable (CLI absent — mock trees, minimal containers)."""
        if self.backend.sysfs_root != "/sys":
            return None  # mock: sysfs availability is the only source
        if self._acc_profiles is None:
            from .acceleratorprofiles import read_accelerator_profiles

            self._acc_profiles = read_accelerator_profiles() or {}
        profiles = self._acc_profiles.get(gpu.index)
        if not profiles:
            return None
        from .acceleratorprofiles import profile_for_mode

        prof = profile_for_mode(profiles, mode)
        if prof is None:
            return (
                f"mode {mode} not in the platform's accelerator profile table "
                f"({[p.type for p in profiles]})"
            )
        if memory_mode and prof.memory_caps and memory_mode.upper() not in prof.memory_caps:
            return (
                f"memory mode {memory_mode} not supported by the {mode} "
                f"profile (caps: {prof.memory_caps})"
            )
        return None

    def possible_partitions(self, gpu: GpuInfo) -> List[PartitionSpec]:
        """All partition placements this GPU could host (for ResourceSlice
        publication; analog of inspectMigProfilesAndPlacements,
        nvlib.go:1202-1277)."""
        specs = []
        for mode in self.supported_compute_modes(gpu):
            n = COMPUTE_MODE_PARTITIONS[mode]
            if n <= 1:
                continue
            for i in range(n):
                specs.append(PartitionSpec(gpu.uuid, mode, i))
        return specs

    # -- partition lifecycle -----------------------------------------------

    def create_partition(
        self, spec: PartitionSpec, memory_mode: str = ""
    ) -> PartitionInfo:
        """Ensure the parent GPU is in `spec.compute_mode` and return the live
        partition for `spec.index`.

        The caller (DeviceState) must hold the node prepare lock and must have
        verified no OTHER prepared device overlaps this GPU in a different
        mode — this method enforces it again defensively.
        """
        with self._lock:
            gpu = self.gpu_by_uuid(spec.parent_uuid)
            if gpu is None:
                raise DeviceError(f"no GPU with uuid {spec.parent_uuid}")
            nparts = COMPUTE_MODE_PARTITIONS.get(spec.compute_mode)
            if nparts is None:
                raise DeviceError(f"unknown compute mode {spec.compute_mode}")
            if not (0 <= spec.index < nparts):
                raise DeviceError(
                    f"partition index {spec.index} out of range for {spec.compute_mode}"
                )
            if spec.compute_mode not in self.supported_compute_modes(gpu):
                raise DeviceError(
                    f"GPU {gpu.canonical_name} does not support mode {spec.compute_mode}"
                )
            if gpu.compute_partition != spec.compute_mode:
                if gpu.compute_partition != SPX:
                    raise DeviceError(
                        f"GPU {gpu.canonical_name} is in mode {gpu.compute_partition}; "
                        f"cannot switch to {spec.compute_mode} while partitioned"
                    )
                self._assert_quiesced(gpu)
                perr = self.accelerator_profile_error(gpu, spec.compute_mode, memory_mode)
                if perr:
                    raise DeviceError(f"partition profile check: {perr}")
                if memory_mode and memory_mode != gpu.memory_partition:
                    min_parts = MEMORY_MODE_MIN_PARTITIONS.get(memory_mode, 1)
                    if nparts < min_parts:
                        raise DeviceError(
                            f"memory mode {memory_mode} requires >= {min_parts}-way compute split"
                        )
                    self.backend.set_memory_partition(gpu.minor, memory_mode)
                try:
                    self.backend.set_compute_partition(gpu.minor, spec.compute_mode)
                except SysfsError:
                    # rollback the memory-mode change (partial-prepare rollback,
                    # ref device_state.go:338-387)
                    if memory_mode and memory_mode != gpu.memory_partition:
                        try:
                            self.backend.set_memory_partition(gpu.minor, gpu.memory_partition)
                        except SysfsError:
                            pass
                    raise
                self.invalidate()
            part = self.find_partition(spec)
            if part is None:
                raise DeviceError(
                    f"partition {spec.index} of {gpu.canonical_name} not exposed after "
                    f"switch to {spec.compute_mode}"
                )
            return part

    def find_partition(self, spec: PartitionSpec) -> Optional[PartitionInfo]:
        """FindMigDevBySpec analog (nvlib.go:1284-1363).

        Index 0 of a partitioned GPU is the primary card itself (the kernel
        re-purposes it as partition 0); indexes >= 1 are the extra cards.
        """
        gpu = self.gpu_by_uuid(spec.parent_uuid)
        if gpu is None or gpu.compute_partition != spec.compute_mode:
            return None
        nparts = COMPUTE_MODE_PARTITIONS.get(spec.compute_mode, 1)
        if spec.index == 0:
            return PartitionInfo(
                parent_uuid=gpu.uuid,
                parent_minor=gpu.minor,
                compute_mode=spec.compute_mode,
                memory_mode=gpu.memory_partition,
                index=0,
                uuid=f"{gpu.uuid}-p0",
                render_minor=gpu.render_minor,
                render_path=gpu.render_path,
                vram_bytes=gpu.vram_bytes // nparts,
                xcd_count=gpu.xcd_count // nparts,
            )
        for p in self.live_partitions():
            if p.parent_uuid == spec.parent_uuid and p.index == spec.index:
                return p
        return None

    def _assert_quiesced(self, gpu: GpuInfo) -> None:
        """Refuse a whole-GPU mode switch while KFD processes hold it."""
        try:
            pids = self.backend.gpu_busy_pids(gpu.minor)
        except Exception:
            pids = []  # unreadable proc tree: the kernel's EBUSY is the backstop
        if pids:
            raise DeviceBusyError(
                f"GPU {gpu.canonical_name} busy (pids {pids}): quiesce all "
                f"workloads before a partition-mode switch"
            )

    def maybe_reset_partition_mode(self, parent_uuid: str) -> bool:
        """Return the GPU to SPX/NPS1 (deleteMigDevice +
        maybeDisableMigMode analog, nvlib.go:1056-1197). Caller guarantees no
        partition of this GPU is still prepared. Returns True if a switch
        happened."""
        with self._lock:
            gpu = self.gpu_by_uuid(parent_uuid)
            if gpu is None:
                raise DeviceError(f"no GPU with uuid {parent_uuid}")
            switched = False
            if gpu.compute_partition != SPX:
                self._assert_quiesced(gpu)
                self.backend.set_compute_partition(gpu.minor, SPX)
                switched = True
            if gpu.memory_partition != NPS1:
                self.backend.set_memory_partition(gpu.minor, NPS1)
                switched = True
            if switched:
                self.invalidate()
            return switched
