"""GPU sharing managers: TimeSlicing and XCD spatial confinement.

The reference implements sharing via ``nvidia-smi compute-policy
--set-timeslice`` + compute mode, and an MPS control-daemon Deployment
(``cmd/gpu-kubelet-plugin/sharing.go:75-178,214-436``).  MI355X has neither
MPS nor a per-device timeslice knob, so the AMD-native strategies are:

* **TimeSlicing** — amdgpu's hardware scheduler (HWS) time-slices compute
  queues between processes by default; there is **no per-device interval
  knob** on ROCm (verified against amd-smi 26.2.1: ``amd-smi set`` exposes
  ``--compute-partition`` but no timeslice option; the KFD quantum is a
  node-global module parameter).  The requested interval is therefore
  surfaced to the container via the namespaced ``AMDDRA_SHARING`` env var
  for observability, and sharing works because HWS already round-robins
  queues — matching the reference's semantic of "multiple containers of one
  claim share the GPU" (gpu-test2) without any device mutation.

* **SpatialPartitioning** — confine the claim's containers to a subset of
  the chip's XCDs using ``ROC_GLOBAL_CU_MASK``, the documented HIP/ROCclr
  environment variable (present and parsed by this image's
  ``libamdhip64.so``: "Setting CU mask 0x%s for hardware queue"): a hex
  bitmask with one bit per CU applied to every queue the process creates.
  MI355X has 256 CUs in 8 XCDs (32 CUs/XCD), so ``xcdCount=k`` emits a mask
  with the low ``k*32`` bits set.  Hard partition isolation (separate
  memory, separate device nodes) is the CPX partition path, not this.

Every env var emitted here is either documented ROCm surface
(``ROC_GLOBAL_CU_MASK``) or clearly ours (``AMDDRA_*``) — no pseudo-knobs.
"""

from __future__ import annotations

import logging
from typing import List, Optional

from ..api.configs import DEFAULT_INTERVAL, GpuSharing

logger = logging.getLogger("amddra.sharing")

# MI355X: 8 XCDs x 32 CUs (aid: gfx950). Masks are sized for this geometry;
# other CDNA parts would inject their own totals via the constructor.
XCD_COUNT = 8
CUS_PER_XCD = 32


def cu_mask_hex(num_cus: int) -> str:
    """Hex bitmask (``0x…``) with the low `num_cus` bits set — the format
    ROC_GLOBAL_CU_MASK parses (one bit per CU, applied to all queues)."""
    if num_cus <= 0:
        num_cus = 1
    return hex((1 << num_cus) - 1)


class SharingManager:
    """Applies a sharing config for one prepared device and returns container
    env edits. All strategies resolve to env only: ROCm has no per-device
    sharing mutation to perform or undo (unlike the reference's
    nvidia-smi/MPS paths), so prepare stays exec-free and remove() is a
    no-op kept for interface parity with the reference's teardown hook."""

    def __init__(self, xcds: int = XCD_COUNT, cus_per_xcd: int = CUS_PER_XCD):
        self.xcds = xcds
        self.cus_per_xcd = cus_per_xcd

    def apply(self, cfg, gpu=None, partition=None) -> List[str]:
        sharing: Optional[GpuSharing] = getattr(cfg, "sharing", None)
        env: List[str] = []
        if gpu is not None and partition is None:
            env.append(f"AMDDRA_GPU_UUID={gpu.uuid}")
        if partition is not None:
            env.append(f"AMDDRA_PARTITION={partition.canonical_name}")
        if sharing is None:
            return env
        if sharing.is_time_slicing():
            interval = (
                sharing.time_slicing_config.interval
                if sharing.time_slicing_config
                else DEFAULT_INTERVAL
            )
            # observability only: HWS time-slices by default; there is no
            # per-device interval knob on ROCm (see module docstring)
            env.append(f"AMDDRA_SHARING=TimeSlicing:{interval}")
        elif sharing.is_spatial():
            sc = sharing.spatial_partitioning_config
            if sc and sc.xcd_count:
                n_xcds = min(sc.xcd_count, self.xcds)
            else:
                pct = sc.default_xcd_percentage if sc else 100
                # XCD granularity: round down, at least one XCD
                n_xcds = max(1, (pct * self.xcds) // 100)
            env.append(f"AMDDRA_SHARING=Spatial:xcd={n_xcds}")
            if n_xcds < self.xcds:
                env.append(
                    f"ROC_GLOBAL_CU_MASK={cu_mask_hex(n_xcds * self.cus_per_xcd)}"
                )
            # n_xcds == full chip: no mask needed (and none emitted)
        return env

    def remove(self, prepared_device) -> None:
        """No device-level state to undo: sharing is env-only on ROCm (env
        dies with the container)."""
        return None
