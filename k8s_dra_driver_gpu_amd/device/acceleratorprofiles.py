"""amd-smi accelerator-partition profile parsing.

The MI355X analog of the reference's MIG profile/placement inspection
(``nvlib.go:1202-1277 inspectMigProfilesAndPlacements``): ``amd-smi
partition --accelerator`` lists, per GPU, every accelerator partition
profile the platform supports (SPX/DPX/QPX/CPX), which memory-partition
modes each is compatible with, how many partitions it yields, and the
resource split (XCC/DECODER/DMA/JPEG instances, shared counts). The row
whose type carries a ``*`` is the currently active profile.

This is an *enrichment and cross-check* source on top of the sysfs
``available_compute_partition`` file the device layer primarily reads:
DeviceLib consults it (when the CLI is present) to validate a requested
mode switch against the platform's profile table before writing, and the
ResourceSlice can attribute resource counts per partition from it.

Output format (captured from a real MI355X, amd-smi 26.2.1 —
tests/fixtures/amd_smi_partition_accelerator.txt):

    ACCELERATOR_PARTITION_PROFILES:
    GPU_ID  PROFILE_INDEX  MEMORY_PARTITION_CAPS  ACCELERATOR_TYPE  PARTITION_ID ...
    0       0              NPS1                   SPX*              0 ...
            1              NPS1                   DPX               N/A ...
                                                  ...continuation resource rows...
"""

from __future__ import annotations

import logging
import re
import shutil
import subprocess
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

logger = logging.getLogger("amddra.accprofiles")


@dataclass
class AcceleratorProfile:
    index: int
    memory_caps: List[str]
    type: str  # SPX/DPX/QPX/CPX
    current: bool  # the '*'-marked active profile
    num_partitions: int
    # resource type -> (instances, shared)
    resources: Dict[str, Tuple[int, int]] = field(default_factory=dict)


_HEADER_RE = re.compile(r"^GPU_ID\s+PROFILE_INDEX\s+MEMORY_PARTITION_CAPS")
_PROFILE_RE = re.compile(
    r"^(?P<gpu>\d+)?\s+(?P<idx>\d+)\s+(?P<mem>[A-Z0-9/,]+)\s+"
    r"(?P<type>[A-Z]+)(?P<star>\*)?\s+(?P<pid>\S+)\s+(?P<nparts>\d+)\s+"
    r"(?P<nres>\d+)\s+(?P<ridx>\d+)\s+(?P<rtype>[A-Z_]+)\s+"
    r"(?P<rinst>\d+)\s+(?P<rshared>\d+)"
)
_RESOURCE_RE = re.compile(
    r"^\s+(?P<ridx>\d+)\s+(?P<rtype>[A-Z_]+)\s+(?P<rinst>\d+)\s+(?P<rshared>\d+)\s*$"
)


def parse_accelerator_profiles(text: str) -> Dict[int, List[AcceleratorProfile]]:
    """Parse `amd-smi partition --accelerator` output -> {gpu_id: profiles}.

    Tolerates truncated output (profiles parsed so far are returned)."""
    out: Dict[int, List[AcceleratorProfile]] = {}
    gpu: Optional[int] = None
    prof: Optional[AcceleratorProfile] = None
    in_table = False
    for line in text.splitlines():
        if "ACCELERATOR_PARTITION_PROFILES" in line:
            in_table = True
            continue
        if not in_table or not line.strip():
            continue
        if _HEADER_RE.match(line.strip()):
            continue
        m = _PROFILE_RE.match(line)
        if m:
            if m.group("gpu") is not None:
                gpu = int(m.group("gpu"))
            if gpu is None:
                continue
            prof = AcceleratorProfile(
                index=int(m.group("idx")),
                memory_caps=[c for c in re.split(r"[/,]", m.group("mem")) if c],
                type=m.group("type"),
                current=m.group("star") is not None,
                num_partitions=int(m.group("nparts")),
            )
            prof.resources[m.group("rtype")] = (
                int(m.group("rinst")), int(m.group("rshared"))
            )
            out.setdefault(gpu, []).append(prof)
            continue
        m = _RESOURCE_RE.match(line)
        if m and prof is not None:
            prof.resources[m.group("rtype")] = (
                int(m.group("rinst")), int(m.group("rshared"))
            )
    return out


def read_accelerator_profiles(
    cli: str = "", timeout: float = 30.0
) -> Optional[Dict[int, List[AcceleratorProfile]]]:
    """Run the CLI and parse; None when amd-smi is unavailable/fails (mock
    trees and minimal containers)."""
    cli = cli or shutil.which("amd-smi") or ""
    if not cli:
        return None
    try:
        r = subprocess.run(
            [cli, "partition", "--accelerator"],
            capture_output=True, text=True, timeout=timeout, check=False,
        )
    except Exception:
        logger.debug("amd-smi partition --accelerator failed", exc_info=True)
        return None
    if r.returncode != 0:
        logger.debug("amd-smi partition rc=%s: %s", r.returncode, r.stderr[:200])
        return None
    return parse_accelerator_profiles(r.stdout)


def profile_for_mode(
    profiles: List[AcceleratorProfile], mode: str
) -> Optional[AcceleratorProfile]:
    for p in profiles:
        if p.type == mode.upper():
            return p
    return None
