"""CPX compute-partition control — the MI355X hard-isolation story.

KFD silently ignores per-queue CU masks on multi-XCD gfx9 parts, so the
only HARD core partition on MI355X is amdgpu's compute partitioning: in
CPX mode each of the 8 XCDs becomes its own KFD node (32 CUs, own render
node), and a container that is only given that partition's /dev/dri nodes
physically cannot touch the other 224 CUs.  This is the MIG-strategy
analog (reference rm/device_map.go:37-118 builds MIG device maps the same
way: an external mode switch, then the plugin advertises what exists) and
the vdev cu_mask contract of the DCU plugin (hygon/dcu/corealloc.go:8-77)
with the enforcement moved into the hardware.

The mode is switched by writing ``current_compute_partition`` under the
card's sysfs device dir.  The write fails with EBUSY while any process
holds the GPU, so the plugin applies it at startup (before any HIP use)
when --compute-partition is not "keep"; operators can also pre-partition
with amd-smi.  Memory partitioning (NPS) is left alone.
"""
from __future__ import annotations

import glob
import logging
import os
from typing import List, Optional

log = logging.getLogger(__name__)

VALID_MODES = ("SPX", "DPX", "QPX", "CPX")
DRM_CLASS = "/sys/class/drm"


def partition_files(drm_root: str = DRM_CLASS) -> List[str]:
    """current_compute_partition files of all amdgpu cards (one per
    physical card; CPX partitions share their parent's file)."""
    out = []
    for path in sorted(glob.glob(
            os.path.join(drm_root, "card*", "device",
                         "current_compute_partition"))):
        out.append(path)
    return out


def read_mode(path: str) -> Optional[str]:
    try:
        with open(path) as f:
            return f.read().strip() or None
    except OSError:
        return None


def write_mode(path: str, mode: str) -> bool:
    """Returns True when the card now reports the requested mode."""
    mode = mode.upper()
    if mode not in VALID_MODES:
        raise ValueError(f"invalid compute partition mode {mode!r}")
    if read_mode(path) == mode:
        return True
    try:
        with open(path, "w") as f:
            f.write(mode)
    except OSError as e:
        log.error("cannot set compute partition %s on %s: %s", mode, path, e)
        return False
    return read_mode(path) == mode


def apply_mode(mode: str, drm_root: str = DRM_CLASS) -> bool:
    """Apply the desired mode to every card; True iff all succeeded."""
    mode = mode.upper()
    if mode == "KEEP":
        return True
    files = partition_files(drm_root)
    if not files:
        log.warning("no compute-partition sysfs files under %s", drm_root)
        return False
    ok = True
    for path in files:
        before = read_mode(path)
        if write_mode(path, mode):
            log.info("compute partition %s: %s -> %s", path, before, mode)
        else:
            ok = False
    return ok
