"""AMD MI355X device semantics.

MI355X analogue of the reference's NVIDIA vendor module
(/root/reference/pkg/device/nvidia/device.go:15-177): resource names, the
use/nouse GPU-type filters, NUMA binding assertion, and per-container
resource-request extraction with the default-mem / percentage fallback rules.

Resource API (pod spec `resources.limits`):
    amd.com/gpu                number of vGPU slices
    amd.com/gpumem             HBM MiB per slice (whole MI355X = 294912)
    amd.com/gpumem-percentage  HBM percent per slice
    amd.com/gpucores           CU percent per slice (100 = all 256 CUs)
    amd.com/priority           task priority (0 high, 1 low)
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Tuple

from ..utils.types import (
    IN_REQUEST_DEVICES,
    SUPPORT_DEVICES,
    ContainerDeviceRequest,
    ContainerSpec,
    DeviceUsage,
)

AMD_DEVICE_TYPE = "AMD"          # request Type discriminator (like "NVIDIA")
AMD_COMMON_WORD = "GPU"

HANDSHAKE_ANNO = "amd.io/node-handshake"
REGISTER_ANNO = "amd.io/node-amd-register"

GPU_IN_USE_ANNO = "amd.com/use-gputype"
GPU_NO_USE_ANNO = "amd.com/nouse-gputype"
NUMA_BIND_ANNO = "amd.com/numa-bind"

# env var the webhook injects for the interceptor's priority gate
TASK_PRIORITY_ENV = "VGPU_TASK_PRIORITY"
CORE_LIMIT_SWITCH_ENV = "GPU_CORE_UTILIZATION_POLICY"


@dataclass
class AMDConfig:
    resource_name: str = "amd.com/gpu"
    resource_mem: str = "amd.com/gpumem"
    resource_mem_percentage: str = "amd.com/gpumem-percentage"
    resource_cores: str = "amd.com/gpucores"
    resource_priority: str = "amd.com/priority"
    default_mem: int = 0      # MiB; 0 => percentage fallback to 100%
    default_cores: int = 0    # percent


class AMDDevices:
    def __init__(self, config: AMDConfig = None):
        self.config = config or AMDConfig()
        IN_REQUEST_DEVICES[AMD_DEVICE_TYPE] = "vgpu.amd.com/devices-to-allocate"
        SUPPORT_DEVICES[AMD_DEVICE_TYPE] = "vgpu.amd.com/devices-allocated"

    # -- webhook ----------------------------------------------------------
    def mutate_admission(self, ctr: ContainerSpec) -> bool:
        """True if this container requests our resource; injects the priority
        env for the interceptor (reference nvidia/device.go:51-62)."""
        prio = ctr.limits.get(self.config.resource_priority)
        if prio is None:
            prio = ctr.requests.get(self.config.resource_priority)
        if prio is not None:
            ctr.env[TASK_PRIORITY_ENV] = str(prio)
        return self.config.resource_name in ctr.limits or (
            self.config.resource_name in ctr.requests
        )

    # -- scheduler type check ---------------------------------------------
    def check_type(
        self, annos: Dict[str, str], d: DeviceUsage, req: ContainerDeviceRequest
    ) -> Tuple[bool, bool, bool]:
        """(recognized, type passes, numa assert) — nvidia/device.go:109-114."""
        if req.type == AMD_DEVICE_TYPE:
            return True, _check_gpu_type(annos, d.type), _assert_numa(annos)
        return False, False, False

    # -- request extraction -----------------------------------------------
    def generate_resource_requests(self, ctr: ContainerSpec) -> ContainerDeviceRequest:
        c = self.config
        v = ctr.limits.get(c.resource_name, ctr.requests.get(c.resource_name))
        if v is None:
            return ContainerDeviceRequest()
        memnum = int(ctr.limits.get(c.resource_mem, ctr.requests.get(c.resource_mem, 0)))
        mempnum = int(
            ctr.limits.get(
                c.resource_mem_percentage, ctr.requests.get(c.resource_mem_percentage, 101)
            )
        )
        # Fallback rules identical to the reference (nvidia/device.go:149-155):
        # neither mem nor percentage set -> default_mem if configured, else 100%.
        if mempnum == 101 and memnum == 0:
            if c.default_mem != 0:
                memnum = c.default_mem
            else:
                mempnum = 100
        corenum = int(
            ctr.limits.get(c.resource_cores, ctr.requests.get(c.resource_cores, c.default_cores))
        )
        return ContainerDeviceRequest(
            nums=int(v),
            type=AMD_DEVICE_TYPE,
            memreq=memnum,
            mem_percentage_req=mempnum,
            coresreq=corenum,
        )


def _check_gpu_type(annos: Dict[str, str], cardtype: str) -> bool:
    """use-gputype whitelist / nouse-gputype blacklist by substring match
    (reference nvidia/device.go:64-96)."""
    card = cardtype.upper()
    inuse = annos.get(GPU_IN_USE_ANNO)
    if inuse is not None:
        return any(val and val.upper() in card for val in inuse.split(","))
    nouse = annos.get(GPU_NO_USE_ANNO)
    if nouse is not None:
        return not any(val and val.upper() in card for val in nouse.split(","))
    return True


def _assert_numa(annos: Dict[str, str]) -> bool:
    v = annos.get(NUMA_BIND_ANNO, "")
    return v.strip().lower() in ("1", "true", "yes", "on")
