"""AmdSmiOperations — the interface every component programs against.

Mirrors the shape of the reference's NvmlOperations seam
(/root/reference/pkg/gpu/nvidia/nvmlutil/nvmlutil.go:30-37) but with
AMD-native semantics: render-node minors instead of nvidia minors, BDF
instead of PciInfo, SPX/DPX/QPX/CPX compute-partition info instead of MIG mode,
RAS/ECC/thermal events instead of Xids.
"""
from __future__ import annotations

import dataclasses
import os
import re
from abc import ABC, abstractmethod
from typing import List, Optional

# Event codes published on the health channel.  These generalize NVIDIA Xids
# (reference health_checker.go:91-97) to AMD-SMI event notifications; numeric
# values follow amdsmi_evt_notification_type_t (amdsmi.h) so a ConfigMap can
# name them by number the same way XID_CONFIG does.
EVT_VMFAULT = 1            # AMDSMI_EVT_NOTIF_VMFAULT — GPU page fault
EVT_THERMAL_THROTTLE = 2   # AMDSMI_EVT_NOTIF_THERMAL_THROTTLE
EVT_GPU_PRE_RESET = 3      # AMDSMI_EVT_NOTIF_GPU_PRE_RESET
EVT_GPU_POST_RESET = 4     # AMDSMI_EVT_NOTIF_GPU_POST_RESET
EVT_PAGE_FAULT_START = 7   # AMDSMI_EVT_NOTIF_PAGE_FAULT_START
# NOTE: this ROCm's amdsmi enum has no RING_HANG; 5/6 are MIGRATE_START/END
# (benign memory-migration traffic) and are deliberately neither armed nor
# monitored.
# Synthetic codes (outside the amdsmi notification enum) raised by the
# polling side of the health checker:
EVT_ECC_UNCORRECTABLE = 48  # uncorrectable ECC count increased (parity with
                            # the reference's default critical Xid 48,
                            # health_checker.go:97)
EVT_XGMI_ERROR = 63         # xGMI link error state (parity: row-remap Xid 63)
EVT_LOST = 79               # device unreachable / fell off the bus (Xid 79)

DEFAULT_HEALTH_CRITICAL_EVENTS = {EVT_ECC_UNCORRECTABLE}
# Events that only raise the Node condition, never device health
# (parity: monitorCriticalXid, health_checker.go:91).
MONITOR_CRITICAL_EVENTS = {
    EVT_VMFAULT,
    EVT_GPU_PRE_RESET,
    EVT_PAGE_FAULT_START,
    EVT_ECC_UNCORRECTABLE,
    EVT_XGMI_ERROR,
    EVT_LOST,
}


@dataclasses.dataclass
class DeviceInfo:
    index: int              # enumeration index within amdsmi
    uuid: str               # amdsmi_get_gpu_device_uuid
    name: str               # market name, e.g. "AMD Instinct MI355X"
    bdf: str                # "0000:0c:00.0"
    render_minor: int       # N of /dev/dri/renderD<N>
    card_minor: int = -1    # N of /dev/dri/card<N>, -1 if unknown
    vram_total_bytes: int = 0
    compute_partition: str = "SPX"   # SPX | DPX | CPX
    memory_partition: str = "NPS1"
    partition_id: int = 0   # which partition of the physical die this
                            # enumerated device is (0 for SPX)
    physical_index: int = 0  # index of the physical die (partitions share it)


@dataclasses.dataclass
class GpuActivity:
    gfx_percent: float
    umc_percent: float      # memory-controller busy
    mm_percent: float


@dataclasses.dataclass
class MemoryInfo:
    total_bytes: int
    used_bytes: int


@dataclasses.dataclass
class Event:
    device_uuid: str        # "" => unattributed, applies to all devices
    code: int               # EVT_* code
    message: str = ""


class AmdSmiOperations(ABC):
    """The seam.  All methods raise RuntimeError on library failure."""

    @abstractmethod
    def init(self) -> None: ...

    @abstractmethod
    def shutdown(self) -> None: ...

    @abstractmethod
    def device_count(self) -> int: ...

    @abstractmethod
    def device_info(self, index: int) -> DeviceInfo: ...

    @abstractmethod
    def memory_info(self, index: int) -> MemoryInfo: ...

    @abstractmethod
    def gpu_activity(self, index: int) -> GpuActivity: ...

    @abstractmethod
    def average_gfx_utilization(self, index: int, window_seconds: float) -> float:
        """Windowed-average GFX busy %, parity with the reference's 16 s
        NVML sample average (metrics/util.go:34-88)."""

    @abstractmethod
    def ecc_uncorrectable_count(self, index: int) -> int: ...

    @abstractmethod
    def xgmi_error_status(self, index: int) -> int:
        """xGMI link error state: 0 = none, 1 = errors, 2 = multiple
        (amdsmi_xgmi_status_t).  Polled by the health watchdog for the
        synthetic code-63 event."""

    @abstractmethod
    def driver_version(self) -> str:
        """amdgpu KMD version string, e.g. '6.10.5'."""

    @abstractmethod
    def rocm_version(self) -> str: ...

    @abstractmethod
    def wait_events(self, timeout_ms: int) -> List[Event]:
        """Blocking dequeue of device events (vm fault / reset / thermal /
        page fault), parity with nvml.WaitForEvent (health_checker.go:461)."""


def numa_node_for_bdf(bdf: str, sysfs_root: str = "/sys") -> Optional[int]:
    """PCI BDF -> NUMA node via sysfs, same trick as the reference
    (nvmlutil.go:114-151: /sys/bus/pci/devices/<busid>/numa_node).
    Returns None when the node is unknown/negative."""
    bdf = bdf.lower()
    # normalize "0c:00.0" -> "0000:0c:00.0"
    if re.fullmatch(r"[0-9a-f]{2}:[0-9a-f]{2}\.[0-9a-f]", bdf):
        bdf = "0000:" + bdf
    path = os.path.join(sysfs_root, "bus", "pci", "devices", bdf, "numa_node")
    try:
        with open(path) as f:
            node = int(f.read().strip())
    except (OSError, ValueError):
        return None
    return node if node >= 0 else None
