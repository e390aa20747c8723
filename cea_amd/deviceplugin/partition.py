"""Compute-partition device manager: SPX / DPX / CPX.

Role parity: /root/reference/pkg/gpu/nvidia/mig/mig.go — but the mechanism is
redesigned for MI355X.  MIG exposes partitions through
/proc/driver/nvidia/capabilities files (mig.go:158-266); AMD compute
partitioning (amd-smi set --compute-partition CPX) re-enumerates the die as
N independent KFD agents, each with its OWN /dev/dri/renderD* node.  So
discovery here is render-node enumeration through the amdsmi seam: logical
devices that share a physical die are grouped by physical_index, advertised
as `amdgpu<die>/xcd<pid>`, and a partition's DeviceSpec is just its render
node (plus the shared /dev/kfd that the manager adds as a default device).

Max-count table (parity: gpuPartitionSizeMaxCount, mig.go:36-82; the mode
set MI355X advertises in sysfs available_compute_partition):
  SPX -> 1 partition/die (the whole GPU, 8 XCDs, 256 CUs)
  DPX -> 2 partitions/die (4 XCDs, 128 CUs each)
  QPX -> 4 partitions/die (2 XCDs, 64 CUs each)
  CPX -> 8 partitions/die (1 XCD, 32 CUs, 36 GB HBM3E each under NPS1)
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

from .. import amdsmi
from ..amdsmi.iface import numa_node_for_bdf
from ..kube import protos as api

log = logging.getLogger(__name__)

# mode -> partitions per physical die
PARTITION_MODE_COUNT = {"SPX": 1, "DPX": 2, "QPX": 4, "CPX": 8}
# accepted config spellings, e.g. "cpx", "cpx-nps1"
def parse_partition_mode(s: str) -> str:
    mode = s.strip().upper().split("-")[0]
    if mode not in PARTITION_MODE_COUNT:
        raise ValueError(
            f"invalid compute partition {s!r}: want one of "
            f"{sorted(PARTITION_MODE_COUNT)} (optionally with -npsN suffix)"
        )
    return mode


class PartitionDeviceManager:
    """Maps partition device ids `amdgpu<die>/xcd<pid>` to DeviceSpecs and
    physical-die UUIDs (for health attribution: one die's RAS event must mark
    all of its partitions unhealthy — the analog of the reference's MIG
    UUID->GI/CI matching, health_checker.go:426-445)."""

    def __init__(self, mode: str, dev_root: str = "/dev"):
        self.mode = parse_partition_mode(mode)
        self.dev_root = dev_root
        # device id -> [DeviceSpec-like dicts]
        self.device_specs: Dict[str, List[dict]] = {}
        # device id -> DeviceInfo
        self.devices: Dict[str, "amdsmi.DeviceInfo"] = {}
        # physical die uuid -> [device ids]
        self.die_to_devices: Dict[str, List[str]] = {}

    def start(self, sysfs_root: str = "/sys") -> None:
        """Validates uniform partitioning across all dies, else errors
        (parity: mig.go:121-155 uniform-partitioning check)."""
        ops = amdsmi.get_ops()
        want = PARTITION_MODE_COUNT[self.mode]
        n = ops.device_count()
        per_die: Dict[int, List] = {}
        for i in range(n):
            info = ops.device_info(i)
            if info.compute_partition != self.mode:
                raise RuntimeError(
                    f"device {i} is in compute partition "
                    f"{info.compute_partition}, config wants {self.mode}; run "
                    "the partition_gpu job first"
                )
            per_die.setdefault(info.physical_index, []).append(info)
        for die, infos in per_die.items():
            if len(infos) != want:
                raise RuntimeError(
                    f"physical GPU {die} has {len(infos)} {self.mode} "
                    f"partitions, expected {want}: non-uniform partitioning "
                    "is unsupported"
                )
        self.device_specs.clear()
        self.devices.clear()
        self.die_to_devices.clear()
        for die, infos in sorted(per_die.items()):
            for info in sorted(infos, key=lambda d: d.partition_id):
                dev_id = self.device_id(die, info.partition_id)
                path = f"{self.dev_root}/dri/renderD{info.render_minor}"
                self.device_specs[dev_id] = [
                    {"host_path": path, "container_path": path, "permissions": "mrw"}
                ]
                self.devices[dev_id] = info
                self.die_to_devices.setdefault(info.uuid, []).append(dev_id)
        log.info(
            "partition manager: %d %s partitions on %d dies",
            len(self.devices), self.mode, len(per_die),
        )

    def device_id(self, die: int, partition_id: int) -> str:
        if self.mode == "SPX":
            return f"amdgpu{die}"
        return f"amdgpu{die}/xcd{partition_id}"

    def list_devices(self, sysfs_root: str = "/sys") -> Dict[str, api.Device]:
        out = {}
        for dev_id, info in self.devices.items():
            d = api.Device(ID=dev_id, health=api.HEALTHY)
            numa = numa_node_for_bdf(info.bdf, sysfs_root)
            if numa is not None:
                d.topology.nodes.add(ID=numa)
            out[dev_id] = d
        return out

    def device_spec(self, dev_id: str) -> Optional[List[dict]]:
        return self.device_specs.get(dev_id)

    def devices_for_die_uuid(self, uuid: str) -> List[str]:
        # normalized lookup, same policy as the health checker's matching
        want = uuid.strip().lower()
        for k, ids in self.die_to_devices.items():
            if k.strip().lower() == want:
                return ids
        return []
