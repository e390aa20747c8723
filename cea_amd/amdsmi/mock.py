"""Mock AmdSmiOperations driven by a fake /dev directory.

Parity with the reference's MockDeviceInfo
(/root/reference/pkg/gpu/nvidia/nvmlutil/nvml_mock.go:28-70), which derives
DeviceCount from fake `nvidiaN` files in a temp dir and returns canned
PciInfo.  Here the fake /dev tree contains `dri/renderD<N>` files plus `kfd`;
partition topology is configurable so CPX tests can model 8 logical devices
per physical die.
"""
from __future__ import annotations

import os
import queue
import re
import threading
from typing import List

from .iface import (
    AmdSmiOperations,
    DeviceInfo,
    Event,
    GpuActivity,
    MemoryInfo,
)

RENDERD_RE = re.compile(r"^renderD(\d+)$")

MI355X_VRAM_BYTES = 288 * 1024**3  # 288 GB HBM3E


class MockAmdSmi(AmdSmiOperations):
    def __init__(
        self,
        dev_dir: str,
        compute_partition: str = "SPX",
        partitions_per_gpu: int = 1,
        vram_total_bytes: int = MI355X_VRAM_BYTES,
    ):
        self.dev_dir = dev_dir
        self.compute_partition = compute_partition
        self.partitions_per_gpu = max(1, partitions_per_gpu)
        self.vram_total_bytes = vram_total_bytes
        self.inited = False
        self._events: "queue.Queue[Event]" = queue.Queue()
        self._ecc_counts = {}
        self._xgmi_status = {}
        self._lock = threading.Lock()

    # -- discovery ----------------------------------------------------------
    def _render_minors(self) -> List[int]:
        dri = os.path.join(self.dev_dir, "dri")
        if not os.path.isdir(dri):
            return []
        minors = []
        for name in os.listdir(dri):
            m = RENDERD_RE.match(name)
            if m:
                minors.append(int(m.group(1)))
        return sorted(minors)

    def init(self) -> None:
        if not os.path.exists(os.path.join(self.dev_dir, "kfd")):
            raise RuntimeError("amdsmi mock: /dev/kfd missing")
        self.inited = True

    def shutdown(self) -> None:
        self.inited = False

    def device_count(self) -> int:
        return len(self._render_minors())

    def device_info(self, index: int) -> DeviceInfo:
        minors = self._render_minors()
        if index >= len(minors):
            raise RuntimeError(f"amdsmi mock: no device {index}")
        minor = minors[index]
        ppg = self.partitions_per_gpu
        return DeviceInfo(
            index=index,
            uuid=f"mock-uuid-{index // ppg}",  # partitions of one die share UUID
            name="AMD Instinct MI355X",
            bdf=f"0000:{index:02x}:00.0",
            render_minor=minor,
            card_minor=minor - 128 if minor >= 128 else -1,
            vram_total_bytes=self.vram_total_bytes // ppg,
            compute_partition=self.compute_partition,
            memory_partition="NPS1",
            partition_id=index % ppg,
            physical_index=index // ppg,
        )

    def memory_info(self, index: int) -> MemoryInfo:
        total = self.vram_total_bytes // self.partitions_per_gpu
        return MemoryInfo(total_bytes=total, used_bytes=total // 10)

    def gpu_activity(self, index: int) -> GpuActivity:
        return GpuActivity(gfx_percent=42.0, umc_percent=21.0, mm_percent=0.0)

    def average_gfx_utilization(self, index: int, window_seconds: float) -> float:
        return 42.0

    def ecc_uncorrectable_count(self, index: int) -> int:
        with self._lock:
            return self._ecc_counts.get(index, 0)

    def xgmi_error_status(self, index: int) -> int:
        with self._lock:
            return self._xgmi_status.get(index, 0)

    def driver_version(self) -> str:
        return "6.10.5"

    def rocm_version(self) -> str:
        return "7.2.0"

    def wait_events(self, timeout_ms: int) -> List[Event]:
        out: List[Event] = []
        try:
            out.append(self._events.get(timeout=timeout_ms / 1000.0))
        except queue.Empty:
            return out
        while True:
            try:
                out.append(self._events.get_nowait())
            except queue.Empty:
                return out

    # -- test hooks ---------------------------------------------------------
    def inject_event(self, ev: Event) -> None:
        self._events.put(ev)

    def set_ecc_count(self, index: int, count: int) -> None:
        with self._lock:
            self._ecc_counts[index] = count

    def set_xgmi_status(self, index: int, status: int) -> None:
        with self._lock:
            self._xgmi_status[index] = status


def make_fake_dev(dev_dir: str, num_gpus: int, first_minor: int = 128) -> None:
    """Populate a fake /dev tree: kfd + dri/renderD<N>, the AMD analog of the
    reference's fake nvidia0/nvidiactl/nvidia-uvm files
    (beta_plugin_test.go:246-262)."""
    os.makedirs(os.path.join(dev_dir, "dri"), exist_ok=True)
    open(os.path.join(dev_dir, "kfd"), "w").close()
    for i in range(num_gpus):
        open(os.path.join(dev_dir, "dri", f"renderD{first_minor + i}"), "w").close()
        open(os.path.join(dev_dir, "dri", f"card{i}"), "w").close()
