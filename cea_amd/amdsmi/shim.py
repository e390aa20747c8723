"""ShimAmdSmi — the production AmdSmiOperations backend.

ctypes bindings over the in-tree native shim csrc/amdsmi_shim.cpp
(libceaamd_smi.so), which links libamd_smi directly and runs the windowed
utilization sampler in C++ (parity: the reference's cgo shim,
pkg/gpu/nvidia/metrics/util.go:17-88).

Fails loudly if the native library is missing: on a GPU node the plugin must
never silently degrade to the mock.
"""
from __future__ import annotations

import ctypes
import os
from typing import List

from .iface import (
    AmdSmiOperations,
    DeviceInfo,
    Event,
    GpuActivity,
    MemoryInfo,
)

_LIB_NAME = "libceaamd_smi.so"


class _CDeviceInfo(ctypes.Structure):
    _fields_ = [
        ("index", ctypes.c_int),
        ("uuid", ctypes.c_char * 256),
        ("name", ctypes.c_char * 256),
        ("bdf", ctypes.c_char * 32),
        ("render_minor", ctypes.c_int),
        ("card_minor", ctypes.c_int),
        ("vram_total", ctypes.c_ulonglong),
        ("compute_partition", ctypes.c_char * 16),
        ("memory_partition", ctypes.c_char * 16),
        ("partition_id", ctypes.c_uint),
        ("physical_index", ctypes.c_int),
    ]


class _CEvent(ctypes.Structure):
    _fields_ = [
        ("uuid", ctypes.c_char * 256),
        ("code", ctypes.c_int),
        ("message", ctypes.c_char * 256),
    ]


def load_library() -> ctypes.CDLL:
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)), _LIB_NAME)
    if not os.path.exists(path):
        raise RuntimeError(
            f"native amdsmi shim not built: {path} missing — run `make smi` "
            "(the device plugin must not run without its native backend)"
        )
    lib = ctypes.CDLL(path)
    lib.cea_smi_last_error.restype = ctypes.c_char_p
    return lib


class ShimAmdSmi(AmdSmiOperations):
    # Fallback event-notification mask, used only when the loaded shim .so
    # predates cea_smi_default_event_mask (AMDSMI_EVENT_MASK_FROM_INDEX(i)
    # = 1<<(i-1)): VMFAULT(1) | THERMAL_THROTTLE(2) | GPU_PRE_RESET(3) |
    # GPU_POST_RESET(4) | PAGE_FAULT_START(7) | PAGE_FAULT_END(8).  The
    # authoritative mask comes from the shim, which builds it from the
    # amdsmi header it was compiled against — enum indices differ across
    # amdsmi versions, so Python must not hardcode them.
    FALLBACK_EVENT_MASK = (
        (1 << 0) | (1 << 1) | (1 << 2) | (1 << 3) | (1 << 6) | (1 << 7)
    )

    def default_event_mask(self) -> int:
        fn = getattr(self.lib, "cea_smi_default_event_mask", None)
        if fn is None:
            return self.FALLBACK_EVENT_MASK
        fn.restype = ctypes.c_ulonglong
        return int(fn())

    def __init__(self, sampler_interval_ms: int = 160):
        self.lib = load_library()
        self.sampler_interval_ms = sampler_interval_ms
        self._events_inited = False

    def _check(self, rc: int, what: str) -> None:
        if rc != 0:
            err = self.lib.cea_smi_last_error().decode(errors="replace")
            raise RuntimeError(f"amdsmi shim {what} failed (rc={rc}): {err}")

    def init(self) -> None:
        self._check(self.lib.cea_smi_init(), "init")
        self._check(
            self.lib.cea_smi_start_sampler(self.sampler_interval_ms), "sampler"
        )

    def shutdown(self) -> None:
        if self._events_inited:
            self.lib.cea_smi_event_stop()
            self._events_inited = False
        self.lib.cea_smi_shutdown()

    def device_count(self) -> int:
        return self.lib.cea_smi_device_count()

    def device_info(self, index: int) -> DeviceInfo:
        out = _CDeviceInfo()
        self._check(
            self.lib.cea_smi_device_info(index, ctypes.byref(out)),
            f"device_info({index})",
        )
        return DeviceInfo(
            index=index,
            uuid=out.uuid.decode(errors="replace"),
            name=out.name.decode(errors="replace"),
            bdf=out.bdf.decode(errors="replace"),
            render_minor=out.render_minor,
            card_minor=out.card_minor,
            vram_total_bytes=out.vram_total,
            compute_partition=out.compute_partition.decode(errors="replace") or "SPX",
            memory_partition=out.memory_partition.decode(errors="replace") or "NPS1",
            partition_id=out.partition_id,
            physical_index=out.physical_index,
        )

    def memory_info(self, index: int) -> MemoryInfo:
        total = ctypes.c_ulonglong()
        used = ctypes.c_ulonglong()
        self._check(
            self.lib.cea_smi_memory_info(
                index, ctypes.byref(total), ctypes.byref(used)
            ),
            f"memory_info({index})",
        )
        return MemoryInfo(total_bytes=total.value, used_bytes=used.value)

    def gpu_activity(self, index: int) -> GpuActivity:
        gfx = ctypes.c_double()
        umc = ctypes.c_double()
        mm = ctypes.c_double()
        self._check(
            self.lib.cea_smi_gpu_activity(
                index, ctypes.byref(gfx), ctypes.byref(umc), ctypes.byref(mm)
            ),
            f"gpu_activity({index})",
        )
        return GpuActivity(gfx.value, umc.value, mm.value)

    def average_gfx_utilization(self, index: int, window_seconds: float) -> float:
        avg = ctypes.c_double()
        count = ctypes.c_int()
        self._check(
            self.lib.cea_smi_average_utilization(
                index,
                ctypes.c_double(window_seconds),
                ctypes.byref(avg),
                ctypes.byref(count),
            ),
            f"average_utilization({index})",
        )
        return avg.value

    def ecc_uncorrectable_count(self, index: int) -> int:
        corr = ctypes.c_ulonglong()
        uncorr = ctypes.c_ulonglong()
        deferred = ctypes.c_ulonglong()
        self._check(
            self.lib.cea_smi_ecc_count(
                index,
                ctypes.byref(corr),
                ctypes.byref(uncorr),
                ctypes.byref(deferred),
            ),
            f"ecc_count({index})",
        )
        return uncorr.value

    def xgmi_error_status(self, index: int) -> int:
        status = ctypes.c_int()
        self._check(
            self.lib.cea_smi_xgmi_error_status(index, ctypes.byref(status)),
            f"xgmi_error_status({index})",
        )
        return status.value

    def driver_version(self) -> str:
        buf = ctypes.create_string_buffer(256)
        self._check(self.lib.cea_smi_driver_version(buf, 256), "driver_version")
        return buf.value.decode(errors="replace")

    def rocm_version(self) -> str:
        # ROCm release of the userspace this shim is linked against
        for p in ("/opt/rocm/.info/version", "/usr/local/amd/.info/version"):
            try:
                with open(p) as f:
                    return f.read().strip()
            except OSError:
                continue
        buf = ctypes.create_string_buffer(64)
        if self.lib.cea_smi_lib_version(buf, 64) == 0:
            return "amdsmi-" + buf.value.decode(errors="replace")
        return "unknown"

    def wait_events(self, timeout_ms: int) -> List[Event]:
        if not self._events_inited:
            self._check(
                self.lib.cea_smi_event_init(
                    ctypes.c_ulonglong(self.default_event_mask())
                ),
                "event_init",
            )
            self._events_inited = True
        max_ev = 64
        arr = (_CEvent * max_ev)()
        n = ctypes.c_int()
        self._check(
            self.lib.cea_smi_wait_events(timeout_ms, arr, max_ev, ctypes.byref(n)),
            "wait_events",
        )
        return [
            Event(
                device_uuid=arr[i].uuid.decode(errors="replace"),
                code=arr[i].code,
                message=arr[i].message.decode(errors="replace"),
            )
            for i in range(n.value)
        ]
