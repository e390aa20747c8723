"""AMD-SMI abstraction layer.

Role parity: pkg/gpu/nvidia/nvmlutil (reference nvmlutil.go:30-151) — the single
seam through which every other component touches the GPU management library,
with a package-global injection point swapped for a mock in unit tests
(reference nvmlutil.go:42, nvml_mock.go:28-70).

The real backend is the in-tree native C++ shim over libamd_smi
(csrc/amdsmi_shim.cpp, the analog of the reference's cgo shim
pkg/gpu/nvidia/metrics/util.go:17-88); the mock backend drives everything
from a fake /dev directory so all control-plane logic is testable on CPU.
"""
from .iface import AmdSmiOperations, DeviceInfo, GpuActivity, MemoryInfo, Event
from .mock import MockAmdSmi

# Package-global injection point, same pattern as nvmlutil.NvmlDeviceInfo
# (reference nvmlutil.go:42).  Production entrypoints set this to ShimAmdSmi;
# tests set it to MockAmdSmi.
ops: AmdSmiOperations = None


def set_ops(o: AmdSmiOperations) -> None:
    global ops
    ops = o


def get_ops() -> AmdSmiOperations:
    if ops is None:
        raise RuntimeError("cea_amd.amdsmi.ops not initialised — call set_ops()")
    return ops
