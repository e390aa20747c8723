"""GPU-sharing policy.

Parity: /root/reference/pkg/gpu/nvidia/gpusharing/gpusharing.go (strategy
enum :23-29, package-global SharingStrategy :31, ValidateRequest :40-50,
virtual-ID scheme and parsing :53-77).

AMD mapping: the `mps` strategy has no daemon analog on ROCm; the MPS-like
strategy here is `cu-fencing` — containers receive HSA_CU_MASK /
GPU_MAX_HW_QUEUES env fencing computed by the manager (manager.Envs()).
Memory isolation is advisory only (documented gap: ROCm has no enforced
per-process VRAM limit; hard isolation is what CPX partitioning is for).
"""
from __future__ import annotations

from typing import List, Tuple

TIME_SHARING = "time-sharing"
CU_FENCING = "cu-fencing"   # the MPS-analog strategy
VALID_STRATEGIES = (TIME_SHARING, CU_FENCING)

# Package-global, set by GPUConfig.add_defaults_and_validate()
# (parity: gpusharing.go:31 + manager.go:107-115).
sharing_strategy: str = ""


class SharingError(ValueError):
    pass


def validate_request(requested_ids: List[str], num_physical: int) -> None:
    """Parity: ValidateRequest (gpusharing.go:40-50).

    time-sharing: at most one virtual device per container request.
    cu-fencing: multiple virtual devices allowed only on single-GPU nodes
    (same restriction the reference applies to MPS).
    """
    virtual = [d for d in requested_ids if is_virtual_id(d)]
    if not virtual:
        return
    if sharing_strategy == TIME_SHARING and len(requested_ids) > 1:
        raise SharingError(
            "invalid request for time-sharing GPUs: at most 1 shared GPU per "
            f"container, got {len(requested_ids)}"
        )
    if sharing_strategy == CU_FENCING and len(requested_ids) > 1 and num_physical > 1:
        raise SharingError(
            "invalid request for cu-fencing GPUs: multiple shared GPUs per "
            "container are only supported on single-GPU nodes"
        )


def is_virtual_id(device_id: str) -> bool:
    return "/vgpu" in device_id


def virtual_to_physical(device_id: str) -> str:
    """amdgpu0/vgpu1 -> amdgpu0; amdgpu0/xcd2/vgpu1 -> amdgpu0/xcd2.
    Parity: VirtualToPhysicalDeviceID (gpusharing.go:53-77)."""
    parts = device_id.split("/")
    if len(parts) < 2 or not parts[-1].startswith("vgpu"):
        raise SharingError(f"not a virtual device id: {device_id}")
    suffix = parts[-1][len("vgpu"):]
    if not suffix.isdigit():
        raise SharingError(f"bad virtual device index in id: {device_id}")
    return "/".join(parts[:-1])


def virtual_id(physical_id: str, index: int) -> str:
    return f"{physical_id}/vgpu{index}"


def split_physical(device_id: str) -> Tuple[str, str]:
    """amdgpu0/xcd2 -> ('amdgpu0', 'xcd2'); amdgpu0 -> ('amdgpu0', '')."""
    parts = device_id.split("/")
    if len(parts) == 1:
        return parts[0], ""
    return parts[0], parts[1]
