"""Driver-version visibility: publish amdgpu/ROCm versions as node
annotations.

Parity: /root/reference/pkg/gpu/nvidia/version_visibility/version_visibility.go
(:30-46, :67-86) — nvml.SystemGetDriverVersion ->
cloud.google.com/cuda.driver-version.* annotations via server-side apply.
AMD mapping: amdsmi driver info (amdgpu KMD) + ROCm release ->
amd.com/rocm.driver-version.* annotations, same field-manager pattern.
"""
from __future__ import annotations

import logging
import re
from typing import Dict

from .. import amdsmi

log = logging.getLogger(__name__)

ANNOTATION_PREFIX = "amd.com/rocm.driver-version"
FIELD_MANAGER = "amd-gpu-device-plugin"   # parity: version_visibility.go:67


def parse_version(version: str) -> Dict[str, str]:
    """'6.10.5' -> {major: 6, minor: 10, revision: 5, full: 6.10.5}
    (parity: the major/minor/revision/full split, version_visibility.go:30-35).
    Tolerates suffixes like '6.10.5-2009582.22.04'."""
    out = {f"{ANNOTATION_PREFIX}.full": version}
    m = re.match(r"^(\d+)\.(\d+)(?:\.(\d+))?", version)
    if m:
        out[f"{ANNOTATION_PREFIX}.major"] = m.group(1)
        out[f"{ANNOTATION_PREFIX}.minor"] = m.group(2)
        out[f"{ANNOTATION_PREFIX}.revision"] = m.group(3) or "0"
    return out


def rccl_version() -> str:
    """RCCL release of the node userspace, via ncclGetVersion (version code
    major*10000 + minor*100 + patch).  Empty string when librccl is absent
    (CPU-only control-plane nodes)."""
    import ctypes

    for name in ("librccl.so", "librccl.so.1"):
        try:
            lib = ctypes.CDLL(name)
            v = ctypes.c_int()
            if lib.ncclGetVersion(ctypes.byref(v)) == 0 and v.value > 0:
                code = v.value
                return f"{code // 10000}.{code % 10000 // 100}.{code % 100}"
        except OSError:
            continue
    return ""


def partition_capabilities(sysfs_root: str = "/sys") -> Dict[str, str]:
    """Partition-mode capability annotations from the KMD's sysfs
    interface: which compute-partition modes this node's GPUs support
    (`available_compute_partition`, real format "SPX, DPX, QPX, CPX" —
    profiles/pool_probe_r02.log) and the current mode.  Operators check
    these BEFORE applying a gpu_config.json asking for CPX.  Beyond the
    reference (which has no MIG-capability annotation analog)."""
    import glob

    out: Dict[str, str] = {}
    avail: set = set()
    current: set = set()
    for path in glob.glob(
            f"{sysfs_root}/class/drm/card*/device/available_compute_partition"):
        try:
            with open(path) as f:
                avail.update(m.strip().upper()
                             for m in f.read().strip().split(","))
            with open(path.replace("available_", "current_")) as f:
                current.add(f.read().strip().upper())
        except OSError:
            continue
    if avail:
        out["amd.com/gpu.partition-modes"] = ",".join(sorted(avail - {""}))
    if len(current) == 1:
        out["amd.com/gpu.compute-partition"] = next(iter(current))
    elif current:
        out["amd.com/gpu.compute-partition"] = "mixed"
    return out


def publish_driver_version_annotations(kube_client, node_name: str,
                                       sysfs_root: str = "/sys") -> None:
    """Parity: PublishDriverVersionAnnotations (version_visibility.go:38-46);
    extended with the RCCL release (the transport the L4 layer enables)
    and the node's partition-mode capabilities."""
    ops = amdsmi.get_ops()
    version = ops.driver_version()
    annotations = parse_version(version)
    annotations["amd.com/rocm.release"] = ops.rocm_version()
    rccl = rccl_version()
    if rccl:
        annotations["amd.com/rccl.version"] = rccl
    annotations.update(partition_capabilities(sysfs_root))
    kube_client.apply_node_annotations(node_name, annotations, FIELD_MANAGER)
    log.info("published driver version annotations: %s", annotations)
