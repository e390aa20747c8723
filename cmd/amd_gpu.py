#!/usr/bin/env python3
"""amd_gpu device-plugin entrypoint.

Parity: /root/reference/cmd/nvidia_gpu/nvidia_gpu.go (flags :50-61, main
:110-226): config load -> manager construction -> device-path retry loop ->
amdsmi init -> start retry loop -> optional metrics server, health checker,
driver-version publisher -> serve (blocks forever).
"""
from __future__ import annotations

import argparse
import json
import logging
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

log = logging.getLogger("amd_gpu")


def parse_args(argv=None):
    p = argparse.ArgumentParser(description="AMD GPU kubelet device plugin")
    # flag parity with nvidia_gpu.go:50-61
    p.add_argument("--host-path", default="/home/kubernetes/bin/amd")
    p.add_argument("--container-path", default="/usr/local/amd")
    p.add_argument("--plugin-directory", default="/device-plugin")
    p.add_argument("--dev-directory", default="/dev")
    p.add_argument("--enable-container-gpu-metrics", action="store_true")
    p.add_argument("--enable-health-monitoring", action="store_true")
    p.add_argument("--gpu-metrics-port", type=int, default=2112)
    p.add_argument("--gpu-metrics-collection-interval", type=int, default=30000,
                   help="milliseconds")
    p.add_argument("--gpu-config", default="/etc/amd/gpu_config.json")
    p.add_argument("--gpu-fraction-divisor-file",
                   default="/etc/amd/gpu-fraction-divisor.txt")
    p.add_argument("--opencl-icd-host-path", default="/etc/OpenCL/vendors",
                   help="host dir with the OpenCL vendor ICD (the Vulkan-ICD"
                        " analog, nvidia_gpu.go:50-61); mounted read-only "
                        "into GPU containers when present")
    p.add_argument("--opencl-icd-container-path", default="/etc/OpenCL/vendors")
    p.add_argument("--publish-driver-version", action="store_true")
    p.add_argument("--mock-amdsmi", action="store_true",
                   help="use the mock AMD-SMI backend (CPU e2e testing; the "
                        "seam the reference swaps via nvmlutil.NvmlDeviceInfo)")
    return p.parse_args(argv)


def parse_gpu_config(path):
    """Parity: parseGPUConfig (nvidia_gpu.go:64-81)."""
    from cea_amd.deviceplugin.manager import GPUConfig, GPUSharingConfig

    if not os.path.exists(path):
        log.info("no gpu config at %s; using defaults", path)
        return GPUConfig()
    with open(path) as f:
        raw = json.load(f) if f else {}
    sharing = raw.get("GPUSharingConfig") or {}
    cfg = GPUConfig(
        compute_partition=raw.get("ComputePartition", raw.get("GPUPartitionSize", "")),
        gpu_sharing_config=GPUSharingConfig(
            gpu_sharing_strategy=sharing.get("GPUSharingStrategy", ""),
            max_shared_clients_per_gpu=int(sharing.get("MaxSharedClientsPerGPU", 0)),
        ) if sharing else None,
    )
    return cfg


def parse_gpu_fraction_divisor(path):
    """Parity: parseGPUFractionDivisor (nvidia_gpu.go:85-108): any value <=1
    (or missing file) => 1."""
    try:
        with open(path) as f:
            v = int(f.read().strip())
        return v if v > 1 else 1
    except (OSError, ValueError):
        return 1


def main(argv=None):
    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )
    args = parse_args(argv)

    import cea_amd.amdsmi as amdsmi
    from cea_amd.amdsmi.shim import ShimAmdSmi
    from cea_amd.deviceplugin.manager import AmdGPUManager

    config = parse_gpu_config(args.gpu_config)
    config.gpu_fraction_divisor = parse_gpu_fraction_divisor(
        args.gpu_fraction_divisor_file
    )
    # parity: XID_CONFIG env from ConfigMap (nvidia_gpu.go:136)
    config.add_health_critical_events(os.environ.get("EVENT_CONFIG", ""))
    config.add_defaults_and_validate()

    mgr = AmdGPUManager(
        config,
        dev_directory=args.dev_directory,
        host_path=args.host_path,
        container_path=args.container_path,
        plugin_directory=args.plugin_directory,
        extra_mounts=[{
            "host_path": args.opencl_icd_host_path,
            "container_path": args.opencl_icd_container_path,
        }],
    )

    # device-path retry loop (parity nvidia_gpu.go:146-154): wait for L0
    # driver install to finish
    while True:
        try:
            mgr.check_device_paths()
            break
        except FileNotFoundError as e:
            log.info("waiting for amdgpu driver: %s", e)
            time.sleep(5)

    if args.mock_amdsmi:
        from cea_amd.amdsmi.mock import MockAmdSmi

        smi = MockAmdSmi(args.dev_directory)
    else:
        smi = ShimAmdSmi()
    smi.init()   # parity: nvml.Init (nvidia_gpu.go:156)
    amdsmi.set_ops(smi)

    # start retry loop (parity nvidia_gpu.go:161-169)
    while True:
        try:
            mgr.start()
            break
        except Exception as e:  # noqa: BLE001
            log.error("manager start failed (retrying in 10s): %s", e)
            time.sleep(10)

    node_name = os.environ.get("NODE_NAME", "")
    kube = None
    if args.enable_health_monitoring or args.publish_driver_version:
        try:
            from cea_amd.kube.client import build_kube_client
            kube = build_kube_client()
        except Exception as e:  # noqa: BLE001
            log.error("in-cluster kube client unavailable: %s", e)

    if args.enable_container_gpu_metrics and not mgr.partition_manager:
        # metrics disabled under partitioning, parity nvidia_gpu.go:172-174
        from cea_amd.deviceplugin.metrics import MetricServer

        ms = MetricServer(
            mgr,
            port=args.gpu_metrics_port,
            collection_interval_s=args.gpu_metrics_collection_interval / 1000.0,
        )
        ms.start()

    if args.enable_health_monitoring:
        from cea_amd.deviceplugin.health import GPUHealthChecker

        hc = GPUHealthChecker(mgr, kube, node_name=node_name)
        hc.start()

    if args.publish_driver_version and kube and node_name:
        import threading

        from cea_amd.deviceplugin.version_visibility import (
            publish_driver_version_annotations,
        )

        def _publish():
            for _ in range(10):
                try:
                    publish_driver_version_annotations(kube, node_name)
                    return
                except Exception as e:  # noqa: BLE001
                    log.error("publish driver version failed: %s", e)
                    time.sleep(30)

        threading.Thread(target=_publish, daemon=True).start()

    # Graceful pod termination: kubelet sends SIGTERM; stop the serve loop
    # so the gRPC server closes and the plugin socket is released cleanly
    # (kubelet re-registers us on restart either way).
    import signal

    def _terminate(signum, frame):  # noqa: ARG001
        log.info("received signal %d; shutting down", signum)
        mgr.stop()

    signal.signal(signal.SIGTERM, _terminate)
    signal.signal(signal.SIGINT, _terminate)

    mgr.serve()   # blocks until stopped (parity nvidia_gpu.go:225)


if __name__ == "__main__":
    main()
