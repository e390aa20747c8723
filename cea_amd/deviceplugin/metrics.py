"""Prometheus metrics server + container->GPU attribution.

Parity: /root/reference/pkg/gpu/nvidia/metrics/ (metrics.go 257 LoC,
devices.go 140, util.go 113):
  * the same 7 gauges, node-level {duty_cycle_gpu_node, memory_total_gpu_node,
    memory_used_gpu_node} and container-level {duty_cycle, memory_total,
    memory_used, request} with the same label sets (metrics.go:59-115),
    make="amd";
  * collection ticker (default 30 s) with a full gauge reset every 60 s to
    purge dead containers (metrics.go:117,241-253);
  * container->device attribution via the kubelet PodResources API over
    /var/lib/kubelet/pod-resources/kubelet.sock, filtering the amd.com/gpu
    resource and skipping virtual ids (devices.go:51-101, devices.go:90);
  * duty-cycle = the native shim's windowed average over ~16 s of sampled
    GFX activity (the cgo nvmlDeviceGetAverageUsage analog, util.go:37-87).
"""
from __future__ import annotations

import logging
import threading
import time
from typing import Dict, List, Optional

import grpc
from prometheus_client import Gauge, start_http_server

from .. import amdsmi
from ..kube import protos as api
from . import RESOURCE_NAME
from .sharing import is_virtual_id, virtual_to_physical

log = logging.getLogger(__name__)

POD_RESOURCES_SOCKET = "/var/lib/kubelet/pod-resources/kubelet.sock"
UTILIZATION_WINDOW_S = 16.0  # parity: ~100 NVML samples ≈ 16 s (util.go:34)

NODE_LABELS = ["make", "accelerator_id", "model"]
CONTAINER_LABELS = [
    "namespace", "pod", "container", "make", "accelerator_id", "model",
]

# Gauges are module-level singletons like the reference's promauto vars
# (metrics.go:59-115).
DutyCycleNodeGpu = Gauge(
    "duty_cycle_gpu_node",
    "GPU utilization of the node's GPU, in percent",
    NODE_LABELS,
)
MemoryTotalNodeGpu = Gauge(
    "memory_total_gpu_node",
    "Total VRAM of the node's GPU, in bytes",
    NODE_LABELS,
)
MemoryUsedNodeGpu = Gauge(
    "memory_used_gpu_node",
    "Used VRAM of the node's GPU, in bytes",
    NODE_LABELS,
)
DutyCycle = Gauge(
    "duty_cycle",
    "GPU utilization attributed to a container, in percent",
    CONTAINER_LABELS,
)
MemoryTotal = Gauge(
    "memory_total",
    "Total VRAM of a container's GPU, in bytes",
    CONTAINER_LABELS,
)
MemoryUsed = Gauge(
    "memory_used",
    "Used VRAM of a container's GPU, in bytes",
    CONTAINER_LABELS,
)
AcceleratorRequests = Gauge(
    "request",
    "Number of accelerator devices requested by the container",
    ["namespace", "pod", "container", "resource_name"],
)

ALL_GAUGES = [
    DutyCycleNodeGpu, MemoryTotalNodeGpu, MemoryUsedNodeGpu,
    DutyCycle, MemoryTotal, MemoryUsed, AcceleratorRequests,
]


class PodResourcesClient:
    """kubelet PodResources List() over the unix socket (devices.go:51-101)."""

    def __init__(self, socket_path: str = POD_RESOURCES_SOCKET):
        self.socket_path = socket_path

    def devices_for_all_containers(self) -> Dict[tuple, List[str]]:
        """(namespace, pod, container) -> [physical device ids]."""
        out: Dict[tuple, List[str]] = {}
        with grpc.insecure_channel(f"unix://{self.socket_path}") as channel:
            list_rpc = channel.unary_unary(
                api.PODRESOURCES_LIST,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=api.ListPodResourcesResponse.FromString,
            )
            resp = list_rpc(api.ListPodResourcesRequest(), timeout=10)
        for pod in resp.pod_resources:
            for container in pod.containers:
                ids: List[str] = []
                for devs in container.devices:
                    if devs.resource_name != RESOURCE_NAME:
                        continue
                    for dev_id in devs.device_ids:
                        # virtual ids attribute to their physical device but
                        # the shared usage is not split (parity devices.go:90)
                        if is_virtual_id(dev_id):
                            dev_id = virtual_to_physical(dev_id)
                        if dev_id not in ids:
                            ids.append(dev_id)
                if ids:
                    out[(pod.namespace, pod.name, container.name)] = ids
        return out


class AmdSmiCollector:
    """metricsCollector implementation over the amdsmi seam (the interface
    the reference mocks in metrics_test.go:26)."""

    def collect(self, device_ids: List[str]) -> Dict[str, dict]:
        ops = amdsmi.get_ops()
        out = {}
        by_name = {f"amdgpu{i}": i for i in range(ops.device_count())}
        for dev_id in device_ids:
            base = dev_id.split("/")[0]
            idx = by_name.get(base)
            if idx is None:
                continue
            try:
                info = ops.device_info(idx)
                mem = ops.memory_info(idx)
                duty = ops.average_gfx_utilization(idx, UTILIZATION_WINDOW_S)
            except Exception as e:  # noqa: BLE001
                log.error("metrics collect failed for %s: %s", dev_id, e)
                continue
            out[dev_id] = {
                "duty_cycle": duty,
                "memory_total": mem.total_bytes,
                "memory_used": mem.used_bytes,
                "accelerator_id": info.uuid,
                "model": info.name,
            }
        return out


class MetricServer:
    """Parity: MetricServer (metrics.go:120-161)."""

    def __init__(
        self,
        manager,
        port: int = 2112,
        collection_interval_s: float = 30.0,
        reset_interval_s: float = 60.0,
        collector=None,
        pod_resources: Optional[PodResourcesClient] = None,
    ):
        self.manager = manager
        self.port = port
        self.collection_interval_s = collection_interval_s
        self.reset_interval_s = reset_interval_s
        self.collector = collector or AmdSmiCollector()
        self.pod_resources = pod_resources or PodResourcesClient()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._last_reset = time.monotonic()

    def start(self) -> None:
        # start_http_server returns (WSGIServer, Thread); keep them so
        # stop() can actually close the listening socket — without this a
        # restart inside one process leaks the port binding.
        self._httpd, self._http_thread = start_http_server(self.port)
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        log.info("metrics server on :%d (collect every %.0fs)",
                 self.port, self.collection_interval_s)

    def stop(self) -> None:
        self._stop.set()
        httpd = getattr(self, "_httpd", None)
        if httpd is not None:
            try:
                httpd.shutdown()
                httpd.server_close()
            except Exception as e:  # noqa: BLE001
                log.error("metrics HTTP listener close failed: %s", e)
            self._httpd = None
        t = getattr(self, "_http_thread", None)
        if t is not None:
            t.join(timeout=5)
            self._http_thread = None

    def _loop(self) -> None:
        while not self._stop.wait(self.collection_interval_s):
            try:
                self.collect_once()
            except Exception as e:  # noqa: BLE001
                log.error("metrics collection failed: %s", e)

    def collect_once(self) -> None:
        """One collection tick (parity: collectMetrics, metrics.go:163-180),
        with the periodic full reset purging dead containers
        (metrics.go:241-253)."""
        if time.monotonic() - self._last_reset >= self.reset_interval_s:
            self.reset_all()
            self._last_reset = time.monotonic()

        # node-level: every physical device
        all_ids = list(self.manager.devices.keys())
        if self.manager.partition_manager:
            all_ids = list(self.manager.partition_manager.devices.keys())
        node_stats = self.collector.collect(all_ids)
        for dev_id, s in node_stats.items():
            labels = dict(make="amd", accelerator_id=s["accelerator_id"],
                          model=s["model"])
            DutyCycleNodeGpu.labels(**labels).set(s["duty_cycle"])
            MemoryTotalNodeGpu.labels(**labels).set(s["memory_total"])
            MemoryUsedNodeGpu.labels(**labels).set(s["memory_used"])

        # container-level: attribution via pod-resources
        try:
            per_container = self.pod_resources.devices_for_all_containers()
        except Exception as e:  # noqa: BLE001
            log.error("pod-resources list failed: %s", e)
            return
        for (ns, pod, container), dev_ids in per_container.items():
            stats = self.collector.collect(dev_ids)
            AcceleratorRequests.labels(
                namespace=ns, pod=pod, container=container,
                resource_name=RESOURCE_NAME,
            ).set(len(dev_ids))
            for dev_id, s in stats.items():
                labels = dict(
                    namespace=ns, pod=pod, container=container,
                    make="amd", accelerator_id=s["accelerator_id"],
                    model=s["model"],
                )
                DutyCycle.labels(**labels).set(s["duty_cycle"])
                MemoryTotal.labels(**labels).set(s["memory_total"])
                MemoryUsed.labels(**labels).set(s["memory_used"])

    @staticmethod
    def reset_all() -> None:
        for g in ALL_GAUGES:
            g.clear()
