"""Metrics tests — parity with metrics_test.go's mockCollector pattern
(:26-118) plus a real fake pod-resources gRPC server over a unix socket."""
import os
import threading
from concurrent import futures

import grpc
import pytest

import cea_amd.amdsmi as amdsmi
from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
from cea_amd.deviceplugin import metrics as m
from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig
from cea_amd.kube import protos as api


class FakePodResources:
    """In-process kubelet PodResources server."""

    def __init__(self, socket_path, pods):
        self.socket_path = socket_path
        self.pods = pods
        self.server = None

    def _list(self, request, context):
        resp = api.ListPodResourcesResponse()
        for (ns, pod, container, resource, ids) in self.pods:
            p = resp.pod_resources.add(name=pod, namespace=ns)
            c = p.containers.add(name=container)
            d = c.devices.add(resource_name=resource)
            d.device_ids.extend(ids)
        return resp

    def start(self):
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        handler = grpc.method_handlers_generic_handler(
            "v1.PodResourcesLister",
            {
                "List": grpc.unary_unary_rpc_method_handler(
                    self._list,
                    request_deserializer=api.ListPodResourcesRequest.FromString,
                    response_serializer=lambda msg: msg.SerializeToString(),
                )
            },
        )
        server.add_generic_rpc_handlers((handler,))
        server.add_insecure_port(f"unix://{self.socket_path}")
        server.start()
        self.server = server

    def stop(self):
        if self.server:
            self.server.stop(grace=0)


def make_mgr(tmp_path, num_gpus=2):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, num_gpus)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    mgr = AmdGPUManager(GPUConfig(), dev_directory=dev,
                        sysfs_root=str(tmp_path / "sys"))
    mgr.config.add_defaults_and_validate()
    mgr.start()
    return mgr


def gauge_value(gauge, **labels):
    for metric in gauge.collect():
        for sample in metric.samples:
            if all(sample.labels.get(k) == v for k, v in labels.items()):
                return sample.value
    return None


def test_collect_node_and_container_metrics(tmp_path):
    mgr = make_mgr(tmp_path)
    sock = str(tmp_path / "podres.sock")
    fake = FakePodResources(sock, [
        ("ns1", "pod1", "ctr1", "amd.com/gpu", ["amdgpu0"]),
        ("ns1", "pod2", "ctr2", "amd.com/gpu", ["amdgpu1/vgpu0"]),   # virtual
        ("ns2", "pod3", "ctr3", "other.com/dev", ["x"]),             # filtered
    ])
    fake.start()
    try:
        server = m.MetricServer(
            mgr, pod_resources=m.PodResourcesClient(sock)
        )
        m.MetricServer.reset_all()
        server.collect_once()
        # node-level gauges: one per device, labeled by uuid
        assert gauge_value(m.DutyCycleNodeGpu, accelerator_id="mock-uuid-0") == 42.0
        assert gauge_value(m.MemoryTotalNodeGpu, accelerator_id="mock-uuid-1") > 0
        # container-level: virtual id collapsed to physical amdgpu1
        assert gauge_value(m.DutyCycle, pod="pod2", container="ctr2") == 42.0
        assert gauge_value(m.AcceleratorRequests, pod="pod1") == 1
        # non-amd resource is not attributed
        assert gauge_value(m.DutyCycle, pod="pod3") is None
    finally:
        fake.stop()


def test_reset_purges_gauges(tmp_path):
    mgr = make_mgr(tmp_path)
    m.DutyCycleNodeGpu.labels(make="amd", accelerator_id="u", model="m").set(5)
    m.MetricServer.reset_all()
    assert gauge_value(m.DutyCycleNodeGpu, accelerator_id="u") is None


def test_collector_unknown_device(tmp_path):
    make_mgr(tmp_path)
    stats = m.AmdSmiCollector().collect(["amdgpu0", "amdgpu9"])
    assert "amdgpu0" in stats and "amdgpu9" not in stats
    assert stats["amdgpu0"]["model"] == "AMD Instinct MI355X"


def test_metrics_http_endpoint_cpu(tmp_path):
    """Full /metrics HTTP scrape on CPU (mock amdsmi + fake pod-resources):
    node AND container gauges served with the reference's label sets."""
    import socket
    import urllib.request

    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 2)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    mgr = AmdGPUManager(GPUConfig(), dev_directory=dev,
                        sysfs_root=str(tmp_path / "sys"))
    mgr.start()

    sock_path = os.path.join(str(tmp_path), "podres.sock")
    fake = FakePodResources(sock_path, [
        ("nsA", "pod1", "train", "amd.com/gpu", ["amdgpu0"]),
    ])
    fake.start()
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    srv = m.MetricServer(
        mgr, port=port, collection_interval_s=3600,
        pod_resources=m.PodResourcesClient(sock_path))
    srv.start()
    try:
        srv.collect_once()
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
        assert 'duty_cycle_gpu_node{' in body
        assert 'make="amd"' in body
        assert 'pod="pod1"' in body and 'container="train"' in body
        assert 'request{' in body
    finally:
        srv.stop()
        fake.stop()
        mgr.stop()


def test_metric_server_stop_releases_port(tmp_path):
    """stop() must close the HTTP listener so the same port can be rebound
    in-process (prometheus start_http_server is
    fire-and-forget; a restart leaked the binding)."""
    import socket
    import urllib.request
    import urllib.error

    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 1)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    mgr = AmdGPUManager(GPUConfig(), dev_directory=dev,
                        sysfs_root=str(tmp_path / "sys"))
    mgr.start()
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    srv = m.MetricServer(mgr, port=port, collection_interval_s=3600)
    srv.start()
    urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics", timeout=10)
    srv.stop()
    with pytest.raises((urllib.error.URLError, ConnectionError, OSError)):
        urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics", timeout=2)

    # restart on the SAME port succeeds because the listener was closed
    srv2 = m.MetricServer(mgr, port=port, collection_interval_s=3600)
    srv2.start()
    try:
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
        assert "duty_cycle" in body or body is not None
    finally:
        srv2.stop()
        mgr.stop()
