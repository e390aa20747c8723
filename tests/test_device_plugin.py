"""End-to-end device-plugin tests over a real in-process gRPC unix socket.

Parity with the reference's beta_plugin_test.go:72-240: register with a stub
kubelet, stream ListAndWatch, Allocate valid/used/invalid device sets, across
four configs (plain, time-sharing, CPX partitioned, CPX+time-sharing).
"""
import os
import threading
import time

import grpc
import pytest

import cea_amd.amdsmi as amdsmi
from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
from cea_amd.deviceplugin import RESOURCE_NAME, sharing
from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig, GPUSharingConfig

from helpers import KubeletStub, PluginClient


def make_manager(tmp_path, config=None, num_gpus=2, partitions_per_gpu=1,
                 compute_partition="SPX"):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, num_gpus * partitions_per_gpu)
    mock = MockAmdSmi(dev, compute_partition=compute_partition,
                      partitions_per_gpu=partitions_per_gpu)
    mock.init()
    amdsmi.set_ops(mock)
    plugin_dir = str(tmp_path / "device-plugin")
    os.makedirs(plugin_dir, exist_ok=True)
    mgr = AmdGPUManager(
        config or GPUConfig(),
        dev_directory=dev,
        plugin_directory=plugin_dir,
        sysfs_root=str(tmp_path / "sys"),
    )
    mgr.config.add_defaults_and_validate()
    mgr.start()
    return mgr, mock


def serve_in_thread(mgr):
    t = threading.Thread(target=mgr.serve, daemon=True)
    t.start()
    socket_path = os.path.join(mgr.plugin_directory, mgr.socket_name)
    deadline = time.time() + 5
    while not os.path.exists(socket_path) and time.time() < deadline:
        time.sleep(0.02)
    assert os.path.exists(socket_path), "plugin socket never appeared"
    return t, socket_path


def test_register_and_list(tmp_path):
    mgr, _ = make_manager(tmp_path, num_gpus=2)
    stub = KubeletStub(mgr.plugin_directory)
    stub.start()
    try:
        t, socket_path = serve_in_thread(mgr)
        assert stub.registered.wait(5)
        req = stub.requests[0]
        assert req.resource_name == RESOURCE_NAME
        assert req.version == "v1beta1"
        assert req.endpoint == mgr.socket_name

        client = PluginClient(socket_path)
        stream = client.list_and_watch_once()
        resp = next(iter(stream))
        ids = sorted(d.ID for d in resp.devices)
        assert ids == ["amdgpu0", "amdgpu1"]
        assert all(d.health == "Healthy" for d in resp.devices)
        stream.cancel()
        client.close()
    finally:
        mgr.stop()
        stub.stop()


def test_allocate_plain(tmp_path):
    mgr, _ = make_manager(tmp_path, num_gpus=2)
    stub = KubeletStub(mgr.plugin_directory)
    stub.start()
    try:
        _, socket_path = serve_in_thread(mgr)
        client = PluginClient(socket_path)
        resp = client.allocate([["amdgpu0"], ["amdgpu1"]])
        assert len(resp.container_responses) == 2
        c0 = resp.container_responses[0]
        host_paths = [d.host_path for d in c0.devices]
        # render node for amdgpu0 + shared /dev/kfd
        assert any(p.endswith("renderD128") for p in host_paths)
        assert any(p.endswith("/kfd") for p in host_paths)
        assert all(d.permissions == "mrw" for d in c0.devices)
        assert c0.mounts[0].container_path == "/usr/local/amd"
        assert c0.mounts[0].read_only

        # invalid device id
        with pytest.raises(grpc.RpcError) as ei:
            client.allocate([["amdgpu7"]])
        assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT
        client.close()
    finally:
        mgr.stop()
        stub.stop()


def test_time_sharing_fanout_and_validation(tmp_path):
    cfg = GPUConfig(
        gpu_sharing_config=GPUSharingConfig(
            gpu_sharing_strategy="time-sharing", max_shared_clients_per_gpu=3
        )
    )
    mgr, _ = make_manager(tmp_path, config=cfg, num_gpus=2)
    assert sharing.sharing_strategy == "time-sharing"
    devs = mgr.list_devices()
    assert len(devs) == 6
    assert sorted(d.ID for d in devs)[0] == "amdgpu0/vgpu0"

    stub = KubeletStub(mgr.plugin_directory)
    stub.start()
    try:
        _, socket_path = serve_in_thread(mgr)
        client = PluginClient(socket_path)
        resp = client.allocate([["amdgpu0/vgpu1"]])
        paths = [d.host_path for d in resp.container_responses[0].devices]
        assert any(p.endswith("renderD128") for p in paths)
        # >1 shared device per container is rejected under time-sharing
        with pytest.raises(grpc.RpcError) as ei:
            client.allocate([["amdgpu0/vgpu0", "amdgpu0/vgpu2"]])
        assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT
        client.close()
    finally:
        mgr.stop()
        stub.stop()


def test_cpx_partitions(tmp_path):
    cfg = GPUConfig(compute_partition="cpx-nps1")
    mgr, _ = make_manager(tmp_path, config=cfg, num_gpus=1,
                          partitions_per_gpu=8, compute_partition="CPX")
    devs = mgr.list_devices()
    assert len(devs) == 8
    assert sorted(d.ID for d in devs)[0] == "amdgpu0/xcd0"

    stub = KubeletStub(mgr.plugin_directory)
    stub.start()
    try:
        _, socket_path = serve_in_thread(mgr)
        client = PluginClient(socket_path)
        resp = client.allocate([["amdgpu0/xcd3"]])
        paths = [d.host_path for d in resp.container_responses[0].devices]
        assert any(p.endswith("renderD131") for p in paths)
        assert any(p.endswith("/kfd") for p in paths)
        client.close()
    finally:
        mgr.stop()
        stub.stop()


def test_cpx_plus_time_sharing(tmp_path):
    cfg = GPUConfig(
        compute_partition="cpx",
        gpu_sharing_config=GPUSharingConfig(
            gpu_sharing_strategy="time-sharing", max_shared_clients_per_gpu=2
        ),
    )
    mgr, _ = make_manager(tmp_path, config=cfg, num_gpus=1,
                          partitions_per_gpu=8, compute_partition="CPX")
    devs = mgr.list_devices()
    assert len(devs) == 16
    ids = {d.ID for d in devs}
    assert "amdgpu0/xcd0/vgpu0" in ids
    # virtual partition id resolves to the partition's render node
    specs = mgr.device_spec("amdgpu0/xcd2/vgpu1")
    assert specs[0]["host_path"].endswith("renderD130")


def test_health_event_triggers_resend(tmp_path):
    from cea_amd.kube import protos as api

    mgr, _ = make_manager(tmp_path, num_gpus=2)
    stub = KubeletStub(mgr.plugin_directory)
    stub.start()
    try:
        _, socket_path = serve_in_thread(mgr)
        client = PluginClient(socket_path)
        stream = client.list_and_watch_once(timeout=10)
        it = iter(stream)
        first = next(it)
        assert all(d.health == "Healthy" for d in first.devices)
        mgr.health.put(api.Device(ID="amdgpu1", health=api.UNHEALTHY))
        second = next(it)
        by_id = {d.ID: d.health for d in second.devices}
        assert by_id["amdgpu1"] == "Unhealthy"
        assert by_id["amdgpu0"] == "Healthy"
        stream.cancel()
        client.close()
    finally:
        mgr.stop()
        stub.stop()


def test_hot_add_restart_trigger(tmp_path):
    mgr, _ = make_manager(tmp_path, num_gpus=1)
    assert not mgr.has_additional_gpus_installed()
    open(os.path.join(mgr.dev_directory, "dri", "renderD200"), "w").close()
    assert mgr.has_additional_gpus_installed()


def test_concurrent_allocate_and_health_churn(tmp_path):
    """Stress the serve path: parallel Allocate callers while the health
    channel churns resends on the ListAndWatch stream — no deadlock, no
    error, stream keeps flowing (guards the grpc thread-pool sizing:
    streams pin worker threads)."""
    import queue as _queue

    from cea_amd.kube import protos as api

    mgr, _ = make_manager(tmp_path, num_gpus=4)
    serve_in_thread(mgr)
    sock = os.path.join(mgr.plugin_directory, mgr.socket_name)
    client = PluginClient(sock)
    stream = client.list_and_watch_once()
    it = iter(stream)
    next(it)  # initial send

    errors: "_queue.Queue[Exception]" = _queue.Queue()

    def churn_health():
        for i in range(20):
            mgr.health.put(api.Device(
                ID=f"amdgpu{i % 4}",
                health=api.UNHEALTHY if i % 2 else api.HEALTHY))
            time.sleep(0.01)

    def allocate_loop():
        c = PluginClient(sock)
        try:
            for _ in range(25):
                resp = c.allocate([["amdgpu1"]])
                assert resp.container_responses[0].devices
        except Exception as e:  # noqa: BLE001
            errors.put(e)
        finally:
            c.close()

    threads = [threading.Thread(target=churn_health)] + [
        threading.Thread(target=allocate_loop) for _ in range(4)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
        assert not t.is_alive(), "worker hung"
    assert errors.empty(), errors.get()
    # the stream must still deliver resends after the churn
    resp = next(it)
    assert len(resp.devices) == 4
    stream.cancel()
    client.close()
    mgr.stop()


def test_preferred_allocation_packs_dies(tmp_path):
    """GetPreferredAllocation (implemented here; the reference error-stubs
    it) must pack a request onto as few dies as possible, honor
    must_include, and exact-fit small dies before breaking big ones."""
    from cea_amd.kube import protos as api

    cfg = GPUConfig(compute_partition="cpx")
    mgr, _ = make_manager(tmp_path, config=cfg, num_gpus=2,
                          partitions_per_gpu=8, compute_partition="CPX")
    serve_in_thread(mgr)
    sock = os.path.join(mgr.plugin_directory, mgr.socket_name)

    channel = grpc.insecure_channel(f"unix://{sock}")
    options = channel.unary_unary(
        api.DP_GET_OPTIONS,
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=api.DevicePluginOptions.FromString,
    )(api.Empty(), timeout=5)
    assert options.get_preferred_allocation_available

    preferred = channel.unary_unary(
        api.DP_GET_PREFERRED_ALLOCATION,
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=api.PreferredAllocationResponse.FromString,
    )

    # die0 has 3 free partitions, die1 has 8: a request for 3 should take
    # die0 whole (exact fit) instead of breaking die1
    req = api.PreferredAllocationRequest()
    c = req.container_requests.add()
    c.available_deviceIDs.extend(
        [f"amdgpu0/xcd{i}" for i in (1, 4, 6)]
        + [f"amdgpu1/xcd{i}" for i in range(8)])
    c.allocation_size = 3
    got = list(preferred(req, timeout=5).container_responses[0].device_ids)
    assert got == ["amdgpu0/xcd1", "amdgpu0/xcd4", "amdgpu0/xcd6"]

    # must_include on die1 pulls the rest from die1
    req = api.PreferredAllocationRequest()
    c = req.container_requests.add()
    c.available_deviceIDs.extend(
        [f"amdgpu0/xcd{i}" for i in (1, 4, 6)]
        + [f"amdgpu1/xcd{i}" for i in range(8)])
    c.must_include_deviceIDs.append("amdgpu1/xcd5")
    c.allocation_size = 4
    got = list(preferred(req, timeout=5).container_responses[0].device_ids)
    assert got[0] == "amdgpu1/xcd5"
    assert all(d.startswith("amdgpu1/") for d in got), got
    assert len(got) == 4

    # need spanning dies: take the big die whole, then the small one
    req = api.PreferredAllocationRequest()
    c = req.container_requests.add()
    c.available_deviceIDs.extend(
        [f"amdgpu0/xcd{i}" for i in (1, 4)]
        + [f"amdgpu1/xcd{i}" for i in range(8)])
    c.allocation_size = 9
    got = list(preferred(req, timeout=5).container_responses[0].device_ids)
    assert len(got) == 9
    assert sum(d.startswith("amdgpu1/") for d in got) == 8
    channel.close()
    mgr.stop()
