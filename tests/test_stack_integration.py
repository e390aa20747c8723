"""Cross-component CPU integration: BASELINE configs 3+5 simulated in one
test — a CPX-partitioned node pool, the topology gang scheduler binding a
job onto it, and the device plugin serving that job's allocation (with
die-aware preferred allocation) over a real unix-socket gRPC round trip."""
import os
import threading
import time

import grpc

import cea_amd.amdsmi as amdsmi
from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig
from cea_amd.kube import protos as api
from cea_amd.kube.client import FakeKubeClient
from cea_amd.scheduler import daemon as sched_daemon

from helpers import KubeletStub, PluginClient


def _node(name, block, gpus):
    return {
        "metadata": {"name": name, "labels": {
            "topology.cea-amd.io/block": block,
            "topology.cea-amd.io/subblock": "s1",
            "topology.cea-amd.io/host": name,
        }},
        "spec": {"taints": []},
        "status": {
            "conditions": [{"type": "Ready", "status": "True"}],
            "allocatable": {"cpu": "64", "memory": "512Gi",
                            "amd.com/gpu": str(gpus)},
        },
    }


def _pod(name, idx, gpus):
    return {
        "metadata": {"name": name, "namespace": "default", "labels": {
            "job-name": "cpxjob",
            "batch.kubernetes.io/job-completion-index": str(idx),
        }},
        "spec": {
            "containers": [{"name": "main", "resources": {
                "requests": {"amd.com/gpu": str(gpus), "cpu": "4",
                             "memory": "16Gi"}}}],
            "schedulingGates": [{"name": "gke.io/topology-aware-auto-cpx"}],
        },
        "status": {"phase": "Pending"},
    }


def test_cpx_gang_schedule_then_allocate(tmp_path):
    # --- cluster side: 2-pod gang of 8 partitions each over CPX nodes ------
    # (each CPX MI355X node advertises 16 = 2 dies x 8 partitions)
    kube = FakeKubeClient(
        nodes=[_node("gpu-a", "b1", 16), _node("gpu-b", "b1", 16),
               _node("far-c", "b9", 16)],
        pods=[_pod("cpxjob-0", 0, 8), _pod("cpxjob-1", 1, 8)],
    )
    sched = sched_daemon.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 2
    chosen = set()
    for name in ("cpxjob-0", "cpxjob-1"):
        p = kube.get_pod("default", name)
        assert p["spec"]["schedulingGates"] == []
        chosen.add(p["spec"]["affinity"]["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"
        ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"][0])
    assert chosen <= {"gpu-a", "gpu-b"}, chosen  # same block, never far-c

    # --- node side: the plugin on a CPX node serves the allocation ---------
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 16)
    mock = MockAmdSmi(dev, compute_partition="CPX", partitions_per_gpu=8)
    mock.init()
    amdsmi.set_ops(mock)
    plugin_dir = str(tmp_path / "plugin")
    os.makedirs(plugin_dir)
    cfg = GPUConfig(compute_partition="cpx")
    cfg.add_defaults_and_validate()
    mgr = AmdGPUManager(cfg, dev_directory=dev, plugin_directory=plugin_dir,
                        sysfs_root=str(tmp_path / "sys"))
    mgr.start()
    stub = KubeletStub(plugin_dir)
    stub.start()
    t = threading.Thread(target=mgr.serve, daemon=True)
    t.start()
    sock = os.path.join(plugin_dir, mgr.socket_name)
    deadline = time.time() + 5
    while not os.path.exists(sock) and time.time() < deadline:
        time.sleep(0.02)
    try:
        assert stub.registered.wait(10)
        # registration advertises preferred allocation
        assert stub.requests[0].options.get_preferred_allocation_available

        client = PluginClient(sock)
        stream = client.list_and_watch_once()
        resp = next(iter(stream))
        assert len(resp.devices) == 16  # 2 dies x 8 CPX partitions
        stream.cancel()

        # kubelet asks for the preferred 8-of-16 -> one whole die
        channel = grpc.insecure_channel(f"unix://{sock}")
        preferred = channel.unary_unary(
            api.DP_GET_PREFERRED_ALLOCATION,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.PreferredAllocationResponse.FromString,
        )
        req = api.PreferredAllocationRequest()
        c = req.container_requests.add()
        c.available_deviceIDs.extend(d.ID for d in resp.devices)
        c.allocation_size = 8
        got = list(preferred(req, timeout=5).container_responses[0].device_ids)
        dies = {d.split("/")[0] for d in got}
        assert len(got) == 8 and len(dies) == 1, got

        # ... then allocates exactly those; specs carry kfd + render nodes
        alloc = client.allocate([got])
        cresp = alloc.container_responses[0]
        paths = [d.host_path for d in cresp.devices]
        assert sum("renderD" in p for p in paths) == 8
        assert any(p.endswith("/kfd") for p in paths)
        channel.close()
        client.close()
    finally:
        mgr.stop()
        stub.stop()
