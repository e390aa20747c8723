"""Health-checker tests — parity with the reference's
health_checker_test.go (TestCatchError:44, TestMonitorXidevent:359,
TestUpdateLastHeartbeatTime:266, TestResetXIDConditionWithBackoff:297),
driven by hand-built events + FakeKubeClient."""
import json

import pytest

import cea_amd.amdsmi as amdsmi
from cea_amd.amdsmi.iface import EVT_ECC_UNCORRECTABLE, EVT_THERMAL_THROTTLE, EVT_VMFAULT, Event
from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
from cea_amd.deviceplugin.health import CONDITION_TYPE, GPUHealthChecker
from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig
from cea_amd.kube.client import FakeKubeClient, KubeError


def make_env(tmp_path, num_gpus=2, partitions_per_gpu=1, compute_partition="SPX",
             config=None, boot_id="boot-1"):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, num_gpus * partitions_per_gpu)
    mock = MockAmdSmi(dev, compute_partition=compute_partition,
                      partitions_per_gpu=partitions_per_gpu)
    mock.init()
    amdsmi.set_ops(mock)
    mgr = AmdGPUManager(config or GPUConfig(), dev_directory=dev,
                        sysfs_root=str(tmp_path / "sys"))
    mgr.config.add_defaults_and_validate()
    mgr.start()
    kube = FakeKubeClient(nodes=[{
        "metadata": {"name": "node1", "labels": {}},
        "status": {"conditions": []},
    }])
    boot_path = tmp_path / "boot_id"
    boot_path.write_text(boot_id)
    hc = GPUHealthChecker(mgr, kube, node_name="node1",
                          boot_id_path=str(boot_path))
    return mgr, mock, kube, hc


def drain(q):
    out = []
    while not q.empty():
        out.append(q.get_nowait())
    return out


def test_critical_event_with_uuid_marks_matching_device(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_ECC_UNCORRECTABLE))
    unhealthy = drain(mgr.health)
    assert [d.ID for d in unhealthy] == ["amdgpu1"]
    assert unhealthy[0].health == "Unhealthy"
    # node condition carries the code set + boot id
    cond = [c for c in kube.nodes["node1"]["status"]["conditions"]
            if c["type"] == CONDITION_TYPE][0]
    assert json.loads(cond["reason"]) == [EVT_ECC_UNCORRECTABLE]
    assert cond["message"] == "boot-1"
    assert len(kube.events) == 1


def test_critical_event_without_uuid_marks_all(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=3)
    hc.catch_error(Event(device_uuid="", code=EVT_ECC_UNCORRECTABLE))
    ids = sorted(d.ID for d in drain(mgr.health))
    assert ids == ["amdgpu0", "amdgpu1", "amdgpu2"]


def test_monitor_only_event_sets_condition_not_health(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path)
    hc.catch_error(Event(device_uuid="mock-uuid-0", code=EVT_VMFAULT))
    assert drain(mgr.health) == []
    cond = [c for c in kube.nodes["node1"]["status"]["conditions"]
            if c["type"] == CONDITION_TYPE][0]
    assert json.loads(cond["reason"]) == [EVT_VMFAULT]


def test_non_critical_event_ignored(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path)
    hc.catch_error(Event(device_uuid="", code=EVT_THERMAL_THROTTLE))
    assert drain(mgr.health) == []
    assert kube.nodes["node1"]["status"]["conditions"] == []


def test_condition_merges_codes(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path)
    hc.catch_error(Event(device_uuid="", code=EVT_VMFAULT))
    hc.catch_error(Event(device_uuid="", code=EVT_ECC_UNCORRECTABLE))
    cond = [c for c in kube.nodes["node1"]["status"]["conditions"]
            if c["type"] == CONDITION_TYPE][0]
    assert json.loads(cond["reason"]) == sorted([EVT_VMFAULT, EVT_ECC_UNCORRECTABLE])


def test_cpx_die_event_marks_all_partitions(tmp_path):
    cfg = GPUConfig(compute_partition="cpx")
    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=1, partitions_per_gpu=8,
                                   compute_partition="CPX", config=cfg)
    hc.catch_error(Event(device_uuid="mock-uuid-0", code=EVT_ECC_UNCORRECTABLE))
    ids = sorted(d.ID for d in drain(mgr.health))
    assert ids == [f"amdgpu0/xcd{i}" for i in range(8)]


def test_reset_condition_on_boot_id_change(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path, boot_id="boot-2")
    kube.nodes["node1"]["status"]["conditions"] = [{
        "type": CONDITION_TYPE, "status": "True",
        "reason": "[48]", "message": "boot-1",  # stale: previous boot
    }]
    assert hc.try_reset_condition()
    assert kube.nodes["node1"]["status"]["conditions"] == []


def test_reset_keeps_condition_same_boot(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path, boot_id="boot-1")
    kube.nodes["node1"]["status"]["conditions"] = [{
        "type": CONDITION_TYPE, "status": "True",
        "reason": "[48]", "message": "boot-1",
    }]
    assert hc.try_reset_condition()
    assert len(kube.nodes["node1"]["status"]["conditions"]) == 1


def test_reset_retries_on_api_error(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path, boot_id="boot-2")
    kube.nodes["node1"]["status"]["conditions"] = [{
        "type": CONDITION_TYPE, "status": "True",
        "reason": "[48]", "message": "boot-1",
    }]
    calls = {"n": 0}

    def reactor(verb, resource, obj):
        if verb == "get" and resource == "nodes":
            calls["n"] += 1
            if calls["n"] == 1:
                raise KubeError(500, "boom")

    kube.prepend_reactor(reactor)
    assert not hc.try_reset_condition()
    assert hc.try_reset_condition()
    assert kube.nodes["node1"]["status"]["conditions"] == []


def test_heartbeat_updates_timestamp(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path)
    kube.nodes["node1"]["status"]["conditions"] = [{
        "type": CONDITION_TYPE, "status": "True",
        "reason": "[48]", "message": "boot-1",
        "lastHeartbeatTime": "2000-01-01T00:00:00Z",
    }]
    hc.update_heartbeat()
    cond = kube.nodes["node1"]["status"]["conditions"][0]
    assert cond["lastHeartbeatTime"] != "2000-01-01T00:00:00Z"


def test_ecc_poll_synthesizes_event(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path)
    hc.ecc_poll_interval_s = 0.05  # speed the 10 s poll up for the test
    hc.start()
    try:
        mock.set_ecc_count(1, 5)
        import time
        deadline = time.time() + 3
        found = []
        while time.time() < deadline and not found:
            found = drain(mgr.health)
            time.sleep(0.05)
        assert found, "ECC increase did not surface as a health event"
        assert found[0].ID == "amdgpu1"
    finally:
        hc.stop()


def test_event_loop_consumes_injected_events(tmp_path):
    mgr, mock, kube, hc = make_env(tmp_path)
    hc.start()
    try:
        import time
        mock.inject_event(Event(device_uuid="mock-uuid-0",
                                code=EVT_ECC_UNCORRECTABLE))
        deadline = time.time() + 3
        found = []
        while time.time() < deadline and not found:
            found = drain(mgr.health)
            time.sleep(0.05)
        assert [d.ID for d in found] == ["amdgpu0"]
    finally:
        hc.stop()


def test_xgmi_error_poll_raises_code_63(tmp_path):
    from cea_amd.amdsmi.iface import EVT_XGMI_ERROR

    mgr, mock, kube, hc = make_env(tmp_path)
    hc.ecc_poll_interval_s = 0.05
    hc.start()
    try:
        import time
        mock.set_xgmi_status(0, 1)
        deadline = time.time() + 3
        cond = None
        while time.time() < deadline and cond is None:
            conds = kube.nodes["node1"]["status"]["conditions"]
            cond = next((c for c in conds if c["type"] == CONDITION_TYPE), None)
            time.sleep(0.05)
        assert cond is not None, "xGMI error never reached the node condition"
        assert EVT_XGMI_ERROR in json.loads(cond["reason"])
        # monitor-only by default: no device goes Unhealthy
        assert drain(mgr.health) == []
        # only raised once per 0->error transition
        time.sleep(0.3)
        n_events = len(kube.events)
        time.sleep(0.3)
        assert len(kube.events) == n_events
    finally:
        hc.stop()


def test_critical_event_with_unknown_uuid_marks_all(tmp_path):
    """A critical event whose UUID matches no enumerated device marks every
    device: never silently drop a node-level critical signal over an id
    formatting mismatch between the event source and enumeration."""
    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2)
    hc.catch_error(Event(device_uuid="GPU-UNKNOWN-FORMAT",
                         code=EVT_ECC_UNCORRECTABLE))
    ids = sorted(d.ID for d in drain(mgr.health))
    assert ids == ["amdgpu0", "amdgpu1"]


def test_post_reset_recovers_unhealthy_device(tmp_path):
    """GPU_POST_RESET re-marks a previously Unhealthy device Healthy when it
    answers a probe again — capacity returns without a plugin restart
    (an improvement over the reference, which stays Unhealthy forever)."""
    from cea_amd.amdsmi.iface import EVT_GPU_POST_RESET

    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_ECC_UNCORRECTABLE))
    for d in drain(mgr.health):
        mgr.set_device_health(d.ID, d.health)   # what ListAndWatch does
    assert mgr.device_health["amdgpu1"] == "Unhealthy"

    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    recovered = drain(mgr.health)
    assert [d.ID for d in recovered] == ["amdgpu1"]
    assert recovered[0].health == "Healthy"

    # a healthy device's reset event pushes nothing
    hc.catch_error(Event(device_uuid="mock-uuid-0", code=EVT_GPU_POST_RESET))
    assert drain(mgr.health) == []


def test_post_reset_unmatched_uuid_recovers_nothing(tmp_path):
    """Recovery is strictly UUID-matched: a GPU_POST_RESET with an empty or
    unmatched UUID must NOT re-mark Unhealthy devices Healthy (the
    all-devices fallback is for *marking* Unhealthy only) — otherwise a
    vague reset signal returns persistently-bad devices to the pool
    (ADVICE r01 medium)."""
    from cea_amd.amdsmi.iface import EVT_GPU_POST_RESET

    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_ECC_UNCORRECTABLE))
    drain(mgr.health)

    hc.catch_error(Event(device_uuid="", code=EVT_GPU_POST_RESET))
    assert drain(mgr.health) == []
    hc.catch_error(Event(device_uuid="GPU-UNKNOWN-FORMAT",
                         code=EVT_GPU_POST_RESET))
    assert drain(mgr.health) == []
    # a correctly-attributed reset still recovers it
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    assert [d.ID for d in drain(mgr.health)] == ["amdgpu1"]


def test_post_reset_keeps_device_with_persistent_ecc(tmp_path):
    """A device whose uncorrectable-ECC counter is still above baseline
    after the reset stays Unhealthy: the reset did not clear the fault."""
    from cea_amd.amdsmi.iface import EVT_GPU_POST_RESET

    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2)
    hc._ecc_baseline = {0: 0, 1: 0}
    mock.set_ecc_count(1, 3)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_ECC_UNCORRECTABLE))
    drain(mgr.health)

    # counter still elevated after the reset => stays Unhealthy
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    assert drain(mgr.health) == []

    # reset actually cleared the counter => recovers, baseline resynced
    mock.set_ecc_count(1, 0)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    recovered = drain(mgr.health)
    assert [d.ID for d in recovered] == ["amdgpu1"]
    assert recovered[0].health == "Healthy"
    assert hc._ecc_baseline[1] == 0


def test_post_reset_keeps_device_with_persistent_xgmi_error(tmp_path):
    """Same for xGMI: link still in error state after reset => Unhealthy."""
    from cea_amd.amdsmi.iface import EVT_GPU_POST_RESET, EVT_XGMI_ERROR

    config = GPUConfig(health_critical_events={EVT_XGMI_ERROR})
    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2, config=config)
    mock.set_xgmi_status(1, 2)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_XGMI_ERROR))
    assert [d.ID for d in drain(mgr.health)] == ["amdgpu1"]

    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    assert drain(mgr.health) == []

    mock.set_xgmi_status(1, 0)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    assert [d.ID for d in drain(mgr.health)] == ["amdgpu1"]
    assert hc._xgmi_state.get(1, 0) == 0


def test_post_reset_recovery_without_listandwatch_drain(tmp_path):
    """Recovery must work even when kubelet never drained the health queue
    (manager.device_health empty): the checker tracks its own Unhealthy
    set."""
    from cea_amd.amdsmi.iface import EVT_GPU_POST_RESET

    mgr, mock, kube, hc = make_env(tmp_path, num_gpus=2)
    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_ECC_UNCORRECTABLE))
    # deliberately do NOT drain/sync mgr.device_health (kubelet disconnected);
    # the Unhealthy entry is still queued
    assert mgr.device_health.get("amdgpu1") is None

    hc.catch_error(Event(device_uuid="mock-uuid-1", code=EVT_GPU_POST_RESET))
    msgs = drain(mgr.health)
    assert msgs[-1].ID == "amdgpu1" and msgs[-1].health == "Healthy"
