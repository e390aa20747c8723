"""Serve/restart state-machine tests — the subtle part of the reference
(manager.go:442-549, three concurrent restart triggers): socket vanish,
kubelet restart, GPU hot-add must each restart + re-register the plugin."""
import os
import threading
import time

import cea_amd.amdsmi as amdsmi
from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
from cea_amd.deviceplugin import manager as mgr_mod
from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig

from helpers import KubeletStub


def make_mgr(tmp_path, monkeypatch):
    # speed up the watchdog cadences for the test
    monkeypatch.setattr(mgr_mod, "SOCKET_CHECK_INTERVAL_S", 0.05)
    monkeypatch.setattr(mgr_mod, "GPU_CHECK_INTERVAL_S", 0.2)
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 1)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    plugin_dir = str(tmp_path / "plugin")
    os.makedirs(plugin_dir)
    m = AmdGPUManager(GPUConfig(), dev_directory=dev,
                      plugin_directory=plugin_dir,
                      sysfs_root=str(tmp_path / "sys"))
    m.config.add_defaults_and_validate()
    m.start()
    return m


def wait_for(cond, timeout=5):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if cond():
            return True
        time.sleep(0.02)
    return False


def test_socket_removal_triggers_restart_and_reregister(tmp_path, monkeypatch):
    m = make_mgr(tmp_path, monkeypatch)
    stub = KubeletStub(m.plugin_directory)
    stub.start()
    try:
        t = threading.Thread(target=m.serve, kwargs={"max_restarts": 5},
                             daemon=True)
        t.start()
        sock = os.path.join(m.plugin_directory, m.socket_name)
        assert wait_for(lambda: len(stub.requests) >= 1)
        os.unlink(sock)   # trigger (a): plugin socket vanished
        assert wait_for(lambda: len(stub.requests) >= 2), \
            "no re-registration after socket removal"
    finally:
        m.stop()
        stub.stop()


def test_kubelet_restart_triggers_restart(tmp_path, monkeypatch):
    m = make_mgr(tmp_path, monkeypatch)
    stub = KubeletStub(m.plugin_directory)
    stub.start()
    try:
        t = threading.Thread(target=m.serve, kwargs={"max_restarts": 5},
                             daemon=True)
        t.start()
        assert wait_for(lambda: len(stub.requests) >= 1)
        # trigger (c): kubelet.sock recreated (kubelet restart)
        stub.stop()
        if os.path.exists(stub.socket_path):
            os.unlink(stub.socket_path)
        stub2 = KubeletStub(m.plugin_directory)
        stub2.start()
        try:
            assert wait_for(lambda: len(stub2.requests) >= 1, timeout=8), \
                "no re-registration after kubelet restart"
        finally:
            stub2.stop()
    finally:
        m.stop()


def test_gpu_hot_add_triggers_restart(tmp_path, monkeypatch):
    m = make_mgr(tmp_path, monkeypatch)
    stub = KubeletStub(m.plugin_directory)
    stub.start()
    try:
        t = threading.Thread(target=m.serve, kwargs={"max_restarts": 5},
                             daemon=True)
        t.start()
        assert wait_for(lambda: len(stub.requests) >= 1)
        # trigger (b): a new render node appears
        open(os.path.join(m.dev_directory, "dri", "renderD200"), "w").close()
        assert wait_for(lambda: len(stub.requests) >= 2, timeout=8), \
            "no re-registration after GPU hot-add"
        # after restart, discovery picked up the new device
        assert wait_for(lambda: len(m.devices) == 2, timeout=5)
    finally:
        m.stop()
        stub.stop()
