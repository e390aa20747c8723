"""CPU end-to-end of the real entrypoint binary (cmd/amd_gpu.py) with the
mock amdsmi backend — the whole startup sequence (config load, driver wait,
amdsmi init, discovery, kubelet registration, serve) as one subprocess."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from cea_amd.amdsmi.mock import make_fake_dev  # noqa: E402
from helpers import KubeletStub, PluginClient  # noqa: E402


def run_entrypoint(tmp_path, num_gpus=2, config=None, extra_args=()):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, num_gpus)
    plugin_dir = str(tmp_path / "plugin")
    os.makedirs(plugin_dir, exist_ok=True)
    cfg_path = str(tmp_path / "gpu_config.json")
    if config is not None:
        with open(cfg_path, "w") as f:
            json.dump(config, f)
    stub = KubeletStub(plugin_dir)
    stub.start()
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.Popen(
        [sys.executable, os.path.join(REPO, "cmd", "amd_gpu.py"),
         "--mock-amdsmi",
         "--dev-directory", dev,
         "--plugin-directory", plugin_dir,
         "--gpu-config", cfg_path, *extra_args],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    return proc, stub, plugin_dir


def stop(proc, stub):
    proc.terminate()
    try:
        proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        proc.kill()
    stub.stop()


def test_entrypoint_plain(tmp_path):
    proc, stub, plugin_dir = run_entrypoint(tmp_path, num_gpus=2)
    try:
        assert stub.registered.wait(30), "entrypoint never registered"
        assert stub.requests[0].resource_name == "amd.com/gpu"
        client = PluginClient(os.path.join(plugin_dir, "amdgpu.sock"))
        stream = client.list_and_watch_once(timeout=10)
        resp = next(iter(stream))
        assert sorted(d.ID for d in resp.devices) == ["amdgpu0", "amdgpu1"]
        stream.cancel()
        alloc = client.allocate([["amdgpu1"]])
        paths = [d.host_path for d in alloc.container_responses[0].devices]
        assert any("renderD129" in p for p in paths)
        client.close()
    finally:
        stop(proc, stub)


def test_entrypoint_time_sharing_config(tmp_path):
    cfg = {"GPUSharingConfig": {"GPUSharingStrategy": "time-sharing",
                                "MaxSharedClientsPerGPU": 2}}
    proc, stub, plugin_dir = run_entrypoint(tmp_path, num_gpus=1, config=cfg)
    try:
        assert stub.registered.wait(30)
        client = PluginClient(os.path.join(plugin_dir, "amdgpu.sock"))
        stream = client.list_and_watch_once(timeout=10)
        resp = next(iter(stream))
        assert sorted(d.ID for d in resp.devices) == [
            "amdgpu0/vgpu0", "amdgpu0/vgpu1"]
        stream.cancel()
        client.close()
    finally:
        stop(proc, stub)


def test_entrypoint_graceful_sigterm(tmp_path):
    """SIGTERM (kubelet pod termination) must exit 0 within the watchdog
    poll interval, with the serve loop shut down cleanly."""
    import signal

    proc, stub, plugin_dir = run_entrypoint(tmp_path, num_gpus=1)
    try:
        assert stub.registered.wait(30), "entrypoint never registered"
        proc.send_signal(signal.SIGTERM)
        rc = proc.wait(timeout=15)
        assert rc == 0, proc.stdout.read()[-1000:]
    finally:
        stop(proc, stub)
