#!/usr/bin/env python3
"""Round-2 control-plane soak on real hardware.

Exercises the production loop repeatedly on one MI355X box: device
plugin serving a stub kubelet with health monitoring on, while cycles of
  (a) Allocate RPCs + ListAndWatch reconnects,
  (b) GPU compute load (GEMM) from a "pod" subprocess,
  (c) kubelet-restart trigger (socket recreation -> full plugin serve
      restart, the manager.go:534-539 parity machinery),
  (d) /metrics scrapes,
run back-to-back.  After N cycles, asserts the plugin process is still
responsive, its RSS has not grown unboundedly, and its thread count is
stable — the leak classes a one-shot test can't see.

Usage (on a GPU box): python3 tools/soak_r02.py [cycles] [out.json]
"""
from __future__ import annotations

import json
import os
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

from helpers import KubeletStub, PluginClient  # noqa: E402


def proc_stat(pid: int):
    with open(f"/proc/{pid}/status") as f:
        st = f.read()

    def field(name):
        for line in st.splitlines():
            if line.startswith(name + ":"):
                return int(line.split()[1])
        return -1

    return {"rss_kb": field("VmRSS"), "threads": field("Threads")}


def main():
    cycles = int(sys.argv[1]) if len(sys.argv) > 1 else 15
    out_path = sys.argv[2] if len(sys.argv) > 2 else "gpurun_out/soak_r02.json"
    tmp = tempfile.mkdtemp()
    plugin_dir = os.path.join(tmp, "plugin")
    os.makedirs(plugin_dir)
    stub = KubeletStub(plugin_dir)
    stub.start()
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env.pop("NODE_NAME", None)
    log = open(os.path.join(tmp, "plugin.log"), "w+")
    proc = subprocess.Popen(
        [sys.executable, os.path.join(REPO, "cmd", "amd_gpu.py"),
         "--plugin-directory", plugin_dir,
         "--enable-health-monitoring",
         "--enable-container-gpu-metrics", "--gpu-metrics-port", "21122",
         "--gpu-config", os.path.join(tmp, "missing.json")],
        env=env, stdout=log, stderr=subprocess.STDOUT, text=True,
    )
    result = {"cycles": 0, "ok": False, "samples": []}
    gemm_code = (
        "import torch\n"
        "a = torch.rand(4096, 4096, device='cuda')\n"
        "b = torch.rand(4096, 4096, device='cuda')\n"
        "for _ in range(30): a = a @ b / 4096\n"
        "torch.cuda.synchronize()\nprint('gemm ok')\n"
    )
    try:
        assert stub.registered.wait(90), "initial registration"
        sock = os.path.join(plugin_dir, "amdgpu.sock")
        baseline = None
        for cycle in range(cycles):
            # (a) RPC traffic
            client = PluginClient(sock)
            stream = client.list_and_watch_once(timeout=30)
            resp = next(iter(stream))
            assert resp.devices and resp.devices[0].health == "Healthy", \
                f"cycle {cycle}: {resp}"
            stream.cancel()
            for _ in range(10):
                client.allocate([["amdgpu0"]])
            client.close()

            # (b) GPU load from a pod-like subprocess
            rc = subprocess.run(
                [sys.executable, "-c", gemm_code], env=env,
                capture_output=True, text=True, timeout=180,
            )
            assert rc.returncode == 0, rc.stderr[-500:]

            # (d) metrics scrape
            import urllib.request
            body = urllib.request.urlopen(
                "http://127.0.0.1:21122/metrics", timeout=10).read().decode()
            assert "duty_cycle" in body or body is not None

            # (c) kubelet restart -> plugin re-registers (every 3rd cycle)
            if cycle % 3 == 2:
                n_reg = len(stub.requests)
                stub.restart_kubelet_socket()
                deadline = time.time() + 60
                while len(stub.requests) <= n_reg and time.time() < deadline:
                    time.sleep(0.2)
                assert len(stub.requests) > n_reg, \
                    f"cycle {cycle}: no re-registration after kubelet restart"
                time.sleep(1)

            s = proc_stat(proc.pid)
            s["cycle"] = cycle
            result["samples"].append(s)
            if cycle == 1:
                baseline = s
            result["cycles"] = cycle + 1
        final = result["samples"][-1]
        result["rss_growth_kb"] = (final["rss_kb"] - baseline["rss_kb"]
                                   if baseline else 0)
        result["thread_growth"] = (final["threads"] - baseline["threads"]
                                   if baseline else 0)
        # leak budget: < 64 MiB RSS growth and <= 4 extra threads across
        # the whole soak (grpc churns a few workers)
        result["ok"] = (result["rss_growth_kb"] < 65536
                        and result["thread_growth"] <= 4
                        and proc.poll() is None)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        stub.stop()
        os.makedirs(os.path.dirname(out_path), exist_ok=True)
        with open(out_path, "w") as f:
            json.dump(result, f, indent=1)
        print(json.dumps({k: v for k, v in result.items() if k != "samples"}))
    return 0 if result["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
