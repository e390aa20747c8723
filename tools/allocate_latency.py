#!/usr/bin/env python3
"""Allocate() latency distribution: serial p50/p90/p99 plus tail
behavior under concurrent clients — the full picture behind the single
p50 the bench reports (BASELINE metric; VERDICT r01 tracked item).

Pure control plane (mock amdsmi, unix-socket gRPC against the real
plugin service); runs on CPU or GPU boxes alike.

Usage: python3 tools/allocate_latency.py [iters] [out.json]
"""
from __future__ import annotations

import json
import os
import statistics
import sys
import tempfile
import threading
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pct(sorted_vals, p):
    i = min(len(sorted_vals) - 1, int(round(p / 100 * (len(sorted_vals) - 1))))
    return sorted_vals[i]


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 2000
    out_path = sys.argv[2] if len(sys.argv) > 2 else \
        "gpurun_out/allocate_latency.json"

    import grpc

    import cea_amd.amdsmi as amdsmi
    from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
    from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig
    from cea_amd.kube import protos as api

    with tempfile.TemporaryDirectory() as tmp:
        dev = os.path.join(tmp, "dev")
        make_fake_dev(dev, 8)
        mock = MockAmdSmi(dev)
        mock.init()
        amdsmi.set_ops(mock)
        plugin_dir = os.path.join(tmp, "plugin")
        os.makedirs(plugin_dir)
        mgr = AmdGPUManager(GPUConfig(), dev_directory=dev,
                            plugin_directory=plugin_dir)
        mgr.config.add_defaults_and_validate()
        mgr.start()
        t = threading.Thread(target=mgr.serve, daemon=True)
        t.start()
        sock = os.path.join(plugin_dir, mgr.socket_name)
        deadline = time.time() + 10
        while not os.path.exists(sock) and time.time() < deadline:
            time.sleep(0.01)

        def client_loop(n, out, dev_id="amdgpu0"):
            channel = grpc.insecure_channel(f"unix://{sock}")
            allocate = channel.unary_unary(
                api.DP_ALLOCATE,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=api.AllocateResponse.FromString,
            )
            req = api.AllocateRequest()
            req.container_requests.add(devices_ids=[dev_id])
            allocate(req, timeout=5)  # warm the channel
            for _ in range(n):
                t0 = time.perf_counter()
                allocate(req, timeout=5)
                out.append((time.perf_counter() - t0) * 1e6)
            channel.close()

        # serial distribution
        serial = []
        client_loop(iters, serial)
        serial.sort()

        # 8 concurrent clients (one per fake GPU), tail under contention
        threads = []
        per = [[] for _ in range(8)]
        for i in range(8):
            th = threading.Thread(
                target=client_loop, args=(iters // 4, per[i], f"amdgpu{i}"))
            threads.append(th)
        t0 = time.time()
        for th in threads:
            th.start()
        for th in threads:
            th.join()
        conc_wall = time.time() - t0
        conc = sorted(v for lst in per for v in lst)

        result = {
            "iters": iters,
            "serial_us": {
                "p50": round(pct(serial, 50), 1),
                "p90": round(pct(serial, 90), 1),
                "p99": round(pct(serial, 99), 1),
                "mean": round(statistics.mean(serial), 1),
            },
            "concurrent8_us": {
                "p50": round(pct(conc, 50), 1),
                "p90": round(pct(conc, 90), 1),
                "p99": round(pct(conc, 99), 1),
                "throughput_rps": round(len(conc) / conc_wall, 0),
            },
        }
        mgr.stop()
        amdsmi.ops = None
    os.makedirs(os.path.dirname(out_path), exist_ok=True)
    json.dump(result, open(out_path, "w"), indent=1)
    print(json.dumps(result))
    return 0


if __name__ == "__main__":
    sys.exit(main())
