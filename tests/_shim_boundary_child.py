"""Child process for the C-boundary shim tests (test_shim_c_boundary.py).

Runs with LD_PRELOAD=tests/_build/libamd_smi.so so the REAL native shim
(cea_amd/amdsmi/libceaamd_smi.so) executes against the fake amdsmi
library.  Prints one JSON result document on stdout.

Scenarios (argv[1]):
  enumerate  — device discovery incl. CPX die attribution through the
               shim's serial->physical_index logic
  events     — thermal/pre-reset/post-reset/vmfault notifications flow
               through the real event-mask arming + wait path
  health     — full stack: ECC bump -> Unhealthy -> GPU_POST_RESET with
               clean counters -> Healthy (the recovery contract), driven
               by the real shim, manager, and health checker
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import cea_amd.amdsmi as amdsmi  # noqa: E402
from cea_amd.amdsmi.shim import ShimAmdSmi  # noqa: E402

CTRL = os.environ["CEA_FAKE_SMI_DIR"]


def append_event(idx: int, code: int, msg: str = "injected") -> None:
    with open(os.path.join(CTRL, "events"), "a") as f:
        f.write(f"{idx} {code} {msg}\n")


def set_ecc(idx: int, count: int) -> None:
    with open(os.path.join(CTRL, f"ecc_{idx}"), "w") as f:
        f.write(str(count))


def scenario_enumerate(smi):
    out = {"count": smi.device_count(), "devices": []}
    for i in range(smi.device_count()):
        d = smi.device_info(i)
        out["devices"].append({
            "uuid": d.uuid, "name": d.name, "bdf": d.bdf,
            "render_minor": d.render_minor,
            "compute_partition": d.compute_partition,
            "partition_id": d.partition_id,
            "physical_index": d.physical_index,
            "vram_total_bytes": d.vram_total_bytes,
        })
    out["ecc0"] = smi.ecc_uncorrectable_count(0)
    out["driver"] = smi.driver_version()
    mem = smi.memory_info(0)
    out["mem_total"] = mem.total_bytes
    act = smi.gpu_activity(0)
    out["gfx"] = act.gfx_percent
    return out


def scenario_events(smi):
    # drain anything stale, then inject the four notification classes
    smi.wait_events(50)
    append_event(0, 2, "thermal throttle")
    append_event(1, 3, "pre reset")
    append_event(1, 4, "post reset")
    append_event(0, 1, "vmfault gpu page fault")
    events = []
    deadline = time.time() + 5
    while len(events) < 4 and time.time() < deadline:
        events.extend(
            {"uuid": e.device_uuid, "code": e.code, "message": e.message}
            for e in smi.wait_events(200)
        )
    return {"events": events}


def scenario_health(smi):
    from cea_amd.amdsmi.mock import make_fake_dev
    from cea_amd.deviceplugin.health import GPUHealthChecker
    from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig
    from cea_amd.kube.client import FakeKubeClient

    tmp = os.environ["CEA_TEST_TMP"]
    dev = os.path.join(tmp, "dev")
    make_fake_dev(dev, smi.device_count())
    mgr = AmdGPUManager(GPUConfig(), dev_directory=dev,
                        sysfs_root=os.path.join(tmp, "sys"))
    mgr.config.add_defaults_and_validate()
    mgr.start()
    kube = FakeKubeClient(nodes=[{
        "metadata": {"name": "node1", "labels": {}},
        "status": {"conditions": []},
    }])
    hc = GPUHealthChecker(mgr, kube, node_name="node1",
                          ecc_poll_interval_s=0.1)
    hc.start()
    out = {}
    try:
        # 1. ECC bump on device 1 -> Unhealthy via the polling watchdog
        set_ecc(1, 5)
        deadline = time.time() + 5
        unhealthy = []
        while time.time() < deadline and not unhealthy:
            while not mgr.health.empty():
                unhealthy.append(mgr.health.get_nowait())
            time.sleep(0.05)
        out["unhealthy"] = [(d.ID, d.health) for d in unhealthy]

        # 2. node condition carries the code
        conds = kube.nodes["node1"]["status"].get("conditions", [])
        cond = next((c for c in conds if c["type"] == "GPUCriticalError"), None)
        out["condition_reason"] = cond["reason"] if cond else None

        # 3. reset clears the counter; POST_RESET event via the real event
        #    path -> recovery re-verifies ECC through the shim and re-marks
        #    Healthy
        set_ecc(1, 0)
        append_event(1, 4, "post reset recovery")
        deadline = time.time() + 5
        recovered = []
        while time.time() < deadline and not recovered:
            while not mgr.health.empty():
                msg = mgr.health.get_nowait()
                if msg.health == "Healthy":
                    recovered.append(msg)
            time.sleep(0.05)
        out["recovered"] = [(d.ID, d.health) for d in recovered]

        # 4. events recorded against the node
        out["n_events"] = len(kube.events)
    finally:
        hc.stop()
        mgr.stop()
    return out


def main():
    scenario = sys.argv[1]
    smi = ShimAmdSmi(sampler_interval_ms=50)
    smi.init()
    amdsmi.set_ops(smi)
    fn = {"enumerate": scenario_enumerate, "events": scenario_events,
          "health": scenario_health}[scenario]
    out = fn(smi)
    smi.shutdown()
    print("RESULT:" + json.dumps(out))


if __name__ == "__main__":
    main()
