"""C-boundary tests for the native amdsmi shim.

A fake libamd_smi (csrc/fake_amdsmi.cpp, built to tests/_build/) is
LD_PRELOADed into a child python process so the REAL production shim
(cea_amd/amdsmi/libceaamd_smi.so) runs end-to-end on CPU: enumeration,
die attribution, the armed event-notification path and ECC polling.
This covers the layer the Python mock seam bypasses — the health-event
codes exercised here (thermal throttle, GPU pre/post reset, ECC) could
previously only be validated on real hardware.

Event-code parity table: docs/health-events.md.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
FAKE_SO = os.path.join(REPO, "tests", "_build", "libamd_smi.so")
SHIM_SO = os.path.join(REPO, "cea_amd", "amdsmi", "libceaamd_smi.so")
CHILD = os.path.join(REPO, "tests", "_shim_boundary_child.py")

pytestmark = pytest.mark.skipif(
    not os.path.exists("/opt/rocm/include/amd_smi/amdsmi.h"),
    reason="needs the amd_smi headers to build the shim + fake library",
)


@pytest.fixture(scope="module")
def fake_lib():
    if not os.path.exists(SHIM_SO):
        subprocess.run(["make", "smi"], cwd=REPO, check=True)
    subprocess.run(["make", "fake-smi"], cwd=REPO, check=True)
    return FAKE_SO


def run_child(fake_lib, tmp_path, scenario, devices=2, partitions=1,
              mode=None):
    ctrl = tmp_path / "ctrl"
    ctrl.mkdir(exist_ok=True)
    env = dict(os.environ)
    env.update({
        "LD_PRELOAD": fake_lib,
        "CEA_FAKE_SMI_DIR": str(ctrl),
        "CEA_FAKE_SMI_DEVICES": str(devices),
        "CEA_FAKE_SMI_PARTITIONS": str(partitions),
        "CEA_TEST_TMP": str(tmp_path),
    })
    if mode:
        env["CEA_FAKE_SMI_PARTITION_MODE"] = mode
    proc = subprocess.run(
        [sys.executable, CHILD, scenario],
        env=env, capture_output=True, text=True, timeout=120,
    )
    assert proc.returncode == 0, (proc.stdout, proc.stderr)
    line = next(l for l in proc.stdout.splitlines() if l.startswith("RESULT:"))
    return json.loads(line[len("RESULT:"):])


def test_shim_enumeration_spx(fake_lib, tmp_path):
    out = run_child(fake_lib, tmp_path, "enumerate", devices=2)
    assert out["count"] == 2
    d0, d1 = out["devices"]
    assert d0["uuid"] == "fake-uuid-0" and d1["uuid"] == "fake-uuid-1"
    assert d0["render_minor"] == 128 and d1["render_minor"] == 129
    assert d0["compute_partition"] == "SPX"
    # distinct serials => distinct dies
    assert d0["physical_index"] != d1["physical_index"]
    assert out["mem_total"] == 288 << 30
    assert out["driver"].startswith("6.")
    assert out["gfx"] == 42.0
    assert out["ecc0"] == 0


def test_shim_cpx_die_attribution(fake_lib, tmp_path):
    """16 enumerated devices, 8 partitions per die (CPX on a 2-die view):
    the shim's serial-based physical_index groups them correctly — the
    attribution key the health checker uses to mark every partition of a
    faulting die (analog of MIG UUID->GI/CI, health_checker.go:426-445)."""
    out = run_child(fake_lib, tmp_path, "enumerate", devices=16, partitions=8)
    devs = out["devices"]
    assert len(devs) == 16
    assert all(d["compute_partition"] == "CPX" for d in devs)
    for i, d in enumerate(devs):
        assert d["physical_index"] == i // 8, (i, d)
        assert d["partition_id"] == i % 8, (i, d)
    # per-partition VRAM = die total / 8
    assert out["mem_total"] == (288 << 30) // 8


def test_shim_event_path(fake_lib, tmp_path):
    """Thermal / pre-reset / post-reset / vmfault notifications flow
    through the real shim's mask arming + amdsmi_get_gpu_event_notification
    wait, with device UUID attribution."""
    out = run_child(fake_lib, tmp_path, "events", devices=2)
    evs = {(e["uuid"], e["code"]) for e in out["events"]}
    assert ("fake-uuid-0", 2) in evs   # THERMAL_THROTTLE
    assert ("fake-uuid-1", 3) in evs   # GPU_PRE_RESET
    assert ("fake-uuid-1", 4) in evs   # GPU_POST_RESET
    assert ("fake-uuid-0", 1) in evs   # VMFAULT
    msgs = {e["code"]: e["message"] for e in out["events"]}
    assert "thermal" in msgs[2]


def test_shim_default_mask_from_header(fake_lib, tmp_path):
    """The shim exports the header-derived default event mask (vmfault|
    thermal|pre/post reset|page fault start/end with this ROCm's enum)."""
    import ctypes
    lib = ctypes.CDLL(SHIM_SO)
    fn = lib.cea_smi_default_event_mask
    fn.restype = ctypes.c_ulonglong
    mask = int(fn())
    # AMDSMI_EVENT_MASK_FROM_INDEX(i) = 1 << (i-1); this ROCm: 1,2,3,4,7,8
    assert mask == (1 << 0) | (1 << 1) | (1 << 2) | (1 << 3) | (1 << 6) | (1 << 7)


def test_health_stack_ecc_to_recovery(fake_lib, tmp_path):
    """Full stack through the real C shim: ECC counter bump -> polling
    watchdog raises synthetic code 48 -> device Unhealthy + node
    condition; counter cleared + GPU_POST_RESET through the real event
    path -> fault-class re-verification passes -> device Healthy."""
    out = run_child(fake_lib, tmp_path, "health", devices=2)
    assert out["unhealthy"] == [["amdgpu1", "Unhealthy"]]
    assert out["condition_reason"] is not None
    assert 48 in json.loads(out["condition_reason"])
    assert out["recovered"] == [["amdgpu1", "Healthy"]]
    assert out["n_events"] >= 1
