"""Table-driven unit tests: sharing policy, util, partition parsing, version
visibility, entrypoint config parsing — parity with the reference's
gpusharing_test.go, util_test.go, version_visibility_test.go and
nvidia_gpu.go flag parsing tests."""
import json
import os

import pytest

import cea_amd.amdsmi as amdsmi
from cea_amd.amdsmi.iface import numa_node_for_bdf
from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
from cea_amd.deviceplugin import sharing
from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig, GPUSharingConfig
from cea_amd.deviceplugin.partition import PartitionDeviceManager, parse_partition_mode
from cea_amd.deviceplugin.util import FileWatcher, device_name_from_path
from cea_amd.deviceplugin.version_visibility import parse_version
from cea_amd.kube.client import FakeKubeClient


# -- sharing (gpusharing_test.go:24,80 parity) ------------------------------
@pytest.mark.parametrize("dev_id,expect", [
    ("amdgpu0/vgpu1", "amdgpu0"),
    ("amdgpu3/vgpu0", "amdgpu3"),
    ("amdgpu0/xcd2/vgpu7", "amdgpu0/xcd2"),
])
def test_virtual_to_physical(dev_id, expect):
    assert sharing.virtual_to_physical(dev_id) == expect


@pytest.mark.parametrize("bad", ["amdgpu0", "amdgpu0/vgpuX", "vgpu1", ""])
def test_virtual_to_physical_invalid(bad):
    with pytest.raises(sharing.SharingError):
        sharing.virtual_to_physical(bad)


def test_validate_time_sharing():
    sharing.sharing_strategy = sharing.TIME_SHARING
    sharing.validate_request(["amdgpu0/vgpu0"], 2)
    with pytest.raises(sharing.SharingError):
        sharing.validate_request(["amdgpu0/vgpu0", "amdgpu0/vgpu1"], 2)


def test_validate_cu_fencing_multi_gpu_node():
    sharing.sharing_strategy = sharing.CU_FENCING
    sharing.validate_request(["amdgpu0/vgpu0", "amdgpu0/vgpu1"], 1)
    with pytest.raises(sharing.SharingError):
        sharing.validate_request(["amdgpu0/vgpu0", "amdgpu0/vgpu1"], 2)


# -- util (util_test.go:23 parity) ------------------------------------------
@pytest.mark.parametrize("path,expect", [
    ("/dev/dri/renderD128", "renderD128"),
    ("/dev/dri/renderD129", "renderD129"),
    ("/dev/kfd", None),
    ("/dev/dri/card0", None),
])
def test_device_name_from_path(path, expect):
    assert device_name_from_path(path) == expect


def test_file_watcher(tmp_path):
    target = tmp_path / "kubelet.sock"
    target.write_text("x")
    w = FileWatcher(str(target))
    assert w.changed() == []
    os.unlink(str(target))
    target.write_text("y")   # recreate = kubelet restart
    assert w.changed() == [str(target)]
    assert w.changed() == []


# -- numa sysfs (nvmlutil.go:114-151 parity) --------------------------------
def test_numa_node_for_bdf(tmp_path):
    d = tmp_path / "bus" / "pci" / "devices" / "0000:0c:00.0"
    d.mkdir(parents=True)
    (d / "numa_node").write_text("1\n")
    assert numa_node_for_bdf("0000:0c:00.0", str(tmp_path)) == 1
    assert numa_node_for_bdf("0c:00.0", str(tmp_path)) == 1
    (d / "numa_node").write_text("-1\n")
    assert numa_node_for_bdf("0000:0c:00.0", str(tmp_path)) is None
    assert numa_node_for_bdf("0000:ff:00.0", str(tmp_path)) is None


# -- partition ----------------------------------------------------------------
@pytest.mark.parametrize("s,expect", [
    ("cpx", "CPX"), ("CPX-NPS1", "CPX"), ("dpx-nps2", "DPX"), ("spx", "SPX"),
])
def test_parse_partition_mode(s, expect):
    assert parse_partition_mode(s) == expect


def test_parse_partition_mode_invalid():
    with pytest.raises(ValueError):
        parse_partition_mode("mig-1g.5gb")


def test_partition_nonuniform_rejected(tmp_path):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 7)  # 7 devices cannot be uniform CPX (needs 8/die)
    mock = MockAmdSmi(dev, compute_partition="CPX", partitions_per_gpu=8)
    mock.init()
    amdsmi.set_ops(mock)
    pm = PartitionDeviceManager("cpx", dev_root=dev)
    with pytest.raises(RuntimeError, match="non-uniform|partitions"):
        pm.start(str(tmp_path / "sys"))


def test_partition_wrong_mode_rejected(tmp_path):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 2)
    mock = MockAmdSmi(dev, compute_partition="SPX")
    mock.init()
    amdsmi.set_ops(mock)
    pm = PartitionDeviceManager("cpx", dev_root=dev)
    with pytest.raises(RuntimeError, match="partition_gpu"):
        pm.start(str(tmp_path / "sys"))


# -- config validation (manager.go:92-139 parity) ----------------------------
def test_config_validation():
    cfg = GPUConfig(gpu_sharing_config=GPUSharingConfig(
        gpu_sharing_strategy="bogus", max_shared_clients_per_gpu=2))
    with pytest.raises(ValueError):
        cfg.add_defaults_and_validate()
    cfg = GPUConfig(gpu_sharing_config=GPUSharingConfig(
        gpu_sharing_strategy="time-sharing", max_shared_clients_per_gpu=0))
    with pytest.raises(ValueError):
        cfg.add_defaults_and_validate()


def test_event_config_env_parsing():
    cfg = GPUConfig()
    cfg.add_health_critical_events("48, 63,79")
    assert cfg.health_critical_events == {48, 63, 79}
    with pytest.raises(ValueError):
        cfg.add_health_critical_events("48,abc")
    cfg2 = GPUConfig()
    cfg2.add_health_critical_events("")
    assert cfg2.health_critical_events == {48}


# -- cu-fencing envs ----------------------------------------------------------
def test_cu_fencing_envs(tmp_path):
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 1)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    cfg = GPUConfig(gpu_sharing_config=GPUSharingConfig(
        gpu_sharing_strategy="cu-fencing", max_shared_clients_per_gpu=4))
    cfg.add_defaults_and_validate()
    mgr = AmdGPUManager(cfg, dev_directory=dev, sysfs_root=str(tmp_path / "sys"))
    mgr.start()
    envs = mgr.envs(1)
    # 1 of 4 clients on a 256-CU die -> 64 CUs
    assert envs["HSA_CU_MASK"] == "0:0-63"
    envs = mgr.envs(2)
    assert envs["HSA_CU_MASK"] == "0:0-127"
    assert int(envs["CEA_AMD_VRAM_LIMIT_BYTES"]) == mock.vram_total_bytes // 2


# -- version visibility (version_visibility_test.go:26,86 parity) ------------
@pytest.mark.parametrize("ver,major,minor,rev", [
    ("6.10.5", "6", "10", "5"),
    ("6.8", "6", "8", "0"),
    ("6.10.5-2009582.22.04", "6", "10", "5"),
])
def test_parse_version(ver, major, minor, rev):
    out = parse_version(ver)
    assert out["amd.com/rocm.driver-version.major"] == major
    assert out["amd.com/rocm.driver-version.minor"] == minor
    assert out["amd.com/rocm.driver-version.revision"] == rev
    assert out["amd.com/rocm.driver-version.full"] == ver


def test_publish_annotations(tmp_path):
    from cea_amd.deviceplugin.version_visibility import (
        publish_driver_version_annotations,
    )

    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 1)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    kube = FakeKubeClient(nodes=[{"metadata": {"name": "n1"}, "status": {}}])
    publish_driver_version_annotations(kube, "n1")
    ann = kube.nodes["n1"]["metadata"]["annotations"]
    assert ann["amd.com/rocm.driver-version.full"] == "6.10.5"
    assert ann["amd.com/rocm.release"] == "7.2.0"
    # librccl ships in the ROCm image this runs in, so the RCCL release
    # annotation must be present and dotted (e.g. "2.27.7")
    from cea_amd.deviceplugin import version_visibility as vv
    if vv.rccl_version():
        assert ann["amd.com/rccl.version"].count(".") == 2


# -- entrypoint config parsing (nvidia_gpu.go:64-108 parity) ------------------
def test_parse_gpu_config_and_divisor(tmp_path):
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "cmd"))
    import amd_gpu

    cfg_path = tmp_path / "gpu_config.json"
    cfg_path.write_text(json.dumps({
        "ComputePartition": "cpx-nps1",
        "GPUSharingConfig": {
            "GPUSharingStrategy": "time-sharing",
            "MaxSharedClientsPerGPU": 2,
        },
    }))
    cfg = amd_gpu.parse_gpu_config(str(cfg_path))
    assert cfg.compute_partition == "cpx-nps1"
    assert cfg.gpu_sharing_config.gpu_sharing_strategy == "time-sharing"
    cfg.add_defaults_and_validate()

    assert amd_gpu.parse_gpu_config(str(tmp_path / "missing.json")).compute_partition == ""

    div = tmp_path / "div.txt"
    div.write_text("4\n")
    assert amd_gpu.parse_gpu_fraction_divisor(str(div)) == 4
    div.write_text("0")
    assert amd_gpu.parse_gpu_fraction_divisor(str(div)) == 1
    assert amd_gpu.parse_gpu_fraction_divisor(str(tmp_path / "nope")) == 1


def test_shell_scripts_parse():
    """Every installer/demo shell script must pass `bash -n`."""
    import glob
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    scripts = glob.glob(os.path.join(repo, "amd-driver-installer", "**", "*.sh"),
                        recursive=True) + \
        glob.glob(os.path.join(repo, "demo", "**", "*.sh"), recursive=True)
    assert len(scripts) >= 3, scripts
    for s in scripts:
        subprocess.run(["bash", "-n", s], check=True)


def test_gpu_readiness_helpers(tmp_path):
    """Readiness gate: devices_ready needs kfd + renderD* + amdgpu module;
    confidential type file is optional (persistenced-installer parity)."""
    import importlib.util
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    spec = importlib.util.spec_from_file_location(
        "gpu_readiness", os.path.join(repo, "cmd", "gpu_readiness.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)

    dev = tmp_path / "dev"
    sysfs = tmp_path / "sys"
    (dev / "dri").mkdir(parents=True)
    (sysfs / "module").mkdir(parents=True)
    assert not mod.devices_ready(str(dev), str(sysfs))
    (dev / "kfd").touch()
    (dev / "dri" / "renderD128").touch()
    assert not mod.devices_ready(str(dev), str(sysfs))  # module missing
    (sysfs / "module" / "amdgpu").mkdir()
    assert mod.devices_ready(str(dev), str(sysfs))

    assert mod.confidential_node_type(str(tmp_path / "absent")) == ""
    f = tmp_path / "ctype.txt"
    f.write_text("SEV-SNP\n")
    assert mod.confidential_node_type(str(f)) == "sev-snp"


def test_extra_mounts_only_when_host_path_exists(tmp_path):
    """OpenCL-ICD mount (the Vulkan-ICD analog): included read-only when the
    host dir exists, omitted otherwise."""
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 1)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    icd = tmp_path / "OpenCL" / "vendors"
    mgr = AmdGPUManager(
        GPUConfig(), dev_directory=dev, sysfs_root=str(tmp_path / "sys"),
        extra_mounts=[{"host_path": str(icd),
                       "container_path": "/etc/OpenCL/vendors"}])
    mgr.start()
    assert len(mgr.mounts()) == 1  # host dir absent
    icd.mkdir(parents=True)
    mounts = mgr.mounts()
    assert len(mounts) == 2
    assert mounts[1]["container_path"] == "/etc/OpenCL/vendors"
    assert mounts[1]["read_only"] is True


def test_gpu_doctor_mock(tmp_path, capsys):
    """gpu_doctor triage over the mock backend: healthy node exits 0; a
    driverless node (no kfd) exits 1."""
    import importlib.util
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    spec = importlib.util.spec_from_file_location(
        "gpu_doctor", os.path.join(repo, "cmd", "gpu_doctor.py"))
    doctor = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(doctor)

    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 2)
    assert doctor.main(["--mock-amdsmi", "--dev-directory", dev]) == 0
    out = capsys.readouterr().out
    assert "node HEALTHY" in out and out.count("[  ok]") >= 8

    empty = str(tmp_path / "nodev")
    os.makedirs(empty)
    assert doctor.main(["--mock-amdsmi", "--dev-directory", empty]) == 1


def test_partition_capability_annotations(tmp_path):
    """Node annotations advertise the KMD's supported partition modes +
    current mode, using the real sysfs format (pool_probe_r02.log)."""
    from cea_amd.deviceplugin.version_visibility import (
        partition_capabilities,
        publish_driver_version_annotations,
    )

    sys_root = tmp_path / "sys"
    d = sys_root / "class" / "drm" / "card0" / "device"
    os.makedirs(d)
    (d / "available_compute_partition").write_text("SPX, DPX, QPX, CPX\n")
    (d / "current_compute_partition").write_text("SPX\n")

    caps = partition_capabilities(str(sys_root))
    assert caps == {
        "amd.com/gpu.partition-modes": "CPX,DPX,QPX,SPX",
        "amd.com/gpu.compute-partition": "SPX",
    }

    # mixed current modes across cards are flagged
    d2 = sys_root / "class" / "drm" / "card1" / "device"
    os.makedirs(d2)
    (d2 / "available_compute_partition").write_text("SPX, DPX, QPX, CPX\n")
    (d2 / "current_compute_partition").write_text("CPX\n")
    caps = partition_capabilities(str(sys_root))
    assert caps["amd.com/gpu.compute-partition"] == "mixed"

    # end-to-end through the publisher
    dev = str(tmp_path / "dev")
    make_fake_dev(dev, 1)
    mock = MockAmdSmi(dev)
    mock.init()
    amdsmi.set_ops(mock)
    kube = FakeKubeClient(nodes=[{"metadata": {"name": "n1"}, "status": {}}])
    publish_driver_version_annotations(kube, "n1", sysfs_root=str(sys_root))
    ann = kube.nodes["n1"]["metadata"]["annotations"]
    assert ann["amd.com/gpu.partition-modes"] == "CPX,DPX,QPX,SPX"

    # no sysfs knobs (e.g. CPU CI container) => annotations simply absent
    assert partition_capabilities(str(tmp_path / "nosys")) == {}
