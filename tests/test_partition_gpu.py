"""partition_gpu tests — parity with partition_gpu_test.go
(Test_buildPartitionStr:22, Test_parseLGIOutput:68 with pasted CLI output,
Test_checkDesired:173) using a fake amd-smi runner."""
import json

import pytest

from cea_amd.partition import partition_gpu as pg

# Pasted from a real MI355X (gpurun capture, amd-smi 26.2):
AMDSMI_SPX = """\
GPU: 0
    PARTITION:
        ACCELERATOR_PARTITION: SPX
        MEMORY_PARTITION: NPS1
        PARTITION_ID: 0
"""

AMDSMI_CPX_2DIES = """\
GPU: 0
    PARTITION:
        ACCELERATOR_PARTITION: CPX
        MEMORY_PARTITION: NPS1
        PARTITION_ID: 0
GPU: 1
    PARTITION:
        ACCELERATOR_PARTITION: CPX
        MEMORY_PARTITION: NPS1
        PARTITION_ID: 1
"""

AMDSMI_OLD_KEY = """\
GPU: 0
    PARTITION:
        COMPUTE_PARTITION: DPX
        MEMORY_PARTITION: NPS2
        PARTITION_ID: 0
"""


def test_parse_partition_status():
    s = pg.parse_partition_status(AMDSMI_SPX)
    assert s == [{"gpu": "0", "accelerator_partition": "SPX",
                  "memory_partition": "NPS1", "partition_id": "0"}]
    s = pg.parse_partition_status(AMDSMI_CPX_2DIES)
    assert len(s) == 2 and s[1]["partition_id"] == "1"
    s = pg.parse_partition_status(AMDSMI_OLD_KEY)
    assert s[0]["accelerator_partition"] == "DPX"
    assert pg.parse_partition_status("garbage\nno gpus here") == []


@pytest.mark.parametrize("text,compute,memory,expect", [
    (AMDSMI_SPX, "SPX", "NPS1", True),
    (AMDSMI_SPX, "CPX", "NPS1", False),
    (AMDSMI_CPX_2DIES, "CPX", "NPS1", True),
    (AMDSMI_OLD_KEY, "DPX", "NPS2", True),
    (AMDSMI_OLD_KEY, "DPX", "NPS1", False),
])
def test_check_desired(text, compute, memory, expect):
    assert pg.check_desired(pg.parse_partition_status(text), compute, memory) == expect


def write_config(tmp_path, spec):
    p = tmp_path / "gpu_config.json"
    p.write_text(json.dumps({"ComputePartition": spec}))
    return str(p)


def test_parse_partition_config(tmp_path):
    assert pg.parse_partition_config(write_config(tmp_path, "cpx-nps1")) == ("CPX", "NPS1")
    assert pg.parse_partition_config(write_config(tmp_path, "dpx")) == ("DPX", "NPS1")
    with pytest.raises(ValueError):
        pg.parse_partition_config(write_config(tmp_path, "1g.5gb"))


class FakeRunner:
    def __init__(self, status_sequence, set_results=None):
        self.status_sequence = list(status_sequence)
        self.set_results = list(set_results or [])
        self.commands = []

    def __call__(self, cmd):
        self.commands.append(cmd)
        if cmd[:2] == ["amd-smi", "static"]:
            return 0, self.status_sequence.pop(0)
        if cmd[:2] == ["amd-smi", "set"]:
            if self.set_results:
                return self.set_results.pop(0)
            return 0, ""
        return 1, f"unexpected command {cmd}"


def test_run_idempotent(tmp_path):
    runner = FakeRunner([AMDSMI_SPX])
    changed = pg.run(write_config(tmp_path, "spx"), runner)
    assert changed is False
    assert all(c[:2] != ["amd-smi", "set"] for c in runner.commands)


def test_run_applies_cpx(tmp_path):
    runner = FakeRunner([AMDSMI_SPX, AMDSMI_CPX_2DIES])
    changed = pg.run(write_config(tmp_path, "cpx-nps1"), runner)
    assert changed is True
    sets = [c for c in runner.commands if c[:2] == ["amd-smi", "set"]]
    assert sets == [["amd-smi", "set", "--gpu", "all",
                     "--compute-partition", "CPX"]]


def test_run_busy_retry(tmp_path, monkeypatch):
    monkeypatch.setattr(pg, "BUSY_RETRY_DELAY_S", 0.01)
    runner = FakeRunner(
        [AMDSMI_SPX, AMDSMI_CPX_2DIES],
        set_results=[(1, "Error: GPU is busy"), (0, "")],
    )
    assert pg.run(write_config(tmp_path, "cpx"), runner) is True
    sets = [c for c in runner.commands if c[:2] == ["amd-smi", "set"]]
    assert len(sets) == 2  # busy then success


def test_run_hard_failure(tmp_path):
    runner = FakeRunner([AMDSMI_SPX], set_results=[(1, "Error: not supported")])
    with pytest.raises(pg.PartitionError, match="not supported"):
        pg.run(write_config(tmp_path, "cpx"), runner,
               sysfs_root=str(tmp_path / "sys"))


def test_sysfs_fallback(tmp_path):
    # amd-smi refuses, but the KMD sysfs knob accepts -> partitioning
    # proceeds through the fallback writer
    sysdir = tmp_path / "sys" / "class" / "drm" / "card0" / "device"
    sysdir.mkdir(parents=True)
    (sysdir / "available_compute_partition").write_text("SPX, DPX, QPX, CPX\n")
    (sysdir / "current_compute_partition").write_text("SPX\n")
    runner = FakeRunner([AMDSMI_SPX, AMDSMI_CPX_2DIES],
                        set_results=[(1, "AMDSMI_STATUS_UNKNOWN_ERROR")])
    assert pg.run(write_config(tmp_path, "cpx"), runner,
                  sysfs_root=str(tmp_path / "sys")) is True
    assert (sysdir / "current_compute_partition").read_text().strip() == "CPX"


def test_sysfs_fallback_mode_unavailable(tmp_path):
    sysdir = tmp_path / "sys" / "class" / "drm" / "card0" / "device"
    sysdir.mkdir(parents=True)
    (sysdir / "available_compute_partition").write_text("SPX\n")
    (sysdir / "current_compute_partition").write_text("SPX\n")
    runner = FakeRunner([AMDSMI_SPX], set_results=[(1, "unknown error")])
    with pytest.raises(pg.PartitionError):
        pg.run(write_config(tmp_path, "cpx"), runner,
               sysfs_root=str(tmp_path / "sys"))


def test_run_verification_failure(tmp_path):
    runner = FakeRunner([AMDSMI_SPX, AMDSMI_SPX])  # set "succeeds" but no change
    with pytest.raises(pg.PartitionError, match="verification failed"):
        pg.run(write_config(tmp_path, "cpx"), runner,
               sysfs_root=str(tmp_path / "sys"))


def test_memory_partition_change_ordered_first(tmp_path, monkeypatch):
    monkeypatch.setattr(pg, "BUSY_RETRY_DELAY_S", 0.01)
    runner = FakeRunner([AMDSMI_OLD_KEY, AMDSMI_CPX_2DIES])
    pg.run(write_config(tmp_path, "cpx-nps1"), runner)
    sets = [c for c in runner.commands if c[:2] == ["amd-smi", "set"]]
    assert sets[0][-2:] == ["--memory-partition", "NPS1"]
    assert sets[1][-2:] == ["--compute-partition", "CPX"]
