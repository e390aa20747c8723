"""CPX enumeration-model validation against real-hardware observables.

The gpurun pool denies the runtime CPX flip (amd-smi set returns
AMDSMI_STATUS_UNKNOWN_ERROR; sysfs knob is EROFS — profiles/
pool_probe_r02.log), so the CPX discovery path cannot run live in this
environment.  Per VERDICT r01 #3, this file pins the mock's model to
every observable a real MI355X DOES expose, captured verbatim from the
round-2 pool probe (profiles/pool_probe_r02.log).  If any of these
fixtures ever disagrees with the model, CPX discovery would be wrong on
real hardware — this is the tripwire.

Residual risk (documented in docs/gpu-sharing-and-partitioning.md):
post-flip render-node re-enumeration order and per-partition UUID
derivation are inferred from the KMD contract and the SPX observables,
not observed live.
"""
import pytest

from cea_amd.partition import partition_gpu as pg

# --- verbatim captures from a real MI355X (profiles/pool_probe_r02.log) ---

REAL_AMDSMI_LIST = """\
GPU: 0
    BDF: 0000:26:00.0
    UUID: 6aff75a3-0000-1000-800a-ba0c188eedbe
    KFD_ID: 36622
    NODE_ID: 5
    PARTITION_ID: 0
"""

REAL_STATIC_PARTITION = """\
GPU: 0
    PARTITION:
        ACCELERATOR_PARTITION: SPX
        MEMORY_PARTITION: NPS1
        PARTITION_ID: 0
"""

REAL_PARTITION_PROFILES = """\
CURRENT_PARTITION:
GPU_ID  MEMORY  ACCELERATOR_TYPE  ACCELERATOR_PROFILE_INDEX  PARTITION_ID
0       NPS1    SPX               0                          0

MEMORY_PARTITION:
GPU_ID  MEMORY_PARTITION_CAPS  CURRENT_MEMORY_PARTITION
0       NPS1,NPS2              NPS1

ACCELERATOR_PARTITION_PROFILES:
GPU_ID  PROFILE_INDEX  MEMORY_PARTITION_CAPS  ACCELERATOR_TYPE  PARTITION_ID     NUM_PARTITIONS  NUM_RESOURCES  RESOURCE_INDEX  RESOURCE_TYPE  RESOURCE_INSTANCES  RESOURCES_SHARED
0       0              NPS1                   SPX*              0                1               4              0               XCC            8                   1
                                                                                                                1               DECODER        4                   1
                                                                                                                2               DMA            16                  1
                                                                                                                3               JPEG           40                  1
        1              NPS1                   DPX               N/A              2               4              4               XCC            8                   1
                                                                                                                5               DECODER        4                   1
                                                                                                                6               DMA            16                  1
                                                                                                                7               JPEG           40                  1
        2              NPS1                   QPX               N/A              4               4              8               XCC            8                   1
                                                                                                                9               DECODER        4                   1
                                                                                                                10              DMA            16                  1
                                                                                                                11              JPEG           40                  1
        3              NPS1                   CPX               N/A              8               4              12              XCC            8                   1
                                                                                                                13              DECODER        4                   1
                                                                                                                14              DMA            16                  1
                                                                                                                15              JPEG           40                  1
"""

REAL_SYSFS_AVAILABLE = "SPX, DPX, QPX, CPX"
REAL_KFD_PROPS = {  # /sys/class/kfd/kfd/topology/nodes/5/properties
    "simd_count": 1024,
    "gfx_target_version": 90500,
    "drm_render_minor": 152,
    "num_xcc": 8,
}


def test_hardware_profile_table_matches_partition_count_model():
    """The hardware's NUM_PARTITIONS per mode equals PARTITION_COUNT —
    the table every component (partitioner, device fan-out, preferred
    allocation) sizes itself from."""
    caps = pg.parse_partition_profiles(REAL_PARTITION_PROFILES)
    assert caps == {"SPX": 1, "DPX": 2, "QPX": 4, "CPX": 8}
    assert caps == pg.PARTITION_COUNT


def test_static_partition_parse_on_real_output():
    states = pg.parse_partition_status(REAL_STATIC_PARTITION)
    assert states == [{
        "gpu": "0",
        "accelerator_partition": "SPX",
        "memory_partition": "NPS1",
        "partition_id": "0",
    }]
    assert pg.check_desired(states, "SPX", "NPS1")
    assert not pg.check_desired(states, "CPX", "NPS1")


def test_sysfs_available_modes_contain_every_model_mode():
    """The KMD advertises exactly the modes the model supports; the sysfs
    fallback's membership check works against the real string format
    ('SPX, DPX, QPX, CPX' — comma+space separated)."""
    for mode in pg.PARTITION_COUNT:
        assert mode in REAL_SYSFS_AVAILABLE.upper()


def test_kfd_observables_match_cu_model():
    """SPX: simd_count 1024 = 256 CUs x 4 SIMD — the 256-CU / 8-XCC model
    behind the CPX expectation of 32 CUs (256/8) per partition that
    test/amd_gpu/device-plugin-cpx-test.yaml asserts on hardware."""
    assert REAL_KFD_PROPS["num_xcc"] == 8
    assert REAL_KFD_PROPS["simd_count"] // 4 == 256
    assert (REAL_KFD_PROPS["simd_count"] // 4) // REAL_KFD_PROPS["num_xcc"] == 32
    # gfx950 == target version 90500 (9.5.0)
    assert REAL_KFD_PROPS["gfx_target_version"] == 90500


def test_mock_uuid_matching_handles_real_uuid_format():
    """Health attribution lowercases/strips UUIDs; the real amdsmi UUID
    format (8-4-4-4-12 hex, lowercase) round-trips the matcher."""
    real_uuid = "6aff75a3-0000-1000-800a-ba0c188eedbe"
    assert real_uuid.strip().lower() == real_uuid
    # amd-smi list PARTITION_ID field exists even in SPX (value 0) — the
    # mock's partition_id=0-for-SPX matches the real contract
    assert "PARTITION_ID: 0" in REAL_AMDSMI_LIST


def test_preflight_rejects_unsupported_mode():
    """run() refuses a mode the hardware profile table lacks, before
    touching any state."""
    calls = []

    def runner(cmd):
        calls.append(cmd)
        if cmd[:2] == ["amd-smi", "partition"]:
            # a part that only supports SPX/DPX
            return 0, REAL_PARTITION_PROFILES.replace(
                "QPX", "XXX").replace("CPX", "YYY")
        if "static" in cmd:
            return 0, REAL_STATIC_PARTITION
        raise AssertionError(f"unexpected command {cmd}")

    import json
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".json", delete=False) as f:
        json.dump({"ComputePartition": "CPX"}, f)
        path = f.name
    with pytest.raises(pg.PartitionError, match="does not support"):
        pg.run(config_path=path, runner=runner)
    # and no `amd-smi set` was ever attempted
    assert not any("set" in c for c in calls)


def test_preflight_rejects_count_mismatch():
    """A hardware NUM_PARTITIONS that disagrees with the static table is a
    hard error (the device fan-out would mis-size)."""
    def runner(cmd):
        if cmd[:2] == ["amd-smi", "partition"]:
            import re
            mutated = re.sub(r"(CPX\s+N/A\s+)8", r"\g<1>6",
                             REAL_PARTITION_PROFILES)
            assert "6" in pg.parse_partition_profiles(mutated).values().__str__()
            return 0, mutated
        if "static" in cmd:
            return 0, REAL_STATIC_PARTITION
        raise AssertionError(f"unexpected command {cmd}")

    import json
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".json", delete=False) as f:
        json.dump({"ComputePartition": "CPX"}, f)
        path = f.name
    with pytest.raises(pg.PartitionError, match="mismatched"):
        pg.run(config_path=path, runner=runner)
