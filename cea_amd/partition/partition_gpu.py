"""partition_gpu — one-shot node job that drives MI355X compute/memory
partition modes.

Role parity: /root/reference/partition_gpu/partition_gpu.go (467 LoC), which
drives `nvidia-smi mig` (enable MIG + reboot on Ampere, destroy/create
GI/CI profiles, idempotency via parsing `nvidia-smi mig -lgi`).  The MI355X
mechanism is simpler and runtime-switchable: `amd-smi set --gpu all
--compute-partition {SPX|DPX|CPX}` (+ `--memory-partition NPS1/NPS2`), no
reboot, but the set can fail with "busy" while KFD processes hold the GPU —
handled with a bounded retry instead of the reference's SIGRTMIN+5 reboot
(partition_gpu.go:297-300).

Idempotency: parse `amd-smi static --gpu all --partition` (the analog of the
parseLGIOutput table state machine, partition_gpu.go:394-444) and exit 0 if
every GPU already shows the desired accelerator + memory partition.
"""
from __future__ import annotations

import json
import logging
import re
import subprocess
import time
from typing import Callable, Dict, List, Optional, Tuple

log = logging.getLogger(__name__)

DEFAULT_CONFIG_PATH = "/etc/amd/gpu_config.json"

VALID_COMPUTE = ("SPX", "DPX", "QPX", "CPX")
VALID_MEMORY = ("NPS1", "NPS2", "NPS4", "NPS8")

# partitions per die by mode — parity with partitionSizeMaxCount
# (partition_gpu.go:91-139)
PARTITION_COUNT = {"SPX": 1, "DPX": 2, "QPX": 4, "CPX": 8}

BUSY_RETRIES = 12
BUSY_RETRY_DELAY_S = 10.0

Runner = Callable[[List[str]], Tuple[int, str]]


def default_runner(cmd: List[str]) -> Tuple[int, str]:
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=300)
    return r.returncode, r.stdout + r.stderr


def parse_partition_config(path: str) -> Tuple[str, str]:
    """'cpx-nps1' (or ComputePartition/MemoryPartition keys) -> ('CPX','NPS1').
    Parity: the config read at partition_gpu.go:157-178."""
    with open(path) as f:
        raw = json.load(f)
    spec = raw.get("ComputePartition", raw.get("GPUPartitionSize", "")) or "spx"
    parts = spec.strip().upper().split("-")
    compute = parts[0]
    memory = parts[1] if len(parts) > 1 else raw.get("MemoryPartition", "NPS1").upper()
    if compute not in VALID_COMPUTE:
        raise ValueError(f"invalid compute partition {compute!r}: want {VALID_COMPUTE}")
    if memory not in VALID_MEMORY:
        raise ValueError(f"invalid memory partition {memory!r}: want {VALID_MEMORY}")
    return compute, memory


GPU_RE = re.compile(r"^GPU:\s*(\d+)\s*$")
KV_RE = re.compile(r"^\s+([A-Z_]+):\s*(\S+)\s*$")


def parse_partition_status(text: str) -> List[Dict[str, str]]:
    """Parse `amd-smi static --gpu all --partition` human output into
    [{gpu, accelerator_partition, memory_partition, partition_id}].
    The regex state machine is the analog of parseLGIOutput
    (partition_gpu.go:394-444)."""
    out: List[Dict[str, str]] = []
    cur: Optional[Dict[str, str]] = None
    for line in text.splitlines():
        m = GPU_RE.match(line.strip()) or GPU_RE.match(line)
        if line.strip().startswith("GPU:"):
            gpu = line.split(":", 1)[1].strip()
            cur = {"gpu": gpu}
            out.append(cur)
            continue
        m = KV_RE.match(line)
        if m and cur is not None:
            key = m.group(1).lower()
            if key in ("accelerator_partition", "compute_partition",
                       "memory_partition", "partition_id"):
                cur[key] = m.group(2)
    # normalize: older amd-smi prints COMPUTE_PARTITION
    for d in out:
        if "accelerator_partition" not in d and "compute_partition" in d:
            d["accelerator_partition"] = d["compute_partition"]
    return [d for d in out if "accelerator_partition" in d]


def parse_partition_profiles(text: str) -> Dict[str, int]:
    """Parse `amd-smi partition` ACCELERATOR_PARTITION_PROFILES output into
    {accelerator_type: num_partitions} — the hardware's own statement of
    which modes it supports and how many logical devices each produces.
    Format captured from a real MI355X (profiles/pool_probe_r02.log):

        ACCELERATOR_PARTITION_PROFILES:
        GPU_ID  PROFILE_INDEX  MEMORY_PARTITION_CAPS  ACCELERATOR_TYPE  \
PARTITION_ID  NUM_PARTITIONS  ...
        0       0              NPS1                   SPX*              \
0             1               ...
                1              NPS1                   DPX               \
N/A           2               ...

    The currently-active profile is marked with ``*``."""
    out: Dict[str, int] = {}
    in_profiles = False
    for line in text.splitlines():
        s = line.strip()
        if s.startswith("ACCELERATOR_PARTITION_PROFILES"):
            in_profiles = True
            continue
        if in_profiles and s.endswith(":") and s == s.upper():
            break  # next section header
        if not in_profiles or not s:
            continue
        m = re.search(
            r"\b(SPX|DPX|QPX|TPX|CPX)\*?\s+(?:\d+|N/A)\s+(\d+)\b", s)
        if m:
            out[m.group(1)] = int(m.group(2))
    return out


def hardware_partition_capabilities(runner: Runner) -> Dict[str, int]:
    """Query `amd-smi partition` for the supported mode->count table;
    empty dict when the CLI subcommand is unavailable (older amd-smi)."""
    rc, out = runner(["amd-smi", "partition"])
    if rc != 0:
        return {}
    return parse_partition_profiles(out)


def check_desired(states: List[Dict[str, str]], compute: str, memory: str) -> bool:
    """Uniformity + desired-mode check (parity: checkDesired,
    partition_gpu.go:446-458)."""
    if not states:
        return False
    for s in states:
        if s.get("accelerator_partition", "").upper() != compute:
            return False
        mp = s.get("memory_partition", "").upper()
        if mp and mp != memory:
            return False
    return True


class PartitionError(RuntimeError):
    pass


def current_partition_status(runner: Runner) -> List[Dict[str, str]]:
    rc, out = runner(["amd-smi", "static", "--gpu", "all", "--partition"])
    if rc != 0:
        raise PartitionError(f"amd-smi static failed (rc={rc}): {out[:500]}")
    return parse_partition_status(out)


def _set_with_busy_retry(runner: Runner, cmd: List[str]) -> None:
    """amd-smi set fails while KFD processes hold the GPU; bounded retry
    (the drain-and-retry replacing the reference's node reboot)."""
    for attempt in range(BUSY_RETRIES):
        rc, out = runner(cmd)
        if rc == 0:
            return
        lowered = out.lower()
        if "busy" in lowered or "in use" in lowered:
            log.warning("GPU busy (attempt %d/%d): waiting for workloads to "
                        "drain", attempt + 1, BUSY_RETRIES)
            time.sleep(BUSY_RETRY_DELAY_S)
            continue
        raise PartitionError(f"{' '.join(cmd)} failed (rc={rc}): {out[:500]}")
    raise PartitionError(f"{' '.join(cmd)}: GPU still busy after "
                         f"{BUSY_RETRIES} attempts")


def set_compute_partition_sysfs(compute: str, sysfs_root: str = "/sys",
                                bdfs: Optional[List[str]] = None) -> bool:
    """Fallback setter: write the mode to the amdgpu KMD's sysfs interface
    (/sys/class/drm/card*/device/current_compute_partition — the same knob
    amd-smi drives).  Returns True if at least one card accepted the write.
    Used when the amd-smi CLI refuses (e.g. library/platform mismatch) but
    the KMD advertises the mode in available_compute_partition.

    `bdfs`: restrict writes to cards whose PCI address is in the list (so a
    tenant holding a subset of a shared host's GPUs never touches foreign
    devices).  None = all cards (single-tenant node job)."""
    import glob as _glob
    import os as _os

    want = {b.lower() for b in bdfs} if bdfs is not None else None
    wrote = False
    for card in sorted(_glob.glob(
            _os.path.join(sysfs_root, "class", "drm", "card*", "device",
                          "current_compute_partition"))):
        dev_dir = _os.path.dirname(card)
        if want is not None:
            try:
                pci = _os.path.basename(_os.path.realpath(dev_dir)).lower()
            except OSError:
                continue
            if pci not in want:
                continue
        avail_path = _os.path.join(dev_dir, "available_compute_partition")
        try:
            with open(avail_path) as f:
                if compute not in f.read().upper():
                    continue
            with open(card, "w") as f:
                f.write(compute + "\n")
            wrote = True
        except OSError as e:
            log.warning("sysfs partition write failed for %s: %s", card, e)
    return wrote


def run(config_path: str = DEFAULT_CONFIG_PATH,
        runner: Runner = default_runner, sysfs_root: str = "/sys") -> bool:
    """Main flow (parity: main, partition_gpu.go:157-236).  Returns True if a
    mode change was applied, False if already in the desired state."""
    compute, memory = parse_partition_config(config_path)
    states = current_partition_status(runner)
    if not states:
        raise PartitionError("amd-smi reported no GPUs")
    log.info("desired: %s/%s; current: %s", compute, memory,
             [(s.get("accelerator_partition"), s.get("memory_partition"))
              for s in states])
    if check_desired(states, compute, memory):
        log.info("GPUs already partitioned as %s/%s; nothing to do",
                 compute, memory)
        return False

    # pre-flight: if the hardware publishes its profile table, confirm the
    # requested mode is supported and our count table agrees with the
    # hardware's NUM_PARTITIONS (catches e.g. TPX-capable parts or future
    # asymmetric profiles before any state is touched)
    caps = hardware_partition_capabilities(runner)
    if caps:
        if compute not in caps:
            raise PartitionError(
                f"hardware does not support compute partition {compute}; "
                f"supported: {sorted(caps)}")
        if caps[compute] != PARTITION_COUNT.get(compute):
            raise PartitionError(
                f"hardware reports {caps[compute]} partitions for {compute}, "
                f"static table says {PARTITION_COUNT.get(compute)} — "
                "refusing to proceed with a mismatched device-count model")

    # memory partition first (it implies a KFD re-enumeration), then compute
    needs_mem = any(
        s.get("memory_partition", "").upper() not in ("", memory) for s in states
    )
    if needs_mem:
        _set_with_busy_retry(
            runner, ["amd-smi", "set", "--gpu", "all",
                     "--memory-partition", memory])
    try:
        _set_with_busy_retry(
            runner, ["amd-smi", "set", "--gpu", "all",
                     "--compute-partition", compute])
    except PartitionError as e:
        # amd-smi CLI/library refusals happen on platforms whose CLI lags
        # the KMD; fall back to the KMD sysfs knob it fronts.
        log.warning("amd-smi set failed (%s); trying sysfs fallback", e)
        if not set_compute_partition_sysfs(compute, sysfs_root):
            raise

    states = current_partition_status(runner)
    if not check_desired(states, compute, memory):
        raise PartitionError(
            f"partitioning applied but verification failed: {states}")
    log.info("partitioned all GPUs to %s/%s (%d logical devices/die)",
             compute, memory, PARTITION_COUNT[compute])
    return True
