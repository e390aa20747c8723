#!/usr/bin/env python3
"""gpu_doctor — one-shot MI355X node triage.

Checks, in dependency order: device nodes -> amdsmi enumeration ->
per-device probe (VRAM/partition/BDF/NUMA) -> activity sampler -> RCCL
userspace -> (optionally) a short event drain.  Prints one line per check
and exits nonzero if any FAIL.  Run on a node shell or via
`kubectl debug node/...`; the GPU test tier runs it end-to-end on
hardware.  (No reference analog — triage there is `nvidia-smi` by hand.)
"""
from __future__ import annotations

import argparse
import glob
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

OK, WARN, FAIL = "ok", "warn", "FAIL"


def check(name, status, detail=""):
    print(f"[{status:>4}] {name}" + (f": {detail}" if detail else ""))
    return status != FAIL


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--dev-directory", default="/dev")
    p.add_argument("--mock-amdsmi", action="store_true")
    p.add_argument("--drain-events-ms", type=int, default=0,
                   help="also drain event notifications for N ms")
    args = p.parse_args(argv)
    healthy = True

    kfd = os.path.join(args.dev_directory, "kfd")
    renders = sorted(glob.glob(os.path.join(args.dev_directory, "dri",
                                            "renderD*")))
    healthy &= check("/dev/kfd", OK if os.path.exists(kfd) else FAIL)
    healthy &= check("render nodes", OK if renders else FAIL,
                     f"{len(renders)} found")
    if not healthy:
        print("amdgpu driver not ready; stopping")
        return 1

    import cea_amd.amdsmi as amdsmi

    if args.mock_amdsmi:
        from cea_amd.amdsmi.mock import MockAmdSmi

        smi = MockAmdSmi(args.dev_directory)
    else:
        from cea_amd.amdsmi.shim import ShimAmdSmi

        smi = ShimAmdSmi()
    try:
        smi.init()
        amdsmi.set_ops(smi)
        healthy &= check("amdsmi init", OK)
    except Exception as e:  # noqa: BLE001
        check("amdsmi init", FAIL, str(e))
        return 1

    try:
        n = smi.device_count()
        healthy &= check("enumeration", OK if n else FAIL, f"{n} device(s)")
        for i in range(n):
            info = smi.device_info(i)
            mem = smi.memory_info(i)
            detail = (f"{info.name} bdf={info.bdf} renderD{info.render_minor} "
                      f"{info.compute_partition}/{info.memory_partition} "
                      f"vram={mem.total_bytes / 2**30:.0f}GiB "
                      f"used={mem.used_bytes / 2**30:.1f}GiB")
            status = OK
            if mem.total_bytes == 0:
                status = FAIL
            elif mem.used_bytes > 0.95 * mem.total_bytes:
                status = WARN
            healthy &= check(f"device {i}", status, detail)
            try:
                ecc = smi.ecc_uncorrectable_count(i)
                healthy &= check(f"device {i} ECC", OK if ecc == 0 else WARN,
                                 f"uncorrectable={ecc}")
            except Exception as e:  # noqa: BLE001
                check(f"device {i} ECC", WARN, f"unavailable ({e})")
        act = smi.gpu_activity(0)
        healthy &= check("activity", OK, f"gfx={act.gfx_percent:.0f}% "
                                         f"umc={act.umc_percent:.0f}%")
        healthy &= check("driver", OK, smi.driver_version() or "unknown")

        from cea_amd.deviceplugin.version_visibility import (
            partition_capabilities,
            rccl_version,
        )

        rccl = rccl_version()
        healthy &= check("librccl", OK if rccl else WARN, rccl or "not found")

        caps = partition_capabilities()
        healthy &= check(
            "partition modes",
            OK if caps else WARN,
            (f"{caps.get('amd.com/gpu.partition-modes', '?')} "
             f"(current {caps.get('amd.com/gpu.compute-partition', '?')})"
             if caps else "sysfs knobs not visible"),
        )

        if args.drain_events_ms > 0:
            evs = smi.wait_events(args.drain_events_ms)
            healthy &= check("events", OK, f"{len(evs)} in "
                                           f"{args.drain_events_ms} ms")
    finally:
        smi.shutdown()
        amdsmi.ops = None

    print("node", "HEALTHY" if healthy else "UNHEALTHY")
    return 0 if healthy else 1


if __name__ == "__main__":
    sys.exit(main())
