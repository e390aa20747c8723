#!/usr/bin/env python3
"""GPU readiness sidecar — the persistenced-installer analog.

Parity: /root/reference/nvidia-persistenced-installer/ (190 LoC).  amdgpu
has no persistenced daemon (the KMD keeps device state), so the analog is a
readiness gate: verify the amdgpu module is loaded and /dev/kfd + render
nodes exist, handle the confidential-node gate
(/etc/amd/confidential_node_type.txt ∈ {sev, sev-snp, tdx} — parity with
nvidia_persistenced_installer.go:172-185), mark ready, then block on
SIGINT/SIGTERM (parity :86-94).  If devices never appear, exits nonzero so
the DaemonSet restarts it (replacing the reference's SIGRTMIN+5 node reboot,
:70-80).
"""
from __future__ import annotations

import glob
import logging
import os
import signal
import sys
import threading
import time

log = logging.getLogger("gpu_readiness")

CONFIDENTIAL_TYPE_FILE = "/etc/amd/confidential_node_type.txt"
READY_FILE = "/run/cea-amd/ready"
WAIT_DEADLINE_S = 600


def confidential_node_type(path: str = CONFIDENTIAL_TYPE_FILE) -> str:
    try:
        with open(path) as f:
            return f.read().strip().lower()
    except OSError:
        return ""


def devices_ready(dev_dir: str = "/dev", sysfs: str = "/sys") -> bool:
    if not os.path.exists(os.path.join(dev_dir, "kfd")):
        return False
    if not glob.glob(os.path.join(dev_dir, "dri", "renderD*")):
        return False
    if not os.path.isdir(os.path.join(sysfs, "module", "amdgpu")):
        return False
    return True


def main() -> int:
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s: %(message)s")
    ctype = confidential_node_type()
    if ctype:
        log.info("confidential node type: %s (SEV-SNP attestation is "
                 "platform-managed on MI355X; no conf-compute ready-state "
                 "flip needed)", ctype)
    deadline = time.time() + WAIT_DEADLINE_S
    while not devices_ready():
        if time.time() > deadline:
            log.error("amdgpu devices never appeared; exiting for restart")
            return 1
        log.info("waiting for amdgpu devices (/dev/kfd, renderD*)")
        time.sleep(10)
    os.makedirs(os.path.dirname(READY_FILE), exist_ok=True)
    with open(READY_FILE, "w") as f:
        f.write("ready\n")
    log.info("GPU node ready; blocking until terminated")
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    stop.wait()
    return 0


if __name__ == "__main__":
    sys.exit(main())
