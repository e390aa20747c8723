"""Node topology labeler daemon.

Parity: /root/reference/gke-topology-scheduler/label-nodes-daemon.py
(69 LoC): every 600 s read the physical-topology identity and patch node
labels.  Sources, in order:
  * env TOPOLOGY_BLOCK/TOPOLOGY_SUBBLOCK/TOPOLOGY_HOST (downward API /
    operator-provided — the cloud-agnostic path),
  * GCE metadata `physical_host` ("/block/subblock/host",
    label-nodes-daemon.py:29) when the metadata server is reachable,
  * a host file (e.g. dropped by a fabric-discovery agent for RoCE rails).
"""
from __future__ import annotations

import logging
import os
import time
from typing import Dict, Optional

from .topology import LABEL_BLOCK, LABEL_HOST, LABEL_SUBBLOCK

log = logging.getLogger(__name__)

GCE_METADATA_URL = (
    "http://metadata.google.internal/computeMetadata/v1/instance/attributes/"
    "physical_host"
)
TOPOLOGY_FILE = "/etc/cea-amd/topology"
UPDATE_INTERVAL_S = 600  # parity label-nodes-daemon.py poll cadence


def topology_from_env(env=os.environ) -> Optional[Dict[str, str]]:
    block = env.get("TOPOLOGY_BLOCK", "")
    if not block:
        return None
    return {
        "block": block,
        "subblock": env.get("TOPOLOGY_SUBBLOCK", ""),
        "host": env.get("TOPOLOGY_HOST", ""),
    }


def parse_physical_host(s: str) -> Optional[Dict[str, str]]:
    """'/block/subblock/host' -> levels (parity label-nodes-daemon.py:29-44)."""
    parts = [p for p in s.strip().split("/") if p]
    if not parts:
        return None
    parts += [""] * (3 - len(parts))
    return {"block": parts[0], "subblock": parts[1], "host": parts[2]}


def topology_from_gce_metadata(timeout: float = 2.0) -> Optional[Dict[str, str]]:
    try:
        import requests

        r = requests.get(GCE_METADATA_URL,
                         headers={"Metadata-Flavor": "Google"}, timeout=timeout)
        if r.status_code == 200:
            return parse_physical_host(r.text)
    except Exception:  # noqa: BLE001
        pass
    return None


def topology_from_file(path: str = TOPOLOGY_FILE) -> Optional[Dict[str, str]]:
    try:
        with open(path) as f:
            return parse_physical_host(f.read())
    except OSError:
        return None


def discover_topology() -> Optional[Dict[str, str]]:
    return (topology_from_env() or topology_from_file()
            or topology_from_gce_metadata())


def label_node(kube, node_name: str, topo: Dict[str, str]) -> None:
    labels = {
        LABEL_BLOCK: topo["block"],
        LABEL_SUBBLOCK: topo["subblock"],
        LABEL_HOST: topo["host"],
    }
    kube.patch_node(node_name, {"metadata": {"labels": labels}})
    log.info("labeled %s with %s", node_name, labels)


def run_forever(kube, node_name: str,
                interval_s: float = UPDATE_INTERVAL_S) -> None:
    while True:
        topo = discover_topology()
        if topo:
            try:
                label_node(kube, node_name, topo)
            except Exception as e:  # noqa: BLE001
                log.error("labeling failed: %s", e)
        else:
            log.warning("no topology source available")
        time.sleep(interval_s)
