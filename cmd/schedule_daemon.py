#!/usr/bin/env python3
"""Topology-aware gang scheduler entrypoint.
Parity: /root/reference/gke-topology-scheduler/schedule-daemon.py main loop."""
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cea_amd.kube.client import build_kube_client  # noqa: E402
from cea_amd.scheduler.daemon import TopologyScheduler  # noqa: E402


def main():
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s: %(message)s")
    interval = float(os.environ.get("SCHEDULE_INTERVAL_S", "5"))
    warmup = float(os.environ.get("SCHEDULE_WARMUP_S", "90"))
    kube = build_kube_client()
    pod_inf = node_inf = None
    if os.environ.get("USE_INFORMERS", "0") == "1":
        # watch-synced caches instead of a full list per pass — for
        # clusters past a few thousand pods (docs/architecture.md)
        from cea_amd.kube.informer import node_informer, pod_informer
        pod_inf = pod_informer(kube)
        node_inf = node_informer(kube)
        pod_inf.start()
        node_inf.start()
        pod_inf.wait_synced()
        node_inf.wait_synced()
    TopologyScheduler(kube, interval_s=interval, pod_informer=pod_inf,
                      node_informer=node_inf).run_forever(warmup_s=warmup)


if __name__ == "__main__":
    main()
