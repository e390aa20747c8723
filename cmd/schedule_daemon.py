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
    TopologyScheduler(build_kube_client(), interval_s=interval).run_forever()


if __name__ == "__main__":
    main()
