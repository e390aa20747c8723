#!/usr/bin/env python3
"""Node topology labeler entrypoint.
Parity: /root/reference/gke-topology-scheduler/label-nodes-daemon.py."""
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cea_amd.kube.client import build_kube_client  # noqa: E402
from cea_amd.scheduler import labeler  # noqa: E402


def main():
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s: %(message)s")
    node_name = os.environ["NODE_NAME"]
    labeler.run_forever(build_kube_client(), node_name)


if __name__ == "__main__":
    main()
