#!/usr/bin/env python3
"""partition_gpu entrypoint — deployed as an init container before the
device plugin (parity: /root/reference/partition_gpu/partition_gpu.yaml)."""
import argparse
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cea_amd.partition import partition_gpu  # noqa: E402


def main():
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s: %(message)s")
    p = argparse.ArgumentParser()
    p.add_argument("--gpu-config", default=partition_gpu.DEFAULT_CONFIG_PATH)
    args = p.parse_args()
    try:
        changed = partition_gpu.run(args.gpu_config)
    except Exception as e:  # noqa: BLE001
        logging.error("partitioning failed: %s", e)
        sys.exit(1)
    logging.info("done (changed=%s)", changed)


if __name__ == "__main__":
    main()
