#!/usr/bin/env python3
"""NRI device-injector entrypoint.
Parity: /root/reference/nri_device_injector/nri_device_injector.go main."""
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cea_amd.nri.injector import InjectorPlugin  # noqa: E402
from cea_amd.nri import protos as api  # noqa: E402


def main():
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s: %(message)s")
    socket_path = os.environ.get("NRI_SOCKET", api.DEFAULT_SOCKET_PATH)
    InjectorPlugin(socket_path).run_forever()


if __name__ == "__main__":
    main()
