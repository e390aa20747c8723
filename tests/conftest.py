"""Shared pytest config: gpu marker registration + package path + global
injection-point hygiene."""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X (run via gpurun)"
    )


@pytest.fixture(autouse=True)
def _reset_globals():
    """Reset package-global injection points between tests (the same hygiene
    the reference needs around gpusharing.SharingStrategy and
    nvmlutil.NvmlDeviceInfo)."""
    import cea_amd.amdsmi as amdsmi
    from cea_amd.deviceplugin import sharing

    yield
    amdsmi.ops = None
    sharing.sharing_strategy = ""
