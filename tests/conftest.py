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


def pytest_collection_modifyitems(config, items):
    """Default per-test timeout on the GPU tier (pytest-timeout): a hung
    kernel or event wait must fail the one test, not stall the whole
    hardware run."""
    if not config.pluginmanager.hasplugin("timeout"):
        return
    for item in items:
        if item.get_closest_marker("gpu") and not item.get_closest_marker("timeout"):
            item.add_marker(pytest.mark.timeout(600, method="thread"))


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
