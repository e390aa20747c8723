"""NRI device injector plugin.

Role parity: /root/reference/nri_device_injector/nri_device_injector.go
(199 LoC): a containerd NRI plugin (registration index "10",
nri_device_injector.go:35) whose CreateContainer hook parses YAML device
lists from the pod annotation `devices.gke.io/container.<ctrName>`
(first-path-wins dedup, :126-155), lstats each path to derive
major/minor/type (block/char/fifo, :158-199) and injects them via
ContainerAdjustment.  Purpose on MI355X nodes: unprivileged pods get
/dev/kfd, /dev/dri/renderD* and RoCE /dev/infiniband/uverbs* without
device-plugin resources.

Transport: the in-repo minimal ttrpc endpoint (cea_amd/nri/ttrpc.py) —
this image has no containerd NRI stub library.
"""
from __future__ import annotations

import logging
import os
import socket
import stat as stat_mod
import threading
from typing import Dict, List, Optional

import yaml

from . import protos as api
from .ttrpc import TtrpcEndpoint

log = logging.getLogger(__name__)

DEVICE_KEY_PREFIX = "devices.gke.io"
CTR_DEVICE_KEY_PREFIX = DEVICE_KEY_PREFIX + "/container."
PLUGIN_NAME = "device_injector_nri"
PLUGIN_IDX = "10"   # parity: nri_device_injector.go:35

BLOCK_DEVICE = "b"
CHAR_DEVICE = "c"
FIFO_DEVICE = "p"


class DeviceError(ValueError):
    pass


def get_devices(ctr_name: str, pod_annotations: Dict[str, str]) -> List[dict]:
    """Parse the YAML device list for a container; first-path-wins dedup
    (parity getDevices, nri_device_injector.go:126-155)."""
    value = pod_annotations.get(CTR_DEVICE_KEY_PREFIX + ctr_name)
    if not value:
        return []
    try:
        parsed = yaml.safe_load(value)
    except yaml.YAMLError as e:
        raise DeviceError(f"invalid device annotation for {ctr_name}: {e}")
    if parsed is None:
        return []
    if not isinstance(parsed, list):
        raise DeviceError(f"device annotation for {ctr_name} must be a list")
    seen = set()
    out = []
    for entry in parsed:
        if isinstance(entry, str):
            entry = {"path": entry}
        if not isinstance(entry, dict) or not entry.get("path"):
            raise DeviceError(f"device entry missing path: {entry!r}")
        path = entry["path"]
        if not isinstance(path, str):
            # YAML can produce ints/lists/dicts here; anything non-string
            # is a malformed annotation, not a device path
            raise DeviceError(f"device path must be a string: {path!r}")
        if path in seen:
            continue
        seen.add(path)
        out.append(entry)
    return out


def to_nri_device(entry: dict) -> api.LinuxDevice:
    """lstat the path, derive type + major/minor (parity toNRIDevice,
    nri_device_injector.go:158-199)."""
    path = entry["path"]
    try:
        st = os.lstat(path)
    except OSError as e:
        raise DeviceError(f"failed to get info from device path {path}: {e}")
    mode = st.st_mode
    if stat_mod.S_ISBLK(mode):
        dev_type = BLOCK_DEVICE
    elif stat_mod.S_ISCHR(mode):
        dev_type = CHAR_DEVICE
    elif stat_mod.S_ISFIFO(mode):
        dev_type = FIFO_DEVICE
    else:
        raise DeviceError(f"invalid device type {oct(mode)} from device path {path}")
    dev = api.LinuxDevice(
        path=path,
        type=dev_type,
        major=os.major(st.st_rdev),
        minor=os.minor(st.st_rdev),
    )
    # explicit None checks: uid/gid/file_mode 0 are valid values
    if entry.get("file_mode") is not None:
        dev.file_mode.value = int(entry["file_mode"])
    if entry.get("uid") is not None:
        dev.uid.value = int(entry["uid"])
    if entry.get("gid") is not None:
        dev.gid.value = int(entry["gid"])
    return dev


def build_adjustment(ctr_name: str, pod_annotations: Dict[str, str]
                     ) -> Optional[api.ContainerAdjustment]:
    entries = get_devices(ctr_name, pod_annotations)
    if not entries:
        return None
    adjust = api.ContainerAdjustment()
    for entry in entries:
        dev = to_nri_device(entry)
        adjust.linux.devices.add().CopyFrom(dev)
        log.info("injecting device %s (%s %d:%d) into %s",
                 dev.path, dev.type, dev.major, dev.minor, ctr_name)
    return adjust


class InjectorPlugin:
    """The NRI plugin endpoint: registers with the runtime, subscribes to
    CreateContainer, serves adjustments."""

    def __init__(self, socket_path: str = api.DEFAULT_SOCKET_PATH):
        self.socket_path = socket_path
        self.endpoint: Optional[TtrpcEndpoint] = None
        self.configured = threading.Event()

    # -- plugin service handlers ---------------------------------------------
    def _configure(self, payload: bytes) -> bytes:
        req = api.ConfigureRequest.FromString(payload)
        log.info("configured by %s %s", req.runtime_name, req.runtime_version)
        self.configured.set()
        return api.ConfigureResponse(
            events=api.event_mask(api.EVENT_CREATE_CONTAINER)
        ).SerializeToString()

    def _synchronize(self, payload: bytes) -> bytes:
        return api.SynchronizeResponse().SerializeToString()

    def _create_container(self, payload: bytes) -> bytes:
        req = api.CreateContainerRequest.FromString(payload)
        resp = api.CreateContainerResponse()
        adjust = build_adjustment(req.container.name, dict(req.pod.annotations))
        if adjust is not None:
            resp.adjust.CopyFrom(adjust)
        return resp.SerializeToString()

    def _shutdown(self, payload: bytes) -> bytes:
        log.info("runtime requested shutdown")
        return api.Empty().SerializeToString()

    def _empty(self, payload: bytes) -> bytes:
        return api.Empty().SerializeToString()

    # -- lifecycle -------------------------------------------------------------
    def connect(self, sock: Optional[socket.socket] = None) -> None:
        if sock is None:
            sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            sock.connect(self.socket_path)
        ep = TtrpcEndpoint(sock)
        ep.register(api.PLUGIN_SERVICE, "Configure", self._configure)
        ep.register(api.PLUGIN_SERVICE, "Synchronize", self._synchronize)
        ep.register(api.PLUGIN_SERVICE, "CreateContainer", self._create_container)
        ep.register(api.PLUGIN_SERVICE, "Shutdown", self._shutdown)
        ep.register(api.PLUGIN_SERVICE, "UpdateContainer",
                    lambda p: api.UpdateContainerResponse().SerializeToString())
        ep.register(api.PLUGIN_SERVICE, "StopContainer",
                    lambda p: api.StopContainerResponse().SerializeToString())
        ep.register(api.PLUGIN_SERVICE, "StateChange", self._empty)
        ep.start()
        self.endpoint = ep
        ep.call(
            api.RUNTIME_SERVICE, "RegisterPlugin",
            api.RegisterPluginRequest(
                plugin_name=PLUGIN_NAME, plugin_idx=PLUGIN_IDX
            ).SerializeToString(),
        )
        log.info("registered NRI plugin %s (idx %s)", PLUGIN_NAME, PLUGIN_IDX)

    def run_forever(self) -> None:
        self.connect()
        self.endpoint._closed.wait()
        log.warning("NRI connection closed")
