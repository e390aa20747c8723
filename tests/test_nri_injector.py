"""NRI injector tests — parity with nri_device_injector_test.go (:25,95):
annotation -> LinuxDevice round-trip against real device nodes (fifo always;
mknod char device when running as root), plus a full ttrpc session against
a fake containerd runtime over a socketpair."""
import os
import socket
import stat
import threading

import pytest

from cea_amd.nri import protos as api
from cea_amd.nri.injector import (
    CTR_DEVICE_KEY_PREFIX,
    DeviceError,
    InjectorPlugin,
    build_adjustment,
    get_devices,
    to_nri_device,
)
from cea_amd.nri.ttrpc import Request, Response, TtrpcEndpoint, HEADER, MESSAGE_TYPE_REQUEST, MESSAGE_TYPE_RESPONSE


def test_get_devices_parsing():
    ann = {CTR_DEVICE_KEY_PREFIX + "ctr1": "- path: /dev/kfd\n- path: /dev/kfd\n- path: /dev/dri/renderD128\n  uid: 1000\n"}
    devs = get_devices("ctr1", ann)
    assert [d["path"] for d in devs] == ["/dev/kfd", "/dev/dri/renderD128"]
    assert devs[1]["uid"] == 1000
    assert get_devices("other", ann) == []
    assert get_devices("ctr1", {}) == []
    with pytest.raises(DeviceError):
        get_devices("c", {CTR_DEVICE_KEY_PREFIX + "c": "path: notalist"})
    with pytest.raises(DeviceError):
        get_devices("c", {CTR_DEVICE_KEY_PREFIX + "c": "- {type: c}"})


def test_to_nri_device_fifo(tmp_path):
    fifo = str(tmp_path / "myfifo")
    os.mkfifo(fifo)
    dev = to_nri_device({"path": fifo})
    assert dev.type == "p"


def test_to_nri_device_char():
    dev = to_nri_device({"path": "/dev/null", "file_mode": 0o666, "uid": 0})
    assert dev.type == "c"
    assert dev.major == 1 and dev.minor == 3
    assert dev.file_mode.value == 0o666


def test_to_nri_device_mknod(tmp_path):
    """Root-gated: create a char device like /dev/kfd (major 241)."""
    path = str(tmp_path / "kfd")
    try:
        os.mknod(path, 0o600 | stat.S_IFCHR, os.makedev(241, 0))
    except PermissionError:
        pytest.skip("needs CAP_MKNOD")
    dev = to_nri_device({"path": path})
    assert (dev.type, dev.major, dev.minor) == ("c", 241, 0)


def test_to_nri_device_rejects_regular_file(tmp_path):
    f = tmp_path / "plain"
    f.write_text("x")
    with pytest.raises(DeviceError, match="invalid device type"):
        to_nri_device({"path": str(f)})
    with pytest.raises(DeviceError, match="failed to get info"):
        to_nri_device({"path": str(tmp_path / "missing")})


def test_build_adjustment(tmp_path):
    fifo = str(tmp_path / "f")
    os.mkfifo(fifo)
    ann = {CTR_DEVICE_KEY_PREFIX + "c1": f"- path: {fifo}\n- path: /dev/null\n"}
    adjust = build_adjustment("c1", ann)
    assert len(adjust.linux.devices) == 2
    assert build_adjustment("nope", ann) is None


class FakeRuntime:
    """Fake containerd NRI side: accepts RegisterPlugin, then drives
    Configure and CreateContainer requests."""

    def __init__(self, sock):
        self.ep = TtrpcEndpoint(sock)
        self.registered = threading.Event()
        self.register_req = None
        self.ep.register(api.RUNTIME_SERVICE, "RegisterPlugin", self._register)
        self.ep.start()

    def _register(self, payload):
        self.register_req = api.RegisterPluginRequest.FromString(payload)
        self.registered.set()
        return api.Empty().SerializeToString()

    def configure(self):
        out = self.ep.call(api.PLUGIN_SERVICE, "Configure",
                           api.ConfigureRequest(runtime_name="containerd",
                                                runtime_version="2.0").SerializeToString())
        return api.ConfigureResponse.FromString(out)

    def create_container(self, pod_annotations, ctr_name):
        req = api.CreateContainerRequest()
        req.pod.name = "pod1"
        for k, v in pod_annotations.items():
            req.pod.annotations[k] = v
        req.container.name = ctr_name
        out = self.ep.call(api.PLUGIN_SERVICE, "CreateContainer",
                           req.SerializeToString())
        return api.CreateContainerResponse.FromString(out)


def test_full_nri_session(tmp_path):
    a, b = socket.socketpair(socket.AF_UNIX, socket.SOCK_STREAM)
    runtime = FakeRuntime(a)
    plugin = InjectorPlugin()
    plugin.connect(b)
    try:
        assert runtime.registered.wait(5)
        assert runtime.register_req.plugin_idx == "10"

        resp = runtime.configure()
        assert resp.events == 1 << (api.EVENT_CREATE_CONTAINER - 1)

        fifo = str(tmp_path / "uverbs0")
        os.mkfifo(fifo)
        ann = {CTR_DEVICE_KEY_PREFIX + "rccl": f"- path: {fifo}\n- path: /dev/null\n"}
        out = runtime.create_container(ann, "rccl")
        devs = out.adjust.linux.devices
        assert [d.path for d in devs] == [fifo, "/dev/null"]
        assert devs[1].type == "c"

        # container without annotation -> no adjustment
        out = runtime.create_container({}, "plain")
        assert len(out.adjust.linux.devices) == 0
    finally:
        plugin.endpoint.close()
        runtime.ep.close()


def test_ttrpc_error_propagation():
    a, b = socket.socketpair(socket.AF_UNIX, socket.SOCK_STREAM)
    server = TtrpcEndpoint(a)
    client = TtrpcEndpoint(b)
    server.start()
    client.start()
    try:
        from cea_amd.nri.ttrpc import TtrpcError
        with pytest.raises(TtrpcError, match="unimplemented"):
            client.call("no.such.Service", "Nope", b"")
    finally:
        server.close()
        client.close()


def test_ttrpc_wire_format():
    """Frame header must match containerd's ttrpc: be32 len, be32 stream,
    u8 type, u8 flags (channel.go:63-90)."""
    a, b = socket.socketpair(socket.AF_UNIX, socket.SOCK_STREAM)
    ep = TtrpcEndpoint(a)
    ep.start()
    result = {}
    t = threading.Thread(
        target=lambda: result.update(
            out=ep.call("svc", "M", b"hello", timeout=5)),
        daemon=True,
    )
    t.start()
    hdr = b.recv(10)
    length, stream_id, mtype, flags = HEADER.unpack(hdr)
    body = b.recv(length)
    req = Request.FromString(body)
    assert (req.service, req.method, req.payload) == ("svc", "M", b"hello")
    assert mtype == MESSAGE_TYPE_REQUEST and flags == 0 and stream_id % 2 == 1
    resp = Response(payload=b"world").SerializeToString()
    b.sendall(HEADER.pack(len(resp), stream_id, MESSAGE_TYPE_RESPONSE, 0) + resp)
    t.join(timeout=5)
    assert not t.is_alive()
    assert result.get("out") == b"world"
    ep.close()
    b.close()


def test_ttrpc_rejects_oversized_and_malformed_frames():
    """Wire robustness: a frame claiming >4 MiB drops the link (no unbounded
    read); a malformed Response body fails the pending call with a status
    instead of killing the read loop."""
    import socket
    import threading
    import time

    from cea_amd.nri import ttrpc as t

    # oversized frame: peer sends a huge length; endpoint must close
    a, b = socket.socketpair()
    ep = t.TtrpcEndpoint(a)
    ep.start()
    b.sendall(t.HEADER.pack(t.MAX_FRAME_BYTES + 1, 1, t.MESSAGE_TYPE_REQUEST, 0))
    deadline = time.time() + 5
    while not ep._closed.is_set() and time.time() < deadline:
        time.sleep(0.02)
    assert ep._closed.is_set()
    ep.close()
    b.close()

    # malformed response: pending call gets an error status, loop survives
    a, b = socket.socketpair()
    ep = t.TtrpcEndpoint(a)
    ep.start()
    result = {}

    def caller():
        try:
            ep.call("svc", "M", b"", timeout=5)
        except t.TtrpcError as e:
            result["err"] = e

    th = threading.Thread(target=caller)
    th.start()
    time.sleep(0.2)
    # stream id 1 is the first client call; reply with garbage protobuf
    b.sendall(t.HEADER.pack(3, 1, t.MESSAGE_TYPE_RESPONSE, 0) + b"\xff\xff\xff")
    th.join(timeout=5)
    assert not th.is_alive()
    assert "err" in result and result["err"].code == 13
    assert not ep._closed.is_set()  # read loop still alive
    ep.close()
    b.close()


def test_ttrpc_survives_garbage_frames():
    """Adversarial wire input: random headers/bodies, oversized frame
    lengths, and truncated streams must never hang the read loop or crash
    the endpoint — it either ignores the junk or closes the link cleanly
    (the containerd socket is a trust boundary for the injector)."""
    import random
    import socket
    import time

    rng = random.Random(1234)
    for trial in range(30):
        a, b = socket.socketpair()
        ep = TtrpcEndpoint(a)
        ep.register("svc", "m", lambda payload: b"")
        ep.start()
        try:
            kind = trial % 3
            if kind == 0:          # pure garbage bytes
                b.sendall(rng.randbytes(rng.randint(1, 256)))
            elif kind == 1:        # valid header, oversized length
                b.sendall(HEADER.pack(1 << 30, 1, MESSAGE_TYPE_REQUEST, 0))
            else:                  # valid header, malformed protobuf body
                body = rng.randbytes(rng.randint(1, 64))
                b.sendall(HEADER.pack(len(body), 3,
                                      MESSAGE_TYPE_REQUEST, 0) + body)
            b.shutdown(socket.SHUT_WR)
            deadline = time.time() + 5
            while not ep._closed.is_set() and time.time() < deadline:
                time.sleep(0.01)
            assert ep._closed.is_set() or kind == 2, \
                f"trial {trial}: read loop did not terminate"
        finally:
            ep.close()
            b.close()


def test_get_devices_rejects_non_string_paths():
    """Malformed annotations with non-string path values (ints, lists,
    dicts from YAML) raise DeviceError — not TypeError (fuzz-found:
    unhashable paths crashed the dedup set)."""
    from cea_amd.nri.injector import DeviceError, get_devices

    for val in ("- path: [a, b]", "- path: {x: 1}", "- path: 3"):
        with pytest.raises(DeviceError):
            get_devices("c", {"devices.gke.io/container.c": val})
