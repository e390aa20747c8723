"""Informer (list+watch cache) tests — stub client driving the sync loop:
initial list, ADDED/MODIFIED/DELETED deltas, bookmark handling, 410-Gone
relist, and scheduler integration (passes read the cache, not the API)."""
import threading
import time

import pytest

from cea_amd.kube.client import FakeKubeClient, KubeError
from cea_amd.kube.informer import Informer
from cea_amd.scheduler import daemon as d


def pod(name, phase="Pending", rv="1", ns="default"):
    return {
        "metadata": {"name": name, "namespace": ns, "resourceVersion": rv,
                     "labels": {"job-name": "j1"}},
        "spec": {},
        "status": {"phase": phase},
    }


class StubClient:
    """list_raw + watch against scripted event batches."""

    def __init__(self, initial, batches):
        self.initial = initial
        self.batches = list(batches)   # each: list of (type, obj) or KubeError
        self.lists = 0
        self.watches = 0
        self._consumed = threading.Event()

    def list_raw(self, path, params=None):
        self.lists += 1
        return {"metadata": {"resourceVersion": "10"},
                "items": list(self.initial)}

    def watch(self, path, resource_version, params=None, timeout_s=300):
        self.watches += 1
        if self.batches:
            batch = self.batches.pop(0)
            if isinstance(batch, KubeError):
                raise batch
            for ev in batch:
                yield ev
        else:
            self._consumed.set()
            time.sleep(0.2)  # idle stream, then server closes

    def wait_consumed(self, timeout=5):
        return self._consumed.wait(timeout)


def test_informer_list_and_deltas():
    stub = StubClient(
        initial=[pod("a"), pod("b")],
        batches=[[
            ("ADDED", pod("c", rv="11")),
            ("MODIFIED", pod("a", phase="Running", rv="12")),
            ("DELETED", pod("b", rv="13")),
            ("BOOKMARK", {"metadata": {"resourceVersion": "14"}}),
        ]],
    )
    seen = []
    inf = Informer(stub, "/api/v1/pods",
                   on_update=lambda t, o: seen.append((t, o["metadata"]["name"])))
    inf.start()
    try:
        assert inf.wait_synced(5)
        assert stub.wait_consumed()
        names = sorted(p["metadata"]["name"] for p in inf.items())
        assert names == ["a", "c"]
        a = next(p for p in inf.items() if p["metadata"]["name"] == "a")
        assert a["status"]["phase"] == "Running"
        assert ("DELETED", "b") in seen and ("ADDED", "c") in seen
    finally:
        inf.stop()


def test_informer_relists_on_410():
    stub = StubClient(
        initial=[pod("a")],
        batches=[KubeError(410, "Gone")],
    )
    inf = Informer(stub, "/api/v1/pods")
    inf.start()
    try:
        assert inf.wait_synced(5)
        deadline = time.time() + 5
        while stub.lists < 2 and time.time() < deadline:
            time.sleep(0.02)
        assert stub.lists >= 2, "410 did not trigger a relist"
    finally:
        inf.stop()


def test_scheduler_reads_informer_cache():
    """With informers set, schedule_once never calls list_pods/list_nodes."""
    from tests.test_scheduler import make_node, make_pod, topo_labels

    kube = FakeKubeClient(nodes=[], pods=[])

    def boom(*a, **k):
        raise AssertionError("scheduler hit the API instead of the cache")
    kube.list_pods = boom
    kube.list_nodes = boom

    class CacheStub:
        def __init__(self, objs):
            self.objs = objs

        def items(self):
            return self.objs

    node = make_node("n1", topo_labels("b", "s", "h"))
    gated = make_pod("j1-0", idx=0)
    kube.pods = {("default", "j1-0"): gated}  # replace_pod target
    sched = d.TopologyScheduler(
        kube, gate_cooloff_s=0,
        pod_informer=CacheStub([gated]),
        node_informer=CacheStub([node]),
    )
    assert sched.schedule_once() == 1
    assert not d.has_topology_gate(kube.get_pod("default", "j1-0"))


def test_informer_over_fake_kube_client():
    """Informer end-to-end over FakeKubeClient's list/watch surface."""
    kube = FakeKubeClient(pods=[pod("seed")])
    inf = Informer(kube, "/api/v1/pods")
    inf.start()
    try:
        assert inf.wait_synced(5)
        assert [p["metadata"]["name"] for p in inf.items()] == ["seed"]
        kube.push_watch_event("ADDED", pod("late"))
        deadline = time.time() + 5
        while len(inf) < 2 and time.time() < deadline:
            time.sleep(0.02)
        assert sorted(p["metadata"]["name"] for p in inf.items()) == \
            ["late", "seed"]
        kube.push_watch_event("DELETED", pod("seed"))
        deadline = time.time() + 5
        while len(inf) > 1 and time.time() < deadline:
            time.sleep(0.02)
        assert [p["metadata"]["name"] for p in inf.items()] == ["late"]
    finally:
        inf.stop()
