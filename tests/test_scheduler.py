"""Topology scheduler tests: distance model, contiguous-window assignment,
gang scheduling end-to-end against FakeKubeClient."""
import pytest

from cea_amd.kube.client import FakeKubeClient
from cea_amd.scheduler import daemon as d
from cea_amd.scheduler import topology as topo
from cea_amd.scheduler.labeler import label_node, parse_physical_host, topology_from_env
from cea_amd.scheduler.topology import CandidateNode, TopoKey, assign_pods


def K(b, s="", h=""):
    return TopoKey(b, s, h)


def test_distance_weights():
    assert topo.distance(K("b1", "s1", "h1"), K("b1", "s1", "h1")) == 0
    assert topo.distance(K("b1", "s1", "h1"), K("b1", "s1", "h2")) == 100
    assert topo.distance(K("b1", "s1", "h1"), K("b1", "s2", "h2")) == 10100
    assert topo.distance(K("b1", "s1", "h1"), K("b2", "s2", "h2")) == 1010100


def test_assign_prefers_same_host_block():
    nodes = [
        CandidateNode("far", K("b2", "s1", "h9"), 1),
        CandidateNode("n1", K("b1", "s1", "h1"), 1),
        CandidateNode("n2", K("b1", "s1", "h1"), 1),
        CandidateNode("n3", K("b1", "s2", "h2"), 1),
    ]
    out = assign_pods(2, nodes)
    assert sorted(out) == ["n1", "n2"]
    # 3 pods: must include n3 (same block) over `far` (other block)
    out = assign_pods(3, nodes)
    assert sorted(out) == ["n1", "n2", "n3"]


def test_assign_packs_one_node_when_possible():
    # a single node with spare capacity beats spanning: distance 0
    nodes = [
        CandidateNode("big", K("b2", "s1", "h9"), 8),
        CandidateNode("n1", K("b1", "s1", "h1"), 1),
        CandidateNode("n2", K("b1", "s1", "h2"), 1),
    ]
    assert assign_pods(3, nodes) == ["big"] * 3


def test_assign_uses_capacity_and_fails_when_too_big():
    nodes = [CandidateNode("n1", K("b1"), 4)]
    assert assign_pods(4, nodes) == ["n1"] * 4
    assert assign_pods(5, nodes) is None
    assert assign_pods(0, nodes) is None


def test_quantity_parsing():
    assert d.parse_quantity("500m") == 0.5
    assert d.parse_quantity("2") == 2
    assert d.parse_quantity("1Gi") == 2**30
    assert d.parse_quantity("8") == 8
    assert d.parse_quantity(4) == 4


def make_node(name, labels=None, gpus=8, cpu="64", mem="512Gi", taints=None):
    return {
        "metadata": {"name": name, "labels": labels or {}},
        "spec": {"taints": taints or []},
        "status": {
            "conditions": [{"type": "Ready", "status": "True"}],
            "allocatable": {"cpu": cpu, "memory": mem, "amd.com/gpu": str(gpus)},
        },
    }


def make_pod(name, ns="default", job="j1", idx=None, gpus=8, gated=True,
             node=None, phase="Pending"):
    labels = {"job-name": job}
    if idx is not None:
        labels["batch.kubernetes.io/job-completion-index"] = str(idx)
    pod = {
        "metadata": {"name": name, "namespace": ns, "labels": labels},
        "spec": {
            "containers": [{
                "name": "main",
                "resources": {"requests": {"amd.com/gpu": str(gpus),
                                           "cpu": "4", "memory": "16Gi"}},
            }],
            "schedulingGates": (
                [{"name": "gke.io/topology-aware-auto-j1"}] if gated else []
            ),
        },
        "status": {"phase": phase},
    }
    if node:
        pod["spec"]["nodeName"] = node
    return pod


def topo_labels(b, s, h):
    return {
        "topology.cea-amd.io/block": b,
        "topology.cea-amd.io/subblock": s,
        "topology.cea-amd.io/host": h,
    }


def test_gang_schedules_to_closest_nodes():
    kube = FakeKubeClient(
        nodes=[
            make_node("nodeA", topo_labels("b1", "s1", "h1")),
            make_node("nodeB", topo_labels("b1", "s1", "h2")),
            make_node("nodeC", topo_labels("b2", "s9", "h9")),
        ],
        pods=[make_pod("j1-0", idx=0), make_pod("j1-1", idx=1)],
    )
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 2
    p0 = kube.get_pod("default", "j1-0")
    p1 = kube.get_pod("default", "j1-1")
    assert p0["spec"]["schedulingGates"] == []
    chosen = {
        p["spec"]["affinity"]["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"
        ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"][0]
        for p in (p0, p1)
    }
    assert chosen == {"nodeA", "nodeB"}  # same subblock, not nodeC


def test_gang_waits_when_not_fitting():
    kube = FakeKubeClient(
        nodes=[make_node("nodeA", topo_labels("b1", "s1", "h1"))],
        pods=[make_pod("j1-0", idx=0), make_pod("j1-1", idx=1)],
    )
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 0  # 2x8 GPUs don't fit on one 8-GPU node
    assert d.has_topology_gate(kube.get_pod("default", "j1-0"))


def test_running_pods_consume_capacity():
    kube = FakeKubeClient(
        nodes=[
            make_node("nodeA", topo_labels("b1", "s1", "h1")),
            make_node("nodeB", topo_labels("b1", "s1", "h2")),
        ],
        pods=[
            make_pod("busy-0", job="other", gated=False, node="nodeA",
                     phase="Running"),
            make_pod("j1-0", idx=0),
        ],
    )
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 1
    p = kube.get_pod("default", "j1-0")
    vals = p["spec"]["affinity"]["nodeAffinity"][
        "requiredDuringSchedulingIgnoredDuringExecution"
    ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"]
    assert vals == ["nodeB"]  # nodeA's GPUs are taken by busy-0


def test_taints_respected():
    kube = FakeKubeClient(
        nodes=[
            make_node("tainted", topo_labels("b1", "s1", "h1"),
                      taints=[{"key": "dedicated", "value": "x",
                               "effect": "NoSchedule"}]),
            make_node("open", topo_labels("b1", "s1", "h2")),
        ],
        pods=[make_pod("j1-0", idx=0)],
    )
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 1
    p = kube.get_pod("default", "j1-0")
    vals = p["spec"]["affinity"]["nodeAffinity"][
        "requiredDuringSchedulingIgnoredDuringExecution"
    ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"]
    assert vals == ["open"]


def test_completion_index_order():
    assert d.completion_index(make_pod("x", idx=7)) == ("", 7)
    assert d.completion_index(make_pod("worker-12")) == ("worker-", 12)
    # kubeflow replica-index label honored (parity schedule-daemon.py:373-377)
    kf = make_pod("kf")
    kf["metadata"]["labels"][d.KUBEFLOW_REPLICA_INDEX_LABEL] = "3"
    assert d.completion_index(kf) == ("", 3)
    # natural numeric ordering: pod2 before pod10
    pods = [make_pod("w-pod10"), make_pod("w-pod2")]
    pods.sort(key=d.completion_index)
    assert [p["metadata"]["name"] for p in pods] == ["w-pod2", "w-pod10"]


def test_job_grouping():
    p1 = make_pod("a", job="jobX")
    p2 = make_pod("b", job="jobX")
    p3 = {"metadata": {"name": "c", "namespace": "default",
                       "ownerReferences": [{"uid": "u1"}]}, "spec": {}}
    assert d.job_key(p1) == d.job_key(p2)
    assert d.job_key(p3) == "default/u1"


def test_labeler():
    assert parse_physical_host("/blk/sb/h") == {
        "block": "blk", "subblock": "sb", "host": "h"}
    assert parse_physical_host("") is None
    assert topology_from_env({"TOPOLOGY_BLOCK": "b", "TOPOLOGY_SUBBLOCK": "s",
                              "TOPOLOGY_HOST": "h"}) == {
        "block": "b", "subblock": "s", "host": "h"}
    kube = FakeKubeClient(nodes=[{"metadata": {"name": "n1", "labels": {}},
                                  "status": {}}])
    label_node(kube, "n1", {"block": "b", "subblock": "s", "host": "h"})
    assert kube.nodes["n1"]["metadata"]["labels"][
        "topology.cea-amd.io/block"] == "b"


def test_assign_scales_to_large_clusters():
    """Control-plane perf: a 512-pod gang over 2048 nodes must place in
    seconds (the reference's exhaustive combinatorial search,
    schedule-daemon.py:500-544, cannot) and must still pick one packed
    block when one exists."""
    import time as _time

    nodes = []
    # 256 blocks x 8 hosts, 1 pod capacity each; block 77 has 64 hosts
    # with capacity 8 = the only single-block fit for 512 pods.
    for b in range(256):
        for h in range(8):
            nodes.append(CandidateNode(
                f"n-{b}-{h}", K(f"b{b:03d}", "s0", f"h{h}"), 1))
    for h in range(64):
        nodes.append(CandidateNode(
            f"big-{h}", K("b077x", "s0", f"H{h}"), 8))
    t0 = _time.perf_counter()
    got = assign_pods(512, nodes)
    elapsed = _time.perf_counter() - t0
    assert got is not None and len(got) == 512
    assert elapsed < 10.0, f"placement took {elapsed:.1f}s"
    assert all(n.startswith("big-") for n in got), \
        "did not pick the packed block"


def test_pairwise_score_matches_bruteforce():
    import random
    rng = random.Random(7)
    for _ in range(20):
        topos = [K(f"b{rng.randint(0, 3)}", f"s{rng.randint(0, 2)}",
                   f"h{rng.randint(0, 4)}") for _ in range(rng.randint(1, 12))]
        brute = sum(topo.distance(topos[i], topos[j])
                    for i in range(len(topos))
                    for j in range(i + 1, len(topos)))
        assert topo._pairwise_score(topos) == brute


def test_node_selector_feasibility():
    """A pod's spec.nodeSelector restricts candidate nodes (parity
    schedule-daemon.py:421-436) — even when a non-matching node is
    topologically closer."""
    kube = FakeKubeClient(nodes=[
        make_node("plain1", topo_labels("b1", "s1", "h1")),
        make_node("plain2", topo_labels("b1", "s1", "h1")),
        make_node("labeled", {**topo_labels("b9", "s9", "h9"),
                              "pool": "mi355x"}),
    ])
    p = make_pod("sel-0", idx=0)
    p["spec"]["nodeSelector"] = {"pool": "mi355x"}
    kube.pods = {("default", "sel-0"): p}
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 1
    bound = kube.get_pod("default", "sel-0")
    vals = bound["spec"]["affinity"]["nodeAffinity"][
        "requiredDuringSchedulingIgnoredDuringExecution"
    ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"]
    assert vals == ["labeled"]


def test_heterogeneous_gang_per_pod_resources():
    """Pods of one gang with different GPU requests are placed with their
    OWN requests (job_pods[0] was used for every pod).  A
    6-GPU and a 2-GPU pod pack into one 8-GPU node — something the
    homogeneous model could never see."""
    kube = FakeKubeClient(nodes=[
        make_node("packed", topo_labels("b1", "s1", "h1"), gpus=8),
        make_node("spare", topo_labels("b2", "s2", "h2"), gpus=8),
    ])
    big = make_pod("het-0", idx=0, gpus=6)
    small = make_pod("het-1", idx=1, gpus=2)
    kube.pods = {("default", "het-0"): big, ("default", "het-1"): small}
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 2
    for name in ("het-0", "het-1"):
        bound = kube.get_pod("default", name)
        vals = bound["spec"]["affinity"]["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"
        ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"]
        assert vals == ["packed"], (name, vals)


def test_heterogeneous_gang_does_not_oversubscribe():
    """Two 6-GPU pods cannot share an 8-GPU node; gang spans two nodes."""
    kube = FakeKubeClient(nodes=[
        make_node("n1", topo_labels("b1", "s1", "h1"), gpus=8),
        make_node("n2", topo_labels("b1", "s1", "h2"), gpus=8),
    ])
    a = make_pod("two-0", idx=0, gpus=6)
    b = make_pod("two-1", idx=1, gpus=6)
    # make requests differ slightly so the hetero path is taken
    b["spec"]["containers"][0]["resources"]["requests"]["cpu"] = "3"
    kube.pods = {("default", "two-0"): a, ("default", "two-1"): b}
    sched = d.TopologyScheduler(kube, gate_cooloff_s=0)
    assert sched.schedule_once() == 2
    bound_nodes = set()
    for name in ("two-0", "two-1"):
        bound = kube.get_pod("default", name)
        vals = bound["spec"]["affinity"]["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"
        ]["nodeSelectorTerms"][0]["matchExpressions"][0]["values"]
        bound_nodes.add(vals[0])
    assert bound_nodes == {"n1", "n2"}


def test_run_forever_warmup_delays_first_pass():
    """run_forever waits warmup_s before the first pass (parity
    schedule-daemon.py:777-807: racing a deployment's pod creation can
    split a gang)."""
    import threading
    import time as _t

    kube = FakeKubeClient(nodes=[make_node("n1", topo_labels("b", "s", "h"))])
    kube.pods = {("default", "w-0"): make_pod("w-0", idx=0)}
    sched = d.TopologyScheduler(kube, interval_s=0.05, gate_cooloff_s=0)
    t = threading.Thread(
        target=sched.run_forever, kwargs={"warmup_s": 0.5, "settle_s": 0},
        daemon=True)
    t.start()
    _t.sleep(0.2)
    # still in warmup: nothing bound yet
    assert d.has_topology_gate(kube.get_pod("default", "w-0"))
    deadline = _t.time() + 5
    while _t.time() < deadline and d.has_topology_gate(
            kube.get_pod("default", "w-0")):
        _t.sleep(0.05)
    sched._stop = True
    assert not d.has_topology_gate(kube.get_pod("default", "w-0"))
