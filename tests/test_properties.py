"""Property-based tests (hypothesis) for the pure seams: virtual-ID
round-trips, version parsing, topology scoring, partition-table parsing.
These are the functions whose inputs come from external text (amd-smi output,
kubelet requests, driver version strings) — fuzz them."""

from hypothesis import given, settings, strategies as st

from cea_amd.deviceplugin import sharing
from cea_amd.deviceplugin.version_visibility import parse_version, ANNOTATION_PREFIX
from cea_amd.partition.partition_gpu import parse_partition_status
from cea_amd.scheduler import topology as topo


dev_ids = st.one_of(
    st.integers(0, 63).map(lambda i: f"amdgpu{i}"),
    st.tuples(st.integers(0, 7), st.integers(0, 7)).map(
        lambda t: f"amdgpu{t[0]}/xcd{t[1]}"
    ),
)


@given(dev_ids, st.integers(0, 255))
def test_virtual_id_round_trip(phys, idx):
    v = sharing.virtual_id(phys, idx)
    assert sharing.is_virtual_id(v)
    assert not sharing.is_virtual_id(phys)
    assert sharing.virtual_to_physical(v) == phys


@given(st.text(max_size=40))
def test_is_virtual_id_never_raises(s):
    sharing.is_virtual_id(s)


@given(st.text(alphabet=st.characters(blacklist_categories=("Cs",)), max_size=40))
def test_parse_version_total(s):
    """parse_version must never raise and always include 'full'."""
    out = parse_version(s)
    assert out[f"{ANNOTATION_PREFIX}.full"] == s
    if f"{ANNOTATION_PREFIX}.major" in out:
        assert out[f"{ANNOTATION_PREFIX}.major"].isdigit()


@given(st.text(max_size=2000))
def test_parse_partition_status_never_raises(s):
    out = parse_partition_status(s)
    for d in out:
        assert "accelerator_partition" in d


@given(
    st.lists(
        st.tuples(st.integers(0, 3), st.integers(0, 2), st.integers(0, 4)),
        min_size=1,
        max_size=10,
    )
)
@settings(max_examples=50)
def test_pairwise_score_matches_bruteforce_property(keys):
    topos = [topo.TopoKey(f"b{a}", f"s{b}", f"h{c}") for a, b, c in keys]
    brute = sum(
        topo.distance(topos[i], topos[j])
        for i in range(len(topos))
        for j in range(i + 1, len(topos))
    )
    assert topo._pairwise_score(topos) == brute


@given(
    st.integers(1, 30),
    st.lists(
        st.tuples(st.integers(0, 5), st.integers(0, 4)), min_size=1, max_size=20
    ),
)
@settings(max_examples=50)
def test_assign_pods_fits_or_none(num_pods, node_spec):
    nodes = [
        topo.CandidateNode(f"n{i}", topo.TopoKey(f"b{b}", "s", f"h{i}"), cap)
        for i, (b, cap) in enumerate(node_spec)
    ]
    got = topo.assign_pods(num_pods, nodes)
    total_cap = sum(c for _, c in node_spec)
    if total_cap < num_pods:
        assert got is None
    else:
        assert got is not None and len(got) == num_pods
        # no node exceeds its capacity
        from collections import Counter
        counts = Counter(got)
        caps = {f"n{i}": c for i, (_, c) in enumerate(node_spec)}
        for name, used in counts.items():
            assert used <= caps[name], (name, used)


@given(
    st.lists(st.tuples(st.integers(0, 3), st.integers(0, 7)),
             min_size=0, max_size=24, unique=True),
    st.integers(0, 2),
    st.integers(0, 12),
)
@settings(max_examples=80)
def test_preferred_allocation_invariants(avail_keys, n_must, size):
    from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig

    mgr = AmdGPUManager(GPUConfig())
    available = [f"amdgpu{d}/xcd{x}" for d, x in avail_keys]
    must = available[:n_must]
    got = mgr.preferred_allocation(available, must, size)
    # no duplicates, bounded by size, drawn from the known ids
    assert len(got) == len(set(got))
    assert len(got) == min(size, len(available))
    assert set(got) <= set(available)
    # must_include honored up to size
    for m in must[:size]:
        assert m in got


@given(
    st.integers(1, 5),
    st.lists(
        st.tuples(st.integers(0, 2), st.integers(0, 1), st.integers(0, 2),
                  st.integers(1, 3)),
        min_size=1,
        max_size=6,
    ),
)
@settings(max_examples=60, deadline=None)
def test_assign_pods_matches_exhaustive_optimum(num_pods, node_spec):
    """The contiguous-window placement achieves the same minimal summed
    pairwise distance as exhaustive search over ALL slot combinations —
    the optimality claim in topology.py verified end-to-end on assign_pods
    itself (previously only the scoring function was
    brute-force-checked)."""
    import itertools

    nodes = [
        topo.CandidateNode(f"n{i}", topo.TopoKey(f"b{b}", f"s{s}", f"h{h}"),
                           cap)
        for i, (b, s, h, cap) in enumerate(node_spec)
    ]
    slots = []
    for n in nodes:
        slots.extend([n.topo] * n.capacity)
    got = topo.assign_pods(num_pods, nodes)
    if len(slots) < num_pods:
        assert got is None
        return
    assert got is not None and len(got) == num_pods
    topo_by_name = {n.name: n.topo for n in nodes}
    got_score = topo._pairwise_score([topo_by_name[name] for name in got])
    brute = min(
        topo._pairwise_score([slots[i] for i in combo])
        for combo in itertools.combinations(range(len(slots)), num_pods)
    )
    assert got_score == brute, (got_score, brute, node_spec, num_pods)


@given(
    st.lists(
        st.tuples(st.integers(1, 4), st.booleans()),  # (gpu req, wants label)
        min_size=1, max_size=5,
    ),
    st.lists(
        st.tuples(st.integers(0, 2), st.integers(2, 8), st.booleans()),
        min_size=1, max_size=5,
    ),
)
@settings(max_examples=60, deadline=None)
def test_assign_pods_hetero_respects_resources_and_selectors(pod_spec,
                                                             node_spec):
    """Heterogeneous placement never oversubscribes a node and never
    violates a pod's nodeSelector."""
    from collections import defaultdict

    pods = [
        {"requests": {"amd.com/gpu": float(g)},
         "node_selector": ({"pool": "special"} if sel else {})}
        for g, sel in pod_spec
    ]
    nodes = [
        topo.HeteroNode(
            name=f"n{i}",
            topo=topo.TopoKey(f"b{b}", "s", f"h{i}"),
            free={"amd.com/gpu": float(cap)},
            labels=({"pool": "special"} if lab else {}),
        )
        for i, (b, cap, lab) in enumerate(node_spec)
    ]
    got = topo.assign_pods_hetero(pods, nodes)
    if got is None:
        return
    assert len(got) == len(pods)
    by_name = {n.name: n for n in nodes}
    used = defaultdict(float)
    for pod, name in zip(pods, got):
        node = by_name[name]
        assert topo.selector_matches(pod["node_selector"], node.labels)
        used[name] += pod["requests"]["amd.com/gpu"]
    for name, total in used.items():
        assert total <= by_name[name].free["amd.com/gpu"]


@given(st.text(max_size=64))
@settings(max_examples=200, deadline=None)
def test_parse_quantity_total(q):
    """parse_quantity never raises (fuzz-found: bare 'm' crashed it —
    a malformed quantity in one pod spec must never abort a scheduling
    pass)."""
    from cea_amd.scheduler.daemon import parse_quantity

    v = parse_quantity(q)
    assert isinstance(v, float) and v >= 0.0 or v < 0.0  # just: a float


@given(st.text(max_size=512))
@settings(max_examples=200, deadline=None)
def test_partition_parsers_total(text):
    """amd-smi output parsers never raise on arbitrary text (the job
    must fail with a clear status error, not a traceback, when the CLI
    output format drifts)."""
    from cea_amd.partition import partition_gpu as pg

    assert isinstance(pg.parse_partition_status(text), list)
    assert isinstance(pg.parse_partition_profiles(text), dict)
